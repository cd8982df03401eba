"""Continuous-batching scheduler.

Net-new engine scope (SURVEY.md §7 stage 3c): admission queue + running set
with prefill-priority interleave and LIFO preemption when the paged cache
runs dry. Sized for 288 GB HBM per GPU — on a real MI355X the block pool is
large enough that preemption is the exception, not the steady state.
"""
from __future__ import annotations

import enum
import time
from collections import deque
from dataclasses import dataclass, field
from typing import List, Optional

from .config import SamplingParams
from .kvcache import BlockManager


class SeqState(enum.Enum):
    WAITING = "waiting"
    RUNNING = "running"
    FINISHED = "finished"


@dataclass
class Sequence:
    seq_id: int
    prompt_token_ids: List[int]
    sampling: SamplingParams
    arrival_time: float = field(default_factory=time.monotonic)
    output_token_ids: List[int] = field(default_factory=list)
    block_ids: List[int] = field(default_factory=list)
    state: SeqState = SeqState.WAITING
    finish_reason: Optional[str] = None
    first_token_time: Optional[float] = None
    slot: Optional[int] = None        # persistent batch slot (graph runner)
    in_flight: bool = False           # a decode step's token not yet drained
    sample_tick: int = 0              # per-seq counter for seeded sampling
    cached_prefix: int = 0            # prompt tokens served from the
                                      # prefix cache (block-aligned)
    prefilled: int = 0                # prompt tokens whose KV is already
                                      # computed (cache hits + finished
                                      # prefill chunks)
    # length of the ORIGINAL prompt: preemption folds generated tokens
    # into prompt_token_ids for the re-prefill, so generation length and
    # output reporting must be measured against this, not prompt_len
    orig_prompt_len: int = -1

    def __post_init__(self) -> None:
        self.orig_prompt_len = len(self.prompt_token_ids)

    @property
    def num_generated(self) -> int:
        return (len(self.prompt_token_ids) + len(self.output_token_ids)
                - self.orig_prompt_len)

    @property
    def all_generated_ids(self) -> List[int]:
        return (self.prompt_token_ids
                + self.output_token_ids)[self.orig_prompt_len:]

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_token_ids) + len(self.output_token_ids)

    @property
    def prompt_len(self) -> int:
        return len(self.prompt_token_ids)


@dataclass
class ScheduledBatch:
    is_prefill: bool
    seqs: List[Sequence]
    # decode-only: single-block growths this step, as (slot, col, block_id)
    # rows for DecodeGraphRunner.write_block_appends
    new_blocks: List[tuple] = field(default_factory=list)
    # prefill-only: tokens each seq processes THIS round (chunked
    # prefill: the last entry may be a partial chunk; None = whole
    # remaining prompt for every seq)
    prefill_takes: Optional[List[int]] = None
    # MIXED round: `seqs`/`prefill_takes` are the prefill chunk and
    # decode_seqs each contribute one decode token to the SAME forward
    # (engine runs it eagerly; pure-decode rounds keep the hipGraph)
    decode_seqs: Optional[List[Sequence]] = None

    @property
    def is_mixed(self) -> bool:
        return bool(self.decode_seqs)


class Scheduler:
    def __init__(self, block_manager: BlockManager, max_num_seqs: int,
                 max_batched_tokens: int, max_model_len: int,
                 chunked_prefill: bool = False,
                 prefill_first: bool = False,
                 mixed_batches: bool = False):
        self.bm = block_manager
        self.max_num_seqs = max_num_seqs
        self.max_batched_tokens = max_batched_tokens
        self.max_model_len = max_model_len
        self.waiting: deque[Sequence] = deque()
        self.running: List[Sequence] = []
        self._free_slots: List[int] = list(range(max_num_seqs - 1, -1, -1))
        # bumped whenever the running set changes (admit/finish/preempt) so
        # the graph runner knows when to refresh its active-slot buffer
        self.membership_version = 0
        # prefill/decode interleave: when both prompts and running decodes
        # are pending, alternate rounds so a burst of new prompts does not
        # starve in-flight streams (and in-flight streams do not starve
        # TTFT) — SURVEY.md §7 'scheduler fairness under 64 SSE clients'
        self._last_was_prefill = False
        self.chunked_prefill = chunked_prefill
        # prefill_first: strict admission priority (vLLM-style) — always
        # prefill when prompts are waiting, instead of alternating with
        # decode rounds.  Under a connect burst the alternate policy
        # generates huge decode volume for early streams while late
        # prompts queue (measured r2: a 1024-stream burst spent 41 s
        # ramping and 584k decode tokens before the last admission;
        # prefill-first ramps in ~prefill-compute time).  Decode ITL
        # spikes during admission in exchange — chunked prefill bounds
        # that spike.
        self.prefill_first = prefill_first
        # mixed_batches: when prompts and running decodes are BOTH
        # pending, emit ONE batch carrying a (chunked) prefill segment
        # plus every running sequence's decode token — no whole-round
        # ITL bubble from a prefill burst, and the union raises the
        # skinny decode GEMMs' M for free.  The engine runs mixed
        # rounds eagerly; pure-decode rounds keep the hipGraph.
        self.mixed_batches = mixed_batches
        self.num_mixed_rounds = 0     # cumulative (metrics)
        self.num_preemptions = 0      # cumulative (metrics)

    def add(self, seq: Sequence) -> None:
        if seq.prompt_len > self.max_model_len:
            raise ValueError(
                f"prompt length {seq.prompt_len} exceeds max_model_len "
                f"{self.max_model_len}"
            )
        self.waiting.append(seq)

    @property
    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def schedule(self) -> Optional[ScheduledBatch]:
        # Alternate prefill and decode rounds when both are pending;
        # admit as many waiting prompts as fit the token budget and the
        # block pool.  With mixed_batches the round carries BOTH.
        mixed = (self.mixed_batches and bool(self.waiting)
                 and bool(self.running))
        want_prefill = bool(self.waiting) and (
            mixed or self.prefill_first
            or not (self._last_was_prefill and self.running))
        # a mixed round's token budget is shared with the decode rows
        budget = self.max_batched_tokens - (
            len(self.running) if mixed else 0)
        budget = max(budget, 1)
        chunked = self.chunked_prefill or mixed
        prefills: List[Sequence] = []
        takes: List[int] = []
        tokens = 0
        while (want_prefill and self.waiting
               and len(self.running) + len(prefills) < self.max_num_seqs):
            seq = self.waiting[0]
            remaining = seq.prompt_len - seq.prefilled
            if (tokens + remaining > budget
                    and chunked):
                # CHUNKED: hard token budget — take a partial chunk of
                # the front prompt and leave it at waiting[0]; its whole
                # prompt's blocks were reserved at its first chunk
                take = budget - tokens
                if take <= 0:
                    break
                if seq.prefilled == 0:
                    if not self.bm.can_allocate(seq.prompt_len):
                        break
                    seq.block_ids, seq.cached_prefix = \
                        self.bm.allocate_for_prompt(seq.prompt_token_ids)
                    seq.prefilled = seq.cached_prefix
                remaining = seq.prompt_len - seq.prefilled
                take = min(take, remaining)
                if take <= 0:
                    break
                if take == remaining:
                    # this chunk COMPLETES the prompt: admit now so the
                    # same round samples its first token (a completed
                    # seq must never sit in waiting with take 0)
                    self.waiting.popleft()
                    seq.slot = self._free_slots.pop()
                    seq.state = SeqState.RUNNING
                prefills.append(seq)
                takes.append(take)
                tokens += take
                break
            if tokens + remaining > budget and prefills:
                break
            if seq.prefilled == 0:
                if not self.bm.can_allocate(seq.prompt_len):
                    break
                # a single prompt longer than max_batched_tokens (legal
                # when max_model_len is larger) gets its own oversized
                # batch when chunking is off — breaking unconditionally
                # would starve it forever
                seq.block_ids, seq.cached_prefix = \
                    self.bm.allocate_for_prompt(seq.prompt_token_ids)
                seq.prefilled = seq.cached_prefix
            self.waiting.popleft()
            seq.slot = self._free_slots.pop()
            seq.state = SeqState.RUNNING
            prefills.append(seq)
            takes.append(seq.prompt_len - seq.prefilled)
            tokens += seq.prompt_len - seq.prefilled
        if prefills:
            admitted = [q for q in prefills if q.state == SeqState.RUNNING]
            decode_seqs = None
            if mixed:
                # piggyback every ALREADY-running sequence's decode
                # token onto this round (the just-admitted completers
                # sample via the prefill side); fall back to a plain
                # prefill round if the pool cannot grow every decode row
                # self.running holds only PRIOR seqs here (admitted are
                # appended below)
                prior = list(self.running)
                nb = self._grow_decode_capacity(prior)
                if nb is not None and prior:
                    decode_seqs = prior
                    self.num_mixed_rounds += 1
                    if admitted:
                        self.running.extend(admitted)
                        self.membership_version += 1
                    self._last_was_prefill = True
                    return ScheduledBatch(is_prefill=True, seqs=prefills,
                                          prefill_takes=takes,
                                          decode_seqs=decode_seqs,
                                          new_blocks=nb)
            if admitted:
                self.running.extend(admitted)
                self.membership_version += 1
            self._last_was_prefill = True
            return ScheduledBatch(is_prefill=True, seqs=prefills,
                                  prefill_takes=takes)
        self._last_was_prefill = False

        if not self.running:
            return None

        # Decode: every running sequence generates one token; make room first.
        # LIFO preemption — newest sequence goes back to waiting, whole-prompt
        # recompute (matches the 288GB-HBM design point where this is rare).
        bs = self.bm.block_size
        new_blocks: List[tuple] = []
        while True:
            ok = True
            for seq in self.running:
                # this step writes KV at position num_tokens-1, plus one
                # more when a step's token is still in flight (async
                # pipeline: the host count lags the device position)
                need = seq.num_tokens + (1 if seq.in_flight else 0)
                if need > len(seq.block_ids) * bs:
                    try:
                        self.bm.ensure_capacity(seq.block_ids, need)
                    except RuntimeError:
                        ok = False
                        break
                    new_blocks.append(
                        (seq.slot, len(seq.block_ids) - 1, seq.block_ids[-1]))
            if ok:
                break
            victim = self.running.pop()  # newest arrival
            self.num_preemptions += 1
            self.membership_version += 1
            new_blocks = [nb for nb in new_blocks if nb[0] != victim.slot]
            self.bm.free(victim.block_ids)
            if victim.slot is not None:
                self._free_slots.append(victim.slot)
                victim.slot = None
            # recompute from scratch later, keep generated tokens as part
            # of the new prompt so the request resumes transparently
            victim.prompt_token_ids = (
                victim.prompt_token_ids + victim.output_token_ids
            )
            victim.output_token_ids = []
            victim.prefilled = 0
            victim.cached_prefix = 0
            victim.state = SeqState.WAITING
            self.waiting.appendleft(victim)
            if not self.running:
                return None
        return ScheduledBatch(is_prefill=False, seqs=list(self.running),
                              new_blocks=new_blocks)

    def _grow_decode_capacity(self, seqs) -> Optional[List[tuple]]:
        """Best-effort block growth for a mixed round's decode rows.
        Returns the (slot, col, block) appends, or None when the pool
        cannot cover every row (the caller then runs a plain prefill
        round; LIFO preemption stays a pure-decode-round concern)."""
        bs = self.bm.block_size
        new_blocks: List[tuple] = []
        for seq in seqs:
            need = seq.num_tokens + (1 if seq.in_flight else 0)
            if need > len(seq.block_ids) * bs:
                try:
                    self.bm.ensure_capacity(seq.block_ids, need)
                except RuntimeError:
                    return None
                new_blocks.append(
                    (seq.slot, len(seq.block_ids) - 1, seq.block_ids[-1]))
        return new_blocks

    # ---- forked sequences (OpenAI n>1: one prefill, shared prompt KV) --
    def can_adopt(self) -> bool:
        return bool(self._free_slots) and len(self.running) < self.max_num_seqs

    def adopt(self, seq: Sequence) -> None:
        """Place an already-prefilled forked branch straight into the
        running set (its block table was cloned from the parent)."""
        seq.slot = self._free_slots.pop()
        seq.state = SeqState.RUNNING
        self.running.append(seq)
        self.membership_version += 1

    def finish(self, seq: Sequence, reason: str) -> None:
        seq.state = SeqState.FINISHED
        seq.finish_reason = reason
        if (self.bm.enable_prefix_caching
                and seq.prefilled < seq.prompt_len and seq.block_ids):
            # aborted mid-chunked-prefill: blocks past the written point
            # were hash-indexed at allocation but never got KV — un-index
            # them so the retained pool cannot serve garbage (kvcache.py)
            self.bm.drop_unwritten(seq.block_ids, seq.prefilled)
        self.bm.free(seq.block_ids)
        if seq.slot is not None:
            self._free_slots.append(seq.slot)
            seq.slot = None
        if seq in self.running:
            self.running.remove(seq)
            self.membership_version += 1

    def abort(self, seq: Sequence) -> None:
        if seq.state == SeqState.WAITING:
            try:
                self.waiting.remove(seq)
            except ValueError:
                pass
        self.finish(seq, "abort")
