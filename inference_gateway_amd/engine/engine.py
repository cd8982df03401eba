"""LLMEngine: the synchronous continuous-batching inference core.

Net-new scope (SURVEY.md §7 stages 3-5): this is the 'local provider' the
gateway calls in-process instead of the reference's self-proxy HTTP hop
(SURVEY.md §3.2 — the rebuild short-circuits it by design). One engine per
GPU process; TP ranks run this same loop SPMD with RCCL collectives inside
the model and identical sampling RNG, so no cross-rank token broadcast is
needed on the hot path.
"""
from __future__ import annotations

import itertools
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from .config import EngineConfig, SamplingParams
from .kvcache import BlockManager
from .models.batch import ForwardBatch
from .models.llama import LlamaForCausalLM
from .sampler import Sampler, apply_penalties
from .scheduler import ScheduledBatch, Scheduler, SeqState, Sequence


@dataclass
class StepOutput:
    seq_id: int
    token_id: int
    finished: bool
    finish_reason: Optional[str] = None
    # set when SamplingParams.logprobs >= 0: log p(sampled token) and the
    # top-k alternatives as [(token_id, logprob), ...] (k may be 0)
    logprob: Optional[float] = None
    top_logprobs: Optional[list] = None


@dataclass
class RequestOutput:
    seq_id: int
    token_ids: List[int] = field(default_factory=list)
    finished: bool = False
    finish_reason: Optional[str] = None


MAX_TOP_LOGPROBS = 20      # OpenAI's cap on top_logprobs


def _extract_logprobs(logits, seqs, tokens):
    """For every batch row whose sequence requested logprobs, compute
    (log p(sampled token), [(token_id, logprob) x k]) from this step's
    logits.  Returns {row_index: (...)} or None when nobody asked — the
    hot path pays nothing."""
    rows = [i for i, s in enumerate(seqs) if s.sampling.logprobs >= 0]
    if not rows:
        return None
    import torch as _t

    dev = logits.device
    rows_t = _t.tensor(rows, dtype=_t.int64, device=dev)
    lsm = logits.index_select(0, rows_t).float().log_softmax(-1)
    toks = _t.tensor([tokens[i] for i in rows], dtype=_t.int64, device=dev)
    samp = lsm.gather(1, toks.unsqueeze(1)).squeeze(1)
    kmax = min(MAX_TOP_LOGPROBS,
               max(seqs[i].sampling.logprobs for i in rows))
    if kmax > 0:
        top_v, top_i = lsm.topk(kmax, dim=-1)
        top_v, top_i = top_v.tolist(), top_i.tolist()
    samp = samp.tolist()
    out = {}
    for n, i in enumerate(rows):
        k = min(seqs[i].sampling.logprobs, MAX_TOP_LOGPROBS)
        alts = (list(zip(top_i[n][:k], top_v[n][:k])) if kmax > 0 and k > 0
                else [])
        out[i] = (samp[n], alts)
    return out


def _maybe_load_tuned_gemms() -> None:
    """Load the committed TunableOp solution table for the decode/prefill
    GEMM shapes (gfx950), if present.  Tensile's default heuristics pick
    poor tilings for the skinny-M decode GEMMs (profiles/r01: 1.7 TB/s
    aggregate on 13.9 GB of weights/step); the tuned table is produced by
    a one-off PYTORCH_TUNABLEOP_TUNING=1 bench run.  IGW_TUNABLEOP=0
    opts out."""
    import os
    from pathlib import Path

    if os.environ.get("IGW_TUNABLEOP", "1") == "0":
        return
    if not torch.cuda.is_available():
        return
    if os.environ.get("PYTORCH_TUNABLEOP_TUNING", "0") == "1":
        return  # a tuning run manages TunableOp itself via env
    csv = Path(__file__).resolve().parent.parent / "data" / \
        "tunableop_gfx950.csv"
    if not csv.exists():
        return
    import torch.cuda.tunable as tunable

    tunable.enable(True)
    tunable.tuning_enable(False)
    tunable.read_file(str(csv))


def auto_num_blocks(model_cfg, block_size: int, tp: int,
                    free_bytes: int, utilization: float) -> int:
    """KV pool sizing from free HBM AFTER weights (the caller measures
    free memory post-model-init).  Per-rank KV bytes shrink with TP (KV
    heads are sharded); a flat 2 GiB is held back for graph pools,
    activations and allocator slack.  Factored out of LLMEngine so the
    70B TP=8 budget is testable without materializing 70B params
    (tests/test_parallel_cpu.py::test_70b_tp8_memory_budget)."""
    bytes_per_block = (
        model_cfg.kv_bytes_per_token() * block_size // tp)
    budget = int(free_bytes * utilization) - (2 << 30)
    return max(64, budget // bytes_per_block)


class LLMEngine:
    def __init__(self, cfg: EngineConfig, device: str = "cuda:0"):
        self.cfg = cfg
        self.device = torch.device(
            device if torch.cuda.is_available() else "cpu")
        _maybe_load_tuned_gemms()
        self.dtype = torch.bfloat16
        if self.device.type == "cuda" and cfg.model.head_dim != 128:
            raise ValueError(
                f"model head_dim {cfg.model.head_dim} is not supported "
                "on GPU: the MFMA attention kernels are tiled for "
                "head_dim=128 (CPU reference path has no such limit)")
        if cfg.model.is_moe:
            from .models.mixtral import MixtralForCausalLM

            self.model = MixtralForCausalLM(cfg.model, self.device, self.dtype)
        else:
            self.model = LlamaForCausalLM(cfg.model, self.device, self.dtype)
        self.model.eval()
        if cfg.weights_path:
            from .weights import load_weights

            load_weights(self.model, cfg.weights_path)

        num_blocks = cfg.num_gpu_blocks or self._auto_num_blocks()
        self.block_manager = BlockManager(
            num_blocks, cfg.block_size,
            enable_prefix_caching=cfg.enable_prefix_caching)
        self.scheduler = Scheduler(
            self.block_manager, cfg.max_num_seqs, cfg.max_batched_tokens,
            cfg.max_model_len,
            chunked_prefill=cfg.enable_chunked_prefill,
            prefill_first=cfg.scheduler_prefill_first,
            mixed_batches=cfg.enable_mixed_batches)
        self.sampler = Sampler(self.device, cfg.seed)
        self.kv_caches = self._allocate_kv(num_blocks)
        self._seq_counter = itertools.count()
        self._seqs: Dict[int, Sequence] = {}
        from collections import deque

        self._finished_order: "deque[int]" = deque()
        self._fork_pending: Dict[int, List[Sequence]] = {}
        # cumulative serving counters (Prometheus gauges read these)
        self.num_generated_tokens = 0
        self.num_prompt_tokens = 0
        self.num_forks = 0
        self.num_steps = 0        # scheduled engine steps (bench windows)
        self._no_progress = 0     # consecutive unschedulable rounds
        # +1 block of headroom: the async decode pipeline can write KV for
        # one in-flight token past max_model_len before the stop check runs
        self.max_blocks_per_seq = (
            cfg.max_model_len + cfg.block_size - 1) // cfg.block_size + 1
        self.graph_runner = None
        self._gr_version = -1
        self._pending = None   # (seqs, ring, event) of the in-flight decode
        if (self.device.type == "cuda" and cfg.use_hip_graphs
                and not cfg.enforce_eager):
            from .graph_runner import DecodeGraphRunner

            self.graph_runner = DecodeGraphRunner(
                self.model, self.kv_caches, cfg, self.device,
                self.max_blocks_per_seq)
            if cfg.capture_at_init:
                with torch.inference_mode():
                    for b in self.graph_runner.buckets:
                        self.graph_runner._capture(b)

    # ---------------- setup ----------------
    def _auto_num_blocks(self) -> int:
        from ..parallel.comm import tp_world_size

        if self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
        else:
            free = (1 << 30) + (2 << 30)
        return auto_num_blocks(self.cfg.model, self.cfg.block_size,
                               tp_world_size(), free,
                               self.cfg.gpu_memory_utilization)

    def _allocate_kv(self, num_blocks: int):
        from ..parallel.comm import tp_world_size

        m = self.cfg.model
        nkv = max(1, m.num_kv_heads // tp_world_size())
        shape = (num_blocks, nkv, self.cfg.block_size, m.head_dim)
        caches = []
        for _ in range(m.num_layers):
            k = torch.zeros(shape, dtype=self.dtype, device=self.device)
            v = torch.zeros(shape, dtype=self.dtype, device=self.device)
            caches.append((k, v))
        return caches

    # ---------------- request API ----------------
    def add_request(self, prompt_token_ids: List[int],
                    sampling: Optional[SamplingParams] = None,
                    seq_id: Optional[int] = None) -> int:
        if seq_id is None:
            seq_id = next(self._seq_counter)
        bs = self.cfg.block_size
        need = (len(prompt_token_ids) + bs - 1) // bs + 1
        if need > self.block_manager.num_blocks - 1:
            # never-schedulable: reject per-request instead of letting
            # the scheduler discover it as a no-progress spin
            raise ValueError(
                f"prompt needs {need} KV blocks but the pool has only "
                f"{self.block_manager.num_blocks - 1} — raise "
                "num_gpu_blocks / gpu_memory_utilization")
        seq = Sequence(
            seq_id=seq_id,
            prompt_token_ids=list(prompt_token_ids),
            sampling=sampling or SamplingParams(),
        )
        self._seqs[seq_id] = seq
        self.scheduler.add(seq)
        return seq_id

    def add_request_group(self, prompt_token_ids: List[int],
                          sampling: Optional[SamplingParams] = None,
                          n: int = 1) -> List[int]:
        """OpenAI n>1: ONE prefill for the whole group.  The first
        sequence prefills normally; the other n-1 are materialized at its
        prefill completion by forking the prompt KV (full blocks shared
        ref-counted, partial tail block copied) — see _fork_at_prefill."""
        sampling = sampling or SamplingParams()
        parent_id = self.add_request(prompt_token_ids, sampling)
        kids = []
        for _ in range(n - 1):
            sid = next(self._seq_counter)
            child = Sequence(seq_id=sid,
                             prompt_token_ids=list(prompt_token_ids),
                             sampling=sampling)
            self._seqs[sid] = child
            kids.append(child)
        if kids:
            self._fork_pending[parent_id] = kids
        return [parent_id] + [c.seq_id for c in kids]

    def abort_request(self, seq_id: int) -> None:
        seq = self._seqs.get(seq_id)
        if seq is not None and seq.state != SeqState.FINISHED:
            self.scheduler.abort(seq)
            # a parent aborted before its fork point: finish the children
            # too so their consumers see a sentinel instead of hanging
            for child in self._fork_pending.pop(seq_id, []):
                if child.state == SeqState.WAITING:
                    self.scheduler.finish(child, "abort")
        self._note_finished(seq_id)

    _FINISHED_RETENTION = 4096

    def _note_finished(self, seq_id: int) -> None:
        """Bounded retention of finished Sequence bookkeeping: without
        eviction a long-running server retains every prompt/output
        token list ever served.  get_output stays valid for the last
        _FINISHED_RETENTION finished requests (sync callers read right
        after finishing; the async path never calls get_output)."""
        self._finished_order.append(seq_id)
        while len(self._finished_order) > self._FINISHED_RETENTION:
            old = self._finished_order.popleft()
            seq = self._seqs.get(old)
            if seq is not None and seq.state == SeqState.FINISHED:
                del self._seqs[old]

    def is_finished(self) -> bool:
        return not self.scheduler.has_work and self._pending is None

    def get_output(self, seq_id: int) -> RequestOutput:
        seq = self._seqs[seq_id]
        return RequestOutput(
            seq_id=seq_id,
            token_ids=seq.all_generated_ids,
            finished=seq.state == SeqState.FINISHED,
            finish_reason=seq.finish_reason,
        )

    # ---------------- step ----------------
    @torch.inference_mode()
    def step(self) -> List[StepOutput]:
        """One engine step.  Graph-captured decode runs an async pipeline:
        the host schedules and submits step N, THEN drains step N-1's
        sampled tokens (D2H overlapped with step N's GPU work), so outputs
        lag decode by one step.  A sequence whose stop condition is met at
        the drain gets one in-flight step's token discarded — the standard
        async-scheduling trade."""
        batch = self.scheduler.schedule()
        if batch is None:
            # nothing schedulable.  If requests are WAITING with nothing
            # running, no future step can free blocks for them: a prompt
            # (or a preempted sequence's folded prompt) larger than the
            # whole pool would otherwise busy-spin here silently forever.
            # Count consecutive no-progress rounds and fail loudly (the
            # async engine surfaces this as EngineDeadError to clients).
            if self.scheduler.waiting and not self.scheduler.running:
                self._no_progress += 1
                if self._no_progress > 4096:
                    seq = self.scheduler.waiting[0]
                    raise RuntimeError(
                        f"scheduler made no progress for "
                        f"{self._no_progress} rounds: seq "
                        f"{seq.seq_id} needs {seq.prompt_len} prompt "
                        "tokens but the KV pool cannot ever hold them "
                        "— raise num_gpu_blocks or lower max_model_len")
            return self._drain_pending()
        self._no_progress = 0
        self.num_steps += 1
        gr = self.graph_runner
        if not batch.is_prefill and gr is not None:
            gr.write_block_appends(batch.new_blocks)
            ver = self.scheduler.membership_version
            try:
                ring, ev = gr.execute(
                    batch.seqs, membership_changed=ver != self._gr_version)
            except RuntimeError as e:
                # hipGraph capture can fail when a collective cannot be
                # captured on this ROCm build (TP without the one-shot
                # kernel: RCCL-in-graph is version-fragile).  Permanent
                # eager fallback — correctness first, the step still
                # runs below (eager-TP as automatic
                # fallback).
                import warnings

                warnings.warn(
                    f"decode graph capture failed ({e}); falling back "
                    "to eager decode for the rest of this process")
                self.graph_runner = None
                gr = None
            else:
                self._gr_version = ver
                prev = self._drain_pending()  # overlaps this step's GPU
                for seq in batch.seqs:
                    seq.in_flight = True
                self._pending = (list(batch.seqs), ring, ev)
                return prev

        prev = self._drain_pending()
        if batch.is_prefill:
            self.num_prompt_tokens += sum(batch.prefill_takes)
        dec_seqs: List[Sequence] = []
        if batch.is_mixed:
            # the drain above may have finished/preempted decode seqs
            # scheduled into this mixed round — drop them now
            dec_seqs = [s for s in batch.decode_seqs
                        if s.state == SeqState.RUNNING]
            batch.decode_seqs = dec_seqs
            if self.graph_runner is not None:
                # keep the graph's block-table rows in sync for the
                # next pure-decode round
                self.graph_runner.write_block_appends(batch.new_blocks)
        fb = self._build_forward_batch(batch)
        hidden = self.model(fb, self.kv_caches)
        if batch.is_prefill:
            for seq, take in zip(batch.seqs, batch.prefill_takes):
                seq.prefilled += take
            # chunked prefill: only sequences whose prompt COMPLETED this
            # round sample a token; a partial chunk just extends KV.
            # Mixed rounds additionally sample every decode row (their
            # logit rows follow the prefill rows — _build_forward_batch)
            sample_seqs = [s for s in batch.seqs
                           if s.prefilled >= s.prompt_len] + dec_seqs
        else:
            sample_seqs = batch.seqs
        if not sample_seqs:
            return prev                  # pure partial-chunk round
        hidden_last = hidden[fb.last_token_indices]
        logits = self.model.compute_logits(hidden_last)
        apply_penalties(logits, sample_seqs)
        tokens = self.sampler.sample(logits, sample_seqs)
        lp_map = _extract_logprobs(logits, sample_seqs, tokens)
        # fork n>1 groups BEFORE parent bookkeeping (a parent finishing at
        # its first token would free the blocks the fork shares)
        fork_outputs: List[StepOutput] = []
        if batch.is_prefill and self._fork_pending:
            for row, seq in enumerate(sample_seqs):
                if seq.seq_id in self._fork_pending:
                    fork_outputs.extend(
                        self._fork_at_prefill(seq, logits[row]))
        outputs = self._append_tokens(sample_seqs, tokens, lp_map)
        outputs = fork_outputs + outputs
        if batch.is_prefill and gr is not None:
            # seed the persistent decode slot state for newly admitted seqs
            for seq, tok in zip(sample_seqs, tokens):
                if seq.state != SeqState.FINISHED:
                    gr.write_block_row(seq.slot, seq.block_ids)
                    gr.init_slot(seq, tok)
        return prev + outputs

    def _drain_pending(self) -> List[StepOutput]:
        if self._pending is None:
            return []
        seqs, ring, ev = self._pending
        self._pending = None
        ev.synchronize()
        tokens = self.graph_runner.tokens_pinned[ring, : len(seqs)].tolist()
        lp_map = self.graph_runner.read_logprobs(ring)
        outputs = []
        for i, (seq, tok) in enumerate(zip(seqs, tokens)):
            seq.in_flight = False
            if seq.state == SeqState.FINISHED:
                continue  # stop applied while this token was in flight
            if seq.state == SeqState.WAITING:
                # preempted after this token was sampled: keep it as part of
                # the rebuilt prompt so the request resumes transparently
                seq.prompt_token_ids.append(tok)
                continue
            outputs.extend(self._apply_token(
                seq, tok, lp_map.get(i) if lp_map else None))
        return outputs

    def _build_forward_batch(self, batch: ScheduledBatch) -> ForwardBatch:
        dev = self.device
        bs = self.cfg.block_size
        if batch.is_prefill:
            # vectorized assembly: python-per-token loops cost ~30 ms on an
            # 8k-token burst, which lands straight on TTFT.  With prefix
            # caching, a sequence contributes only its SUFFIX rows (the
            # cached prefix is block-aligned, so suffix KV writes land in
            # private blocks) and attention takes the paged-extend path.
            token_ids, pos_parts, slot_parts = [], [], []
            cu, seq_lens, last, prefixes = [0], [], [], []
            for seq, take in zip(batch.seqs, batch.prefill_takes):
                L = seq.prompt_len
                p0 = seq.prefilled        # cache hits + earlier chunks
                p1 = p0 + take
                token_ids.extend(seq.prompt_token_ids[p0:p1])
                pos_parts.append(torch.arange(p0, p1, dtype=torch.int32))
                blocks = torch.tensor(seq.block_ids, dtype=torch.int32)
                slots = (blocks.repeat_interleave(bs) * bs
                         + torch.arange(bs, dtype=torch.int32).repeat(
                             len(seq.block_ids)))[p0:p1]
                slot_parts.append(slots)
                cu.append(cu[-1] + take)
                seq_lens.append(take)
                if p1 >= L:               # completes the prompt: sample
                    last.append(cu[-1] - 1)
                prefixes.append(p0)
            P = cu[-1]
            dseqs = batch.decode_seqs if batch.is_mixed else None
            dec_bt = dec_ctx = None
            if dseqs:
                # mixed round: append one decode row per running seq
                # after the prefill rows; GEMMs run over the union,
                # attention splits at row P (models/llama.py)
                for s in dseqs:
                    token_ids.append(s.output_token_ids[-1]
                                     if s.output_token_ids
                                     else s.prompt_token_ids[-1])
                dpos, dslots = [], []
                maxb = max(len(s.block_ids) for s in dseqs)
                dec_bt = torch.zeros((len(dseqs), maxb), dtype=torch.int32)
                dec_ctx = torch.tensor([s.num_tokens for s in dseqs],
                                       dtype=torch.int32)
                for i, s in enumerate(dseqs):
                    pos = s.num_tokens - 1
                    dpos.append(pos)
                    blk = s.block_ids[pos // bs]
                    dslots.append(blk * bs + pos % bs)
                    dec_bt[i, : len(s.block_ids)] = torch.tensor(
                        s.block_ids, dtype=torch.int32)
                pos_parts.append(torch.tensor(dpos, dtype=torch.int32))
                slot_parts.append(torch.tensor(dslots, dtype=torch.int32))
                last.extend(range(P, P + len(dseqs)))
            fb = ForwardBatch(
                token_ids=torch.tensor(token_ids, dtype=torch.int64, device=dev),
                positions=torch.cat(pos_parts).to(dev),
                slot_mapping=torch.cat(slot_parts).to(dev),
                is_prefill=True,
                cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
                seq_lens=seq_lens,
                last_token_indices=torch.tensor(last, dtype=torch.int64,
                                                device=dev),
            )
            if dseqs:
                fb.num_prefill_tokens = P
                fb.decode_block_table = dec_bt.to(dev)
                fb.decode_ctx_lens = dec_ctx.to(dev)
            if any(prefixes):
                maxb = max(len(s.block_ids) for s in batch.seqs)
                bt = torch.zeros((len(batch.seqs), maxb), dtype=torch.int32)
                for i, s in enumerate(batch.seqs):
                    bt[i, : len(s.block_ids)] = torch.tensor(
                        s.block_ids, dtype=torch.int32)
                fb.block_table = bt.to(dev)
                fb.prefix_lens = torch.tensor(prefixes, dtype=torch.int32,
                                              device=dev)
            return fb
        # decode
        B = len(batch.seqs)
        token_ids = [s.output_token_ids[-1] if s.output_token_ids
                     else s.prompt_token_ids[-1] for s in batch.seqs]
        positions = [s.num_tokens - 1 for s in batch.seqs]
        slots = []
        maxb = max(len(s.block_ids) for s in batch.seqs)
        bt = torch.zeros((B, maxb), dtype=torch.int32)
        ctx = torch.tensor([s.num_tokens for s in batch.seqs],
                           dtype=torch.int32)
        for i, s in enumerate(batch.seqs):
            pos = s.num_tokens - 1
            blk = s.block_ids[pos // bs]
            slots.append(blk * bs + pos % bs)
            bt[i, : len(s.block_ids)] = torch.tensor(s.block_ids,
                                                     dtype=torch.int32)
        return ForwardBatch(
            token_ids=torch.tensor(token_ids, dtype=torch.int64, device=dev),
            positions=torch.tensor(positions, dtype=torch.int32, device=dev),
            slot_mapping=torch.tensor(slots, dtype=torch.int32, device=dev),
            is_prefill=False,
            block_table=bt.to(dev),
            ctx_lens=ctx.to(dev),
            last_token_indices=torch.arange(B, dtype=torch.int64, device=dev),
        )

    def _copy_block(self, src: int, dst: int) -> None:
        for k, v in self.kv_caches:
            k[dst].copy_(k[src])
            v[dst].copy_(v[src])

    def _fork_at_prefill(self, parent, row_logits) -> List[StepOutput]:
        """Materialize a group's n-1 branches from the parent's freshly
        prefilled KV: full prompt blocks are SHARED (ref-counted, decode
        never writes them again), the partially filled tail block is
        copied per branch, and each branch samples its own first token
        from the parent's prefill logits.  If slots or blocks run short a
        branch falls back to an ordinary re-prefilling request."""
        kids = [c for c in self._fork_pending.pop(parent.seq_id, [])
                if c.state == SeqState.WAITING]
        if not kids:
            return []
        bs = self.cfg.block_size
        plen = parent.prompt_len
        full, partial = plen // bs, plen % bs
        rep = row_logits.unsqueeze(0).repeat(len(kids), 1)
        toks = self.sampler.sample(rep, kids)
        lp_map = _extract_logprobs(rep, kids, toks)
        gr = self.graph_runner
        outs: List[StepOutput] = []
        for j, child in enumerate(kids):
            tail: List[int] = []
            ok = self.scheduler.can_adopt()
            if ok and partial:
                try:
                    tail = self.block_manager.allocate(1)
                except RuntimeError:
                    ok = False
            if not ok:
                # not enough slots/blocks right now: plain re-prefill
                # (the sampled first token is discarded — the child's own
                # prefill will sample one, so give back the seed tick the
                # discarded draw consumed or seeded branches would
                # diverge under pressure)
                if child.sampling.seed is not None:
                    child.sample_tick = max(0, child.sample_tick - 1)
                self.scheduler.add(child)
                continue
            if tail:
                self._copy_block(parent.block_ids[full], tail[0])
            child.block_ids = (self.block_manager.share(parent.block_ids[:full]) + tail)
            # the child's prompt KV EXISTS (shared/copied from the
            # parent): mark it prefilled, or Scheduler.finish would
            # un-index the shared blocks as unwritten (the mid-chunk
            # abort guard) and kill prefix-cache reuse
            child.prefilled = plen
            self.scheduler.adopt(child)
            self.num_forks += 1
            if gr is not None:
                gr.write_block_row(child.slot, child.block_ids)
            outs.extend(self._apply_token(
                child, toks[j], lp_map.get(j) if lp_map else None))
            if child.state != SeqState.FINISHED and gr is not None:
                gr.init_slot(child, toks[j])
        return outs

    def _apply_token(self, seq, tok: int, lp=None) -> List[StepOutput]:
        import time

        self.num_generated_tokens += 1
        seq.output_token_ids.append(tok)
        if seq.first_token_time is None:
            seq.first_token_time = time.monotonic()
        reason = None
        sp = seq.sampling
        if not sp.ignore_eos and tok in sp.stop_token_ids:
            reason = "stop"
        elif seq.num_generated >= sp.max_new_tokens:
            reason = "length"
        elif seq.num_tokens >= self.cfg.max_model_len:
            reason = "length"
        if reason:
            self.scheduler.finish(seq, reason)
            self._note_finished(seq.seq_id)
        out = StepOutput(seq.seq_id, tok, reason is not None, reason)
        if lp is not None:
            out.logprob, out.top_logprobs = lp
        return [out]

    def _append_tokens(self, seqs: List[Sequence], tokens: List[int],
                       lp_map=None) -> List[StepOutput]:
        outputs = []
        for i, (seq, tok) in enumerate(zip(seqs, tokens)):
            outputs.extend(self._apply_token(
                seq, tok, lp_map.get(i) if lp_map else None))
        return outputs
