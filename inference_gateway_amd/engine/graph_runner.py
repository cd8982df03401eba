"""hipGraph-captured decode execution (SURVEY.md §7 stage 3d).

Decode steps on MI355X are launch-bound in eager mode (~300 kernel launches
per 8B step; profiles/r01: ~3.8 ms/step of gaps). v2 moves ALL per-step
input preparation onto the device and inside the captured graph:

  * persistent per-SLOT device state — last sampled token, position,
    block-table row, sampling params — written once at admission and then
    advanced on-device each step (one scatter + one index_add after
    sampling), never rebuilt from Python lists;
  * the captured graph gathers {tokens, positions, slot_mapping,
    block_table, ctx_lens} from slot state by a per-bucket active-slot
    index buffer, runs the forward AND the lm_head, and leaves fp32
    logits in a static buffer (torch.cuda.CUDAGraph == hipGraph on ROCm);
  * per step the host does: replay, one fused-sampling kernel, two tiny
    state-advance ops, one async D2H of the sampled tokens — no per-seq
    Python work until the bookkeeping loop.

The active-slot index buffer is refreshed only when batch membership
changes (admission / finish / preemption); in steady-state decode the
host's per-step device interaction is O(1) kernel launches.

Padding rows (B < bucket) point at a dedicated pad slot whose position is
0 and whose block-table row is all zeros, so they run a 1-token attention
against reserved scratch block 0 and write their dummy KV there
(BlockManager never hands out block 0).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from .. import ops
from .config import EngineConfig
from .models.batch import ForwardBatch
from .sampler import apply_penalties, apply_seeded_uniforms
from .scheduler import Sequence

_BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 160, 192, 224, 256,
            320, 384, 448, 512, 640, 768, 896, 1024]


class DecodeGraphRunner:
    def __init__(self, model, kv_caches, cfg: EngineConfig, device,
                 max_blocks_per_seq: int):
        self.model = model
        self.kv_caches = kv_caches
        self.cfg = cfg
        self.device = device
        self.max_slots = cfg.max_num_seqs
        self.maxb = max_blocks_per_seq
        self.buckets = [b for b in _BUCKETS if b <= cfg.max_num_seqs]
        if self.buckets[-1] != cfg.max_num_seqs:
            self.buckets.append(cfg.max_num_seqs)
        bmax = self.buckets[-1]
        S = self.max_slots + 1          # +1: dedicated pad slot (last row)
        self.pad_slot = S - 1

        # ---- persistent per-slot device state ----
        self.slot_token = torch.zeros(S, dtype=torch.int64, device=device)
        self.slot_pos = torch.zeros(S, dtype=torch.int32, device=device)
        self.bt_slots = torch.zeros((S, self.maxb), dtype=torch.int32,
                                    device=device)
        self.slot_temp = torch.ones(S, dtype=torch.float32, device=device)
        self.slot_topk = torch.zeros(S, dtype=torch.int32, device=device)
        self.slot_topp = torch.ones(S, dtype=torch.float32, device=device)

        # ---- per-bucket static graph inputs ----
        self.in_slot_idx = torch.full((bmax,), self.pad_slot,
                                      dtype=torch.int64, device=device)
        self._ones_i32 = torch.ones(bmax, dtype=torch.int32, device=device)

        # pinned host staging for state writes / token readback.  Two
        # buffers + events: the engine runs one decode step ahead of the
        # host bookkeeping (async pipeline), so step N's D2H lands while
        # step N+1 executes.
        self.tokens_pinned = torch.zeros((2, bmax), dtype=torch.int32,
                                         pin_memory=True)
        self._events = [torch.cuda.Event(), torch.cuda.Event()]
        self._ring = 0
        # logprobs readback (lazy — only requests with logprobs pay):
        # pinned ring buffers + per-ring (rows, kmax, per-row k) metadata
        self._lp_meta: List = [None, None]
        self.lp_pinned = None
        self.lp_topv_pinned = None
        self.lp_topi_pinned = None
        self._graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self._logits: Dict[int, torch.Tensor] = {}
        self._pool = None
        self._cur_bucket: Optional[int] = None
        self._cur_B = 0
        self._gen = torch.Generator(device=device)
        self._gen.manual_seed(cfg.seed)

    # ---- slot-state maintenance (admission / growth) --------------------
    def write_block_row(self, slot: int, block_ids: List[int],
                        start: int = 0) -> None:
        if start >= len(block_ids):
            return
        vals = torch.tensor(block_ids[start:], dtype=torch.int32)
        self.bt_slots[slot, start:len(block_ids)] = vals.to(
            self.device, non_blocking=True)

    def write_block_appends(self, items: List) -> None:
        """Batched single-block appends: items = [(slot, col, block_id)]."""
        if not items:
            return
        t = torch.tensor(items, dtype=torch.int64).to(self.device,
                                                      non_blocking=True)
        self.bt_slots[t[:, 0], t[:, 1]] = t[:, 2].to(torch.int32)

    def init_slot(self, seq: Sequence, first_token: int) -> None:
        """After prefill: seed the slot's decode state (token/pos/params)."""
        s = seq.slot
        sp = seq.sampling
        vals = torch.tensor(
            [float(first_token), float(seq.num_tokens - 1),
             sp.temperature, float(sp.top_k), sp.top_p],
            dtype=torch.float64).to(self.device, non_blocking=True)
        self.slot_token[s] = vals[0].to(torch.int64)
        self.slot_pos[s] = vals[1].to(torch.int32)
        self.slot_temp[s] = vals[2].to(torch.float32)
        self.slot_topk[s] = vals[3].to(torch.int32)
        self.slot_topp[s] = vals[4].to(torch.float32)

    def set_batch(self, seqs: List[Sequence]) -> int:
        """Refresh the active-slot index buffer; returns the bucket size."""
        B = len(seqs)
        bucket = self._bucket_for(B)
        slots = [s.slot for s in seqs] + [self.pad_slot] * (bucket - B)
        idx = torch.tensor(slots, dtype=torch.int64).to(self.device,
                                                        non_blocking=True)
        self.in_slot_idx[:bucket].copy_(idx)
        self._cur_bucket = bucket
        self._cur_B = B
        return bucket

    # ---- capture --------------------------------------------------------
    def _build_fb(self, bucket: int):
        idx = self.in_slot_idx[:bucket]
        bs = self.cfg.block_size
        tokens = self.slot_token.index_select(0, idx)
        pos = self.slot_pos.index_select(0, idx)
        bt = self.bt_slots.index_select(0, idx)
        posl = pos.to(torch.int64)
        blk = bt.gather(1, (posl // bs).unsqueeze(1)).squeeze(1)
        slot_map = (blk * bs + (pos % bs)).to(torch.int32)
        fb = ForwardBatch(
            token_ids=tokens,
            positions=pos,
            slot_mapping=slot_map,
            is_prefill=False,
            block_table=bt,
            ctx_lens=pos + 1,
        )
        return fb

    def _forward(self, bucket: int) -> torch.Tensor:
        fb = self._build_fb(bucket)
        hidden = self.model(fb, self.kv_caches)
        return self.model.compute_logits(hidden)

    def _capture(self, bucket: int) -> None:
        # NOTE: warmup/capture forwards run against LIVE slot state, so
        # they write K/V for every active slot at its current position —
        # idempotent by construction (the replay immediately rewrites the
        # same values at the same slot_mapping), so a mid-serving capture
        # of a new bucket is safe.
        # warmup on a side stream (records hipBLASLt workspaces etc.)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                self._forward(bucket)
        torch.cuda.current_stream().wait_stream(side)

        if self._pool is None:
            self._pool = torch.cuda.graph_pool_handle()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=self._pool):
            logits = self._forward(bucket)
        self._graphs[bucket] = g
        self._logits[bucket] = logits

    def _bucket_for(self, B: int) -> int:
        for b in self.buckets:
            if b >= B:
                return b
        return self.buckets[-1]

    # ---- execute --------------------------------------------------------
    def execute(self, seqs: List[Sequence], membership_changed: bool):
        """One decode step; advances slot state (token, pos) on-device,
        queues an async D2H of the sampled tokens, and returns
        (ring_index, event) — the engine drains the tokens later (one-step
        -ahead pipeline), reading self.tokens_pinned[ring_index]."""
        B = len(seqs)
        if membership_changed or self._cur_B != B:
            bucket = self.set_batch(seqs)
        else:
            bucket = self._cur_bucket
        if bucket not in self._graphs:
            self._capture(bucket)
        self._graphs[bucket].replay()

        logits = self._logits[bucket]
        idx = self.in_slot_idx[:bucket]
        apply_penalties(logits, seqs,
                        last_tokens=self.slot_token.index_select(0, idx[:B]))
        temps = self.slot_temp.index_select(0, idx)
        topk = self.slot_topk.index_select(0, idx)
        topp = self.slot_topp.index_select(0, idx)
        u = torch.rand(bucket, generator=self._gen,
                       device=self.device).clamp_(1e-6, 1 - 1e-6)
        apply_seeded_uniforms(u, seqs)
        toks = ops.sample(logits, temps, topk, topp, u)     # [bucket] int32
        act = idx[:B]
        self.slot_token.scatter_(0, act, toks[:B].to(torch.int64))
        self.slot_pos.index_add_(0, act, self._ones_i32[:B])
        ring = self._ring
        self._ring ^= 1
        self._queue_logprobs(ring, seqs, logits, toks)
        self.tokens_pinned[ring, :B].copy_(toks[:B], non_blocking=True)
        self._events[ring].record()
        return ring, self._events[ring]

    # ---- logprobs (opt-in per sequence; the hot path pays nothing) ------
    _LP_KMAX = 20        # OpenAI's top_logprobs cap

    def _queue_logprobs(self, ring: int, seqs, logits, toks) -> None:
        """For rows whose sequence asked for logprobs, compute them from
        this step's logits (still valid: the next replay of this bucket is
        stream-ordered after these ops) and queue an async D2H alongside
        the token readback; read_logprobs(ring) decodes after the event."""
        rows = [i for i, s in enumerate(seqs) if s.sampling.logprobs >= 0]
        self._lp_meta[ring] = None
        if not rows:
            return
        if self.lp_pinned is None:
            bmax = self.tokens_pinned.shape[1]
            self.lp_pinned = torch.zeros((2, bmax), dtype=torch.float32,
                                         pin_memory=True)
            self.lp_topv_pinned = torch.zeros((2, bmax, self._LP_KMAX),
                                              dtype=torch.float32,
                                              pin_memory=True)
            self.lp_topi_pinned = torch.zeros((2, bmax, self._LP_KMAX),
                                              dtype=torch.int32,
                                              pin_memory=True)
        n = len(rows)
        rows_t = torch.tensor(rows, dtype=torch.int64).to(self.device,
                                                          non_blocking=True)
        lsm = logits.index_select(0, rows_t).float().log_softmax(-1)
        samp = lsm.gather(1, toks.index_select(0, rows_t).to(
            torch.int64).unsqueeze(1)).squeeze(1)
        ks = [min(seqs[i].sampling.logprobs, self._LP_KMAX) for i in rows]
        kmax = max(ks)
        if kmax > 0:
            top_v, top_i = lsm.topk(kmax, dim=-1)
            self.lp_topv_pinned[ring, :n, :kmax].copy_(top_v,
                                                       non_blocking=True)
            self.lp_topi_pinned[ring, :n, :kmax].copy_(
                top_i.to(torch.int32), non_blocking=True)
        self.lp_pinned[ring, :n].copy_(samp, non_blocking=True)
        self._lp_meta[ring] = (rows, kmax, ks)

    def read_logprobs(self, ring: int):
        """{batch_row: (sampled_logprob, [(token_id, logprob) x k])} for
        the drained step, or None.  Call only after the ring's event."""
        meta = self._lp_meta[ring]
        if meta is None:
            return None
        self._lp_meta[ring] = None
        rows, kmax, ks = meta
        n = len(rows)
        samp = self.lp_pinned[ring, :n].tolist()
        if kmax > 0:
            tv = self.lp_topv_pinned[ring, :n, :kmax].tolist()
            ti = self.lp_topi_pinned[ring, :n, :kmax].tolist()
        out = {}
        for j, i in enumerate(rows):
            alts = (list(zip(ti[j][: ks[j]], tv[j][: ks[j]]))
                    if kmax > 0 and ks[j] > 0 else [])
            out[i] = (samp[j], alts)
        return out
