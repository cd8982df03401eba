"""hipGraph-captured decode execution (SURVEY.md §7 stage 3d).

Decode steps on MI355X are launch-bound in eager mode (~300 kernel launches
per 8B step; profiles/r01: ~3.8 ms/step of gaps). This runner:

  * keeps per-SLOT persistent device block tables (a sequence keeps one slot
    for its whole life; appends are one small index_put per step, not a
    full rebuild);
  * captures the decode forward as a hipGraph per batch-size bucket
    (torch.cuda.CUDAGraph == hipGraph on ROCm) over static input buffers,
    sharing one memory pool across buckets;
  * per step: one pinned H2D copy of the [B] scalars, one index_select to
    assemble the bucket's block-table rows, one graph replay.

Padding rows (B < bucket) run a 1-token attention against a live slot's
first block (read-only) and write their dummy KV to the reserved scratch
block 0 (BlockManager never hands out block 0).
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch

from .config import EngineConfig
from .models.batch import ForwardBatch
from .scheduler import Sequence

_BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 160, 192, 224, 256,
            320, 384, 448, 512]


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

        # persistent per-slot block tables
        self.bt_slots = torch.zeros((self.max_slots, self.maxb),
                                    dtype=torch.int32, device=device)
        # static graph inputs
        self.in_tokens = torch.zeros(bmax, dtype=torch.int64, device=device)
        self.in_positions = torch.zeros(bmax, dtype=torch.int32, device=device)
        self.in_slot_mapping = torch.zeros(bmax, dtype=torch.int32,
                                           device=device)
        self.in_ctx = torch.ones(bmax, dtype=torch.int32, device=device)
        self.in_block_table = torch.zeros((bmax, self.maxb),
                                          dtype=torch.int32, device=device)
        # pinned staging: [4, bmax] int64 rows: tokens, positions, slots, ctx
        self.staging = torch.zeros((4, bmax), dtype=torch.int64,
                                   pin_memory=True)
        self.slot_idx_staging = torch.zeros(bmax, dtype=torch.int64,
                                            pin_memory=True)

        self._graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self._hidden: Dict[int, torch.Tensor] = {}
        self._pool = None

    # ---- block-table maintenance ---------------------------------------
    def write_block_row(self, slot: int, block_ids: List[int],
                        start: int = 0) -> None:
        if start >= len(block_ids):
            return
        vals = torch.tensor(block_ids[start:], dtype=torch.int32)
        self.bt_slots[slot, start:len(block_ids)] = vals.to(
            self.device, non_blocking=True)

    # ---- capture --------------------------------------------------------
    def _capture(self, B: int) -> None:
        fb = ForwardBatch(
            token_ids=self.in_tokens[:B],
            positions=self.in_positions[:B],
            slot_mapping=self.in_slot_mapping[:B],
            is_prefill=False,
            block_table=self.in_block_table[:B],
            ctx_lens=self.in_ctx[:B],
        )
        # warmup on a side stream (records cuBLAS/hipBLASLt workspaces etc.)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                self.model(fb, self.kv_caches)
        torch.cuda.current_stream().wait_stream(side)

        if self._pool is None:
            self._pool = torch.cuda.graph_pool_handle()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=self._pool):
            hidden = self.model(fb, self.kv_caches)
        self._graphs[B] = g
        self._hidden[B] = hidden

    def _bucket_for(self, B: int) -> int:
        for b in self.buckets:
            if b >= B:
                return b
        return self.buckets[-1]

    # ---- execute --------------------------------------------------------
    def execute(self, seqs: List[Sequence]) -> torch.Tensor:
        """One decode step for seqs; returns hidden states [B, hidden]."""
        B = len(seqs)
        bucket = self._bucket_for(B)
        bs = self.cfg.block_size

        st = self.staging
        pad_slot = seqs[0].slot
        npad = bucket - B
        tokens = [(s.output_token_ids[-1] if s.output_token_ids
                   else s.prompt_token_ids[-1]) for s in seqs] + [0] * npad
        positions = [s.num_tokens - 1 for s in seqs] + [0] * npad
        # padding rows write their dummy KV into reserved scratch block 0
        slot_map = [s.block_ids[(s.num_tokens - 1) // bs] * bs
                    + (s.num_tokens - 1) % bs for s in seqs] + [0] * npad
        ctx = [s.num_tokens for s in seqs] + [1] * npad
        slots = [s.slot for s in seqs] + [pad_slot] * npad
        st[:, :bucket] = torch.tensor(
            [tokens, positions, slot_map, ctx], dtype=torch.int64)
        self.slot_idx_staging[:bucket] = torch.tensor(slots,
                                                      dtype=torch.int64)

        dev_st = st[:, :bucket].to(self.device, non_blocking=True)
        self.in_tokens[:bucket].copy_(dev_st[0])
        self.in_positions[:bucket].copy_(dev_st[1].to(torch.int32))
        self.in_slot_mapping[:bucket].copy_(dev_st[2].to(torch.int32))
        self.in_ctx[:bucket].copy_(dev_st[3].to(torch.int32))
        slot_idx = self.slot_idx_staging[:bucket].to(self.device,
                                                     non_blocking=True)
        torch.index_select(self.bt_slots, 0, slot_idx,
                           out=self.in_block_table[:bucket])

        if bucket not in self._graphs:
            self._capture(bucket)
        self._graphs[bucket].replay()
        return self._hidden[bucket][:B]
