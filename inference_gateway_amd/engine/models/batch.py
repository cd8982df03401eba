"""Forward-batch metadata shared by all models (net-new engine scope,
SURVEY.md §7 stage 3 — the reference gateway has no compute path)."""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch


@dataclass
class ForwardBatch:
    token_ids: torch.Tensor        # [T] int64
    positions: torch.Tensor        # [T] int32
    slot_mapping: torch.Tensor     # [T] int32
    is_prefill: bool
    # prefill-only
    cu_seqlens: Optional[torch.Tensor] = None   # [B+1] int32
    seq_lens: Optional[List[int]] = None
    # prefix-cached prefill: per-seq cached tokens before the q rows
    # (requires block_table; attention takes the paged-extend path)
    prefix_lens: Optional[torch.Tensor] = None  # [B] int32
    # decode-only
    block_table: Optional[torch.Tensor] = None  # [B, max_blocks] int32
    ctx_lens: Optional[torch.Tensor] = None     # [B] int32
    # indices of each sequence's last token within token_ids
    last_token_indices: Optional[torch.Tensor] = None  # [B] int64
    # MIXED batch (same-step prefill chunk + decode rows): token rows
    # [0, num_prefill_tokens) are the prefill chunk (cu_seqlens/
    # seq_lens/prefix_lens/block_table describe them); rows
    # [num_prefill_tokens, T) are one decode token per running sequence
    # (decode_block_table/decode_ctx_lens describe them).  GEMMs and
    # elementwise ops run over the union; attention splits by row kind.
    num_prefill_tokens: int = 0
    decode_block_table: Optional[torch.Tensor] = None
    decode_ctx_lens: Optional[torch.Tensor] = None

    @property
    def is_mixed(self) -> bool:
        return self.decode_ctx_lens is not None

    @property
    def num_tokens(self) -> int:
        return self.token_ids.shape[0]
