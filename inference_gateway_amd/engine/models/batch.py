"""Forward-batch metadata shared by all models."""
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

    @property
    def num_tokens(self) -> int:
        return self.token_ids.shape[0]
