"""Paged KV cache block manager.

Net-new engine scope (SURVEY.md §5 'long context': the rebuild handles long
context with a paged KV cache sized for the MI355X's 288 GB HBM3E). Blocks
are 16 tokens; the physical cache layout is
[num_blocks, n_kv_heads, block_size, head_dim] per layer (see
ops/csrc/rope_cache.hip for why).
"""
from __future__ import annotations

from typing import List


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int):
        self.num_blocks = num_blocks
        self.block_size = block_size
        # block 0 is reserved as the scratch target for hipGraph padding
        # rows (graph_runner.py) and is never handed out
        self._free: List[int] = list(range(num_blocks - 1, 0, -1))

    @property
    def num_free(self) -> int:
        return len(self._free)

    def can_allocate(self, num_tokens: int) -> bool:
        need = (num_tokens + self.block_size - 1) // self.block_size
        return need <= len(self._free)

    def allocate(self, num_tokens: int) -> List[int]:
        need = (num_tokens + self.block_size - 1) // self.block_size
        if need > len(self._free):
            raise RuntimeError("out of KV cache blocks")
        return [self._free.pop() for _ in range(need)]

    def ensure_capacity(self, block_ids: List[int], num_tokens: int) -> List[int]:
        """Grow the block list until it can hold num_tokens KV slots."""
        while len(block_ids) * self.block_size < num_tokens:
            if not self._free:
                raise RuntimeError("out of KV cache blocks")
            block_ids.append(self._free.pop())
        return block_ids

    def free(self, block_ids: List[int]) -> None:
        self._free.extend(reversed(block_ids))
        block_ids.clear()
