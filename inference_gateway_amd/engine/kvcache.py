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
    """Blocks are ref-counted so forked sequences (OpenAI n>1) can SHARE
    their prompt's full blocks read-only: decode only ever appends at
    positions past the fork point, which live in each branch's private
    tail blocks, so no copy-on-write is needed for shared blocks."""

    def __init__(self, num_blocks: int, block_size: int):
        self.num_blocks = num_blocks
        self.block_size = block_size
        # block 0 is reserved as the scratch target for hipGraph padding
        # rows (graph_runner.py) and is never handed out
        self._free: List[int] = list(range(num_blocks - 1, 0, -1))
        self._refs: List[int] = [0] * num_blocks

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
        out = [self._free.pop() for _ in range(need)]
        for b in out:
            self._refs[b] = 1
        return out

    def ensure_capacity(self, block_ids: List[int], num_tokens: int) -> List[int]:
        """Grow the block list until it can hold num_tokens KV slots."""
        while len(block_ids) * self.block_size < num_tokens:
            if not self._free:
                raise RuntimeError("out of KV cache blocks")
            b = self._free.pop()
            self._refs[b] = 1
            block_ids.append(b)
        return block_ids

    def share(self, block_ids: List[int]) -> List[int]:
        """Take an additional reference on every block; returns a new
        list the caller owns (and must eventually free())."""
        for b in block_ids:
            self._refs[b] += 1
        return list(block_ids)

    def free(self, block_ids: List[int]) -> None:
        dead = []
        for b in block_ids:
            self._refs[b] -= 1
            if self._refs[b] == 0:
                dead.append(b)
        self._free.extend(reversed(dead))
        block_ids.clear()
