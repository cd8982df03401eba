"""Paged KV cache block manager.

Net-new engine scope (SURVEY.md §5 'long context': the rebuild handles long
context with a paged KV cache sized for the MI355X's 288 GB HBM3E). Blocks
are 16 tokens; the physical cache layout is
[num_blocks, n_kv_heads, block_size, head_dim] per layer (see
ops/csrc/rope_cache.hip for why).

Prefix caching (opt-in): full PROMPT blocks are indexed by a chained
content hash of their token prefix.  A new prompt whose leading full
blocks hash-match either a LIVE block (shared, ref++) or a RETAINED one
(freed by its last owner but kept in an LRU pool instead of reclaimed)
skips recomputing that prefix — the model extend-prefills only the
suffix through `ops.flash_prefill_paged`.  Correctness hinges on three
invariants: decode only appends PAST a sequence's prompt, so shared full
blocks are never written again; the shared prefix must stay
block-ALIGNED, so the suffix's KV lands in private blocks only; and only
the FRONT of the waiting queue may be mid-chunked-prefill (scheduler.py
takes partial chunks from waiting[0] only), so a later prompt can never
hash-match blocks whose KV a still-chunking sequence has not written
yet.  (Same-round matches are safe: per layer, append_kv for the whole
batch runs before attention reads the cache.)
"""
from __future__ import annotations

import zlib
from collections import OrderedDict
from typing import List, Optional, Tuple


class BlockManager:
    """Blocks are ref-counted so forked sequences (OpenAI n>1) can SHARE
    their prompt's full blocks read-only: decode only ever appends at
    positions past the fork point, which live in each branch's private
    tail blocks, so no copy-on-write is needed for shared blocks."""

    def __init__(self, num_blocks: int, block_size: int,
                 enable_prefix_caching: bool = False):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.enable_prefix_caching = enable_prefix_caching
        # block 0 is reserved as the scratch target for hipGraph padding
        # rows (graph_runner.py) and is never handed out
        self._free: List[int] = list(range(num_blocks - 1, 0, -1))
        self._refs: List[int] = [0] * num_blocks
        # prefix cache state: per-block chained content hash, the
        # hash -> block index (live OR retained), and the LRU pool of
        # RETAINED blocks (ref 0, KV intact, reclaimable)
        self._hash_of: List[Optional[int]] = [None] * num_blocks
        self._by_hash: dict = {}
        self._pool: OrderedDict = OrderedDict()   # block -> None (LRU)
        self.prefix_hit_tokens = 0                 # cumulative (metrics)

    @property
    def num_free(self) -> int:
        return len(self._free) + len(self._pool)

    def can_allocate(self, num_tokens: int) -> bool:
        need = (num_tokens + self.block_size - 1) // self.block_size
        return need <= self.num_free

    # ---- internal ------------------------------------------------------
    def _drop_hash(self, b: int) -> None:
        h = self._hash_of[b]
        if h is not None:
            if self._by_hash.get(h) == b:
                del self._by_hash[h]
            self._hash_of[b] = None

    def _take_block(self) -> int:
        """One fresh block: the free list first, then evict the
        least-recently-retained prefix block."""
        if self._free:
            b = self._free.pop()
        elif self._pool:
            b, _ = self._pool.popitem(last=False)
            self._drop_hash(b)
        else:
            raise RuntimeError("out of KV cache blocks")
        self._refs[b] = 1
        return b

    # ---- allocation ----------------------------------------------------
    def allocate(self, num_tokens: int) -> List[int]:
        need = (num_tokens + self.block_size - 1) // self.block_size
        if need > self.num_free:
            raise RuntimeError("out of KV cache blocks")
        return [self._take_block() for _ in range(need)]

    @staticmethod
    def chain_hashes(prompt_token_ids: List[int], block_size: int
                     ) -> List[tuple]:
        """Identity key per FULL block: (block index, chained crc32 over
        the whole prefix, the block's exact tokens).  The exact last
        chunk + position means a false hit needs a crc collision between
        two prefixes that ALSO agree on their final block — ~2^-32 per
        such pair instead of per any pair (a long-running server sees
        millions of prompts; bare crc chains would eventually collide)."""
        out, h = [], 0
        for i in range(len(prompt_token_ids) // block_size):
            chunk = tuple(
                prompt_token_ids[i * block_size:(i + 1) * block_size])
            h = zlib.crc32(repr(chunk).encode(), h)
            out.append((i, h, chunk))
        return out

    def allocate_for_prompt(self, prompt_token_ids: List[int]
                            ) -> Tuple[List[int], int]:
        """Allocate a prompt's blocks, reusing hash-matched leading full
        blocks (live -> share; retained -> claim).  Returns (block_ids,
        cached_prefix_tokens); the prefix is block-aligned and strictly
        shorter than the prompt (>=1 query token must remain so prefill
        produces logits, and the suffix's KV writes stay private)."""
        plen = len(prompt_token_ids)
        bs = self.block_size
        need = (plen + bs - 1) // bs
        if not self.enable_prefix_caching:
            return self.allocate(plen), 0
        hashes = self.chain_hashes(prompt_token_ids, bs)
        max_shared = min(len(hashes), (plen - 1) // bs)
        blocks: List[int] = []
        for i in range(max_shared):
            b = self._by_hash.get(hashes[i])
            if b is None:
                break
            if self._refs[b] == 0:           # retained: claim from pool
                self._pool.pop(b, None)
                self._refs[b] = 1
            else:                            # live: share
                self._refs[b] += 1
            blocks.append(b)
        prefix = len(blocks) * bs
        self.prefix_hit_tokens += prefix
        try:
            for i in range(len(blocks), need):
                b = self._take_block()
                blocks.append(b)
                if i < len(hashes):          # index fresh FULL blocks
                    self._drop_hash(b)
                    self._hash_of[b] = hashes[i]
                    self._by_hash[hashes[i]] = b
        except RuntimeError:
            # the fresh blocks' KV was never computed: un-index them so
            # the pool can't retain garbage, then release everything
            for b in blocks[prefix // bs:]:
                self._drop_hash(b)
            self.free(blocks)
            raise
        return blocks, prefix

    def drop_unwritten(self, block_ids: List[int], written_tokens: int) -> None:
        """Un-index blocks whose KV was never fully computed.  A sequence
        aborted mid-chunked-prefill has hash-indexed ALL its full prompt
        blocks at allocation time (allocate_for_prompt) but only written KV
        for the first `written_tokens`; freeing it as-is would retain
        garbage-KV blocks in the prefix pool, and a later prompt sharing
        the prefix would silently generate from them.  Blocks below
        written_tokens//block_size are written (cache hits or completed
        chunks) and keep their hashes."""
        for i in range(written_tokens // self.block_size, len(block_ids)):
            self._drop_hash(block_ids[i])

    def ensure_capacity(self, block_ids: List[int], num_tokens: int) -> List[int]:
        """Grow the block list until it can hold num_tokens KV slots."""
        while len(block_ids) * self.block_size < num_tokens:
            block_ids.append(self._take_block())
        return block_ids

    def share(self, block_ids: List[int]) -> List[int]:
        """Take an additional reference on every block; returns a new
        list the caller owns (and must eventually free())."""
        for b in block_ids:
            self._refs[b] += 1
        return list(block_ids)

    def free(self, block_ids: List[int]) -> None:
        for b in block_ids:
            self._refs[b] -= 1
            if self._refs[b] == 0:
                if self._hash_of[b] is not None:
                    self._pool[b] = None      # retain (KV intact)
                    self._pool.move_to_end(b)
                else:
                    self._free.append(b)
        block_ids.clear()
