"""Paged KV cache: block pool + prefix-sharing block manager.

This is the MI355X-native replacement for the reference's remote-API
statelessness (SURVEY.md §2.3 "Paged KV-block manager with shared-prefix
refcounting / copy-on-write"): tree branches forked from one node share
their conversation prefix (ref tree.py:59-65 lineage), sibling rollouts and
repeated per-turn prompts reuse prefill, and identical judge prompts (3
judges per trajectory, ref evaluator.py:171-172) prefill once.

Mechanism: fixed-size token blocks; full blocks are content-addressed by a
chain hash (hash of parent-hash + the block's token ids) and refcounted.
  - allocate(prompt): longest-prefix match against the hash table — hit
    blocks are shared (ref++), only the tail is scheduled for prefill;
  - fork(): child shares all full blocks, copy-on-write of the partial
    tail block (the only mutable block);
  - free(): ref--; zero-ref hashed blocks stay cached in an LRU pool and
    are evicted only when allocation runs dry.

Cache layout (per layer): K and V each [num_blocks, n_kv_heads,
block_size, head_dim] — one (block, kv-head) is a contiguous
[block_size, head_dim] tile (16x128 bf16 = 4 KiB), the unit the CDNA4
decode-attention kernel streams through LDS.
"""

from __future__ import annotations

from collections import OrderedDict
from dataclasses import dataclass
from typing import Optional

import torch

DEFAULT_BLOCK_SIZE = 16


def chain_hash(prev_hash: int, tokens: tuple) -> int:
    return hash((prev_hash, tokens))


@dataclass
class Block:
    id: int
    ref_count: int = 0
    hash: Optional[int] = None  # set once full + registered
    token_ids: tuple = ()  # content of a full registered block (for verify)


class BlockManager:
    """Refcounted block allocator with content-addressed prefix cache."""

    def __init__(self, num_blocks: int, block_size: int = DEFAULT_BLOCK_SIZE) -> None:
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.blocks = [Block(i) for i in range(num_blocks)]
        self.free_ids = list(range(num_blocks - 1, -1, -1))  # pop() -> low ids first
        self.hash_table: dict = {}  # hash -> block_id
        self.evictable: OrderedDict = OrderedDict()  # block_id -> None (LRU)
        # stats
        self.cache_hit_tokens = 0
        self.cache_miss_tokens = 0

    # -- allocation core ------------------------------------------------
    def num_free(self) -> int:
        return len(self.free_ids) + len(self.evictable)

    def _pop_free_block(self) -> Block:
        if self.free_ids:
            return self.blocks[self.free_ids.pop()]
        if self.evictable:
            bid, _ = self.evictable.popitem(last=False)  # LRU
            block = self.blocks[bid]
            if block.hash is not None and self.hash_table.get(block.hash) == bid:
                del self.hash_table[block.hash]
            block.hash = None
            block.token_ids = ()
            return block
        raise MemoryError("KV block pool exhausted")

    def _acquire(self, block: Block) -> None:
        if block.ref_count == 0 and block.id in self.evictable:
            del self.evictable[block.id]
        block.ref_count += 1

    def allocate_fresh(self) -> int:
        block = self._pop_free_block()
        block.ref_count = 1
        return block.id

    def free_block(self, block_id: int) -> None:
        block = self.blocks[block_id]
        assert block.ref_count > 0, "double free"
        block.ref_count -= 1
        if block.ref_count == 0:
            if block.hash is not None:
                self.evictable[block.id] = None  # keep cached, evict LRU
            else:
                self.free_ids.append(block.id)

    # -- prefix cache ----------------------------------------------------
    def match_prefix(self, tokens: list) -> tuple:
        """Longest cached prefix of `tokens` in full-block units.

        Returns (block_ids, n_matched_tokens); matched blocks are acquired.
        """
        matched: list = []
        prev = 0
        n = 0
        for start in range(0, len(tokens) - self.block_size + 1, self.block_size):
            chunk = tuple(tokens[start : start + self.block_size])
            h = chain_hash(prev, chunk)
            bid = self.hash_table.get(h)
            if bid is None or self.blocks[bid].token_ids != chunk:
                break
            self._acquire(self.blocks[bid])
            matched.append(bid)
            prev = h
            n += self.block_size
        return matched, n

    def register_full_block(self, block_id: int, prev_hash: int, tokens: tuple) -> int:
        """Content-register a block that just became full; returns its hash."""
        block = self.blocks[block_id]
        h = chain_hash(prev_hash, tokens)
        block.hash = h
        block.token_ids = tokens
        # first writer wins; a duplicate block stays usable, just unshared
        self.hash_table.setdefault(h, block_id)
        return h


class KVCachePool:
    """Owns the cache tensors for every layer of one model replica."""

    def __init__(
        self,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        num_blocks: int,
        block_size: int = DEFAULT_BLOCK_SIZE,
        dtype: torch.dtype = torch.bfloat16,
        device: str = "cpu",
    ) -> None:
        self.block_size = block_size
        self.num_blocks = num_blocks
        shape = (num_layers, num_blocks, num_kv_heads, block_size, head_dim)
        self.k = torch.zeros(shape, dtype=dtype, device=device)
        self.v = torch.zeros(shape, dtype=dtype, device=device)

    def layer(self, i: int) -> tuple:
        return self.k[i], self.v[i]

    def copy_block(self, src: int, dst: int) -> None:
        """COW copy for fork of a partial tail block."""
        self.k[:, dst].copy_(self.k[:, src])
        self.v[:, dst].copy_(self.v[:, src])

    @staticmethod
    def blocks_for_memory(
        bytes_budget: int,
        num_layers: int,
        num_kv_heads: int,
        head_dim: int,
        block_size: int = DEFAULT_BLOCK_SIZE,
        dtype_bytes: int = 2,
    ) -> int:
        per_block = 2 * num_layers * num_kv_heads * block_size * head_dim * dtype_bytes
        return max(1, bytes_budget // per_block)
