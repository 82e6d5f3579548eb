"""Block manager + prefix cache tests."""

import pytest
import torch

from dts_amd.serving.kv_cache import BlockManager, KVCachePool, chain_hash


@pytest.fixture
def bm():
    return BlockManager(num_blocks=16, block_size=4)


class TestAllocation:
    def test_fresh_alloc_and_free(self, bm):
        b = bm.allocate_fresh()
        assert bm.blocks[b].ref_count == 1
        assert bm.num_free() == 15
        bm.free_block(b)
        assert bm.num_free() == 16

    def test_exhaustion_raises(self):
        bm = BlockManager(num_blocks=2, block_size=4)
        bm.allocate_fresh()
        bm.allocate_fresh()
        with pytest.raises(MemoryError):
            bm.allocate_fresh()


class TestPrefixCache:
    def test_register_and_match(self, bm):
        tokens = list(range(10))  # 2 full blocks + partial
        b0, b1 = bm.allocate_fresh(), bm.allocate_fresh()
        h0 = bm.register_full_block(b0, 0, tuple(tokens[0:4]))
        bm.register_full_block(b1, h0, tuple(tokens[4:8]))

        blocks, n = bm.match_prefix(tokens)
        assert blocks == [b0, b1]
        assert n == 8
        assert bm.blocks[b0].ref_count == 2  # original + match

    def test_no_match_on_divergent_content(self, bm):
        b0 = bm.allocate_fresh()
        bm.register_full_block(b0, 0, (1, 2, 3, 4))
        blocks, n = bm.match_prefix([1, 2, 3, 99, 5, 6, 7, 8])
        assert blocks == [] and n == 0

    def test_match_stops_at_first_miss(self, bm):
        b0, b1 = bm.allocate_fresh(), bm.allocate_fresh()
        h0 = bm.register_full_block(b0, 0, (0, 1, 2, 3))
        bm.register_full_block(b1, h0, (4, 5, 6, 7))
        blocks, n = bm.match_prefix([0, 1, 2, 3, 9, 9, 9, 9, 4, 5, 6, 7])
        assert blocks == [b0] and n == 4

    def test_evictable_revival(self, bm):
        """Freed hashed blocks stay matchable until evicted (LRU)."""
        b0 = bm.allocate_fresh()
        bm.register_full_block(b0, 0, (1, 2, 3, 4))
        bm.free_block(b0)
        assert bm.blocks[b0].ref_count == 0
        blocks, n = bm.match_prefix([1, 2, 3, 4, 5])
        assert blocks == [b0] and n == 4
        assert bm.blocks[b0].ref_count == 1
        assert b0 not in bm.evictable

    def test_eviction_unregisters_hash(self):
        bm = BlockManager(num_blocks=1, block_size=4)
        b0 = bm.allocate_fresh()
        h = bm.register_full_block(b0, 0, (1, 2, 3, 4))
        bm.free_block(b0)
        # pool empty except evictable; fresh allocation must evict b0
        b1 = bm.allocate_fresh()
        assert b1 == b0
        assert h not in bm.hash_table
        blocks, n = bm.match_prefix([1, 2, 3, 4])
        assert n == 0


class TestPool:
    def test_copy_block(self):
        pool = KVCachePool(2, 2, 8, num_blocks=4, block_size=4, dtype=torch.float32)
        pool.k[:, 1].normal_()
        pool.v[:, 1].normal_()
        pool.copy_block(1, 3)
        assert torch.equal(pool.k[:, 1], pool.k[:, 3])
        assert torch.equal(pool.v[:, 1], pool.v[:, 3])

    def test_blocks_for_memory(self):
        n = KVCachePool.blocks_for_memory(
            1 << 30, num_layers=32, num_kv_heads=8, head_dim=128, block_size=16
        )
        # 2*32*8*16*128*2 bytes = 2 MiB per block -> 512 blocks per GiB
        assert n == 512
