"""CPU-testable units of the hipGraph decode runner."""

import pytest
import torch

from dts_amd.serving.batch import ForwardBatch
from dts_amd.serving.graph_runner import BUCKETS, DecodeGraphRunner


class FakePool:
    block_size = 16


def make_runner(max_bucket=256):
    return DecodeGraphRunner(
        model=None,
        kv_pool=FakePool(),
        device="cpu",
        scratch_block=99,
        max_blocks_per_seq=64,
        max_bucket=max_bucket,
    )


def decode_batch(n, table_width=4):
    return ForwardBatch(
        token_ids=torch.zeros(n, dtype=torch.long),
        positions=torch.zeros(n, dtype=torch.long),
        slot_mapping=torch.zeros(n, dtype=torch.long),
        num_decode_seqs=n,
        decode_block_tables=torch.zeros(n, table_width, dtype=torch.int32),
        decode_kv_lens=torch.ones(n, dtype=torch.int32),
        sample_indices=torch.arange(n),
    )


def test_bucket_rounding():
    r = make_runner()
    assert r._bucket_for(1) == 1
    assert r._bucket_for(3) == 4
    assert r._bucket_for(5) == 8
    assert r._bucket_for(16) == 16
    with pytest.raises(ValueError):
        r._bucket_for(BUCKETS[-1] + 1)


def test_can_run_policy():
    r = make_runner()
    assert r.can_run(decode_batch(4))
    assert not r.can_run(decode_batch(BUCKETS[-1] + 1))  # beyond buckets
    # mixed batch (prefill present) is eager
    b = decode_batch(2)
    b.num_prefill_seqs = 1
    b.num_prefill_tokens = 10
    assert not r.can_run(b)
    # oversized block table is eager
    assert not r.can_run(decode_batch(2, table_width=65))


def test_scratch_slot_geometry():
    r = make_runner()
    assert r.scratch_slot == 99 * 16
