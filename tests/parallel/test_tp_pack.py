"""TP batch wire format: pack/unpack round-trip (no process group)."""

import torch

from dts_amd.serving.batch import ForwardBatch
from dts_amd.serving.tp_engine import pack_batch, unpack_batch


def _mixed_batch():
    return ForwardBatch(
        token_ids=torch.tensor([5, 6, 7, 8, 9], dtype=torch.long),
        positions=torch.tensor([0, 1, 2, 7, 3], dtype=torch.long),
        slot_mapping=torch.tensor([10, 11, 12, 99, 40], dtype=torch.long),
        num_prefill_seqs=1,
        num_prefill_tokens=3,
        cu_q=torch.tensor([0, 3], dtype=torch.int32),
        prefill_block_tables=torch.tensor([[1, 2]], dtype=torch.int32),
        prefill_kv_lens=torch.tensor([3], dtype=torch.int32),
        num_decode_seqs=2,
        decode_block_tables=torch.tensor([[6, 0], [2, 5]], dtype=torch.int32),
        decode_kv_lens=torch.tensor([8, 4], dtype=torch.int32),
        sample_indices=torch.tensor([2, 3, 4], dtype=torch.long),
    )


def _assert_batch_equal(a: ForwardBatch, b: ForwardBatch):
    for f in (
        "num_prefill_seqs",
        "num_prefill_tokens",
        "num_decode_seqs",
    ):
        assert getattr(a, f) == getattr(b, f), f
    for f in (
        "token_ids",
        "positions",
        "slot_mapping",
        "cu_q",
        "prefill_block_tables",
        "prefill_kv_lens",
        "decode_block_tables",
        "decode_kv_lens",
        "sample_indices",
    ):
        ta, tb = getattr(a, f), getattr(b, f)
        if ta is None:
            assert tb is None, f
        else:
            assert torch.equal(ta, tb.to(ta.dtype)), f


def test_roundtrip_mixed():
    b = _mixed_batch()
    hdr, payload = pack_batch(b, "cpu")
    out = unpack_batch(hdr, payload)
    _assert_batch_equal(b, out)


def test_roundtrip_decode_only():
    b = ForwardBatch(
        token_ids=torch.tensor([3, 4], dtype=torch.long),
        positions=torch.tensor([9, 10], dtype=torch.long),
        slot_mapping=torch.tensor([90, 91], dtype=torch.long),
        num_decode_seqs=2,
        decode_block_tables=torch.tensor([[1], [2]], dtype=torch.int32),
        decode_kv_lens=torch.tensor([10, 11], dtype=torch.int32),
        sample_indices=torch.tensor([0, 1], dtype=torch.long),
    )
    hdr, payload = pack_batch(b, "cpu")
    out = unpack_batch(hdr, payload)
    _assert_batch_equal(b, out)
    # payload length formula the worker uses must match exactly
    T, P, _, D, pf_w, dc_w, S = (int(x) for x in hdr[1:8])
    plen = 3 * T + ((P + 1) + P * pf_w + P if P else 0) + (D * dc_w + D if D else 0) + S
    assert payload.numel() == plen


def test_roundtrip_prefill_only():
    b = ForwardBatch(
        token_ids=torch.tensor([1, 2, 3, 4], dtype=torch.long),
        positions=torch.tensor([0, 1, 2, 3], dtype=torch.long),
        slot_mapping=torch.tensor([0, 1, 2, 3], dtype=torch.long),
        num_prefill_seqs=1,
        num_prefill_tokens=4,
        cu_q=torch.tensor([0, 4], dtype=torch.int32),
        prefill_block_tables=torch.tensor([[0]], dtype=torch.int32),
        prefill_kv_lens=torch.tensor([4], dtype=torch.int32),
        sample_indices=torch.tensor([3], dtype=torch.long),
    )
    hdr, payload = pack_batch(b, "cpu")
    out = unpack_batch(hdr, payload)
    _assert_batch_equal(b, out)
