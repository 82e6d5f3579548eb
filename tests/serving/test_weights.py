"""Safetensors loader round-trip on the tiny Llama model."""

import torch

from dts_amd.models.config import get_model_spec
from dts_amd.models.llama import LlamaModel
from dts_amd.models.weights import load_llama_safetensors, save_llama_safetensors


def test_safetensors_roundtrip(tmp_path):
    spec = get_model_spec("llama-tiny")
    m1 = LlamaModel(spec, dtype=torch.float32, device="cpu")
    m1.random_init(seed=7)
    save_llama_safetensors(m1, str(tmp_path))

    m2 = LlamaModel(spec, dtype=torch.float32, device="cpu")
    m2.random_init(seed=99)  # different weights first
    n = load_llama_safetensors(m2, str(tmp_path))
    assert n > 0
    for (n1, p1), (n2, p2) in zip(
        m1.named_parameters(), m2.named_parameters()
    ):
        assert n1 == n2
        assert torch.equal(p1, p2), n1
