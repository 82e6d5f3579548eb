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


def test_engine_loads_safetensors_checkpoint(tmp_path):
    """ServingEngine(weights_path=...) serves a real HF-layout checkpoint:
    outputs must match an engine running the source model directly."""
    import torch

    from dts_amd.llm.types import SamplingParams
    from dts_amd.models.config import get_model_spec
    from dts_amd.models.llama import LlamaModel
    from dts_amd.models.weights import save_llama_safetensors
    from dts_amd.serving import ServingEngine

    src = LlamaModel(get_model_spec("llama-tiny"), dtype=torch.float32, device="cpu")
    src.random_init(seed=77)
    save_llama_safetensors(src, str(tmp_path))

    def run(eng):
        f = eng.submit_tokens(
            list(range(1, 40)), SamplingParams(max_tokens=8, temperature=0.0)
        )
        eng.run_until_idle()
        out = f.result(timeout=10).token_ids
        eng.stop()
        return out

    direct = run(
        ServingEngine(
            model_name="llama-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=128,
            block_size=8,
            model=src,
        )
    )
    loaded = run(
        ServingEngine(
            model_name="llama-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=128,
            block_size=8,
            weights_path=str(tmp_path),
        )
    )
    assert direct == loaded


def test_mixtral_safetensors_roundtrip(tmp_path):
    """Save a random-init Mixtral in HF naming, reload via the engine,
    outputs token-identical to the source model."""
    import torch

    from dts_amd.llm.types import SamplingParams
    from dts_amd.models.config import get_model_spec
    from dts_amd.models.mixtral import MixtralModel
    from dts_amd.models.weights import save_mixtral_safetensors
    from dts_amd.serving import ServingEngine

    src = MixtralModel(
        get_model_spec("mixtral-tiny"), dtype=torch.float32, device="cpu"
    )
    src.random_init(seed=5)
    save_mixtral_safetensors(src, str(tmp_path))

    def run(eng):
        f = eng.submit_tokens(
            list(range(1, 30)), SamplingParams(max_tokens=6, temperature=0.0)
        )
        eng.run_until_idle()
        out = f.result(timeout=10).token_ids
        eng.stop()
        return out

    direct = run(
        ServingEngine(
            model_name="mixtral-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=128,
            block_size=8,
            model=src,
        )
    )
    loaded = run(
        ServingEngine(
            model_name="mixtral-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=128,
            block_size=8,
            weights_path=str(tmp_path),
        )
    )
    assert direct == loaded


def test_gpt2_safetensors_roundtrip(tmp_path):
    import torch

    from dts_amd.llm.types import SamplingParams
    from dts_amd.models.config import get_model_spec
    from dts_amd.models.gpt2 import GPT2Model
    from dts_amd.models.weights import save_gpt2_safetensors
    from dts_amd.serving import ServingEngine

    src = GPT2Model(get_model_spec("gpt2-tiny"), dtype=torch.float32, device="cpu")
    src.random_init(seed=11)
    save_gpt2_safetensors(src, str(tmp_path))

    def run(eng):
        f = eng.submit_tokens(
            list(range(1, 30)), SamplingParams(max_tokens=6, temperature=0.0)
        )
        eng.run_until_idle()
        out = f.result(timeout=10).token_ids
        eng.stop()
        return out

    direct = run(
        ServingEngine(
            model_name="gpt2-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=128,
            block_size=8,
            model=src,
        )
    )
    loaded = run(
        ServingEngine(
            model_name="gpt2-tiny",
            device="cpu",
            dtype=torch.float32,
            num_blocks=128,
            block_size=8,
            weights_path=str(tmp_path),
        )
    )
    assert direct == loaded
