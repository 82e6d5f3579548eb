"""Model architecture specs.

The north-star configs (BASELINE.json) name Llama-3-8B / Llama-3-70B
actors+judges, Mixtral-8x7B judges and a GPT-2-small CPU plumbing config;
tiny variants exist for CPU tests. Specs are architecture only — weights
are random-init or loaded from safetensors (dts_amd/models/weights.py).
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ModelSpec:
    name: str
    arch: str = "llama"  # "llama" | "gpt2" | "mixtral"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 8192
    tie_embeddings: bool = False
    # MoE (mixtral)
    num_experts: int = 0
    experts_per_token: int = 2
    # gpt2
    layernorm_eps: float = 1e-5

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    def kv_bytes_per_token(self, dtype_bytes: int = 2) -> int:
        return 2 * self.num_layers * self.kv_size * dtype_bytes


MODEL_REGISTRY = {
    # Llama-3-8B — the flagship bench model (BASELINE.json config 2)
    "llama-3-8b": ModelSpec(
        name="llama-3-8b",
        vocab_size=128256,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        # 32k rope table (Llama-3.1-class context): the comparative judge
        # re-embeds all sibling trajectories in one prompt (SURVEY.md §2.3
        # — the largest prompt in the system, ~13k tokens on the 6x5
        # config) and must also fit its ~3k-token structured answer.
        max_position=32768,
    ),
    # Llama-3-70B — TP=8 config (BASELINE.json config 4)
    "llama-3-70b": ModelSpec(
        name="llama-3-70b",
        vocab_size=128256,
        hidden_size=8192,
        intermediate_size=28672,
        num_layers=80,
        num_heads=64,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=500000.0,
        max_position=8192,
    ),
    # Mixtral-8x7B — MoE judge config (BASELINE.json config 5)
    "mixtral-8x7b": ModelSpec(
        name="mixtral-8x7b",
        arch="mixtral",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=32,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        max_position=8192,
        num_experts=8,
        experts_per_token=2,
    ),
    # GPT-2-small — CPU plumbing config (BASELINE.json config 1)
    "gpt2-small": ModelSpec(
        name="gpt2-small",
        arch="gpt2",
        vocab_size=50257,
        hidden_size=768,
        intermediate_size=3072,
        num_layers=12,
        num_heads=12,
        num_kv_heads=12,
        head_dim=64,
        max_position=1024,
        tie_embeddings=True,
    ),
    # Llama-3-8B dims at depth 2 — for GPU cross-path consistency tests:
    # a random-init 32-layer stack amplifies bf16 reduction-order noise
    # into decorrelated logits, so path-equivalence is asserted at depth 2
    # where drift stays bounded while the kernels run at real shapes.
    "llama-3-8b-2l": ModelSpec(
        name="llama-3-8b-2l",
        vocab_size=128256,
        hidden_size=4096,
        intermediate_size=14336,
        num_layers=2,
        num_heads=32,
        num_kv_heads=8,
        head_dim=128,
        rope_theta=500000.0,
        max_position=16384,
    ),
    # Tiny models for CPU tests
    "llama-tiny": ModelSpec(
        name="llama-tiny",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        max_position=4096,
    ),
    "gpt2-tiny": ModelSpec(
        name="gpt2-tiny",
        arch="gpt2",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=4,
        head_dim=16,
        max_position=2048,
        tie_embeddings=True,
    ),
    # small-vocab GPU test model: D=64/G=1 runs the real HIP kernels,
    # and greedy decode on random weights enters short cycles quickly —
    # which is what the speculative-decode GPU tests need (an 8B-class
    # random model does not repeat a bigram within a short generation)
    "llama-mini-gpu": ModelSpec(
        name="llama-mini-gpu",
        vocab_size=2048,
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=4,
        num_kv_heads=4,
        head_dim=64,
        rope_theta=10000.0,
        max_position=4096,
    ),
    "mixtral-tiny": ModelSpec(
        name="mixtral-tiny",
        arch="mixtral",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        num_layers=2,
        num_heads=4,
        num_kv_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        max_position=4096,
        num_experts=4,
        experts_per_token=2,
    ),
}


def get_model_spec(name: str) -> ModelSpec:
    if name not in MODEL_REGISTRY:
        raise KeyError(f"unknown model '{name}'; known: {sorted(MODEL_REGISTRY)}")
    return MODEL_REGISTRY[name]
