"""Weight I/O: safetensors loading (HF layout) + random-init mode.

SURVEY.md §2.3 inventory row "Tokenizer + safetensors weight loader w/
random-init mode". Maps HuggingFace Llama/Mixtral checkpoint names onto
this framework's fused layouts (q/k/v fused into qkv_proj, gate/up fused
into gate_up_proj) with TP-aware sharding, so a real Llama-3 checkpoint
drops in when one is available; random-init (models/*.random_init) is the
synthetic-bench path since this environment has no network for weights.
"""

from __future__ import annotations

from pathlib import Path

import torch

from dts_amd.utils.logging import logger


def _shard(t: torch.Tensor, dim: int, rank: int, size: int) -> torch.Tensor:
    if size == 1:
        return t
    chunk = t.shape[dim] // size
    return t.narrow(dim, rank * chunk, chunk)


def _shard_kv(
    t: torch.Tensor, num_kv_heads: int, head_dim: int, rank: int, size: int
) -> torch.Tensor:
    """Shard a k/v projection [nkv*hd, H] over TP ranks.

    When size > num_kv_heads the model replicates KV heads
    (LlamaAttention: num_kv_heads_local = max(1, nkv//size)); each rank
    takes the single head it shares — plain narrowing would produce a
    fraction of a head and fail to load.
    """
    if size <= num_kv_heads:
        return _shard(t, 0, rank, size)
    head = rank * num_kv_heads // size
    return t.narrow(0, head * head_dim, head_dim)


def load_llama_safetensors(model, path: str) -> int:
    """Load an HF-format Llama checkpoint directory into a LlamaModel.

    Returns the number of tensors consumed. TP: column-parallel weights are
    sharded on dim 0, row-parallel on dim 1, per the model's tp context.
    """
    from safetensors import safe_open

    spec = model.spec
    tp = model.tp
    files = sorted(Path(path).glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no .safetensors under {path}")

    tensors: dict = {}
    for f in files:
        with safe_open(str(f), framework="pt") as sf:
            for name in sf.keys():
                tensors[name] = sf.get_tensor(name)

    def get(name: str) -> torch.Tensor:
        if name not in tensors:
            raise KeyError(f"missing tensor {name}")
        return tensors[name].to(torch.float32)

    consumed = 0
    with torch.no_grad():
        model.embed.copy_(get("model.embed_tokens.weight").to(model.dtype))
        consumed += 1
        for i, layer in enumerate(model.layers):
            p = f"model.layers.{i}."
            layer.input_norm_w.copy_(get(p + "input_layernorm.weight").to(model.dtype))
            layer.post_norm_w.copy_(
                get(p + "post_attention_layernorm.weight").to(model.dtype)
            )
            q = _shard(get(p + "self_attn.q_proj.weight"), 0, tp.rank, tp.size)
            k = _shard_kv(
                get(p + "self_attn.k_proj.weight"),
                spec.num_kv_heads, spec.head_dim, tp.rank, tp.size,
            )
            v = _shard_kv(
                get(p + "self_attn.v_proj.weight"),
                spec.num_kv_heads, spec.head_dim, tp.rank, tp.size,
            )
            layer.attn.qkv_proj.weight.copy_(
                torch.cat([q, k, v], dim=0).to(model.dtype)
            )
            layer.attn.o_proj.weight.copy_(
                _shard(get(p + "self_attn.o_proj.weight"), 1, tp.rank, tp.size).to(
                    model.dtype
                )
            )
            g = _shard(get(p + "mlp.gate_proj.weight"), 0, tp.rank, tp.size)
            u = _shard(get(p + "mlp.up_proj.weight"), 0, tp.rank, tp.size)
            layer.mlp.gate_up_proj.weight.copy_(
                torch.cat([g, u], dim=0).to(model.dtype)
            )
            layer.mlp.down_proj.weight.copy_(
                _shard(get(p + "mlp.down_proj.weight"), 1, tp.rank, tp.size).to(
                    model.dtype
                )
            )
            consumed += 9
        model.final_norm_w.copy_(get("model.norm.weight").to(model.dtype))
        lm = tensors.get("lm_head.weight", tensors.get("model.embed_tokens.weight"))
        model.lm_head.weight.copy_(
            _shard(lm.to(torch.float32), 0, tp.rank, tp.size).to(model.dtype)
        )
        consumed += 2
    logger.info("loaded %d tensors from %s", consumed, path)
    return consumed


def save_llama_safetensors(model, path: str) -> None:
    """Write the model back out in HF Llama naming (single shard).

    Only valid for tp.size == 1; used by tests to round-trip the loader.
    """
    from safetensors.torch import save_file

    assert model.tp.size == 1
    out: dict = {}
    out["model.embed_tokens.weight"] = model.embed.data.clone()
    spec = model.spec
    q_size = spec.num_heads * spec.head_dim
    kv_size = spec.num_kv_heads * spec.head_dim
    for i, layer in enumerate(model.layers):
        p = f"model.layers.{i}."
        out[p + "input_layernorm.weight"] = layer.input_norm_w.data.clone()
        out[p + "post_attention_layernorm.weight"] = layer.post_norm_w.data.clone()
        qkv = layer.attn.qkv_proj.weight.data
        out[p + "self_attn.q_proj.weight"] = qkv[:q_size].clone()
        out[p + "self_attn.k_proj.weight"] = qkv[q_size : q_size + kv_size].clone()
        out[p + "self_attn.v_proj.weight"] = qkv[q_size + kv_size :].clone()
        out[p + "self_attn.o_proj.weight"] = layer.attn.o_proj.weight.data.clone()
        gu = layer.mlp.gate_up_proj.weight.data
        inter = spec.intermediate_size
        out[p + "mlp.gate_proj.weight"] = gu[:inter].clone()
        out[p + "mlp.up_proj.weight"] = gu[inter:].clone()
        out[p + "mlp.down_proj.weight"] = layer.mlp.down_proj.weight.data.clone()
    out["model.norm.weight"] = model.final_norm_w.data.clone()
    out["lm_head.weight"] = model.lm_head.weight.data.clone()
    Path(path).mkdir(parents=True, exist_ok=True)
    save_file(out, str(Path(path) / "model.safetensors"))


def load_mixtral_safetensors(model, path: str) -> int:
    """Load an HF-format Mixtral checkpoint into a MixtralModel.

    HF names: self_attn.{q,k,v,o}_proj (TP-sharded like Llama),
    block_sparse_moe.gate (router, replicated) and
    block_sparse_moe.experts.{e}.{w1,w3,w2} fused into the grouped
    [E_local, 2I, H] / [E_local, H, I] expert tensors (EP-sharded).
    """
    from safetensors import safe_open

    spec = model.spec
    tp = model.tp
    files = sorted(Path(path).glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no .safetensors under {path}")
    tensors: dict = {}
    for f in files:
        with safe_open(str(f), framework="pt") as sf:
            for name in sf.keys():
                tensors[name] = sf.get_tensor(name)

    def get(name: str) -> torch.Tensor:
        if name not in tensors:
            raise KeyError(f"missing tensor {name}")
        return tensors[name].to(torch.float32)

    consumed = 0
    with torch.no_grad():
        model.embed.copy_(get("model.embed_tokens.weight").to(model.dtype))
        consumed += 1
        for i, layer in enumerate(model.layers):
            p = f"model.layers.{i}."
            layer.input_norm_w.copy_(get(p + "input_layernorm.weight").to(model.dtype))
            layer.post_norm_w.copy_(
                get(p + "post_attention_layernorm.weight").to(model.dtype)
            )
            q = _shard(get(p + "self_attn.q_proj.weight"), 0, tp.rank, tp.size)
            k = _shard_kv(
                get(p + "self_attn.k_proj.weight"),
                spec.num_kv_heads, spec.head_dim, tp.rank, tp.size,
            )
            v = _shard_kv(
                get(p + "self_attn.v_proj.weight"),
                spec.num_kv_heads, spec.head_dim, tp.rank, tp.size,
            )
            layer.attn.qkv_proj.weight.copy_(
                torch.cat([q, k, v], dim=0).to(model.dtype)
            )
            layer.attn.o_proj.weight.copy_(
                _shard(get(p + "self_attn.o_proj.weight"), 1, tp.rank, tp.size).to(
                    model.dtype
                )
            )
            moe = layer.moe
            moe.router_w.copy_(
                get(p + "block_sparse_moe.gate.weight").to(model.dtype)
            )
            consumed += 6
            for el in range(moe.experts_local):
                e = moe.expert_offset + el
                ep = p + f"block_sparse_moe.experts.{e}."
                w1 = get(ep + "w1.weight")  # [I, H] gate
                w3 = get(ep + "w3.weight")  # [I, H] up
                w2 = get(ep + "w2.weight")  # [H, I] down
                moe.gate_up_w[el].copy_(torch.cat([w1, w3], dim=0).to(model.dtype))
                moe.down_w[el].copy_(w2.to(model.dtype))
                consumed += 3
        model.final_norm_w.copy_(get("model.norm.weight").to(model.dtype))
        lm = tensors.get("lm_head.weight", tensors.get("model.embed_tokens.weight"))
        model.lm_head.weight.copy_(
            _shard(lm.to(torch.float32), 0, tp.rank, tp.size).to(model.dtype)
        )
        consumed += 2
    logger.info("loaded %d tensors from %s", consumed, path)
    return consumed


def save_mixtral_safetensors(model, path: str) -> None:
    """Write a MixtralModel out in HF Mixtral naming (single shard,
    TP/EP size 1)."""
    from safetensors.torch import save_file

    out = {"model.embed_tokens.weight": model.embed.detach().cpu()}
    spec = model.spec
    for i, layer in enumerate(model.layers):
        p = f"model.layers.{i}."
        out[p + "input_layernorm.weight"] = layer.input_norm_w.detach().cpu()
        out[p + "post_attention_layernorm.weight"] = layer.post_norm_w.detach().cpu()
        qkv = layer.attn.qkv_proj.weight.detach().cpu()
        qs = spec.num_heads * spec.head_dim
        ks = spec.num_kv_heads * spec.head_dim
        out[p + "self_attn.q_proj.weight"] = qkv[:qs]
        out[p + "self_attn.k_proj.weight"] = qkv[qs : qs + ks]
        out[p + "self_attn.v_proj.weight"] = qkv[qs + ks :]
        out[p + "self_attn.o_proj.weight"] = layer.attn.o_proj.weight.detach().cpu()
        moe = layer.moe
        out[p + "block_sparse_moe.gate.weight"] = moe.router_w.detach().cpu()
        for e in range(spec.num_experts):
            ep = p + f"block_sparse_moe.experts.{e}."
            gu = moe.gate_up_w[e].detach().cpu()
            out[ep + "w1.weight"] = gu[: spec.intermediate_size].clone()
            out[ep + "w3.weight"] = gu[spec.intermediate_size :].clone()
            out[ep + "w2.weight"] = moe.down_w[e].detach().cpu().clone()
    out["model.norm.weight"] = model.final_norm_w.detach().cpu()
    out["lm_head.weight"] = model.lm_head.weight.detach().cpu()
    out = {k: v.contiguous() for k, v in out.items()}
    Path(path).mkdir(parents=True, exist_ok=True)
    save_file(out, str(Path(path) / "model.safetensors"))


def load_gpt2_safetensors(model, path: str) -> int:
    """Load an HF-format GPT-2 checkpoint into a GPT2Model.

    HF GPT-2 stores linear weights as transposed Conv1D ([in, out]);
    they transpose into this framework's [out, in] layout.
    """
    from safetensors import safe_open

    files = sorted(Path(path).glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no .safetensors under {path}")
    tensors: dict = {}
    for f in files:
        with safe_open(str(f), framework="pt") as sf:
            for name in sf.keys():
                tensors[name] = sf.get_tensor(name)

    def get(name: str) -> torch.Tensor:
        # HF publishes both with and without the "transformer." prefix
        for cand in (name, "transformer." + name):
            if cand in tensors:
                return tensors[cand].to(torch.float32)
        raise KeyError(f"missing tensor {name}")

    consumed = 0
    with torch.no_grad():
        model.wte.copy_(get("wte.weight").to(model.dtype))
        model.wpe.copy_(get("wpe.weight").to(model.dtype))
        consumed += 2
        for i, layer in enumerate(model.layers):
            p = f"h.{i}."
            layer.ln1_w.copy_(get(p + "ln_1.weight").to(model.dtype))
            layer.ln1_b.copy_(get(p + "ln_1.bias").to(model.dtype))
            layer.ln2_w.copy_(get(p + "ln_2.weight").to(model.dtype))
            layer.ln2_b.copy_(get(p + "ln_2.bias").to(model.dtype))
            layer.qkv_w.copy_(get(p + "attn.c_attn.weight").t().to(model.dtype))
            layer.qkv_b.copy_(get(p + "attn.c_attn.bias").to(model.dtype))
            layer.o_w.copy_(get(p + "attn.c_proj.weight").t().to(model.dtype))
            layer.o_b.copy_(get(p + "attn.c_proj.bias").to(model.dtype))
            layer.fc_w.copy_(get(p + "mlp.c_fc.weight").t().to(model.dtype))
            layer.fc_b.copy_(get(p + "mlp.c_fc.bias").to(model.dtype))
            layer.proj_w.copy_(get(p + "mlp.c_proj.weight").t().to(model.dtype))
            layer.proj_b.copy_(get(p + "mlp.c_proj.bias").to(model.dtype))
            consumed += 12
        model.lnf_w.copy_(get("ln_f.weight").to(model.dtype))
        model.lnf_b.copy_(get("ln_f.bias").to(model.dtype))
        consumed += 2
    logger.info("loaded %d tensors from %s", consumed, path)
    return consumed


def save_gpt2_safetensors(model, path: str) -> None:
    """Write a GPT2Model out in HF GPT-2 naming (Conv1D-transposed)."""
    from safetensors.torch import save_file

    out = {
        "wte.weight": model.wte.detach().cpu(),
        "wpe.weight": model.wpe.detach().cpu(),
        "ln_f.weight": model.lnf_w.detach().cpu(),
        "ln_f.bias": model.lnf_b.detach().cpu(),
    }
    for i, layer in enumerate(model.layers):
        p = f"h.{i}."
        out[p + "ln_1.weight"] = layer.ln1_w.detach().cpu()
        out[p + "ln_1.bias"] = layer.ln1_b.detach().cpu()
        out[p + "ln_2.weight"] = layer.ln2_w.detach().cpu()
        out[p + "ln_2.bias"] = layer.ln2_b.detach().cpu()
        out[p + "attn.c_attn.weight"] = layer.qkv_w.detach().cpu().t()
        out[p + "attn.c_attn.bias"] = layer.qkv_b.detach().cpu()
        out[p + "attn.c_proj.weight"] = layer.o_w.detach().cpu().t()
        out[p + "attn.c_proj.bias"] = layer.o_b.detach().cpu()
        out[p + "mlp.c_fc.weight"] = layer.fc_w.detach().cpu().t()
        out[p + "mlp.c_fc.bias"] = layer.fc_b.detach().cpu()
        out[p + "mlp.c_proj.weight"] = layer.proj_w.detach().cpu().t()
        out[p + "mlp.c_proj.bias"] = layer.proj_b.detach().cpu()
    out = {k: v.contiguous() for k, v in out.items()}
    Path(path).mkdir(parents=True, exist_ok=True)
    save_file(out, str(Path(path) / "model.safetensors"))
