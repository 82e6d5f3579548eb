"""Real-checkpoint-path smoke (VERDICT round-1 #8, adapted to no-network).

There is no egress to fetch actual Llama-3 weights, so this exercises
every component a real checkpoint would touch with the closest
offline-constructible stand-ins:

  1. train a byte-level BPE tokenizer (HF `tokenizers`, Llama-3-style
     special tokens) on local text — the REAL HFTokenizer/HFChatTemplate
     path, including the byte->token map that guided JSON needs;
  2. save a random-init llama-3-8b to HF-format safetensors (~16 GB) and
     load it back through load_llama_safetensors — the full checkpoint
     loader at real scale, q/k/v + gate/up fusion included;
  3. run a complete dialogue-tree search (2 branches x 2 turns,
     comparative, constrained JSON through the byte->token translation)
     on the loaded engine and save the exploration checkpoint.

Writes a summary JSON + the exploration dict under gpurun_out/.
"""

import asyncio
import json
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

OUT = Path("gpurun_out")
OUT.mkdir(exist_ok=True)


def build_tokenizer(path: Path) -> str:
    from tokenizers import Tokenizer, decoders, models, pre_tokenizers, trainers

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=8000,
        special_tokens=[
            "<|begin_of_text|>",
            "<|end_of_text|>",
            "<|eot_id|>",
            "<|start_header_id|>",
            "<|end_header_id|>",
        ],
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
    )
    corpus = []
    for p in list(Path("dts_amd").rglob("*.py"))[:60]:
        try:
            corpus.append(p.read_text())
        except Exception:  # noqa: BLE001
            pass
    corpus += [
        "Help the user choose a database architecture. " * 20,
        '{"goal": "text", "ranking": [{"rank": 1, "score": 7.5}]} ' * 10,
    ]
    tok.train_from_iterator(corpus, trainer)
    tok.save(str(path))
    return str(path)


async def run_search(backend):
    from dts_amd.llm import LLM
    from dts_amd.search.config import DTSConfig
    from dts_amd.search.engine import DTSEngine

    llm = LLM(backend, default_model="llama-3-8b")
    cfg = DTSConfig(
        goal="Help the user choose a database architecture",
        first_message="Which database should we pick for analytics?",
        init_branches=2,
        turns_per_branch=2,
        scoring_mode="comparative",
        seed=11,
    )
    engine = DTSEngine(llm=llm, config=cfg)
    return await engine.run(rounds=1), engine


def main():
    t0 = time.time()
    tok_path = build_tokenizer(OUT / "llama3_style_bpe.json")
    print(f"tokenizer trained: {time.time()-t0:.1f}s", flush=True)

    from dts_amd.models.config import get_model_spec
    from dts_amd.models.llama import LlamaModel
    from dts_amd.models.weights import save_llama_safetensors
    from dts_amd.serving import LocalBackend, ServingEngine

    ckpt_dir = Path("/tmp/llama3_8b_randinit")
    spec = get_model_spec("llama-3-8b")
    t1 = time.time()
    src = LlamaModel(spec, dtype=torch.bfloat16, device="cuda")
    src.random_init(seed=1234)
    src_cpu = src.to("cpu")
    save_llama_safetensors(src_cpu, str(ckpt_dir))
    del src, src_cpu
    torch.cuda.empty_cache()
    print(f"checkpoint saved ({time.time()-t1:.1f}s)", flush=True)

    t2 = time.time()
    eng = ServingEngine(
        model_name="llama-3-8b",
        device="cuda:0",
        dtype=torch.bfloat16,
        kv_memory_bytes=32 << 30,
        weights_path=str(ckpt_dir),
        tokenizer_path=tok_path,
    )
    load_s = time.time() - t2
    print(f"engine loaded from safetensors ({load_s:.1f}s)", flush=True)
    from dts_amd.serving.tokenizer import HFTokenizer

    assert isinstance(eng.tokenizer, HFTokenizer), "HF tokenizer path not taken"

    backend = LocalBackend.single(eng, name="llama-3-8b")
    t3 = time.time()
    result, engine = asyncio.run(run_search(backend))
    search_s = time.time() - t3
    backend.shutdown()

    expl = result.to_exploration_dict()
    (OUT / "real_weights_smoke_exploration.json").write_text(
        json.dumps(expl)[:200000]
    )
    summary = {
        "probe": "real_weights_smoke",
        "tokenizer": "HF BPE 8000 (byte-level, llama-3 specials)",
        "checkpoint_load_s": round(load_s, 1),
        "search_s": round(search_s, 1),
        "best_score": result.best_score,
        "branches": len(expl["branches"]),
        "sample_text": (expl["branches"][0]["trajectory"][-1]["content"] or "")[:160]
        if expl["branches"] and expl["branches"][0]["trajectory"]
        else "",
        "engine_stats": eng.cache_stats,
    }
    (OUT / "real_weights_smoke.json").write_text(json.dumps(summary, indent=1))
    print(json.dumps(summary, indent=1))


if __name__ == "__main__":
    main()
