"""Flagship benchmark: scored trajectories/sec on the reference 6x5 search.

One bench step = one COMPLETE dialogue-tree search round (BASELINE.json
config: Llama-3-8B actor+user+judges, init_branches=6, turns_per_branch=5,
comparative scoring) executed end-to-end on the local MI355X serving
engine: strategy generation (constrained JSON), 6 branch rollouts of 5
user+assistant turns each (sequential per branch, batched across
branches), one comparative forced-ranking judge pass, prune + backprop.
Every step uses fresh synthetic prompts (step-salted) so nothing is served
from a previous step's KV.

Weak scaling: with --gpus N the ONE search has init_branches = 6*N,
branches sharded round-robin across ranks (SPMD DP — dts_amd/search/
dist_engine.py); per-GPU work stays 6 branches. value = scored
trajectories per second aggregated over the whole job.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--model", type=str, default="llama-3-8b")
    p.add_argument(
        "--judge-model",
        type=str,
        default=None,
        help="serve judges on a second engine (e.g. mixtral-8x7b — "
        "BASELINE config 5)",
    )
    p.add_argument("--branches-per-gpu", type=int, default=6)
    p.add_argument("--turns", type=int, default=5)
    p.add_argument("--rounds", type=int, default=1)
    p.add_argument("--scoring", type=str, default="comparative",
                   choices=["comparative", "absolute"])
    p.add_argument("--intents", type=int, default=1,
                   help=">1 enables user-intent forking (BASELINE config 3: "
                        "user_variability=True, K intents per branch)")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--kv-frac", type=float, default=0.75)
    return p.parse_args()


GOAL = (
    "Help a mid-level engineer decide between two database architectures "
    "for a new analytics product, covering tradeoffs, costs and migration risk"
)
FIRST_MESSAGE = (
    "We're torn between a columnar warehouse and a distributed row store "
    "for our analytics backend. How should we think about this choice?"
)


def build_config(args, world, seed):
    from dts_amd.search import DTSConfig
    from dts_amd.search.config import GenerationBudget

    return DTSConfig(
        goal=GOAL,
        first_message=FIRST_MESSAGE,
        init_branches=args.branches_per_gpu * world,
        turns_per_branch=args.turns,
        user_intents_per_branch=args.intents,
        user_variability=args.intents > 1,
        scoring_mode=args.scoring,
        prune_threshold=6.5,
        min_survivors=1,
        max_concurrency=64,
        judge_model=args.judge_model,
        seed=seed,
        budget=GenerationBudget(
            strategy=3072,
            intent=1024,
            rephrase=96,
            user=160,
            assistant=256,
            judge=8192,
        ),
    )


async def run_search(llm, cfg, dp, rounds=1):
    from dts_amd.search.dist_engine import DistributedDTSEngine

    engine = DistributedDTSEngine(llm, cfg, dp=dp)
    result = await engine.run(rounds=rounds)
    scored = sum(
        1 for n in result.all_nodes if n.strategy is not None and n.stats.judge_scores
    )
    return scored, engine.phase_times


def main():
    args = parse_args()
    world_env = int(os.environ.get("WORLD_SIZE", "1"))

    from dts_amd.llm import LLM
    from dts_amd.parallel import init_distributed
    from dts_amd.parallel.dp import DPContext
    from dts_amd.serving import LocalBackend, ServingEngine

    world = init_distributed() if world_env > 1 else 1
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    dp = DPContext()

    use_gpu = torch.cuda.is_available()
    device = args.device or (f"cuda:{local_rank}" if use_gpu else "cpu")
    dtype = torch.bfloat16 if use_gpu else torch.float32

    if use_gpu:
        torch.cuda.set_device(device)
        kv_bytes = None  # engines self-size from POST-weights free memory
    else:
        kv_bytes = 512 << 20

    if args.judge_model == args.model:
        args.judge_model = None  # same model: one engine serves both roles

    # KV sizing: each engine measures free memory AFTER its own weights
    # load and takes kv_frac of it (a pre-weights split over-committed:
    # 70B weights + 0.75-of-free pool exceeded the 288 GB card). For
    # two-engine runs (config 5) the actor takes a conservative slice so
    # the judge's 94 GB of weights still fit.
    engine = ServingEngine(
        model_name=args.model,
        device=device,
        dtype=dtype,
        kv_memory_bytes=kv_bytes,
        kv_frac=args.kv_frac * (0.45 if args.judge_model else 1.0),
        max_batch_tokens=16384,
        max_running=512,
        weight_seed=0,
    )
    engines = {args.model: engine}
    if args.judge_model:
        engines[args.judge_model] = ServingEngine(
            model_name=args.judge_model,
            device=device,
            dtype=dtype,
            kv_memory_bytes=None if use_gpu else kv_bytes // 2,
            kv_frac=args.kv_frac,
            max_batch_tokens=16384,
            max_running=512,
            weight_seed=1,
        )
    backend = LocalBackend(engines, default_model=args.model)
    llm = LLM(backend, default_model=args.model)

    def sync():
        if use_gpu:
            torch.cuda.synchronize()
        dp.barrier()

    def one_step(step_idx):
        cfg = build_config(args, world, seed=1000 + step_idx)
        # salt prompts per step so no cross-step KV reuse inflates numbers
        cfg = type(cfg)(**{**cfg.__dict__,
                           "goal": f"{GOAL} (scenario {step_idx})",
                           "first_message": f"{FIRST_MESSAGE} (case {step_idx})"})
        scored, phases = asyncio.run(run_search(llm, cfg, dp, rounds=args.rounds))
        for k, v in phases.items():
            phase_totals[k] = phase_totals.get(k, 0.0) + v
        return scored

    # warmup
    phase_totals: dict = {}
    for w in range(args.warmup):
        one_step(10_000 + w)

    sync()
    phase_totals.clear()  # keep only the timed steps' phase wall-clock
    t0 = time.perf_counter()
    total_scored = 0
    for s in range(args.steps):
        total_scored += one_step(s)
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks (nccl collectives need device tensors)
    if world > 1:
        import torch.distributed as dist

        dev = "cuda" if dist.get_backend() == "nccl" else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu()[0])

    n_gpus = world if use_gpu else args.gpus
    # total_scored counts the WHOLE tree's scored branches (identical on
    # every rank under SPMD DP) — already the job-level aggregate
    traj_per_sec = total_scored / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        stats = engine.cache_stats
        print(
            json.dumps(
                {
                    "metric": "scored_trajectories_per_sec",
                    "value": round(traj_per_sec, 4),
                    "unit": "trajectories/s",
                    "n_gpus": n_gpus,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": round(ms_per_step, 1),
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16" if use_gpu else "fp32",
                    "data": "synthetic (byte-level tokenizer, random-init "
                    "weights, constrained JSON decoding)",
                    "config": {
                        "model": args.model,
                        "init_branches": args.branches_per_gpu * world,
                        "turns_per_branch": args.turns,
                        "rounds_per_step": args.rounds,
                        "scoring_mode": args.scoring,
                        "global_batch": args.branches_per_gpu * world,
                        "seq_len": "chat-scale (user<=160, assistant<=256 "
                        "tok/turn, judge prompt ~order-10k tok)",
                        "parallelism": f"dp{n_gpus}"
                        + (f"+fork{args.intents}" if args.intents > 1 else ""),
                        "engine": stats,
                        # rank-0 wall per search phase over the timed
                        # steps: exposes the init(strategy)/expand/score
                        # split for scaling analysis
                        "search_phases": {
                            k: round(v, 2) for k, v in phase_totals.items()
                        },
                    },
                },
            )
        )
    backend.shutdown()
    if world > 1:
        from dts_amd.parallel.dist import destroy

        destroy()


if __name__ == "__main__":
    main()
