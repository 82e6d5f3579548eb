"""DTS engine — the search orchestrator.

Parity: reference backend/core/dts/engine.py:33-624 — init→rounds(expand→
score→backprop→prune)→result, per-phase model resolution (ref :72-76),
fixed-vs-LLM intent strategy (ref :251-263), prune policy with threshold
filter / top-K cap / min-survivors floor (ref :537-585), the 12-type event
stream (ref :431-434 emission sites), and DTSRunResult assembly.

The engine is backend-agnostic: it drives any `LLM` (dts_amd/llm/backend.py)
— the MI355X serving engine in production, deterministic fakes in tests.
"""

from __future__ import annotations

import asyncio
from typing import Any, Callable, Optional

from dts_amd.llm.backend import LLM
from dts_amd.llm.types import Completion, Message
from dts_amd.search.config import DTSConfig
from dts_amd.search.evaluator import TrajectoryEvaluator
from dts_amd.search.events import emit_event
from dts_amd.search.generator import FIXED_INTENT, StrategyGenerator
from dts_amd.search.simulator import ConversationSimulator
from dts_amd.search.tree import DialogueTree, generate_node_id
from dts_amd.search.types import (
    AggregatedScore,
    DialogueNode,
    DTSRunResult,
    NodeStatus,
    Strategy,
    TokenTracker,
    UserIntent,
)
from dts_amd.utils.logging import log_phase, logger


class DTSEngine:
    """Parallel beam search over multi-turn conversations."""

    def __init__(
        self,
        llm: LLM,
        config: DTSConfig,
        researcher: Optional[object] = None,
    ) -> None:
        self.llm = llm
        self.config = config

        default_model = config.model or getattr(llm, "_default_model", None)
        strategy_model = config.strategy_model or default_model
        simulator_model = config.simulator_model or default_model
        judge_model = config.judge_model or default_model

        self._token_tracker = TokenTracker(model_name=default_model or "unknown")

        self._generator = StrategyGenerator(
            llm=llm,
            goal=config.goal,
            model=strategy_model,
            temperature=config.temperature,
            max_concurrency=config.max_concurrency,
            on_usage=self._track_usage,
            max_tokens=config.budget.strategy,
            intent_max_tokens=config.budget.intent,
            seed=config.seed,
            strategy_split=config.strategy_split,
        )
        self._simulator = ConversationSimulator(
            llm=llm,
            goal=config.goal,
            model=simulator_model,
            temperature=config.temperature,
            max_concurrency=config.max_concurrency,
            on_usage=self._track_usage,
            on_event=self._emit_async,
            budget=config.budget,
            seed=config.seed,
            reasoning_enabled=config.reasoning_enabled,
        )
        self._evaluator = TrajectoryEvaluator(
            llm=llm,
            goal=config.goal,
            model=judge_model,
            judge_temperature=config.judge_temperature,
            prune_threshold=config.prune_threshold,
            max_concurrency=config.max_concurrency,
            on_usage=self._track_usage,
            max_tokens=config.budget.judge,
            seed=config.seed,
            comparative_split=config.comparative_split,
        )
        # optional deep-research provider (dts_amd/search/researcher.py)
        self._researcher = researcher

        self._tree: Optional[DialogueTree] = None
        # wall-clock per search phase (init/expand/score), cumulative over
        # rounds — dumped by bench.py to expose Amdahl pieces at scale
        self.phase_times: dict = {"init_s": 0.0, "expand_s": 0.0, "score_s": 0.0}
        self._event_callback: Optional[Callable] = None
        self._research_report: Optional[str] = None

    # ------------------------------------------------------------------
    def set_event_callback(self, callback: Callable) -> None:
        """Async callback receiving (event_type, data) — ref engine.py:129-141."""
        self._event_callback = callback

    @property
    def tree(self) -> Optional[DialogueTree]:
        return self._tree

    @property
    def token_tracker(self) -> TokenTracker:
        return self._token_tracker

    # ------------------------------------------------------------------
    async def run(self, rounds: int = 1, resume_from=None) -> DTSRunResult:
        """resume_from: a saved exploration dict (or a path to its JSON) —
        rebuilds the tree from the checkpoint instead of generating
        strategies, then continues `rounds` more rounds. The reference has
        no mid-search resume (SURVEY.md §5 Checkpoint/resume); this uses
        the same tree-state JSON it persists."""
        cfg = self.config
        log_phase("INIT", f"Goal: {cfg.goal[:60]}")
        log_phase(
            "INIT",
            f"Branches: {cfg.init_branches} | Turns: {cfg.turns_per_branch} | "
            f"Rounds: {rounds} | Scoring: {cfg.scoring_mode}",
        )
        self._emit(
            "search_started",
            {
                "goal": cfg.goal,
                "first_message": cfg.first_message,
                "total_rounds": rounds,
                "config": {
                    "init_branches": cfg.init_branches,
                    "turns_per_branch": cfg.turns_per_branch,
                    "user_intents_per_branch": cfg.user_intents_per_branch,
                    "scoring_mode": cfg.scoring_mode,
                    "prune_threshold": cfg.prune_threshold,
                },
            },
        )

        self._emit("phase", {"phase": "initializing", "message": "Creating tree structure..."})
        import time as _time

        _t0 = _time.perf_counter()
        if resume_from is not None:
            tree = self._load_tree(resume_from)
        else:
            tree = await self._initialize_tree()
        self.phase_times["init_s"] += _time.perf_counter() - _t0
        self._tree = tree

        total_pruned = 0
        for round_num in range(rounds):
            log_phase("ROUND", f"Round {round_num + 1}/{rounds}")
            self._emit("round_started", {"round": round_num + 1, "total_rounds": rounds})

            active_leaves = tree.active_leaves()
            expandable = [n for n in active_leaves if n.strategy is not None]
            if not expandable:
                logger.warning("No expandable nodes")
                break

            if cfg.user_intents_per_branch > 1 and cfg.user_variability:
                self._emit(
                    "phase",
                    {
                        "phase": "generating_intents",
                        "message": f"Generating {cfg.user_intents_per_branch} user intents per branch...",
                        "intents_per_branch": cfg.user_intents_per_branch,
                        "branch_count": len(expandable),
                    },
                )

            self._emit(
                "phase",
                {
                    "phase": "expanding",
                    "message": f"Expanding {len(expandable)} branches...",
                    "branch_count": len(expandable),
                    "turns_per_branch": cfg.turns_per_branch,
                },
            )

            # fixed persona short-circuits forking (ref engine.py:251-263)
            if cfg.user_variability:
                intents_per_node = cfg.user_intents_per_branch
                generate_intents_fn = self._generator.generate_intents
            else:
                intents_per_node = 1

                async def generate_intents_fn(_history, _count):  # type: ignore[misc]
                    return [FIXED_INTENT]

            _t0 = _time.perf_counter()
            expanded = await self._simulator.expand_nodes(
                expandable,
                turns=cfg.turns_per_branch,
                intents_per_node=intents_per_node,
                tree=tree,
                generate_intents=generate_intents_fn,
            )
            self.phase_times["expand_s"] += _time.perf_counter() - _t0
            log_phase("EXPAND", f"Completed {len(expanded)} expansions", indent=1)

            for node in expanded:
                self._emit("node_added", self._node_event(node))

            self._emit(
                "phase",
                {
                    "phase": "scoring",
                    "message": f"Scoring {len(expanded)} branches...",
                    "node_count": len(expanded),
                    "scoring_mode": cfg.scoring_mode,
                },
            )
            _t0 = _time.perf_counter()
            if cfg.scoring_mode == "comparative":
                scores = await self._evaluator.evaluate_comparative(expanded)
            else:
                scores = await self._evaluator.evaluate_absolute(expanded)
            self.phase_times["score_s"] += _time.perf_counter() - _t0

            for node in expanded:
                if node.id in scores:
                    score = scores[node.id]
                    log_phase(
                        "JUDGE",
                        f"'{node.strategy_label}': {score.aggregated_score:.1f}/10",
                        indent=1,
                    )
                    self._emit(
                        "node_updated",
                        {
                            "id": node.id,
                            "status": "scored",
                            "score": score.aggregated_score,
                            "individual_scores": score.individual_scores,
                            "passed": score.passed,
                        },
                    )

            for node in expanded:
                if node.id in scores:
                    tree.backpropagate(node.id, scores[node.id].aggregated_score)

            self._emit(
                "phase",
                {
                    "phase": "pruning",
                    "message": f"Pruning branches below {cfg.prune_threshold}...",
                    "threshold": cfg.prune_threshold,
                },
            )
            survivors = self._prune(expanded, scores)
            pruned_count = len(expanded) - len(survivors)
            total_pruned += pruned_count
            log_phase("PRUNE", f"Kept {len(survivors)}, pruned {pruned_count}", indent=1)

            pruned_nodes = [n for n in expanded if n.status == NodeStatus.PRUNED]
            if pruned_nodes:
                self._emit(
                    "nodes_pruned",
                    {
                        "ids": [n.id for n in pruned_nodes],
                        "reasons": {n.id: n.prune_reason for n in pruned_nodes},
                    },
                )
            self._emit(
                "token_update",
                {
                    "totals": {
                        "input_tokens": self._token_tracker.total_input_tokens,
                        "output_tokens": self._token_tracker.total_output_tokens,
                        "total_cost_usd": 0.0,
                    }
                },
            )
            self._save_checkpoint(tree, round_num + 1)

        best = tree.best_leaf_by_score()
        log_phase(
            "DONE",
            f"Best: '{best.strategy_label}' score {best.stats.aggregated_score:.1f}/10"
            if best
            else "No surviving branch",
        )
        logger.info("%s", self._token_tracker.summary_str())
        self._emit(
            "phase",
            {
                "phase": "complete",
                "message": "Search complete!",
                "best_score": best.stats.aggregated_score if best else 0.0,
                "best_strategy": best.strategy_label if best else None,
            },
        )
        return DTSRunResult(
            best_node_id=best.id if best else None,
            best_score=best.stats.aggregated_score if best else 0.0,
            best_messages=list(best.messages) if best else [],
            all_nodes=tree.all_nodes(),
            pruned_count=total_pruned,
            token_usage=self._token_tracker.to_dict(),
            total_rounds=rounds,
            research_report=self._research_report,
        )

    def _interim_result(self, tree: DialogueTree, rounds_done: int) -> DTSRunResult:
        best = tree.best_leaf_by_score()
        return DTSRunResult(
            best_node_id=best.id if best else None,
            best_score=best.stats.aggregated_score if best else 0.0,
            best_messages=list(best.messages) if best else [],
            all_nodes=tree.all_nodes(),
            pruned_count=sum(
                1 for n in tree.all_nodes() if n.status == NodeStatus.PRUNED
            ),
            token_usage=self._token_tracker.to_dict(),
            total_rounds=rounds_done,
            research_report=self._research_report,
        )

    def _save_checkpoint(self, tree: DialogueTree, rounds_done: int) -> None:
        """Atomic per-round tree-state write for crash recovery (no
        reference analogue — SURVEY.md §5 Checkpoint: final JSON only)."""
        path = self.config.checkpoint_path
        if not path:
            return
        if getattr(self, "dp", None) is not None and getattr(self.dp, "rank", 0) != 0:
            return  # identical trees on every rank; rank 0 writes
        import json as _json
        import os as _os

        try:
            d = self._interim_result(tree, rounds_done).to_exploration_dict()
            tmp = f"{path}.tmp"
            with open(tmp, "w") as f:
                _json.dump(d, f)
            _os.replace(tmp, path)
            logger.info("checkpoint after round %d -> %s", rounds_done, path)
        except Exception as e:  # noqa: BLE001 — never kill the search
            logger.warning("checkpoint write failed: %s", e)

    # ------------------------------------------------------------------
    async def _initialize_tree(self) -> DialogueTree:
        cfg = self.config
        from dts_amd.search.tree import derive_node_id

        root = DialogueNode(
            id=derive_node_id("dts-root", f"{cfg.goal}::{cfg.first_message}"),
            depth=0,
            messages=[Message.user(cfg.first_message)],
        )
        tree = DialogueTree.create(root)
        self._emit("node_added", self._node_event(root))

        deep_context = None
        if cfg.deep_research and self._researcher is not None:
            self._emit(
                "phase",
                {"phase": "researching", "message": "Conducting deep research on the topic..."},
            )
            try:
                deep_context = await self._researcher.research(
                    goal=cfg.goal, first_message=cfg.first_message
                )
            except Exception as e:  # noqa: BLE001
                logger.warning("Deep research failed: %s", e)
            self._research_report = deep_context
            if deep_context:
                self._evaluator.set_research_context(deep_context)

        self._emit(
            "phase",
            {
                "phase": "generating_strategies",
                "message": f"Generating {cfg.init_branches} conversation strategies...",
                "count": cfg.init_branches,
            },
        )
        strategies = await self._generator.generate_strategies(
            cfg.first_message, cfg.init_branches, deep_context
        )
        for i, strategy in enumerate(strategies, 1):
            log_phase("INIT", f"{i}. {strategy.tagline}", indent=1)
            self._emit(
                "strategy_generated",
                {
                    "index": i,
                    "total": len(strategies),
                    "tagline": strategy.tagline,
                    "description": strategy.description,
                },
            )

        from dts_amd.search.tree import derive_node_id

        for s_idx, strategy in enumerate(strategies):
            child = DialogueNode(
                id=derive_node_id(root.id, f"strategy:{s_idx}"),
                strategy=strategy,
                messages=[Message.user(cfg.first_message)],
            )
            tree.add_child(root.id, child)
            self._emit("node_added", self._node_event(child))
        return tree

    def _load_tree(self, checkpoint) -> DialogueTree:
        """Rebuild a DialogueTree from a saved exploration dict
        (to_exploration_dict output, possibly loaded from disk)."""
        try:
            return self._load_tree_inner(checkpoint)
        except Exception as e:  # noqa: BLE001 — malformed file, not a bug here
            raise ValueError(
                f"invalid checkpoint ({type(e).__name__}: {e}); expected "
                "a to_exploration_dict JSON with a 'branches' list"
            ) from e

    def _load_tree_inner(self, checkpoint) -> DialogueTree:
        import json as _json

        cfg = self.config
        if isinstance(checkpoint, str):
            with open(checkpoint) as f:
                checkpoint = _json.load(f)
        # recover the ORIGINAL root id: depth-1 branches carry it as their
        # parent_id, and ids derived from it must keep matching after
        # resume (resume == uninterrupted run; also every DP rank loads
        # the checkpoint independently and trees must match node-for-node)
        from dts_amd.search.tree import derive_node_id

        branch_ids = {b["id"] for b in checkpoint.get("branches", [])}
        root_id = next(
            (
                b["parent_id"]
                for b in checkpoint.get("branches", [])
                if b.get("parent_id") and b["parent_id"] not in branch_ids
            ),
            derive_node_id("dts-root", f"{cfg.goal}::{cfg.first_message}"),
        )
        root = DialogueNode(
            id=root_id,
            depth=0,
            messages=[Message.user(cfg.first_message)],
        )
        tree = DialogueTree.create(root)
        self._emit("node_added", self._node_event(root))
        self._research_report = checkpoint.get("research_report")
        if self._research_report:
            self._evaluator.set_research_context(self._research_report)
        branches = sorted(
            checkpoint.get("branches", []),
            key=lambda b: (b.get("depth", 1), b["id"]),
        )
        for b in branches:
            strat = b.get("strategy") or {}
            ui = b.get("user_intent")
            node = DialogueNode(
                id=b["id"],
                strategy=Strategy(
                    tagline=strat.get("tagline", ""),
                    description=strat.get("description", ""),
                ),
                user_intent=(
                    UserIntent(
                        id=ui.get("id", "intent_1"),
                        label=ui.get("label", ""),
                        description=ui.get("description", ""),
                        emotional_tone=ui.get("emotional_tone", ""),
                        cognitive_stance=ui.get("cognitive_stance", ""),
                    )
                    if ui
                    else None
                ),
                messages=[
                    Message(role=m["role"], content=m["content"])
                    for m in b.get("trajectory", [])
                ],
                status=NodeStatus(b.get("status", "active")),
                prune_reason=b.get("prune_reason"),
            )
            sc = b.get("scores") or {}
            node.stats.judge_scores = list(sc.get("individual") or [])
            node.stats.aggregated_score = sc.get("aggregated", 0.0) or 0.0
            node.stats.visits = sc.get("visits", 0) or 0
            node.stats.value_mean = sc.get("value_mean", 0.0) or 0.0
            node.stats.value_sum = node.stats.value_mean * node.stats.visits
            node.stats.critiques = dict(sc.get("critiques") or {})
            parent = b.get("parent_id")
            if not parent or parent not in tree.nodes:
                parent = root.id  # pre-extension checkpoints: flat under root
            tree.add_child(parent, node)
            self._emit("node_added", self._node_event(node))
        log_phase("INIT", f"Resumed {len(branches)} branches from checkpoint")
        return tree

    def _prune(self, nodes: list, scores: dict) -> list:
        """Threshold filter → top-K cap → min-survivors floor (ref engine.py:537-585)."""
        cfg = self.config
        if not nodes:
            return []

        survivors = [
            n
            for n in nodes
            if n.id in scores and scores[n.id].aggregated_score >= cfg.prune_threshold
        ]
        if cfg.keep_top_k and len(survivors) > cfg.keep_top_k:
            survivors.sort(key=lambda n: scores[n.id].aggregated_score, reverse=True)
            survivors = survivors[: cfg.keep_top_k]
        if len(survivors) < cfg.min_survivors:
            ranked = sorted(
                nodes,
                key=lambda n: scores.get(
                    n.id, AggregatedScore.zero(cfg.prune_threshold)
                ).aggregated_score,
                reverse=True,
            )
            survivors = ranked[: cfg.min_survivors]

        survivor_ids = {n.id for n in survivors}
        for n in nodes:
            if n.id not in survivor_ids:
                n.status = NodeStatus.PRUNED
                score = scores.get(n.id)
                n.prune_reason = (
                    f"score {score.aggregated_score:.1f} < {cfg.prune_threshold}"
                    if score
                    else "scoring failed"
                )
        return survivors

    # ------------------------------------------------------------------
    def _node_event(self, node: DialogueNode) -> dict:
        return {
            "id": node.id,
            "parent_id": node.parent_id,
            "depth": node.depth,
            "status": node.status.value,
            "strategy": node.strategy.tagline if node.strategy else None,
            "user_intent": node.intent_label,
            "message_count": len(node.messages),
        }

    def _emit(self, event_type: str, data: dict) -> None:
        if self._event_callback is not None:
            asyncio.create_task(
                emit_event(self._event_callback, event_type, data)
            )

    def _emit_async(self, event_type: str, data: dict) -> Any:
        self._emit(event_type, data)

    def _track_usage(self, completion: Completion, phase: str) -> None:
        if not completion.usage:
            return
        model = completion.model or self._token_tracker.model_name
        self._token_tracker.add_usage(model, completion.usage, phase)
