"""SPMD data-parallel DTS engine (branch sharding across GPUs).

Every rank runs the identical search loop; expansion and judging tasks are
executed only on their owner rank (dts_amd/parallel/dp.py) against the
rank-local serving engine, and merged at phase boundaries. All ranks end
each round with identical trees, so prune/backprop/best-leaf need no
coordination. See parallel/dp.py for the ordering rules.
"""

from __future__ import annotations

from typing import Optional

from dts_amd.parallel.dp import (
    DPContext,
    apply_payload,
    node_to_payload,
    payload_to_score,
    score_to_payload,
)
from dts_amd.search.engine import DTSEngine
from dts_amd.search.tree import DialogueTree, generate_node_id
from dts_amd.search.types import DialogueNode, Strategy
from dts_amd.utils.logging import log_phase

# per-rank lenses for sharded strategy generation: each rank generates its
# slice of the initial branches through a distinct angle, so the union
# stays diverse without the single serial mega-call (one 6*world-entry
# form decoded on rank 0 while the other GPUs idle was the Amdahl
# bottleneck of the weak-scaling bench). Shared with the within-rank
# split mode (generator.strategy_split).
from dts_amd.search.generator import DIVERSITY_ANGLES as _DIVERSITY_ANGLES


class _DPSimulator:
    """Shards parent nodes round-robin; gathers full node payloads."""

    def __init__(self, sim, dp: DPContext, tree_ref) -> None:
        self.sim = sim
        self.dp = dp
        self._tree_ref = tree_ref  # callable returning current tree

    async def expand_nodes(self, nodes, turns, intents_per_node=1, tree=None,
                           generate_intents=None):
        if not self.dp.enabled:
            return await self.sim.expand_nodes(
                nodes, turns, intents_per_node, tree, generate_intents
            )
        ordered = sorted(nodes, key=lambda n: n.id)
        mine = [n for i, n in enumerate(ordered) if self.dp.owns(i)]
        expanded_local = await self.sim.expand_nodes(
            mine, turns, intents_per_node, tree, generate_intents
        )
        payloads = [node_to_payload(n) for n in expanded_local]
        gathered = self.dp.all_gather_obj(payloads)
        merged: list = []
        all_payloads = [p for rank_list in gathered for p in rank_list]
        all_payloads.sort(key=lambda p: (p.parent_id or "", p.node_id))
        for p in all_payloads:
            node = apply_payload(tree, p)
            merged.append(node)
        return merged


class _DPEvaluator:
    def __init__(self, ev, dp: DPContext) -> None:
        self.ev = ev
        self.dp = dp

    async def evaluate_absolute(self, nodes):
        return await self._evaluate(nodes, "absolute")

    async def evaluate_comparative(self, nodes):
        return await self._evaluate(nodes, "comparative")

    def set_research_context(self, ctx):
        self.ev.set_research_context(ctx)

    async def _evaluate(self, nodes, mode):
        if not self.dp.enabled:
            fn = (
                self.ev.evaluate_comparative
                if mode == "comparative"
                else self.ev.evaluate_absolute
            )
            return await fn(nodes)
        ordered = sorted(nodes, key=lambda n: n.id)
        if mode == "comparative":
            # shard ranking CHUNKS (sibling groups, oversized ones split by
            # the evaluator) so each forced ranking stays on one rank
            import asyncio as _asyncio

            chunks = self.ev.comparative_chunks(ordered)
            tasks = []
            for i, (parent_id, chunk) in enumerate(chunks):
                if not self.dp.owns(i):
                    continue
                if len(chunk) == 1:
                    tasks.append(self.ev._judge_single_wrapped(chunk[0]))
                else:
                    tasks.append(self.ev._judge_group_comparative(parent_id, chunk))
            local_scores = {}
            if tasks:
                for result in await _asyncio.gather(*tasks, return_exceptions=True):
                    if isinstance(result, dict):
                        local_scores.update(result)
        else:
            mine = [n for i, n in enumerate(ordered) if self.dp.owns(i)]
            local_scores = await self.ev.evaluate_absolute(mine) if mine else {}

        by_id = {n.id: n for n in ordered}
        payloads = [
            score_to_payload(nid, agg, by_id[nid].stats.critiques)
            for nid, agg in local_scores.items()
        ]
        gathered = self.dp.all_gather_obj(payloads)
        scores: dict = {}
        threshold = self.ev.prune_threshold
        flat = [p for rank_list in gathered for p in rank_list]
        flat.sort(key=lambda p: p.node_id)
        for p in flat:
            agg = payload_to_score(p, threshold)
            scores[p.node_id] = agg
            node = by_id.get(p.node_id)
            if node is not None:
                node.stats.judge_scores = agg.individual_scores
                node.stats.aggregated_score = agg.aggregated_score
                if p.critiques:
                    node.stats.critiques = p.critiques
        return scores


class DistributedDTSEngine(DTSEngine):
    """DTSEngine with DP branch sharding over a torch.distributed group."""

    def __init__(self, llm, config, dp: Optional[DPContext] = None,
                 researcher: Optional[object] = None) -> None:
        super().__init__(llm, config, researcher=researcher)
        self.dp = dp or DPContext()
        self._simulator = _DPSimulator(self._simulator, self.dp, lambda: self.tree)
        self._evaluator = _DPEvaluator(self._evaluator, self.dp)

    async def _initialize_tree(self) -> DialogueTree:
        if not self.dp.enabled:
            return await super()._initialize_tree()
        import os as _os

        cfg = self.config
        if (
            self.dp.world >= 2
            and cfg.init_branches >= self.dp.world
            and cfg.init_branches % self.dp.world == 0
            and _os.environ.get("DTS_SHARDED_INIT", "1") != "0"
        ):
            return await self._initialize_tree_sharded()
        return await self._initialize_tree_rank0()

    async def _initialize_tree_sharded(self) -> DialogueTree:
        """Every rank generates init_branches/world strategies on its own
        GPU through a distinct diversity lens; an object all-gather plus
        deterministic child ids (uuid5 over root:index) rebuilds the same
        tree everywhere. Replaces the serial rank-0 mega-form."""
        import uuid as _uuid

        from dts_amd.llm.types import Message

        cfg = self.config
        world, rank = self.dp.world, self.dp.rank
        if rank == 0:
            root_id = generate_node_id()
            deep = None
            if cfg.deep_research and self._researcher is not None:
                try:
                    deep = await self._researcher.research(
                        goal=cfg.goal, first_message=cfg.first_message
                    )
                except Exception:  # noqa: BLE001
                    deep = None
            meta = (root_id, deep)
        else:
            meta = None
        root_id, deep = self.dp.broadcast_obj(meta, src=0)
        self._research_report = deep
        if deep:
            self._evaluator.set_research_context(deep)
        root = DialogueNode(
            id=root_id, depth=0, messages=[Message.user(cfg.first_message)]
        )
        tree = DialogueTree.create(root)
        self._emit("node_added", self._node_event(root))

        local_n = cfg.init_branches // world
        angle = _DIVERSITY_ANGLES[rank % len(_DIVERSITY_ANGLES)]
        if getattr(self._generator, "strategy_split", False):
            # within-rank split mode already applies per-call lenses —
            # rotate them by rank so the global lens set stays disjoint
            self._generator.lens_offset = rank * local_n
            ctx = (f"{deep}" if deep else None)
        else:
            ctx = (f"{deep}\n\n" if deep else "") + (
                "Diversity constraint: approach ALL of your strategies through "
                f"the lens of {angle}; other strategy sets cover other lenses."
            )
        strategies = await self._generator.generate_strategies(
            cfg.first_message, local_n, ctx
        )
        gathered = self.dp.all_gather_obj(
            [(st.tagline, st.description) for st in strategies]
        )
        total = sum(len(p) for p in gathered)
        idx = 0
        for plist in gathered:
            for tagline, desc in plist:
                self._emit(
                    "strategy_generated",
                    {
                        "index": idx + 1,
                        "total": total,
                        "tagline": tagline,
                        "description": desc,
                    },
                )
                child = DialogueNode(
                    id=str(
                        _uuid.uuid5(
                            _uuid.NAMESPACE_URL, f"{root_id}:strategy:{idx}"
                        )
                    ),
                    strategy=Strategy(tagline=tagline, description=str(desc)),
                    messages=[Message.user(cfg.first_message)],
                )
                tree.add_child(root.id, child)
                self._emit("node_added", self._node_event(child))
                idx += 1
        log_phase(
            "INIT",
            f"[rank {rank}] sharded strategy init: {idx} branches "
            f"({local_n}/rank, lens: {angle})",
        )
        return tree

    async def _initialize_tree_rank0(self) -> DialogueTree:
        if self.dp.rank == 0:
            tree = await super()._initialize_tree()
            root = tree.get_root()
            payloads = [node_to_payload(root)] + [
                node_to_payload(tree.get(cid)) for cid in root.children
            ]
        else:
            payloads = None
        payloads = self.dp.broadcast_obj(payloads, src=0)
        if self.dp.rank != 0:
            root_p = payloads[0]
            from dts_amd.llm.types import Message

            root = DialogueNode(
                id=root_p.node_id,
                depth=0,
                messages=[Message(role=r, content=c) for r, c in root_p.messages],
            )
            tree = DialogueTree.create(root)
            for p in payloads[1:]:
                apply_payload(tree, p)
            log_phase(
                "INIT",
                f"[rank {self.dp.rank}] mirrored tree with "
                f"{len(payloads) - 1} branches",
            )
        return tree
