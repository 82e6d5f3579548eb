"""Data-parallel branch sharding for the search engine.

BASELINE.json config 3 / SURVEY.md §2.4: each of N ranks holds a full
model replica (8B fits easily in 288 GB); ONE logical search is executed
SPMD — every rank runs the same engine loop, but each expansion / judging
task is executed only by its owner rank (round-robin by deterministic task
index) on its local serving engine. Results are exchanged with
`all_gather_object` at the phase barrier and applied identically
everywhere, so all ranks hold identical trees without any central
coordinator. Inference-only DP needs no gradient collectives (SURVEY.md
§2.3) — the only traffic is these per-phase object gathers, which ride
gloo on CPU tests and RCCL/xGMI on the node.

Ordering rules that make SPMD safe:
  - task lists are sorted by node id before sharding;
  - collectives happen at fixed phase boundaries (expand end, score end),
    never inside the asyncio fan-out;
  - non-owner ranks receive full node payloads (id + messages + status),
    so owner-generated UUIDs never need cross-rank determinism.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Optional

import torch.distributed as dist


class DPContext:
    """Shard ownership + object collectives over the default group."""

    def __init__(self, group=None) -> None:
        self.group = group
        if dist.is_available() and dist.is_initialized():
            self.rank = dist.get_rank(group)
            self.world = dist.get_world_size(group)
        else:
            self.rank = 0
            self.world = 1

    @property
    def enabled(self) -> bool:
        return self.world > 1

    def owns(self, index: int) -> bool:
        return index % self.world == self.rank

    def all_gather_obj(self, obj: Any) -> list:
        if not self.enabled:
            return [obj]
        out: list = [None] * self.world
        dist.all_gather_object(out, obj, group=self.group)
        return out

    def broadcast_obj(self, obj: Any, src: int = 0) -> Any:
        if not self.enabled:
            return obj
        box = [obj if self.rank == src else None]
        dist.broadcast_object_list(box, src=src, group=self.group)
        return box[0]

    def barrier(self) -> None:
        if self.enabled:
            dist.barrier(group=self.group)


# ---------------------------------------------------------------------------
# Node payloads shipped between ranks
# ---------------------------------------------------------------------------

@dataclass
class NodePayload:
    """Everything a non-owner rank needs to mirror an expanded node."""

    node_id: str
    parent_id: Optional[str]
    status: str
    prune_reason: Optional[str]
    messages: list  # [(role, content)]
    strategy: Optional[tuple]  # (tagline, description)
    user_intent: Optional[tuple]  # (id, label, desc, tone, stance)
    is_new_child: bool = False


def node_to_payload(node, is_new_child: bool = False) -> NodePayload:
    return NodePayload(
        node_id=node.id,
        parent_id=node.parent_id,
        status=node.status.value,
        prune_reason=node.prune_reason,
        messages=[(m.role, m.content) for m in node.messages],
        strategy=(node.strategy.tagline, node.strategy.description)
        if node.strategy
        else None,
        user_intent=(
            node.user_intent.id,
            node.user_intent.label,
            node.user_intent.description,
            node.user_intent.emotional_tone,
            node.user_intent.cognitive_stance,
        )
        if node.user_intent
        else None,
        is_new_child=is_new_child,
    )


def apply_payload(tree, payload: NodePayload):
    """Create or update the node described by `payload` in the local tree.

    Returns the node.
    """
    from dts_amd.llm.types import Message
    from dts_amd.search.types import DialogueNode, NodeStatus, Strategy, UserIntent

    messages = [Message(role=r, content=c) for r, c in payload.messages]
    if payload.node_id in tree.nodes:
        node = tree.get(payload.node_id)
        node.messages = messages
        node.status = NodeStatus(payload.status)
        node.prune_reason = payload.prune_reason
        return node
    node = DialogueNode(
        id=payload.node_id,
        status=NodeStatus(payload.status),
        prune_reason=payload.prune_reason,
        messages=messages,
        strategy=Strategy(*payload.strategy) if payload.strategy else None,
        user_intent=UserIntent(*payload.user_intent) if payload.user_intent else None,
    )
    if payload.parent_id is not None:
        tree.add_child(payload.parent_id, node)
    else:
        tree.add_node(node)
    return node


@dataclass
class ScorePayload:
    node_id: str
    individual_scores: list
    aggregated_score: float
    pass_votes: int
    passed: bool
    critiques: dict


def score_to_payload(node_id, agg, critiques) -> ScorePayload:
    return ScorePayload(
        node_id=node_id,
        individual_scores=list(agg.individual_scores),
        aggregated_score=agg.aggregated_score,
        pass_votes=agg.pass_votes,
        passed=agg.passed,
        critiques=critiques or {},
    )


def payload_to_score(p: ScorePayload, threshold: float):
    from dts_amd.search.types import AggregatedScore

    return AggregatedScore(
        individual_scores=p.individual_scores,
        aggregated_score=p.aggregated_score,
        pass_threshold=threshold,
        pass_votes=p.pass_votes,
        passed=p.passed,
    )
