from dts_amd.search.config import DTSConfig
from dts_amd.search.types import (
    AggregatedScore,
    DialogueNode,
    DTSRunResult,
    NodeStats,
    NodeStatus,
    Strategy,
    TokenTracker,
    UserIntent,
)
from dts_amd.search.tree import DialogueTree, generate_node_id
from dts_amd.search.aggregator import aggregate_majority_vote
from dts_amd.search.engine import DTSEngine

__all__ = [
    "DTSConfig",
    "DTSEngine",
    "DialogueTree",
    "DialogueNode",
    "DTSRunResult",
    "NodeStats",
    "NodeStatus",
    "Strategy",
    "UserIntent",
    "AggregatedScore",
    "TokenTracker",
    "aggregate_majority_vote",
    "generate_node_id",
]
