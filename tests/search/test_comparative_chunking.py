import pytest
from dts_amd.llm import LLM, FakeBackend
from dts_amd.search.evaluator import TrajectoryEvaluator
from dts_amd.search.types import DialogueNode, Strategy
from dts_amd.llm.types import Message

def make_nodes(n, parent="p"):
    return [DialogueNode(id=f"n{i:03d}", parent_id=parent,
                         strategy=Strategy(tagline="t", description="d"),
                         messages=[Message.user("q"), Message.assistant("a")])
            for i in range(n)]

def test_chunking_splits_and_merges_tail():
    ev = TrajectoryEvaluator(LLM(FakeBackend(), default_model="f"), goal="g")
    nodes = make_nodes(17)
    chunks = ev.comparative_chunks(nodes)
    sizes = [len(c) for _, c in chunks]
    assert sum(sizes) == 17
    assert all(2 <= s <= 9 for s in sizes)  # no singletons from splitting

def test_large_group_ranked_in_chunks(run_async):
    ev = TrajectoryEvaluator(LLM(FakeBackend(), default_model="f"), goal="g",
                             prune_threshold=0.0)
    nodes = make_nodes(20)
    scores = run_async(ev.evaluate_comparative(nodes))
    assert set(scores) == {n.id for n in nodes}
    assert all(len(s.individual_scores) == 3 for s in scores.values())
