"""Tree-container tests (parity coverage of ref tests/core/dts/test_tree.py)."""

import pytest

from dts_amd.llm.types import Message
from dts_amd.search import (
    DialogueNode,
    DialogueTree,
    NodeStatus,
    generate_node_id,
)


def make_node(**kw):
    return DialogueNode(id=generate_node_id(), **kw)


@pytest.fixture
def tree():
    root = make_node(messages=[Message.user("hi")])
    return DialogueTree.create(root), root


class TestBasics:
    def test_create_and_get(self, tree):
        t, root = tree
        assert t.get_root() is root
        assert t.get(root.id) is root
        with pytest.raises(KeyError):
            t.get("missing")

    def test_add_child_sets_lineage(self, tree):
        t, root = tree
        child = make_node()
        t.add_child(root.id, child)
        assert child.parent_id == root.id
        assert child.depth == 1
        assert root.children == [child.id]

    def test_remove_node(self, tree):
        t, root = tree
        child = make_node()
        t.add_child(root.id, child)
        t.remove_node(child.id)
        assert child.id not in t.nodes
        assert root.children == []


class TestLeaves:
    def test_active_leaves_excludes_internal_and_pruned(self, tree):
        t, root = tree
        a, b = make_node(), make_node()
        t.add_child(root.id, a)
        t.add_child(root.id, b)
        b.status = NodeStatus.PRUNED
        leaves = t.active_leaves()
        assert leaves == [a]

    def test_best_leaf_by_score(self, tree):
        t, root = tree
        a, b = make_node(), make_node()
        t.add_child(root.id, a)
        t.add_child(root.id, b)
        a.stats.aggregated_score = 4.0
        b.stats.aggregated_score = 8.0
        assert t.best_leaf_by_score() is b

    def test_best_leaf_none_when_empty(self, tree):
        t, root = tree
        root.status = NodeStatus.PRUNED
        assert t.best_leaf_by_score() is None


class TestPaths:
    def test_path_to_root(self, tree):
        t, root = tree
        a = make_node()
        b = make_node()
        t.add_child(root.id, a)
        t.add_child(a.id, b)
        path = t.path_to_root(b.id)
        assert [n.id for n in path] == [b.id, a.id, root.id]
        assert [n.id for n in t.path_from_root(b.id)] == [root.id, a.id, b.id]


class TestBackprop:
    def test_backpropagate_updates_chain(self, tree):
        """Visits/value_sum/value_mean up to root (ref tree.py:109-120)."""
        t, root = tree
        a = make_node()
        b = make_node()
        t.add_child(root.id, a)
        t.add_child(a.id, b)
        t.backpropagate(b.id, 8.0)
        t.backpropagate(b.id, 4.0)
        for n in (b, a, root):
            assert n.stats.visits == 2
            assert n.stats.value_sum == 12.0
            assert n.stats.value_mean == 6.0


class TestPrune:
    def test_prune_subtree_counts(self, tree):
        t, root = tree
        a = make_node()
        b = make_node()
        c = make_node()
        t.add_child(root.id, a)
        t.add_child(a.id, b)
        t.add_child(a.id, c)
        count = t.prune_subtree(a.id, reason="bad")
        assert count == 3
        assert all(n.status == NodeStatus.PRUNED for n in (a, b, c))
        assert a.prune_reason == "bad"

    def test_statistics(self, tree):
        t, root = tree
        a = make_node()
        t.add_child(root.id, a)
        a.status = NodeStatus.PRUNED
        stats = t.statistics()
        assert stats["total_nodes"] == 2
        assert stats["pruned_nodes"] == 1
        assert stats["max_depth"] == 1
