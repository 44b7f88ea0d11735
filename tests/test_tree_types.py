"""Tree-type tests (analog of gpuplugintypes/typeutils_test.go:7-34)."""

from kubegpu_amd.plugintypes import (
    SortedTreeNode,
    add_node_to_sorted_tree_node,
    add_to_sorted_tree_node,
    add_to_sorted_tree_node_with_score,
    compare_tree_node,
    print_tree_node,
)


def test_insertion_descending_order():
    root = SortedTreeNode(val=10)
    add_to_sorted_tree_node(root, 2)
    add_to_sorted_tree_node(root, 8)
    add_to_sorted_tree_node(root, 4)
    assert [c.val for c in root.children] == [8, 4, 2]


def test_insertion_score_tiebreak():
    root = SortedTreeNode(val=10)
    add_to_sorted_tree_node_with_score(root, 4, 1.0)
    add_to_sorted_tree_node_with_score(root, 4, 3.0)
    add_to_sorted_tree_node_with_score(root, 4, 2.0)
    assert [(c.val, c.score) for c in root.children] == [
        (4, 3.0),
        (4, 2.0),
        (4, 1.0),
    ]


def test_insertion_stable_for_equal_keys():
    root = SortedTreeNode()
    a = SortedTreeNode(val=4, score=1.0)
    b = SortedTreeNode(val=4, score=1.0)
    add_node_to_sorted_tree_node(root, a)
    add_node_to_sorted_tree_node(root, b)
    assert root.children[0] is a and root.children[1] is b


def test_compare_tree_node():
    def mk():
        root = SortedTreeNode(val=8)
        h = add_to_sorted_tree_node(root, 8)
        add_to_sorted_tree_node(h, 4)
        add_to_sorted_tree_node(h, 4)
        return root

    assert compare_tree_node(mk(), mk())
    other = mk()
    other.children[0].children[0].val = 3
    assert not compare_tree_node(mk(), other)
    assert not compare_tree_node(mk(), None)
    assert compare_tree_node(None, None)


def test_print_tree_node_renders():
    root = SortedTreeNode(val=2)
    add_to_sorted_tree_node(root, 2)
    out = print_tree_node(root)
    assert "val=2" in out


def test_compare_ignores_score_like_reference():
    """CompareTreeNode (typeutils.go:75-93) checks Val and Child only —
    score is derived, so identically-shaped trees compare equal even
    with different scores."""
    a = SortedTreeNode(val=4, score=1.0)
    b = SortedTreeNode(val=4, score=99.0)
    add_to_sorted_tree_node(a, 4)
    add_to_sorted_tree_node(b, 4)
    assert compare_tree_node(a, b)
