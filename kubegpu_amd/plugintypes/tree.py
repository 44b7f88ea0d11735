"""SortedTreeNode — canonical topology-tree shape.

Behavioral parity with /root/reference/gpuplugintypes/typeutils.go
(insertion point :10-23, add-with-score :27-31, add-node :33-36,
compare :75-93, print/log :38-72), re-implemented from the surveyed
semantics:

* ``val``   = number of leaf GPUs under the node
* ``score`` = tie-breaker used when two children have equal val
* children are kept sorted in **descending** (val, score) order, so a
  left-to-right DFS visits densest/highest-scoring groups first — which is
  exactly the packing order the request synthesizer relies on
  (gpu.go:247-271).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

from ..api import utils


@dataclass
class SortedTreeNode:
    val: int = 0
    score: float = 0.0
    children: List["SortedTreeNode"] = field(default_factory=list)

    # -- structure ---------------------------------------------------------

    def leaf_count(self) -> int:
        if not self.children:
            return self.val
        return sum(c.leaf_count() for c in self.children)


def _find_insertion_point(parent: SortedTreeNode, val: int, score: float) -> int:
    """Index at which a child with (val, score) keeps descending order.

    Equal keys insert after existing equals (stable).
    """
    i = 0
    for i, c in enumerate(parent.children):
        if (val, score) > (c.val, c.score):
            return i
    return len(parent.children)


def add_to_sorted_tree_node_with_score(
    parent: SortedTreeNode, val: int, score: float
) -> SortedTreeNode:
    """Create a child with (val, score), insert in sorted position."""
    child = SortedTreeNode(val=val, score=score)
    parent.children.insert(_find_insertion_point(parent, val, score), child)
    return child


def add_to_sorted_tree_node(parent: SortedTreeNode, val: int) -> SortedTreeNode:
    """Create a child with val (score 0), insert in sorted position."""
    return add_to_sorted_tree_node_with_score(parent, val, 0.0)


def add_node_to_sorted_tree_node(parent: SortedTreeNode, node: SortedTreeNode) -> None:
    """Insert an existing subtree in sorted position."""
    parent.children.insert(_find_insertion_point(parent, node.val, node.score), node)


def compare_tree_node(a: Optional[SortedTreeNode], b: Optional[SortedTreeNode]) -> bool:
    """Structural equality: val + recursively ordered children.

    Scores are deliberately NOT compared — exactly the reference
    (CompareTreeNode, typeutils.go:75-93, checks Val and Child only):
    score is derived from the shape, so two identically-shaped trees
    are the same canonical tree even mid-rescoring."""
    if a is None or b is None:
        return a is b
    if a.val != b.val:
        return False
    if len(a.children) != len(b.children):
        return False
    return all(compare_tree_node(x, y) for x, y in zip(a.children, b.children))


def print_tree_node(node: Optional[SortedTreeNode], indent: int = 0) -> str:
    """Pretty-print a tree to a string."""
    if node is None:
        return " " * indent + "<nil>\n"
    out = " " * indent + f"(val={node.val} score={node.score:g})\n"
    for c in node.children:
        out += print_tree_node(c, indent + 2)
    return out


def log_tree_node(level: int, msg: str, node: Optional[SortedTreeNode]) -> None:
    """Log a tree at the given verbosity level (typeutils.go:66-72)."""
    if utils.logb(level):
        utils.logf(level, "%s:\n%s", msg, print_tree_node(node))
