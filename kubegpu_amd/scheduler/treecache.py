"""Node topology-tree cache.

Parity with the reference's scheduler-side node cache
(gpuschedulerplugin/gpu.go:129-245): a node's advertised resource list is
parsed into a canonical SortedTreeNode, scored, and deduplicated across
the cluster so request synthesis works per tree *shape*, not per node
(NodeCacheMap keyed by canonical tree, NodeLocationMap node->tree,
gpu.go:168-169).  Differences from the reference, deliberate:

* the cache is an explicit, lock-guarded object — the reference's
  package-global unsynchronized maps are called out as a hazard in
  SURVEY.md §5 ("the build should make the cache an explicit, locked
  component");
* alongside the shape we retain per-node *labeled* group layouts so the
  group-scheduler core can map canonical tree positions back to concrete
  groups (the reference split this across two repos).
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..api.resource import CARDS_RE
from ..plugintypes import (
    SortedTreeNode,
    add_node_to_sorted_tree_node,
    log_tree_node,
)


@dataclass
class LabeledLayout:
    """Per-node concrete group layout, in canonical order.

    groups[hi] = (h_label, [(g_label, [gpu ids...]), ...])
    hi / gi index positions in the canonical (sorted) tree, so a
    synthesized request naming position (hi, gi) resolves to the concrete
    group labels (and member GPU ids) here.
    """

    groups: List[Tuple[str, List[Tuple[str, List[str]]]]] = field(default_factory=list)

    def total(self) -> int:
        return sum(len(ids) for _, l0 in self.groups for _, ids in l0)


def parse_node_resources(resources: Dict[str, int]) -> Tuple[Optional[SortedTreeNode], LabeledLayout]:
    """Parse advertised `/cards` names into (canonical tree, labeled layout).

    Reference analog: addToNode's regex parse (gpu.go:129-161) with the
    `.*/gpugrp<L>/(.*?)/.*/cards` grammar.  Names that do not match the
    2-level grammar are ignored (garbage nodes yield an empty tree, cf.
    the garbage-node case in gpu_test.go:46).
    """
    # h_label -> g_label -> [gpu ids]
    found: Dict[str, Dict[str, List[str]]] = {}
    for name, val in resources.items():
        m = CARDS_RE.match(name)
        if not m or val <= 0:
            continue
        found.setdefault(m.group("h"), {}).setdefault(m.group("g"), []).append(m.group("id"))
    if not found:
        return None, LabeledLayout()

    # canonical order: (-val, label) at each level; scores are attached
    # after the shape exists.
    layout = LabeledLayout()
    h_items = []
    for h_label, g_map in found.items():
        g_items = sorted(
            ((g_label, sorted(ids)) for g_label, ids in g_map.items()),
            key=lambda t: (-len(t[1]), t[0]),
        )
        h_items.append((h_label, g_items, sum(len(ids) for _, ids in g_items)))
    h_items.sort(key=lambda t: (-t[2], t[0]))

    root = SortedTreeNode(val=0)
    for h_label, g_items, h_val in h_items:
        h_node = SortedTreeNode(val=h_val)
        for _, ids in g_items:
            add_node_to_sorted_tree_node(h_node, SortedTreeNode(val=len(ids)))
        add_node_to_sorted_tree_node(root, h_node)
        layout.groups.append((h_label, [(g_label, ids) for g_label, ids in g_items]))
        root.val += h_val
    compute_tree_score(root)
    return root, layout


def compute_tree_score(node: SortedTreeNode) -> float:
    """Score a tree: denser, deeper grouping scores higher.

    EXACT reference recursion (computeTreeScoreAtLevel, gpu.go:180-190):

        score(n, level, numChild_of_parent) =
            n.val * level / numChild_of_parent
            + Σ score(child, level + 1, len(n.children))

    entered at the root with level=0 and numChild=len(root.children)
    (so the root's own term is 0).  Deeper levels weigh more and each
    term is divided by its sibling count — fewer, larger groups
    (better interconnect locality) outrank a fragmented shape of equal
    size.  Each node's .score records its own subtree contribution.
    """

    def at_level(n: SortedTreeNode, level: int, num_child: int) -> float:
        s = (n.val * level) / num_child if num_child else 0.0
        for c in n.children:
            s += at_level(c, level + 1, len(n.children))
        n.score = s
        return s

    return at_level(node, 0, len(node.children) or 1)


def tree_key(node: Optional[SortedTreeNode]) -> str:
    """Canonical serialization used as the dedup cache key."""
    if node is None:
        return "nil"
    inner = ",".join(tree_key(c) for c in node.children)
    return f"({node.val}:[{inner}])"


class NodeTreeCache:
    """Lock-guarded cluster-wide canonical-tree cache."""

    def __init__(self) -> None:
        self._lock = threading.RLock()
        self.node_cache_map: Dict[str, SortedTreeNode] = {}  # key -> tree
        self.node_location_map: Dict[str, str] = {}  # node name -> key
        self._layouts: Dict[str, LabeledLayout] = {}  # node name -> layout
        self._refcount: Dict[str, int] = {}

    def add_node_resources(self, node_name: str, resources: Dict[str, int]) -> Optional[SortedTreeNode]:
        """Parse + canonicalize + dedup (gpu.go:192-224)."""
        tree, layout = parse_node_resources(resources)
        with self._lock:
            old_key = self.node_location_map.get(node_name)
            if tree is None:
                # garbage / GPU-less node: drop any previous registration
                if old_key is not None:
                    self._release(old_key)
                    del self.node_location_map[node_name]
                    self._layouts.pop(node_name, None)
                return None
            key = tree_key(tree)
            if old_key is not None and old_key != key:
                self._release(old_key)
            if old_key != key:
                if key not in self.node_cache_map:
                    self.node_cache_map[key] = tree
                    self._refcount[key] = 0
                    log_tree_node(4, f"new canonical tree for {node_name}", tree)
                self._refcount[key] += 1
                self.node_location_map[node_name] = key
            self._layouts[node_name] = layout
            return self.node_cache_map[key]

    def _release(self, key: str) -> None:
        self._refcount[key] -= 1
        if self._refcount[key] <= 0:
            del self._refcount[key]
            self.node_cache_map.pop(key, None)

    def remove_node(self, node_name: str) -> None:
        """gpu.go:226-230."""
        with self._lock:
            key = self.node_location_map.pop(node_name, None)
            self._layouts.pop(node_name, None)
            if key is not None:
                self._release(key)

    def find_best_tree(self, num_gpus: int) -> Optional[SortedTreeNode]:
        """Highest-scoring cached tree with val >= num_gpus
        (gpu.go:232-245)."""
        with self._lock:
            best: Optional[SortedTreeNode] = None
            for tree in self.node_cache_map.values():
                if tree.val < num_gpus:
                    continue
                if best is None or tree.score > best.score:
                    best = tree
            return best

    def layout_for(self, node_name: str) -> Optional[LabeledLayout]:
        with self._lock:
            return self._layouts.get(node_name)

    def tree_for(self, node_name: str) -> Optional[SortedTreeNode]:
        with self._lock:
            key = self.node_location_map.get(node_name)
            return self.node_cache_map.get(key) if key is not None else None

    def __len__(self) -> int:
        with self._lock:
            return len(self.node_cache_map)
