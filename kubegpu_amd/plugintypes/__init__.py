"""Shared plugin types: the sorted resource tree.

Parity with the reference's gpuplugintypes package
(/root/reference/gpuplugintypes/types.go:6-13, typeutils.go:10-93):
``RESOURCE_GPU`` is the flat GPU resource name seen by the stock kubelet
(amd.com/gpu here, nvidia.com/gpu there), and ``SortedTreeNode`` is the
canonical shape of a node's advertised topology tree — children kept in
descending (val, score) order so that walking a tree left-to-right packs
the densest / best-connected groups first.
"""

from .tree import (  # noqa: F401
    SortedTreeNode,
    add_node_to_sorted_tree_node,
    add_to_sorted_tree_node,
    add_to_sorted_tree_node_with_score,
    compare_tree_node,
    print_tree_node,
    log_tree_node,
)

# Flat GPU resource name (reference: ResourceGPU = "nvidia.com/gpu",
# gpuplugintypes/types.go:6).  MI355X nodes advertise amd.com/gpu.
RESOURCE_GPU = "amd.com/gpu"
