"""Node-tree cache tests (analog of gpuschedulerplugin/gpu_test.go:13-58).

Three synthetic 8-GPU nodes with different gpugrp shapes plus a garbage
node, exercising parse/canonicalize/score, dedup, and removal.
"""

from kubegpu_amd.scheduler import (
    NodeTreeCache,
    parse_node_resources,
    tree_key,
)


def _node_resources(shape):
    """shape: {h_label: {g_label: n_gpus}}"""
    res = {}
    k = 0
    for h, gs in shape.items():
        for g, n in gs.items():
            for _ in range(n):
                res[f"resource/group/gpugrp1/{h}/gpugrp0/{g}/gpu/GPU{k}/cards"] = 1
                res[f"resource/group/gpugrp1/{h}/gpugrp0/{g}/gpu/GPU{k}/memory"] = 1 << 30
                k += 1
    return res


DENSE_8 = {"0": {"0": 8}}  # one full hive
SPLIT_4_4 = {"0": {"0": 4, "1": 4}}  # two 4-GPU groups
SPLIT_2222 = {"0": {"0": 2, "1": 2}, "1": {"2": 2, "3": 2}}  # fragmented


def test_parse_builds_sorted_tree():
    tree, layout = parse_node_resources(_node_resources({"0": {"1": 2, "0": 6}}))
    assert tree.val == 8
    assert [c.val for c in tree.children[0].children] == [6, 2]
    # layout order follows the canonical (descending) order
    assert [len(ids) for _, ids in layout.groups[0][1]] == [6, 2]


def test_garbage_node_ignored():
    tree, layout = parse_node_resources({"cpu": 4, "memory": 1 << 35, "junk/name": 1})
    assert tree is None
    assert layout.total() == 0


def test_denser_tree_scores_higher():
    dense, _ = parse_node_resources(_node_resources(DENSE_8))
    split, _ = parse_node_resources(_node_resources(SPLIT_4_4))
    frag, _ = parse_node_resources(_node_resources(SPLIT_2222))
    assert dense.score > split.score > frag.score


def test_cache_dedup_and_remove():
    cache = NodeTreeCache()
    cache.add_node_resources("n1", _node_resources(DENSE_8))
    cache.add_node_resources("n2", _node_resources(DENSE_8))  # same shape
    cache.add_node_resources("n3", _node_resources(SPLIT_4_4))
    cache.add_node_resources("garbage", {"cpu": 4})
    assert len(cache) == 2  # deduped; garbage ignored
    assert cache.node_location_map["n1"] == cache.node_location_map["n2"]

    cache.remove_node("n1")
    assert len(cache) == 2  # n2 still holds the dense shape
    cache.remove_node("n2")
    assert len(cache) == 1  # dense shape gone


def test_find_best_tree_prefers_dense():
    cache = NodeTreeCache()
    cache.add_node_resources("dense", _node_resources(DENSE_8))
    cache.add_node_resources("split", _node_resources(SPLIT_4_4))
    best = cache.find_best_tree(3)
    assert tree_key(best) == cache.node_location_map["dense"]
    # remove the dense node: the split shape is now the best 3-GPU host
    cache.remove_node("dense")
    best = cache.find_best_tree(3)
    assert tree_key(best) == cache.node_location_map["split"]
    # nothing can hold 9 GPUs
    assert cache.find_best_tree(9) is None


def test_node_shape_change_rehomes():
    cache = NodeTreeCache()
    cache.add_node_resources("n", _node_resources(DENSE_8))
    key1 = cache.node_location_map["n"]
    cache.add_node_resources("n", _node_resources(SPLIT_4_4))
    key2 = cache.node_location_map["n"]
    assert key1 != key2
    assert len(cache) == 1  # old shape garbage-collected


def test_score_formula_matches_reference_exactly():
    """Pin the exact reference recursion (computeTreeScoreAtLevel,
    gpu.go:180-190): score = Σ val·level/numChild_of_parent over the
    tree, root entered at level 0.

    Hand-computed: one 8-dense gpugrp0 → 8·1/1 + 8·2/1 = 24;
    two 4-hives → 8·1/1 + 2·(4·2/2) = 16;
    four 2-pairs → 8·1/1 + 4·(2·2/4) = 12."""
    def res_for(group_sizes):
        res, g = {}, 0
        for gi, size in enumerate(group_sizes):
            for _ in range(size):
                res[f"resource/group/gpugrp1/0/gpugrp0/{gi}/gpu/G{g}/cards"] = 1
                g += 1
        return res

    dense, _ = parse_node_resources(res_for([8]))
    split, _ = parse_node_resources(res_for([4, 4]))
    frag, _ = parse_node_resources(res_for([2, 2, 2, 2]))
    assert dense.score == 24.0
    assert split.score == 16.0
    assert frag.score == 12.0
