"""Leaf-path CSR extraction from fitted device forests.

The leaf-path TreeSHAP kernel (ops/hip/treeshap.hip, treeshap_paths_kernel)
wants, for every leaf of every tree, the root->leaf node-id path.  This
builds the CSR (leaf_tree, leaf_off, path_nodes) from the forest's SoA node
arrays with vectorized level-synchronous parent walks.
"""

import numpy as np


def build_leaf_paths(nfeat, nleft, j_node_off, node_alloc):
    """nfeat/nleft: host int32 arrays over all trees' node slots;
    j_node_off: int64 per-tree bases; node_alloc: allocated counts.
    Returns (leaf_tree i32[L], leaf_off i32[L+1], path_nodes i32[P],
    max_depth) with node ids LOCAL to their tree."""
    n_trees = len(node_alloc)

    leaf_tree_parts, leaf_local_parts = [], []
    parent_parts = []   # per-tree local parent arrays
    for t in range(n_trees):
        base = int(j_node_off[t])
        cnt = int(node_alloc[t])
        feat = nfeat[base:base + cnt]
        left = nleft[base:base + cnt]
        internal = np.flatnonzero(feat >= 0)
        parent = np.full(cnt, -1, dtype=np.int64)
        parent[left[internal]] = internal
        parent[left[internal] + 1] = internal
        parent_parts.append(parent)
        leaves = np.flatnonzero(feat < 0)
        leaf_local_parts.append(leaves)
        leaf_tree_parts.append(np.full(len(leaves), t, dtype=np.int32))

    leaf_tree = np.concatenate(leaf_tree_parts)
    leaf_local = np.concatenate(leaf_local_parts).astype(np.int64)
    L = len(leaf_tree)

    # level-synchronous walk up: collect reversed paths column by column
    cur = leaf_local.copy()
    tree_of = leaf_tree.astype(np.int64)
    cols = []
    active = np.ones(L, dtype=bool)
    while active.any():
        cols.append(np.where(active, cur, -1).copy())
        nxt = np.full(L, -1, dtype=np.int64)
        for t in range(n_trees):
            mask = active & (tree_of == t)
            if mask.any():
                nxt[mask] = parent_parts[t][cur[mask]]
        cur = nxt
        active = cur >= 0

    depths = np.zeros(L, dtype=np.int64)   # path length per leaf
    mat = np.stack(cols, axis=1)           # [L, maxD] reversed paths
    depths = (mat >= 0).sum(axis=1)

    leaf_off = np.zeros(L + 1, dtype=np.int32)
    leaf_off[1:] = np.cumsum(depths)
    path_nodes = np.empty(leaf_off[-1], dtype=np.int32)
    # write each leaf's path root->leaf (reverse of collected order)
    maxd = mat.shape[1]
    for d in range(maxd):
        # element at reversed position d exists when depth > d; its
        # forward position is depth-1-d
        has = depths > d
        dst = leaf_off[:-1][has] + (depths[has] - 1 - d)
        path_nodes[dst] = mat[has, d]

    return (leaf_tree, leaf_off, path_nodes, int(depths.max()))
