"""Reference (numpy) implementation of the histogram-forest algorithm.

This is the EXACT specification the HIP kernels implement — same Philox
draws keyed on (tag, node sample-range, draw index), same fp64 split-score
expression and comparison order — so a device-built forest reproduces these
trees bit-for-bit (the hip-vs-ref GPU tests assert it).  It also serves as
the CPU execution path for the scores/shap stages when no GPU is present.

Model semantics follow sklearn 1.0.2 defaults (reference experiment.py:96-98):
  Decision Tree : splitter=best, max_features=all,  1 tree,   no bootstrap
  Random Forest : splitter=best, max_features=sqrt, 100 trees, bootstrap
  Extra Trees   : splitter=random, max_features=sqrt, 100 trees, no bootstrap
Gini impurity, min_samples_split=2, min_samples_leaf=1, unbounded depth.
Ensemble prediction averages per-tree class probabilities (leaf class counts
normalized), argmax with ties to class 0 — sklearn's predict_proba/argmax.

Deviation from sklearn (deliberate): splits are over <=256
quantile-bin boundaries, not exact sorted values; Extra-Trees random
thresholds are uniform over occupied bin boundaries rather than uniform over
the raw value range.  Parity is metric-level, enforced by golden tests.
"""

from collections import deque
from dataclasses import dataclass

import numpy as np

from ..utils.philox import (
    TAG_BOOTSTRAP, TAG_FEATSEL, TAG_THRESH, bounded_int, draws_u32, philox4x32,
)

LEAF = -1


@dataclass
class ForestParams:
    n_trees: int = 100
    bootstrap: bool = True
    splitter: str = "best"        # "best" | "random"
    max_features: str = "sqrt"    # "sqrt" | "all"
    seed: int = 0

    def resolve_max_features(self, n_features):
        if self.max_features == "all":
            return n_features
        return max(1, int(np.sqrt(n_features)))


def params_for_model(model_spec, seed=0):
    kind = model_spec["kind"]
    if kind == "decision_tree":
        return ForestParams(1, False, "best", "all", seed)
    if kind == "random_forest":
        return ForestParams(model_spec["n_estimators"], True, "best", "sqrt", seed)
    if kind == "extra_trees":
        return ForestParams(model_spec["n_estimators"], False, "random", "sqrt", seed)
    raise ValueError(kind)


@dataclass
class Tree:
    """SoA tree storage; node 0 is the root.

    feature[i] == LEAF marks a leaf; children hold node indices; value rows
    are raw class counts (weighted sample counts) at the node.
    """
    feature: np.ndarray      # int32[n_nodes], LEAF for leaves
    split_bin: np.ndarray    # int32[n_nodes]
    threshold: np.ndarray    # float32[n_nodes] raw-space threshold (x < thr)
    left: np.ndarray         # int32[n_nodes]
    right: np.ndarray        # int32[n_nodes]
    count0: np.ndarray       # float64[n_nodes]
    count1: np.ndarray       # float64[n_nodes]

    @property
    def n_nodes(self):
        return len(self.feature)


def _feature_permutation(n_features, start, end, depth, k0, k1):
    """Partial Fisher-Yates permutation of the feature ids, driven by Philox
    draws at counters (TAG_FEATSEL | depth<<8, start, end, i)."""
    perm = np.arange(n_features, dtype=np.int64)
    tag = np.uint32(TAG_FEATSEL | ((depth & 0xFF) << 8))
    i = np.arange(n_features - 1, dtype=np.uint32)
    u, _, _, _ = philox4x32(tag, np.uint32(start), np.uint32(end), i, k0, k1)
    u = np.atleast_1d(u)
    for i in range(n_features - 1):
        j = i + int(bounded_int(u[i], n_features - i))
        perm[i], perm[j] = perm[j], perm[i]
    return perm


def _et_threshold_draws(n_features, start, end, depth, k0, k1):
    """Extra-Trees random draws, one uint32 per feature id f at counter
    (TAG_THRESH | depth<<8, start, end, f).  The split bin for feature f is
    bmin + bounded(draw[f], bmax - bmin)."""
    tag = np.uint32(TAG_THRESH | ((depth & 0xFF) << 8))
    f = np.arange(n_features, dtype=np.uint32)
    u, _, _, _ = philox4x32(tag, np.uint32(start), np.uint32(end), f, k0, k1)
    return np.atleast_1d(u)


def build_tree(codes, y, sidx, params, max_features, k0, k1, cuts=None):
    """Build one tree.  codes: (N,F) uint8; y: (N,) uint8 in {0,1};
    sidx: int64 sample indices (bootstrap or identity) — will be reordered.

    Returns a Tree.  Node ids are allocated in BFS order here; the device
    allocator may order them differently, which is immaterial because all
    randomness is keyed on the node's sample range, not its id.
    """
    n_features = codes.shape[1]
    feature, split_bin, threshold = [], [], []
    left, right, count0, count1 = [], [], [], []

    def alloc():
        feature.append(LEAF)
        split_bin.append(0)
        threshold.append(np.float32(0.0))
        left.append(-1)
        right.append(-1)
        count0.append(0.0)
        count1.append(0.0)
        return len(feature) - 1

    root = alloc()
    queue = deque([(root, 0, len(sidx), 0)])   # (node, start, end, depth)

    while queue:
        node, start, end, depth = queue.popleft()
        n = end - start
        rows = sidx[start:end]
        c1 = int(y[rows].sum())
        c0 = n - c1
        count0[node] = float(c0)
        count1[node] = float(c1)

        if n < 2 or c0 == 0 or c1 == 0:
            continue  # leaf: too small or pure

        # Node histogram: totals and class-1 counts per (feature, bin).
        node_codes = codes[rows]           # (n, F)
        flat = node_codes.astype(np.int64) + \
            np.arange(n_features, dtype=np.int64) * 256
        hist_n = np.bincount(flat.ravel(), minlength=n_features * 256)
        hist_1 = np.bincount(flat[y[rows] == 1].ravel(),
                             minlength=n_features * 256)
        hist_n = hist_n.reshape(n_features, 256)
        hist_1 = hist_1.reshape(n_features, 256)

        perm = _feature_permutation(n_features, start, end, depth, k0, k1)
        if params.splitter == "random":
            thr_draws = _et_threshold_draws(n_features, start, end, depth,
                                            k0, k1)

        best_score = -np.inf   # maximize sum of squared-count/size over children
        best_f = -1
        best_b = -1
        n_evaluated = 0

        for f in perm:
            nz = np.nonzero(hist_n[f])[0]
            bmin, bmax = int(nz[0]), int(nz[-1])
            if bmin == bmax:
                continue  # constant feature: does not count toward max_features

            # Prefix sums over bins (exact integers).
            cn = np.cumsum(hist_n[f]).astype(np.int64)
            c1f = np.cumsum(hist_1[f]).astype(np.int64)

            if params.splitter == "random":
                cand = np.array(
                    [bmin + int(bounded_int(thr_draws[f], bmax - bmin))])
            else:
                cand = np.arange(bmin, bmax)

            # fp64, fixed elementwise op order (HIP side is compiled
            # -ffp-contract=off and evaluates the same expression).
            nL = cn[cand]
            n1L = c1f[cand]
            n0L = nL - n1L
            nR = n - nL
            n1R = c1 - n1L
            n0R = c0 - n0L
            score = ((n0L * n0L + n1L * n1L).astype(np.float64)
                     / nL.astype(np.float64)
                     + (n0R * n0R + n1R * n1R).astype(np.float64)
                     / nR.astype(np.float64))
            i = int(np.argmax(score))  # first-of-ties == sequential scan
            if score[i] > best_score:
                best_score, best_f, best_b = float(score[i]), int(f), int(cand[i])

            n_evaluated += 1
            if n_evaluated >= max_features:
                break

        if best_f < 0:
            continue  # no valid split anywhere: leaf

        # Stable partition of the node's sample slice.
        go_left = codes[rows, best_f] <= best_b
        sidx[start:end] = np.concatenate([rows[go_left], rows[~go_left]])
        mid = start + int(go_left.sum())

        l_id, r_id = alloc(), alloc()
        feature[node] = best_f
        split_bin[node] = best_b
        if cuts is not None:
            threshold[node] = cuts[best_f][best_b]
        left[node] = l_id
        right[node] = r_id
        queue.append((l_id, start, mid, depth + 1))
        queue.append((r_id, mid, end, depth + 1))

    return Tree(
        np.array(feature, dtype=np.int32),
        np.array(split_bin, dtype=np.int32),
        np.array(threshold, dtype=np.float32),
        np.array(left, dtype=np.int32),
        np.array(right, dtype=np.int32),
        np.array(count0, dtype=np.float64),
        np.array(count1, dtype=np.float64),
    )


@dataclass
class Forest:
    trees: list
    params: ForestParams
    n_features: int = 0
    job_base: int = 0


def fit_forest(codes, y, params, job_base=0, cuts=None):
    """Fit params.n_trees trees.  Tree t uses Philox key
    (k0=params.seed, k1=job_base + t)."""
    codes = np.ascontiguousarray(codes, dtype=np.uint8)
    y = np.ascontiguousarray(y, dtype=np.uint8)
    n = len(y)
    max_features = params.resolve_max_features(codes.shape[1])
    trees = []

    for t in range(params.n_trees):
        k0, k1 = np.uint32(params.seed), np.uint32(job_base + t)
        if params.bootstrap:
            u = draws_u32(TAG_BOOTSTRAP, 0, 0, n, k0, k1)
            sidx = bounded_int(u, n)
        else:
            sidx = np.arange(n, dtype=np.int64)
        trees.append(build_tree(codes, y, sidx, params, max_features,
                                k0, k1, cuts=cuts))

    return Forest(trees, params, codes.shape[1], job_base)


def _tree_leaves(tree, codes_test):
    """Vectorized batch traversal: leaf node index per test row."""
    node = np.zeros(codes_test.shape[0], dtype=np.int64)
    while True:
        feat = tree.feature[node]
        active = feat != LEAF
        if not active.any():
            return node
        rows = np.flatnonzero(active)
        f = feat[rows]
        go_left = codes_test[rows, f] <= tree.split_bin[node[rows]]
        node[rows] = np.where(go_left, tree.left[node[rows]],
                              tree.right[node[rows]])


PREDICT_TREE_BLOCK = 10


def predict_forest(forest, codes_test):
    """Ensemble prediction: mean per-tree class probabilities, ties ->
    class 0 (sklearn argmax).  fp64 accumulation is BLOCKED in groups of
    PREDICT_TREE_BLOCK trees (sequential within a block, blocks summed in
    ascending order) — the exact association the device predict kernel
    uses, so predictions stay bit-identical across backends."""
    codes_test = np.asarray(codes_test, dtype=np.uint8)
    m = codes_test.shape[0]
    acc0 = np.zeros(m, dtype=np.float64)
    acc1 = np.zeros(m, dtype=np.float64)

    for b in range(0, len(forest.trees), PREDICT_TREE_BLOCK):
        b0 = np.zeros(m, dtype=np.float64)
        b1 = np.zeros(m, dtype=np.float64)
        for tree in forest.trees[b:b + PREDICT_TREE_BLOCK]:
            leaf = _tree_leaves(tree, codes_test)
            tot = tree.count0[leaf] + tree.count1[leaf]
            b0 += tree.count0[leaf] / tot
            b1 += tree.count1[leaf] / tot
        acc0 += b0
        acc1 += b1

    return (acc1 > acc0).astype(np.uint8)
