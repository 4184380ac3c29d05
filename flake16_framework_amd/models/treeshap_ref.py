"""Path-dependent TreeSHAP — reference (numpy/python) implementation.

Implements the tree_path_dependent algorithm of Lundberg et al.,
"Consistent Individualized Feature Attribution for Tree Ensembles"
(Algorithm 2), which is what shap 0.40.0's TreeExplainer computes for
sklearn forests (reference experiment.py:517).  For a binary classifier the
tree output is the leaf's class-0 probability, ensemble output the mean
over trees, and shap_values(...)[0] is the N x F class-0 attribution matrix
(class-1 is its negation).

The HIP kernel (ops/hip/treeshap.hip) implements the identical recursion
iteratively; it is validated against this module, and this module against a
brute-force Shapley evaluation on small trees (tests/test_treeshap.py).
"""

import numpy as np

from .forest_ref import LEAF


def _extend(d, z, o, w, l, pz, po, pi):
    """EXTEND: append (pi, pz, po) to the path of length l. Returns l+1."""
    d[l] = pi
    z[l] = pz
    o[l] = po
    w[l] = 1.0 if l == 0 else 0.0
    for i in range(l - 1, -1, -1):
        w[i + 1] += po * w[i] * (i + 1) / (l + 1)
        w[i] = pz * w[i] * (l - i) / (l + 1)
    return l + 1


def _unwind(d, z, o, w, l, i):
    """UNWIND: remove path element i (path length l). Returns l-1."""
    lm = l - 1
    n = w[lm]
    if o[i] != 0.0:
        for j in range(lm - 1, -1, -1):
            t = w[j]
            w[j] = n * l / ((j + 1) * o[i])
            n = t - w[j] * z[i] * (lm - j) / l
    else:
        for j in range(lm - 1, -1, -1):
            w[j] = w[j] * l / (z[i] * (lm - j))
    for j in range(i, lm):
        d[j] = d[j + 1]
        z[j] = z[j + 1]
        o[j] = o[j + 1]
    return lm


def _unwound_sum(z, o, w, l, i):
    """Sum of weights after hypothetically unwinding element i."""
    lm = l - 1
    total = 0.0
    if o[i] != 0.0:
        n = w[lm]
        for j in range(lm - 1, -1, -1):
            t = n * l / ((j + 1) * o[i])
            total += t
            n = w[j] - t * z[i] * (lm - j) / l
    else:
        for j in range(lm - 1, -1, -1):
            total += w[j] * l / (z[i] * (lm - j))
    return total


def tree_shap_single(tree, code_row, phi):
    """Accumulate one tree's class-0 SHAP values for one sample into phi.

    tree: forest_ref.Tree; code_row: uint8 bin codes; phi: float64[F].
    """
    max_path = 256
    d = np.zeros(max_path, dtype=np.int64)
    z = np.zeros(max_path)
    o = np.zeros(max_path)
    w = np.zeros(max_path)

    cover = tree.count0 + tree.count1

    def recurse(j, l, pz, po, pi):
        # copies of the path state for this frame
        dd, zz, oo, ww = d.copy(), z.copy(), o.copy(), w.copy()
        l = _extend(dd, zz, oo, ww, l, pz, po, pi)

        if tree.feature[j] == LEAF:
            v = tree.count0[j] / cover[j]   # class-0 probability
            for i in range(1, l):
                phi[dd[i]] += (_unwound_sum(zz, oo, ww, l, i)
                               * (oo[i] - zz[i]) * v)
            return

        f = int(tree.feature[j])
        if code_row[f] <= tree.split_bin[j]:
            hot, cold = tree.left[j], tree.right[j]
        else:
            hot, cold = tree.right[j], tree.left[j]

        iz = io = 1.0
        k = -1
        for i in range(l):
            if dd[i] == f:
                k = i
                break
        if k >= 0:
            iz, io = zz[k], oo[k]
            l = _unwind(dd, zz, oo, ww, l, k)

        # install this frame's path as the shared state for children
        d[:], z[:], o[:], w[:] = dd, zz, oo, ww
        rj = cover[j]
        recurse(hot, l, iz * cover[hot] / rj, io, f)
        d[:], z[:], o[:], w[:] = dd, zz, oo, ww
        recurse(cold, l, iz * cover[cold] / rj, 0.0, f)

    recurse(0, 0, 1.0, 1.0, -1)


def forest_shap(forest, codes, n_features):
    """Class-0 SHAP matrix [N, n_features] for all rows of codes: mean of
    per-tree SHAP over the ensemble (sklearn predict_proba averaging)."""
    codes = np.asarray(codes, dtype=np.uint8)
    n = codes.shape[0]
    out = np.zeros((n, n_features))
    phi = np.zeros(n_features + 1)   # slot for the d=-1 sentinel (index -1)
    for tree in forest.trees:
        for i in range(n):
            phi[:] = 0.0
            tree_shap_single(tree, codes[i], phi)
            out[i] += phi[:n_features]
    return out / len(forest.trees)


def brute_force_shap(tree, code_row, n_features):
    """Exact Shapley values of the tree's path-dependent conditional
    expectation, by enumerating all feature subsets.  Exponential — tests
    only (n_features <= ~12)."""
    import itertools
    from math import factorial

    cover = tree.count0 + tree.count1

    def expect(j, S):
        if tree.feature[j] == LEAF:
            return tree.count0[j] / cover[j]
        f = int(tree.feature[j])
        l, r = tree.left[j], tree.right[j]
        if f in S:
            nxt = l if code_row[f] <= tree.split_bin[j] else r
            return expect(nxt, S)
        return (expect(l, S) * cover[l] + expect(r, S) * cover[r]) / cover[j]

    feats = list(range(n_features))
    phi = np.zeros(n_features)
    M = n_features
    for f in feats:
        others = [g for g in feats if g != f]
        for k in range(len(others) + 1):
            for S in itertools.combinations(others, k):
                S = set(S)
                weight = (factorial(len(S)) * factorial(M - len(S) - 1)
                          / factorial(M))
                phi[f] += weight * (expect(0, S | {f}) - expect(0, S))
    return phi
