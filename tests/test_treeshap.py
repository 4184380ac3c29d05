"""TreeSHAP reference implementation vs exact brute-force Shapley values."""

import numpy as np
import pytest

from flake16_framework_amd.models.binning import bin_codes, compute_bin_cuts
from flake16_framework_amd.models.forest_ref import (
    ForestParams, fit_forest,
)
from flake16_framework_amd.models.treeshap_ref import (
    brute_force_shap, forest_shap, tree_shap_single,
)


def _fit_small(n=80, f=6, seed=0, n_trees=1, splitter="best",
               bootstrap=False):
    rng = np.random.RandomState(seed)
    y = (rng.rand(n) < 0.4).astype(np.uint8)
    X = rng.randn(n, f).astype(np.float32)
    X[y == 1, : f // 2] += 1.0
    cuts = compute_bin_cuts(X)
    codes = bin_codes(X, cuts)
    params = ForestParams(n_trees, bootstrap, splitter,
                          "all" if splitter == "best" else "sqrt", 0)
    forest = fit_forest(codes, y, params, job_base=0, cuts=cuts)
    return forest, codes


class TestTreeShapRef:
    @pytest.mark.parametrize("seed", [0, 1, 2])
    def test_single_tree_matches_brute_force(self, seed):
        forest, codes = _fit_small(n=40, f=5, seed=seed)
        tree = forest.trees[0]
        for i in [0, 7, 19, 33]:
            phi = np.zeros(6)
            tree_shap_single(tree, codes[i], phi)
            exact = brute_force_shap(tree, codes[i], 5)
            np.testing.assert_allclose(phi[:5], exact, atol=1e-9)

    def test_additivity(self):
        """phi sums to f(x) - E[f] for every sample."""
        forest, codes = _fit_small(n=120, f=8, seed=3, n_trees=5,
                                   bootstrap=True)
        shap = forest_shap(forest, codes[:20], 8)
        # model output: mean class-0 prob; base: mean over root covers
        out = np.zeros(20)
        base = 0.0
        for tree in forest.trees:
            cover = tree.count0 + tree.count1
            base += tree.count0[0] / cover[0]
            for i in range(20):
                node = 0
                while tree.feature[node] != -1:
                    if codes[i, tree.feature[node]] <= tree.split_bin[node]:
                        node = tree.left[node]
                    else:
                        node = tree.right[node]
                out[i] += tree.count0[node] / cover[node]
        out /= len(forest.trees)
        base /= len(forest.trees)
        np.testing.assert_allclose(shap.sum(axis=1), out - base, atol=1e-9)

    def test_extra_trees_shap_runs(self):
        forest, codes = _fit_small(n=100, f=6, seed=5, n_trees=3,
                                   splitter="random")
        shap = forest_shap(forest, codes[:10], 6)
        assert shap.shape == (10, 6)
        assert np.isfinite(shap).all()


class TestLeafPaths:
    def test_csr_paths_valid(self):
        import numpy as np
        from flake16_framework_amd.models.leafpaths import build_leaf_paths
        forest, codes = _fit_small(n=200, f=8, seed=4, n_trees=3,
                                   bootstrap=True)
        nfeat = np.concatenate([t.feature for t in forest.trees])
        nleft = np.concatenate([t.left for t in forest.trees])
        offs, alloc = [], []
        base = 0
        for t in forest.trees:
            offs.append(base)
            alloc.append(t.n_nodes)
            base += t.n_nodes
        leaf_tree, leaf_off, path_nodes, max_d = build_leaf_paths(
            nfeat, nleft, np.array(offs), np.array(alloc))

        n_leaves = sum(int((t.feature == -1).sum()) for t in forest.trees)
        assert len(leaf_tree) == n_leaves
        for li in range(len(leaf_tree)):
            t = forest.trees[leaf_tree[li]]
            path = path_nodes[leaf_off[li]:leaf_off[li + 1]]
            assert path[0] == 0
            for i in range(len(path) - 1):
                u, v = path[i], path[i + 1]
                assert t.feature[u] != -1
                assert v in (t.left[u], t.left[u] + 1)
            assert t.feature[path[-1]] == -1
        assert max_d == max(len(path_nodes[leaf_off[i]:leaf_off[i + 1]])
                            for i in range(len(leaf_tree)))
