"""Randomized invariant tests (fixed seeds): structural properties that
must hold for ANY input, complementing the example-based suites."""

import numpy as np

from flake16_framework_amd.constants import FLAKY, NON_FLAKY, OD_FLAKY
from flake16_framework_amd.dataset.collate import RunStats
from flake16_framework_amd.dataset.labeling import classify
from flake16_framework_amd.models.binning import bin_codes, compute_bin_cuts
from flake16_framework_amd.models.forest_ref import (
    LEAF, ForestParams, fit_forest,
)


class TestLabelingInvariants:
    def test_random_run_stats(self):
        rng = np.random.RandomState(0)
        n_runs = {"baseline": 20, "shuffle": 20}
        for _ in range(500):
            stats = {}
            for mode in ("baseline", "shuffle"):
                st = RunStats()
                st.total = int(rng.choice([20, rng.randint(0, 21)]))
                st.failures = int(rng.randint(0, st.total + 1)) \
                    if st.total else 0
                st.first_fail = int(rng.randint(0, 20)) \
                    if st.failures else None
                st.first_pass = int(rng.randint(0, 20)) \
                    if st.failures < st.total else None
                stats[mode] = st
            req, label = classify(stats, n_runs)

            b, s = stats["baseline"], stats["shuffle"]
            if b.total != 20 or s.total != 20:
                assert label is None and req == 0
                continue
            assert label in (NON_FLAKY, OD_FLAKY, FLAKY)
            if 0 < b.failures < b.total:
                assert label == FLAKY            # intermittent baseline
                assert req == max(b.first_fail, b.first_pass)
            elif label == NON_FLAKY:
                assert (b.failures == 0 and s.failures == 0) or \
                       (b.failures == b.total and s.failures == s.total)
                assert req == 0
            else:
                assert label == OD_FLAKY
                assert 0 <= req < 20


class TestForestInvariants:
    def _random_fit(self, seed, splitter, bootstrap):
        rng = np.random.RandomState(seed)
        n = int(rng.randint(30, 300))
        f = int(rng.choice([7, 16]))
        X = rng.randn(n, f).astype(np.float32)
        # duplicate rows + integer columns to exercise ties/constant bins
        X[: n // 4] = X[n // 4: n // 2][: n // 4]
        X[:, 0] = np.floor(X[:, 0] * 2)
        y = (rng.rand(n) < rng.uniform(0.1, 0.9)).astype(np.uint8)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        params = ForestParams(3, bootstrap, splitter,
                              "sqrt" if splitter == "random" else "all", 0)
        return fit_forest(codes, y, params, job_base=seed, cuts=cuts), n, y

    def test_structure_invariants(self):
        for seed in range(12):
            for splitter, bootstrap in (("best", False), ("best", True),
                                        ("random", False)):
                forest, n, y = self._random_fit(seed, splitter, bootstrap)
                for tree in forest.trees:
                    root_total = tree.count0[0] + tree.count1[0]
                    assert root_total == n
                    for node in range(tree.n_nodes):
                        tot = tree.count0[node] + tree.count1[node]
                        assert tot >= 1
                        if tree.feature[node] == LEAF:
                            continue
                        l, r = tree.left[node], tree.right[node]
                        # children partition the parent exactly
                        assert tree.count0[l] + tree.count0[r] == \
                            tree.count0[node]
                        assert tree.count1[l] + tree.count1[r] == \
                            tree.count1[node]
                        assert tree.count0[l] + tree.count1[l] >= 1
                        assert tree.count0[r] + tree.count1[r] >= 1
                    # every leaf with >= 2 samples of both classes must have
                    # had no valid split (all candidate features constant) —
                    # can't check directly, but pure leaves must be leaves:
                    for node in range(tree.n_nodes):
                        if tree.feature[node] != LEAF:
                            assert tree.count0[node] > 0
                            assert tree.count1[node] > 0

    def test_bin_code_monotonicity(self):
        rng = np.random.RandomState(3)
        X = rng.randn(5000, 4).astype(np.float32)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        for f in range(4):
            order = np.argsort(X[:, f], kind="stable")
            assert (np.diff(codes[order, f].astype(np.int32)) >= 0).all()
            assert codes[:, f].max() <= len(cuts[f])
