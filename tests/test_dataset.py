"""Unit tests for the collation / labeling / feature layer.

Mirrors the coverage of the reference's test_experiment.py (its 4 pure
functions) with fresh cases, plus the tests.json round trip.
"""

import io
import json

import numpy as np
import pytest

from flake16_framework_amd.constants import FLAKY, NON_FLAKY, OD_FLAKY
from flake16_framework_amd.dataset.collate import ProjectData, RunStats
from flake16_framework_amd.dataset.features import get_features_cov
from flake16_framework_amd.dataset.labeling import classify
from flake16_framework_amd.dataset.synthetic import make_synthetic_tests
from flake16_framework_amd.dataset.tests_io import load_feat_lab_proj

N_RUNS_SMALL = {"baseline": 4, "shuffle": 4, "testinspect": 1}


def _ingest_runs(lines_per_run, mode):
    data = ProjectData("p")
    for run_n, lines in enumerate(lines_per_run):
        data.add_outcomes(io.StringIO("\n".join(lines)), mode, run_n)
    return data


def _stats(total, failures, first_fail, first_pass):
    s = RunStats()
    s.total = total
    s.failures = failures
    s.first_fail = first_fail
    s.first_pass = first_pass
    return s


class TestRunIngestion:
    def test_counts_and_min_runs(self):
        data = _ingest_runs([
            ["passed\tt::a", "failed\tt::b"],
            ["failed\tt::a", "failed\tt::b"],
            ["passed\tt::a", "passed\tt::b"],
        ], "baseline")
        a = data.tests["t::a"].runs["baseline"]
        b = data.tests["t::b"].runs["baseline"]
        assert (a.total, a.failures, a.first_fail, a.first_pass) == \
            (3, 1, 1, 0)
        assert (b.total, b.failures, b.first_fail, b.first_pass) == \
            (3, 2, 0, 2)

    def test_outcome_substring_failed(self):
        # the contract is substring matching: "xfailed" counts as failed
        data = _ingest_runs([["xfailed\tt::a"]], "shuffle")
        assert data.tests["t::a"].runs["shuffle"].failures == 1

    def test_nodeid_with_tabs_preserved(self):
        # only the FIRST tab splits outcome from nodeid
        data = ProjectData("p")
        data.add_outcomes(io.StringIO("passed\tt::a[x\ty]"), "baseline", 0)
        assert "t::a[x\ty]" in data.tests


class TestLabeling:
    @pytest.mark.parametrize("baseline,shuffle,expect", [
        # incomplete counts -> dropped
        ((3, 0, None, 0), (4, 0, None, 0), (0, None)),
        ((4, 0, None, 0), (3, 0, None, 0), (0, None)),
        # never fails anywhere -> non-flaky
        ((4, 0, None, 0), (4, 0, None, 0), (0, NON_FLAKY)),
        # baseline clean, shuffle fails -> OD, req = first failing shuffle run
        ((4, 0, None, 0), (4, 2, 1, 0), (1, OD_FLAKY)),
        # always fails everywhere -> non-flaky
        ((4, 4, 0, None), (4, 4, 0, None), (0, NON_FLAKY)),
        # always fails in baseline, shuffle passes once -> OD, req = first pass
        ((4, 4, 0, None), (4, 3, 0, 2), (2, OD_FLAKY)),
        # intermittent baseline -> NOD, req = max(first fail, first pass)
        ((4, 2, 1, 0), (4, 0, None, 0), (1, FLAKY)),
        ((4, 1, 3, 0), (4, 4, 0, None), (3, FLAKY)),
    ])
    def test_decision_table(self, baseline, shuffle, expect):
        runs = {"baseline": _stats(*baseline), "shuffle": _stats(*shuffle)}
        assert classify(runs, N_RUNS_SMALL) == expect

    def test_missing_mode_is_incomplete(self):
        assert classify({"baseline": _stats(4, 0, None, 0)},
                        N_RUNS_SMALL) == (0, None)


class TestNumbits:
    """Known-answer vectors for coverage.py's numbits packing (bit i of
    byte b = line b*8+i) — pins the format both the collector's encoder
    and the collation decoder implement (coverage.py itself is not
    installable here)."""

    def test_encode_known_answer(self):
        from flake16_framework_amd.collect.testinspect import (
            nums_to_numbits,
        )
        assert nums_to_numbits({1, 2, 9}) == b"\x06\x02"
        assert nums_to_numbits({0}) == b"\x01"
        assert nums_to_numbits(set()) == b""
        assert nums_to_numbits({15}) == b"\x00\x80"

    def test_decode_known_answer(self):
        from flake16_framework_amd.dataset.collate import _numbits_to_nums
        assert _numbits_to_nums(b"\x06\x02") == [1, 2, 9]
        assert _numbits_to_nums(b"\x00\x80") == [15]
        assert _numbits_to_nums(b"") == []

    def test_roundtrip(self):
        import random
        from flake16_framework_amd.collect.testinspect import (
            nums_to_numbits,
        )
        from flake16_framework_amd.dataset.collate import _numbits_to_nums
        rng = random.Random(5)
        for _ in range(20):
            nums = {rng.randrange(1, 400) for _ in range(rng.randrange(40))}
            assert set(_numbits_to_nums(nums_to_numbits(nums))) == nums


class TestCoverageFeatures:
    def test_basic_counts(self):
        cov = {"src/a.py": {1, 2, 3}, "tests/t.py": {5, 6}}
        churn = {"src/a.py": {1: 4, 3: 2, 9: 7}}
        n_lines, n_changes, n_src = get_features_cov(cov, {"tests/t.py"}, churn)
        assert (n_lines, n_changes, n_src) == (5, 6, 3)

    def test_empty(self):
        assert get_features_cov({}, set(), {}) == (0, 0, 0)


class TestRusage:
    def test_parse(self):
        data = ProjectData("p")
        data.add_rusage(io.StringIO("1.5\t2\t3\t4\t5\t6.25\tt::a"))
        assert data.tests["t::a"].rusage == [1.5, 2.0, 3.0, 4.0, 5.0, 6.25]


class TestSyntheticAndLoad:
    def test_shapes_and_determinism(self):
        tests = make_synthetic_tests(n_tests=500, seed=3)
        tests2 = make_synthetic_tests(n_tests=500, seed=3)
        assert json.dumps(tests) == json.dumps(tests2)
        total = sum(len(v) for v in tests.values())
        assert total == 500

        X, y, proj = load_feat_lab_proj(FLAKY, tuple(range(16)), tests=tests)
        assert X.shape == (500, 16)
        assert y.dtype == bool and 0 < y.sum() < 500
        assert len(proj) == 500

    def test_feature_subset(self):
        tests = make_synthetic_tests(n_tests=200, seed=1)
        X16, _, _ = load_feat_lab_proj(FLAKY, tuple(range(16)), tests=tests)
        X7, _, _ = load_feat_lab_proj(FLAKY, (0, 1, 2, 3, 10, 11, 14),
                                      tests=tests)
        assert X7.shape[1] == 7
        np.testing.assert_array_equal(X7[:, 3], X16[:, 3])
