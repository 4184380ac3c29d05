"""Synthetic dataset generator.

There is no network access (the real study data lives behind an external
link), so the scores/shap/figures stages and the benchmark run on a synthetic
tests.json of the same SHAPE as the real one: 26 projects, 16 features per
test, labels {0=non-flaky, 1=OD-flaky, 2=NOD-flaky} with flaky minorities.

Feature distributions are class-conditional log-normal mixtures so the
classifiers have real signal (class separation is partial, keeping tree
building non-trivial), and feature magnitudes roughly mimic the real
features' scales (line counts, seconds, bytes, complexity metrics).
"""

import json

import numpy as np

from ..constants import FEATURE_NAMES, FLAKY, NON_FLAKY, OD_FLAKY

# 26 synthetic project names mirroring the study's 26 subjects.
DEFAULT_N_PROJECTS = 26

# Class proportions: flaky tests are a minority, as in the study.
DEFAULT_PROPS = {NON_FLAKY: 0.86, OD_FLAKY: 0.08, FLAKY: 0.06}


def _class_profile(rng, n_features):
    """Per-class (mean, std) profile in log space for each feature."""
    mean = rng.uniform(0.5, 4.0, size=n_features)
    std = rng.uniform(0.4, 1.2, size=n_features)
    return mean, std


def make_synthetic_tests(n_tests=10000, n_projects=DEFAULT_N_PROJECTS,
                         seed=0, props=None):
    """Build a synthetic tests.json dict.

    Rows are [req_runs, label, f0..f15].  Deterministic in `seed`.
    """
    props = props or DEFAULT_PROPS
    rng = np.random.RandomState(seed)
    n_features = len(FEATURE_NAMES)

    # Shared base profile plus per-class offsets on a random subset of
    # features: NOD-flakiness correlates with time/memory/IO features,
    # OD-flakiness with coverage/churn features — loosely like the study.
    base_mean, base_std = _class_profile(rng, n_features)
    offsets = {
        NON_FLAKY: np.zeros(n_features),
        OD_FLAKY: rng.uniform(0.0, 1.0, n_features) * (rng.rand(n_features) < 0.5),
        FLAKY: rng.uniform(0.0, 1.2, n_features) * (rng.rand(n_features) < 0.5),
    }

    labels = rng.choice(
        list(props.keys()), size=n_tests, p=list(props.values()))

    # Project sizes: log-uniform, mimicking the real spread of suite sizes.
    raw = np.exp(rng.uniform(0.0, 2.5, size=n_projects))
    proj_sizes = np.maximum(1, (raw / raw.sum() * n_tests).astype(int))
    while proj_sizes.sum() < n_tests:
        proj_sizes[rng.randint(n_projects)] += 1
    while proj_sizes.sum() > n_tests:
        i = rng.randint(n_projects)
        if proj_sizes[i] > 1:
            proj_sizes[i] -= 1

    tests = {}
    row = 0
    for p in range(n_projects):
        proj = f"proj{p:02d}"
        tests_proj = {}
        for t in range(proj_sizes[p]):
            label = int(labels[row])
            mu = base_mean + offsets[label]
            feats = np.exp(rng.normal(mu, base_std))
            # Integer-like columns (counts) are rounded, like the real data.
            for col in (0, 1, 2, 4, 5, 6, 7, 9, 10, 11, 14):
                feats[col] = np.floor(feats[col])
            req_runs = 0
            if label != NON_FLAKY:
                req_runs = int(rng.randint(1, 2500))
            nid = f"tests/test_{proj}.py::test_{t:05d}"
            tests_proj[nid] = [req_runs, label, *[float(f) for f in feats]]
            row += 1
        tests[proj] = tests_proj

    return tests


def write_synthetic_tests(tests_file, n_tests=10000, seed=0):
    tests = make_synthetic_tests(n_tests=n_tests, seed=seed)
    with open(tests_file, "w") as fd:
        json.dump(tests, fd, indent=4)
    return tests
