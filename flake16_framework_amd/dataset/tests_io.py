"""tests.json construction and loading.

Format contract (reference experiment.py:376-427):
  tests.json = { proj: { nodeid: [req_runs, label, f0..f15] } }
  rows keep the canonical 16-column feature order of FEATURE_NAMES; projects
  and nodeids are sorted case-insensitively; tests with any missing part
  (runs / coverage / rusage / static id) or an incomplete run count are
  dropped.
"""

import json

import numpy as np

from ..constants import TESTS_FILE
from .collate import collate
from .features import get_features_cov
from .labeling import classify


def _ci_sorted(names):
    return sorted(names, key=str.lower)


def build_tests(projects):
    """{proj: ProjectData} -> the tests.json dict.

    Projects and nodeids are emitted in case-insensitive sorted order;
    a test contributes a row only when every evidence part is present
    and the labeling rule accepts its run counts.  Row layout:
    [req_runs, label, 3 coverage features, 6 rusage features,
    7 static features] — the canonical 16-column order."""
    tests = {}

    for proj in _ci_sorted(projects):
        data = projects[proj]
        if not data.complete:
            continue

        rows = {}
        for nid in _ci_sorted(data.tests):
            rec = data.tests[nid]
            if not rec.complete:
                continue
            req_runs, label = classify(rec.runs)
            if label is None:
                continue
            rows[nid] = (
                req_runs, label,
                *get_features_cov(rec.coverage, data.test_files,
                                  data.churn),
                *rec.rusage,
                *data.fn_metrics[rec.static_id],
            )

        if rows:
            tests[proj] = rows

    return tests


def write_tests(tests_file=TESTS_FILE):
    tests = build_tests(collate())
    with open(tests_file, "w") as fd:
        json.dump(tests, fd, indent=4)


def load_tests(tests_file=None):
    # None means "the stage artifact in the working directory" — the CLI
    # path passes no explicit file (reference behavior)
    with open(tests_file or TESTS_FILE, "r") as fd:
        return json.load(fd)


def load_feat_lab_proj(flaky_label, feature_set, tests_file=None,
                       tests=None):
    """tests.json -> (features [N x len(feature_set)] float64,
    labels bool[N] (label == flaky_label), projects str[N]).

    Row order is the file's iteration order (projects then nodeids, both
    already sorted case-insensitively at write time), matching the
    reference's load_feat_lab_proj (experiment.py:410-427).
    """
    if tests is None:
        tests = load_tests(tests_file)

    features, labels, projects = [], [], []

    for proj, tests_proj in tests.items():
        projects += [proj] * len(tests_proj)
        for (_, label_nid, *features_nid) in tests_proj.values():
            features.append(features_nid)
            labels.append(label_nid)

    features = np.array(features, dtype=np.float64)[:, feature_set]
    labels = np.array(labels) == flaky_label
    projects = np.array(projects)

    return features, labels, projects
