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
from .collate import get_collated
from .features import get_features_cov
from .labeling import get_req_runs_label


def build_tests(collated):
    """Collated structure -> the tests.json dict."""
    tests = {}

    for proj in sorted(collated.keys(), key=lambda s: s.lower()):
        if not all(collated[proj]):
            continue

        test_data, test_fn_data, test_files, churn = collated[proj]
        tests_proj = {}

        for nid in sorted(test_data.keys(), key=lambda s: s.lower()):
            if not all(test_data[nid]):
                continue

            runs_nid, cov_nid, rusage_nid, fid = test_data[nid]
            req_runs_nid, label_nid = get_req_runs_label(runs_nid)

            if label_nid is None:
                continue

            tests_proj[nid] = (
                req_runs_nid, label_nid,
                *get_features_cov(cov_nid, test_files, churn),
                *rusage_nid,
                *test_fn_data[fid],
            )

        if tests_proj:
            tests[proj] = tests_proj

    return tests


def write_tests(tests_file=TESTS_FILE):
    tests = build_tests(get_collated())
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
