"""Flakiness labeling rule.

Implements the decision table of the reference's labeling logic
(reference experiment.py:339-359) over RunStats records:

  - incomplete run counts (baseline != 2500 or shuffle != 2500) -> (0, None):
    the test is dropped.
  - baseline never fails:
      shuffle never fails          -> (0, NON_FLAKY)
      shuffle fails at least once  -> (first failing shuffle run, OD_FLAKY)
  - baseline always fails:
      shuffle always fails         -> (0, NON_FLAKY)
      shuffle passes at least once -> (first passing shuffle run, OD_FLAKY)
  - baseline intermittently fails  -> (max(first failing, first passing
                                       baseline run), FLAKY)  [NOD-flaky]
"""

from ..constants import FLAKY, NON_FLAKY, N_RUNS, OD_FLAKY
from .collate import RunStats


def classify(runs, n_runs=None):
    """(required_runs, label) for one test.

    runs: {mode: RunStats} with modes "baseline" and "shuffle";
    n_runs: override of the per-mode expected run counts (for tests).
    Label None means the evidence is incomplete and the test is dropped.
    """
    n_runs = n_runs or N_RUNS
    baseline = runs.get("baseline") or RunStats()
    shuffle = runs.get("shuffle") or RunStats()

    if baseline.total != n_runs["baseline"] or \
            shuffle.total != n_runs["shuffle"]:
        return 0, None

    if baseline.never_failed:
        if shuffle.never_failed:
            return 0, NON_FLAKY
        return shuffle.first_fail, OD_FLAKY

    if baseline.always_failed:
        if shuffle.always_failed:
            return 0, NON_FLAKY
        return shuffle.first_pass, OD_FLAKY

    # intermittent baseline failures: order-independent (NOD) flakiness;
    # detection requires seeing both a failure and a pass
    return max(baseline.first_fail, baseline.first_pass), FLAKY
