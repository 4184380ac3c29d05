"""Flakiness labeling rule.

Implements the decision table of the reference's get_req_runs_label_nid
(reference experiment.py:339-359):

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

`runs_mode` accumulators are [n_runs, n_fails, min_failing_run, min_passing_run]
as produced by the run collator.
"""

from ..constants import FLAKY, NON_FLAKY, N_RUNS, OD_FLAKY

EMPTY_RUNS = (0, 0, None, None)


def get_req_runs_label(runs_nid, n_runs=None):
    """Return (required_runs, label) for one test's run statistics.

    runs_nid: {"baseline": [n, fails, min_fail_run, min_pass_run],
               "shuffle":  [...]}
    n_runs: override of the per-mode expected run counts (for tests).
    """
    n_runs = n_runs or N_RUNS
    baseline = runs_nid.get("baseline", list(EMPTY_RUNS))
    shuffle = runs_nid.get("shuffle", list(EMPTY_RUNS))

    if baseline[0] != n_runs["baseline"] or shuffle[0] != n_runs["shuffle"]:
        return 0, None

    if baseline[1] == 0:
        if shuffle[1] == 0:
            return 0, NON_FLAKY
        return shuffle[2], OD_FLAKY

    if baseline[1] == baseline[0]:
        if shuffle[1] == shuffle[0]:
            return 0, NON_FLAKY
        return shuffle[3], OD_FLAKY

    return max(baseline[2], baseline[3]), FLAKY
