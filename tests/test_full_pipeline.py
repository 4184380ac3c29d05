"""Full-pipeline integration: run stage (local mode, no docker) over a
synthetic flaky test suite -> data/ -> collation -> tests.json with the
expected labels and all 16 features.

The fake subject has three tests:
  test_od_victim   passes in original order, fails when the shuffled order
                   runs test_zz_breaker first        -> OD_FLAKY
  test_nod_flaky   fails on every 3rd run (persistent counter) -> FLAKY
  test_stable      always passes                     -> NON_FLAKY
"""

import json
import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SUITE = '''
import os
import sys


def _run_info():
    # container name encodes (mode, run number): proj_<mode>_<runN>
    for a in sys.argv:
        if "--record-file=" in a or "--testinspect=" in a:
            base = os.path.basename(a.split("=", 1)[1])
            parts = base.split("_")
            return parts[1], int(parts[2].split(".")[0])
    return "none", 0


MODE, RUN_N = _run_info()

# per-run marker: local-mode runs share the checkout (real study runs are
# container-isolated), so order-dependence state must not leak across the
# concurrent pool runs
MARKER = os.path.join(os.path.dirname(__file__),
                      "breaker_ran_%s_%d" % (MODE, RUN_N))

if os.path.exists(MARKER):
    os.remove(MARKER)


def test_od_victim():
    assert not os.path.exists(MARKER)


def test_nod_flaky():
    # deterministic intermittent baseline failure -> NOD-flaky
    assert not (MODE == "baseline" and RUN_N % 2 == 1)


def test_stable():
    assert 1 + 1 == 2


def test_zz_breaker():
    with open(MARKER, "w") as fd:
        fd.write("x")
'''

N_RUNS_SMALL = {"baseline": 6, "shuffle": 12, "testinspect": 1}


@pytest.mark.timeout(600)
def test_run_tests_pipeline(tmp_path, monkeypatch):
    from flake16_framework_amd.constants import FLAKY, NON_FLAKY, OD_FLAKY
    from flake16_framework_amd.dataset.collate import collate
    from flake16_framework_amd.dataset.tests_io import build_tests
    import flake16_framework_amd.dataset.labeling as labeling
    from flake16_framework_amd.orchestrate import runner

    # fake subject checkout
    subjects_dir = tmp_path / "subjects"
    proj_dir = subjects_dir / "proj" / "proj"
    proj_dir.mkdir(parents=True)
    (proj_dir / "test_suite.py").write_text(SUITE)

    # a git history so the churn collector has line-change counts
    import subprocess
    git = ["git", "-C", str(proj_dir), "-c", "user.email=t@t",
           "-c", "user.name=t"]
    subprocess.run([*git[:3], "init", "-q"], check=True)
    subprocess.run([*git, "add", "test_suite.py"], check=True)
    subprocess.run([*git, "commit", "-q", "-m", "init"], check=True)
    (proj_dir / "test_suite.py").write_text(SUITE + "\n# touched\n")
    subprocess.run([*git, "commit", "-q", "-am", "touch"], check=True)

    subjects_file = tmp_path / "subjects.txt"
    subjects_file.write_text("local/proj,HEAD,.,python -m pytest -q -p no:cacheprovider test_suite.py\n")

    monkeypatch.chdir(tmp_path)
    monkeypatch.setenv("FLAKE16_LOCAL_RUN", "1")
    monkeypatch.setenv("FLAKE16_SUBJECTS_DIR", str(subjects_dir))
    monkeypatch.setenv("FLAKE16_DATA_DIR", str(tmp_path / "data"))
    monkeypatch.setenv("PYTHONPATH", REPO)
    monkeypatch.setattr(labeling, "N_RUNS", N_RUNS_SMALL)

    # --- run stage (resumable driver, process pool, local runner) --------
    with pytest.raises(SystemExit) as exc:
        runner.drive_runs(
            "baseline", "shuffle", "testinspect",
            subjects_file=str(subjects_file), n_runs=N_RUNS_SMALL)
    assert exc.value.code == 0, open(
        tmp_path / "stdout" / "proj_baseline_0").read()

    # completion log covers every run (crash-restart bookkeeping)
    log = set(runner.read_log(str(tmp_path / "log.txt")))
    assert len(log) == 6 + 12 + 1

    data_files = os.listdir(tmp_path / "data")
    assert len([f for f in data_files if f.endswith(".tsv")]) >= 6 + 12

    # --- collation + labeling -> tests.json ------------------------------
    collated = collate(data_dir=str(tmp_path / "data"),
                       subjects_dir=str(subjects_dir))
    tests = build_tests(collated)
    assert "proj" in tests, list(collated["proj"].tests)
    rows = tests["proj"]

    by_name = {nid.split("::")[-1]: row for nid, row in rows.items()}
    assert by_name["test_stable"][1] == NON_FLAKY
    assert by_name["test_zz_breaker"][1] == NON_FLAKY
    assert by_name["test_od_victim"][1] == OD_FLAKY
    assert by_name["test_nod_flaky"][1] == FLAKY

    # 16 features per row; static/rusage/coverage features populated
    for name, row in by_name.items():
        assert len(row) == 2 + 16
        covered_lines, _, _, t_exec = row[2], row[3], row[4], row[5]
        assert covered_lines > 0, name     # coverage collector worked
        assert t_exec >= 0                 # rusage collector worked
        assert row[2 + 14] >= 2            # Test LoC (static collector)
