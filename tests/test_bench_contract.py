"""Guards for the bench.py driver contract.

The round driver invokes `python bench.py --gpus N --steps K --warmup W`
(N>1 via torch.distributed.run OR directly — bench.py self-launches) and
parses ONE JSON line from rank 0.  These tests run the ref backend on
tiny synthetic data to lock the surface without a GPU.
"""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _run_bench(*args, timeout=420):
    env = dict(os.environ, PYTHONPATH=REPO)
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), *args],
        capture_output=True, text=True, timeout=timeout, env=env, cwd=REPO)
    assert proc.returncode == 0, proc.stdout + proc.stderr
    lines = [ln for ln in proc.stdout.strip().splitlines()
             if ln.startswith("{")]
    assert lines, proc.stdout
    return json.loads(lines[-1])


@pytest.mark.timeout(600)
def test_single_rank_json_contract():
    out = _run_bench("--backend", "ref", "--cells", "3", "--n-tests", "300",
                     "--steps", "1", "--warmup", "0")
    assert REQUIRED_KEYS <= set(out)
    assert out["n_gpus"] == 1
    assert out["steps"] == 1 and out["warmup"] == 0
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["higher_is_better"] is True
    assert out["scaling"] == "strong"
    assert out["data"] == "synthetic"
    assert out["config"]["grid_cells"] == 3
    assert out["config"]["n_tests"] == 300


@pytest.mark.timeout(600)
def test_self_launch_two_ranks():
    """--gpus 2 with no torchrun env must self-launch 2 ranks (gloo on
    CPU) and report n_gpus=2 — the SCALE-run contract."""
    out = _run_bench("--gpus", "2", "--backend", "ref", "--cells", "3",
                     "--n-tests", "300", "--steps", "1", "--warmup", "0")
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"].startswith("cell-sharded dp2")


@pytest.mark.timeout(600)
def test_shap_stage_contract():
    out = _run_bench("--stage", "shap", "--backend", "ref", "--n-tests",
                     "120", "--steps", "1", "--warmup", "0")
    assert REQUIRED_KEYS <= set(out)
    assert out["metric"].startswith("shap-configs/sec")
    assert len(out["config"]["shap_configs"]) == 2
