"""End-to-end tests of the scores stage (CPU reference backend) and the
scores.pkl format contract, plus distributed (gloo, world=2) equivalence."""

import os
import pickle
import subprocess
import sys
import textwrap

import numpy as np
import pytest

from flake16_framework_amd.configgrid import iter_config_keys
from flake16_framework_amd.constants import FLAKY
from flake16_framework_amd.dataset.synthetic import make_synthetic_tests
from flake16_framework_amd.engine.scores import evaluate_cell_ref, run_scores

ALL_KEYS = list(iter_config_keys())


def _small_tests(n=400, seed=2):
    return make_synthetic_tests(n_tests=n, seed=seed)


def _find_cell(flaky, fset, prep, bal, model):
    keys = (flaky, fset, prep, bal, model)
    return ALL_KEYS.index(keys), keys


class TestEvaluateCell:
    @pytest.mark.parametrize("prep,bal,model", [
        ("None", "None", "Decision Tree"),
        ("Scaling", "SMOTE", "Random Forest"),
        ("PCA", "Tomek Links", "Extra Trees"),
        ("None", "ENN", "Decision Tree"),
        ("Scaling", "SMOTE ENN", "Decision Tree"),
        ("None", "SMOTE Tomek", "Decision Tree"),
    ])
    def test_cell_runs_and_format(self, prep, bal, model):
        tests = _small_tests()
        cell_idx, keys = _find_cell("NOD", "Flake16", prep, bal, model)
        out = evaluate_cell_ref(keys, cell_idx, tests=tests)

        t_train, t_test, scores, scores_total = out
        assert t_train >= 0 and t_test >= 0
        assert set(scores.keys()) == set(tests.keys())
        for proj, row in scores.items():
            assert len(row) == 6
            fp, fn, tp = row[:3]
            assert all(isinstance(v, int) and v >= 0 for v in (fp, fn, tp))
        # totals equal sum of per-project counts
        for k in range(3):
            assert scores_total[k] == sum(r[k] for r in scores.values())

    def test_forest_beats_chance_on_synthetic(self):
        tests = _small_tests(600, seed=5)
        cell_idx, keys = _find_cell("NOD", "Flake16", "None", "SMOTE",
                                    "Random Forest")
        _, _, _, total = evaluate_cell_ref(keys, cell_idx, tests=tests)
        f1 = total[5]
        assert f1 is not None and f1 > 0.3, total

    def test_deterministic(self):
        tests = _small_tests()
        cell_idx, keys = _find_cell("OD", "FlakeFlagger", "Scaling", "SMOTE",
                                    "Decision Tree")
        a = evaluate_cell_ref(keys, cell_idx, tests=tests)
        b = evaluate_cell_ref(keys, cell_idx, tests=tests)
        assert a[2] == b[2] and a[3] == b[3]


class TestRunScores:
    def test_subset_and_pickle_roundtrip(self, tmp_path):
        tests = _small_tests()
        cells = [0, 107, 215]
        result = run_scores(tests=tests, backend="ref", cells=cells)
        assert len(result) == 3
        for keys, val in result.items():
            assert keys in ALL_KEYS
            assert len(val) == 4

        p = tmp_path / "scores.pkl"
        with open(p, "wb") as fd:
            pickle.dump(result, fd)
        with open(p, "rb") as fd:
            loaded = pickle.load(fd)
        assert loaded.keys() == result.keys()


WORKER = textwrap.dedent("""
    import json, os, pickle, sys
    import torch.distributed as dist
    sys.path.insert(0, {repo!r})
    from flake16_framework_amd.dataset.synthetic import make_synthetic_tests
    from flake16_framework_amd.engine.scores import run_scores
    from flake16_framework_amd.parallel import comm

    dist.init_process_group("gloo")
    rank, world = comm.rank_world()
    tests = make_synthetic_tests(n_tests=300, seed=2)
    n_cells = int(os.environ.get("TEST_N_CELLS", "6"))
    cells = comm.shard_cells(world, rank, n_cells=n_cells)
    result = run_scores(tests=tests, backend="ref", cells=cells)
    full = comm.gather_scores(result)
    if rank == 0:
        with open({out!r}, "wb") as fd:
            pickle.dump(full, fd)
    dist.destroy_process_group()
""")


SHAP_WORKER = textwrap.dedent("""
    import os, pickle, sys
    import torch.distributed as dist
    sys.path.insert(0, {repo!r})
    from flake16_framework_amd.dataset.synthetic import make_synthetic_tests
    from flake16_framework_amd.engine.shap_stage import write_shap

    dist.init_process_group("gloo")
    tests = make_synthetic_tests(n_tests=150, seed=2)
    write_shap(tests=tests, shap_file={out!r}, backend="ref")
    dist.destroy_process_group()
""")


class TestDistributedGloo:
    def _launch(self, tmp_path, n_cells, port):
        out = str(tmp_path / "dist_scores.pkl")
        script = tmp_path / "worker.py"
        script.write_text(WORKER.format(repo="/root/repo", out=out))
        env = dict(os.environ, MASTER_ADDR="127.0.0.1",
                   MASTER_PORT=str(port), TEST_N_CELLS=str(n_cells))
        procs = []
        for rank in range(2):
            env_r = dict(env, RANK=str(rank), WORLD_SIZE="2",
                         LOCAL_RANK=str(rank))
            procs.append(subprocess.Popen(
                [sys.executable, str(script)], env=env_r,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        for p in procs:
            outp, _ = p.communicate(timeout=300)
            assert p.returncode == 0, outp.decode()
        with open(out, "rb") as fd:
            return pickle.load(fd)

    def test_empty_rank_gathers_clean(self, tmp_path):
        """n_cells=3 puts the whole (single) balance group on rank 0;
        rank 1 owns nothing — the all-reduce must still produce the
        full result (guards the empty-shard path at world=8)."""
        dist_result = self._launch(tmp_path, n_cells=3, port=29612)
        tests = make_synthetic_tests(n_tests=300, seed=2)
        single = run_scores(tests=tests, backend="ref", cells=[0, 1, 2])
        assert set(dist_result) == set(single)
        for k in single:
            assert dist_result[k][2] == single[k][2]
            assert dist_result[k][3] == single[k][3]

    def test_shap_rank_split_equals_single(self, tmp_path):
        """write_shap splits the 2 configs across ranks and all-gathers;
        the merged shap.pkl must equal the single-process result."""
        out = str(tmp_path / "shap.pkl")
        script = tmp_path / "shap_worker.py"
        script.write_text(SHAP_WORKER.format(repo="/root/repo", out=out))
        env = dict(os.environ, MASTER_ADDR="127.0.0.1",
                   MASTER_PORT="29613")
        procs = []
        for rank in range(2):
            env_r = dict(env, RANK=str(rank), WORLD_SIZE="2",
                         LOCAL_RANK=str(rank))
            procs.append(subprocess.Popen(
                [sys.executable, str(script)], env=env_r,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        for p in procs:
            outp, _ = p.communicate(timeout=300)
            assert p.returncode == 0, outp.decode()

        from flake16_framework_amd.engine.shap_stage import write_shap
        tests = make_synthetic_tests(n_tests=150, seed=2)
        single = write_shap(tests=tests,
                            shap_file=str(tmp_path / "single.pkl"),
                            backend="ref")
        with open(out, "rb") as fd:
            dist_nod, dist_od = pickle.load(fd)
        np.testing.assert_array_equal(dist_nod, single[0])
        np.testing.assert_array_equal(dist_od, single[1])

    def test_two_rank_shard_equals_single(self, tmp_path):
        out = str(tmp_path / "dist_scores.pkl")
        script = tmp_path / "worker.py"
        script.write_text(WORKER.format(repo="/root/repo", out=out))

        env = dict(os.environ)
        env.setdefault("MASTER_ADDR", "127.0.0.1")
        env.setdefault("MASTER_PORT", "29611")
        procs = []
        for rank in range(2):
            env_r = dict(env, RANK=str(rank), WORLD_SIZE="2",
                         LOCAL_RANK=str(rank))
            procs.append(subprocess.Popen(
                [sys.executable, str(script)], env=env_r,
                stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        for p in procs:
            outp, _ = p.communicate(timeout=300)
            assert p.returncode == 0, outp.decode()

        with open(out, "rb") as fd:
            dist_result = pickle.load(fd)

        tests = make_synthetic_tests(n_tests=300, seed=2)
        single = run_scores(tests=tests, backend="ref", cells=list(range(6)))
        for keys, val in single.items():
            assert keys in dist_result
            assert dist_result[keys][2] == val[2]
            assert dist_result[keys][3] == val[3]

    def test_shard_cells_partition(self):
        from flake16_framework_amd.parallel.comm import shard_cells
        for world in (1, 2, 4, 8):
            all_cells = sorted(
                c for r in range(world) for c in shard_cells(world, r))
            assert all_cells == list(range(216))
        # LPT balance: within ~12% across 8 ranks on the cost model
        from flake16_framework_amd.configgrid import cell_cost_estimate
        keys = ALL_KEYS
        loads = [sum(cell_cost_estimate(keys[c]) for c in shard_cells(8, r))
                 for r in range(8)]
        assert max(loads) / min(loads) < 1.12

    def test_ref_process_pool_matches_serial(self):
        """run_scores(processes=2) — the reference's Pool execution model
        — must produce the serial results."""
        from flake16_framework_amd.engine.scores import run_scores
        tests = _small_tests(300, seed=4)
        cells = [i for i, k in enumerate(ALL_KEYS)
                 if k[4] == "Decision Tree" and k[3] == "None"][:3]
        serial = run_scores(tests=tests, backend="ref", cells=cells)
        pooled = run_scores(tests=tests, backend="ref", cells=cells,
                            processes=2)
        assert serial.keys() == pooled.keys()
        for k in serial:
            assert pooled[k][2] == serial[k][2]
            assert pooled[k][3] == serial[k][3]

    def test_shard_never_splits_balance_groups(self):
        """The fused mixed-model fit assumes a rank owns WHOLE balance
        groups (the 3 model cells share balanced folds and one fit)."""
        from flake16_framework_amd.configgrid import balance_group_index
        from flake16_framework_amd.parallel.comm import shard_cells
        for world in (2, 4, 8):
            owner = {}
            for r in range(world):
                for c in shard_cells(world, r):
                    g = balance_group_index(ALL_KEYS[c])
                    assert owner.setdefault(g, r) == r, (world, g)

    def test_checkpoint_reshard_drops_foreign_cells(self, tmp_path,
                                                    monkeypatch):
        """A resume under a different world size must not merge cells the
        current shard does not own (they would double-count through the
        all-reduce SUM)."""
        import json
        import flake16_framework_amd.engine.scores as scores_mod
        from flake16_framework_amd.engine.scores import write_scores
        from flake16_framework_amd.parallel import comm

        tests = _small_tests(400, seed=2)
        monkeypatch.chdir(tmp_path)
        with open(tmp_path / "tests.json", "w") as fd:
            json.dump(tests, fd)

        dt = [i for i, k in enumerate(ALL_KEYS)
              if k[4] == "Decision Tree"][:4]
        ckpt = str(tmp_path / "ck")

        # "old world" run owned 4 cells and checkpointed them all
        monkeypatch.setattr(comm, "shard_cells",
                            lambda world, rank, n_cells=None: dt)
        write_scores(tests_file=str(tmp_path / "tests.json"),
                     scores_file=str(tmp_path / "s.pkl"),
                     backend="ref", checkpoint=ckpt)

        # "new world": this rank owns only the first two of them — the
        # other two checkpointed cells must be ignored, not merged
        monkeypatch.setattr(comm, "shard_cells",
                            lambda world, rank, n_cells=None: dt[:2])
        evaluated = []
        real_eval = scores_mod.evaluate_cell_ref

        def spy(keys, ci, **kw):
            evaluated.append(ci)
            return real_eval(keys, ci, **kw)

        monkeypatch.setattr(scores_mod, "evaluate_cell_ref", spy)
        r2 = write_scores(tests_file=str(tmp_path / "tests.json"),
                         scores_file=str(tmp_path / "s.pkl"),
                         backend="ref", checkpoint=ckpt)
        assert not evaluated                 # both owned cells checkpointed
        assert set(r2) == {ALL_KEYS[c] for c in dt[:2]}


class TestCheckpointAndTrace:
    def test_checkpoint_resume_skips_done_cells(self, tmp_path, monkeypatch):
        import flake16_framework_amd.engine.scores as scores_mod
        from flake16_framework_amd.engine.scores import write_scores

        tests = _small_tests(400, seed=2)
        monkeypatch.chdir(tmp_path)

        # restrict the sweep to 4 cheap cells via shard_cells monkeypatch
        cheap = [i for i, k in enumerate(ALL_KEYS)
                 if k[4] == "Decision Tree"][:4]
        from flake16_framework_amd.parallel import comm
        monkeypatch.setattr(comm, "shard_cells",
                            lambda world, rank, n_cells=None: cheap)

        ckpt = str(tmp_path / "ck")
        r1 = write_scores(tests_file=None, scores_file=str(tmp_path / "s.pkl"),
                          backend="ref", checkpoint=ckpt) if False else None
        # write_scores reads tests.json; provide it
        import json
        with open(tmp_path / "tests.json", "w") as fd:
            json.dump(tests, fd)
        r1 = write_scores(tests_file=str(tmp_path / "tests.json"),
                          scores_file=str(tmp_path / "s.pkl"),
                          backend="ref", checkpoint=ckpt)
        assert len(r1) == 4
        ck = scores_mod._load_checkpoint(ckpt + ".rank0")
        assert len(ck) == 4

        # second run: all cells come from the checkpoint (evaluate never
        # called)
        called = []
        monkeypatch.setattr(scores_mod, "evaluate_cell_ref",
                            lambda *a, **k: called.append(1))
        r2 = write_scores(tests_file=str(tmp_path / "tests.json"),
                          scores_file=str(tmp_path / "s.pkl"),
                          backend="ref", checkpoint=ckpt)
        assert not called
        for k in r1:
            assert r2[k][2] == r1[k][2]

    def test_trace_spans_written(self, tmp_path, monkeypatch):
        from flake16_framework_amd.utils import trace
        tpath = str(tmp_path / "trace.jsonl")
        monkeypatch.setenv("FLAKE16_TRACE", tpath)
        monkeypatch.setattr(trace, "_explicit", False)

        tests = _small_tests(400, seed=2)
        run_scores(tests=tests, backend="ref", cells=[2])  # a DT cell

        import json
        lines = [json.loads(l) for l in open(tpath)]
        assert any(r["name"] == "cell" for r in lines)
        assert all("dur_s" in r for r in lines)
