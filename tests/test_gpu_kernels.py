"""GPU tests: HIP kernels vs the numpy/torch reference implementations.

Exactness contract (see models/forest_ref.py, balance/__init__.py):
  - binning, knn, smote, enn/tomek masks, forest trees and predictions are
    BIT-IDENTICAL to the reference given identical input bits;
  - scaler/PCA (parallel fp64 reductions) match within tolerance;
  - full-cell evaluation matches the CPU path exactly for cells without
    preprocessing, and at metric tolerance otherwise.
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ops():
    from flake16_framework_amd.ops.backend import get_ops
    return get_ops()


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda:0")


def _data(n=500, f=16, seed=3, sep=1.4):
    rng = np.random.RandomState(seed)
    y = (rng.rand(n) < 0.25).astype(np.uint8)
    X = rng.randn(n, f).astype(np.float32)
    X[y == 1, : f // 2] += sep
    return X, y


def _pad16(X):
    n, f = X.shape
    out = np.zeros((n, 16), dtype=X.dtype)
    out[:, :f] = X
    return out


class TestBinKnnSmote:
    def test_bin_codes_exact(self, ops, dev):
        from flake16_framework_amd.models.binning import (
            bin_codes, compute_bin_cuts,
        )
        X, _ = _data(700)
        cuts = compute_bin_cuts(X)
        ref = bin_codes(X, cuts)

        flat = np.concatenate(cuts).astype(np.float32)
        off = np.zeros(17, dtype=np.int32)
        off[1:17] = np.cumsum([len(c) for c in cuts] + [0] * (16 - len(cuts)))
        got = ops.bin_codes(torch.from_numpy(_pad16(X)).to(dev),
                            torch.from_numpy(flat).to(dev),
                            torch.from_numpy(off).to(dev), 16)
        np.testing.assert_array_equal(got.cpu().numpy()[:, :16], ref)

    @pytest.mark.parametrize("k,skip", [(1, True), (3, True), (5, True)])
    def test_knn_exact(self, ops, dev, k, skip):
        from flake16_framework_amd.balance import knn_indices
        X, _ = _data(600)
        ref = knn_indices(X, X, k, skip_identity=skip)
        got = ops.knn(torch.from_numpy(_pad16(X)).to(dev), k, skip)
        np.testing.assert_array_equal(got.cpu().numpy(), ref)

    def test_knn_f7_exact(self, ops, dev):
        from flake16_framework_amd.balance import knn_indices
        X, _ = _data(400, f=7)
        ref = knn_indices(X, X, 5, skip_identity=True)
        got = ops.knn(torch.from_numpy(_pad16(X)).to(dev), 5, True)
        np.testing.assert_array_equal(got.cpu().numpy(), ref)

    def test_smote_exact(self, ops, dev):
        from flake16_framework_amd.balance import smote
        X, y = _data(500)
        Xb_ref, yb_ref = smote(X, y, 0, 77)

        n1 = int(y.sum())
        n0 = len(y) - n1
        min_label = 1 if n1 < n0 else 0
        min_rows = np.flatnonzero(y == min_label).astype(np.int32)
        k = min(5, len(min_rows) - 1)
        n_new = abs(n0 - n1)

        Xd = torch.from_numpy(_pad16(X)).to(dev)
        min_rows_d = torch.from_numpy(min_rows).to(dev)
        X_min = Xd.index_select(0, min_rows_d.long()).contiguous()
        nn = ops.knn(X_min, k, True)
        X_new = ops.smote_interpolate(Xd, min_rows_d, nn, n_new, 0, 77)

        np.testing.assert_array_equal(X_new.cpu().numpy()[:, :16],
                                      Xb_ref[len(y):])
        assert (yb_ref[len(y):] == min_label).all()

    def test_enn_tomek_masks_exact(self, ops, dev):
        from flake16_framework_amd.balance import enn_mask, knn_indices, \
            tomek_links_mask
        X, y = _data(400, sep=0.7)   # heavy overlap: many removals
        Xd = torch.from_numpy(_pad16(X)).to(dev)
        yd = torch.from_numpy(y).to(dev)
        maj = 1 if int(y.sum()) > len(y) - int(y.sum()) else 0

        nn3 = ops.knn(Xd, 3, True)
        for clean_all in (False, True):
            ref = enn_mask(X, y, "all" if clean_all else "auto")
            got = ops.enn_keep(yd, nn3, 3, maj, clean_all)
            np.testing.assert_array_equal(got.cpu().numpy().astype(bool), ref)

        nn1 = ops.knn(Xd, 1, True)
        for remove_all in (False, True):
            ref = tomek_links_mask(X, y, "all" if remove_all else "auto")
            got = ops.tomek_keep(yd, nn1[:, 0].contiguous(), maj, remove_all)
            np.testing.assert_array_equal(got.cpu().numpy().astype(bool), ref)


class TestScalerPca:
    def test_scaler_close(self, ops, dev):
        from flake16_framework_amd.preprocess import scaler_fit_transform
        X, _ = _data(800)
        X64 = _pad16(X).astype(np.float64)
        ref = scaler_fit_transform(X.astype(np.float64))
        got = ops.scaler_fit_transform(
            torch.from_numpy(X64).to(dev)).cpu().numpy()
        np.testing.assert_allclose(got[:, :16], ref, atol=1e-10)

    def test_pca_close(self, ops, dev):
        from flake16_framework_amd.preprocess import (
            pca_fit_transform, scaler_fit_transform,
        )
        X, _ = _data(800)
        Xs = scaler_fit_transform(X.astype(np.float64))
        ref = pca_fit_transform(Xs)
        got = ops.pca_fit_transform(
            ops.scaler_fit_transform(
                torch.from_numpy(_pad16(X).astype(np.float64)).to(dev)),
            16).cpu().numpy()[:, :16]
        # same subspace, same ordering, same sign convention
        np.testing.assert_allclose(np.abs(got), np.abs(ref),
                                   rtol=1e-6, atol=1e-7)
        np.testing.assert_allclose(got, ref, rtol=1e-5, atol=1e-6)


class TestForestExact:
    @pytest.mark.parametrize("kind,n_trees", [
        ("decision_tree", 1),
        ("random_forest", 20),
        ("extra_trees", 20),
    ])
    def test_trees_bitwise_vs_ref(self, ops, dev, kind, n_trees):
        from flake16_framework_amd.models.binning import (
            bin_codes, compute_bin_cuts,
        )
        from flake16_framework_amd.models.forest_ref import (
            fit_forest, params_for_model, predict_forest,
        )
        X, y = _data(500)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        spec = {"kind": kind, "n_estimators": n_trees}
        params = params_for_model(spec, seed=0)
        params.n_trees = n_trees
        ref_forest = fit_forest(codes, y, params, job_base=11, cuts=cuts)
        ref_pred = predict_forest(ref_forest, codes)

        codes_d = torch.from_numpy(
            np.ascontiguousarray(_pad16(codes))).to(dev)
        y_d = torch.from_numpy(y).to(dev)
        J = n_trees
        j_row_off = torch.zeros(J, dtype=torch.int32, device=dev)
        j_n = torch.full((J,), len(y), dtype=torch.int32, device=dev)
        j_key = torch.arange(11, 11 + J, dtype=torch.int32, device=dev)
        max_features = 16 if kind == "decision_tree" else 4
        nfeat, nsplit, nleft, ncnt0, ncnt1, j_node_off, node_alloc = \
            ops.forest_fit(codes_d, y_d, j_row_off, j_n, j_key, 16,
                           max_features, params.bootstrap,
                           params.splitter == "random", 0)

        node_alloc_h = node_alloc.cpu().numpy()
        j_node_off_h = j_node_off.cpu().numpy()
        # node counts must match exactly
        for t, tree in enumerate(ref_forest.trees):
            assert node_alloc_h[t] == tree.n_nodes, (t, node_alloc_h[t],
                                                     tree.n_nodes)

        # node numbering may differ (device allocation order): compare
        # structurally by walking both trees from the root.
        nfeat_h = nfeat.cpu().numpy()
        nsplit_h = nsplit.cpu().numpy()
        nleft_h = nleft.cpu().numpy()
        c0_h = ncnt0.cpu().numpy()
        c1_h = ncnt1.cpu().numpy()
        for t, tree in enumerate(ref_forest.trees):
            base = int(j_node_off_h[t])
            stack = [(0, 0)]   # (ref node, dev node)
            while stack:
                r, d = stack.pop()
                assert c0_h[base + d] == tree.count0[r]
                assert c1_h[base + d] == tree.count1[r]
                if tree.feature[r] == -1:
                    assert nfeat_h[base + d] == -1
                    continue
                assert nfeat_h[base + d] == tree.feature[r]
                assert nsplit_h[base + d] == tree.split_bin[r]
                dl = nleft_h[base + d]
                stack.append((tree.left[r], dl))
                stack.append((tree.right[r], dl + 1))

        # predictions bit-identical
        proj_id = torch.zeros(len(y), dtype=torch.int32, device=dev)
        pair_row = torch.arange(len(y), dtype=torch.int32, device=dev)
        pair_fold = torch.zeros(len(y), dtype=torch.int32, device=dev)
        pred, confusion = ops.forest_predict_confusion(
            codes_d, y_d, proj_id, pair_row, pair_fold, j_node_off,
            nfeat, nsplit, nleft, ncnt0, ncnt1, n_trees, 1)
        np.testing.assert_array_equal(pred.cpu().numpy(), ref_pred)

        # confusion bookkeeping
        conf = confusion.cpu().numpy()
        tp = int(((y == 1) & (ref_pred == 1)).sum())
        fp = int(((y == 0) & (ref_pred == 1)).sum())
        fn = int(((y == 1) & (ref_pred == 0)).sum())
        assert conf[1].tolist() == [fp, fn, tp]

    def test_multi_fold_ragged_batch(self, ops, dev):
        """Jobs with different sample counts/offsets in one call."""
        from flake16_framework_amd.models.binning import (
            bin_codes, compute_bin_cuts,
        )
        from flake16_framework_amd.models.forest_ref import (
            ForestParams, fit_forest,
        )
        X, y = _data(400)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        split = [(0, 250), (250, 400)]

        params = ForestParams(3, True, "best", "sqrt", 0)
        refs = []
        for (a, b), jb in zip(split, (0, 128)):
            refs.append(fit_forest(codes[a:b], y[a:b], params, job_base=jb,
                                   cuts=cuts))

        codes_d = torch.from_numpy(
            np.ascontiguousarray(_pad16(codes))).to(dev)
        y_d = torch.from_numpy(y).to(dev)
        j_row_off, j_n, j_key = [], [], []
        for (a, b), jb in zip(split, (0, 128)):
            for t in range(3):
                j_row_off.append(a)
                j_n.append(b - a)
                j_key.append(jb + t)
        out = ops.forest_fit(
            codes_d, y_d,
            torch.tensor(j_row_off, dtype=torch.int32, device=dev),
            torch.tensor(j_n, dtype=torch.int32, device=dev),
            torch.tensor(j_key, dtype=torch.int32, device=dev),
            16, 4, True, False, 0)
        node_alloc_h = out[6].cpu().numpy()
        expected = [t.n_nodes for r in refs for t in r.trees]
        np.testing.assert_array_equal(node_alloc_h, expected)


class TestCellE2E:
    def test_none_preproc_cell_exact_vs_cpu(self, ops):
        from flake16_framework_amd.configgrid import iter_config_keys
        from flake16_framework_amd.dataset.synthetic import (
            make_synthetic_tests,
        )
        from flake16_framework_amd.engine.hip_cell import evaluate_cell_hip
        from flake16_framework_amd.engine.scores import evaluate_cell_ref

        tests = make_synthetic_tests(n_tests=400, seed=2)
        all_keys = list(iter_config_keys())
        for keys in [("NOD", "Flake16", "None", "SMOTE", "Random Forest"),
                     ("NOD", "Flake16", "None", "None", "Decision Tree"),
                     ("OD", "FlakeFlagger", "None", "SMOTE Tomek",
                      "Extra Trees")]:
            ci = all_keys.index(keys)
            ref = evaluate_cell_ref(keys, ci, tests=tests)
            got = evaluate_cell_hip(keys, ci, tests=tests)
            assert got[2] == ref[2], keys
            assert got[3] == ref[3], keys

    def test_fused_group_equals_per_cell(self, ops):
        """The production sweep path fuses a balance group's 3 model
        cells into one mixed-model fit — results must equal per-cell
        evaluation bit-for-bit (same Philox keys per job)."""
        from flake16_framework_amd.configgrid import iter_config_keys
        from flake16_framework_amd.dataset.synthetic import (
            make_synthetic_tests,
        )
        from flake16_framework_amd.engine.hip_cell import SweepContext

        tests = make_synthetic_tests(n_tests=400, seed=2)
        all_keys = list(iter_config_keys())
        group = [k for k in all_keys
                 if k[:4] == ("NOD", "Flake16", "None", "SMOTE")]
        assert len(group) == 3
        cells = [(k, all_keys.index(k)) for k in group]

        ctx = SweepContext(tests=tests)
        fused = ctx.evaluate_group(cells)
        for cell in cells:
            single = SweepContext(tests=tests).evaluate_group([cell])
            keys = cell[0]
            assert fused[keys][2] == single[keys][2], keys
            assert fused[keys][3] == single[keys][3], keys

    def test_preproc_cells_metric_close(self, ops):
        from flake16_framework_amd.configgrid import iter_config_keys
        from flake16_framework_amd.dataset.synthetic import (
            make_synthetic_tests,
        )
        from flake16_framework_amd.engine.hip_cell import evaluate_cell_hip
        from flake16_framework_amd.engine.scores import evaluate_cell_ref

        tests = make_synthetic_tests(n_tests=400, seed=2)
        all_keys = list(iter_config_keys())
        for keys in [("NOD", "Flake16", "Scaling", "SMOTE", "Random Forest"),
                     ("NOD", "Flake16", "PCA", "ENN", "Extra Trees")]:
            ci = all_keys.index(keys)
            ref = evaluate_cell_ref(keys, ci, tests=tests)
            got = evaluate_cell_hip(keys, ci, tests=tests)
            # fp64 reduction order differs: totals agree within tolerance
            rt, gt = ref[3], got[3]
            for a, b in zip(rt[:3], gt[:3]):
                assert abs(a - b) <= max(8, 0.1 * max(a, b)), (keys, rt, gt)


class TestTreeShapGpu:
    def test_treeshap_matches_ref(self, ops, dev):
        from flake16_framework_amd.models.binning import (
            bin_codes, compute_bin_cuts,
        )
        from flake16_framework_amd.models.forest_ref import (
            ForestParams, fit_forest,
        )
        from flake16_framework_amd.models.treeshap_ref import forest_shap

        X, y = _data(300, f=16, seed=9)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        params = ForestParams(10, True, "best", "sqrt", 0)
        forest = fit_forest(codes, y, params, job_base=40, cuts=cuts)
        ref = forest_shap(forest, codes[:50], 16)

        codes_d = torch.from_numpy(np.ascontiguousarray(codes)).to(dev)
        y_d = torch.from_numpy(y).to(dev)
        j_row_off = torch.zeros(10, dtype=torch.int32, device=dev)
        j_n = torch.full((10,), len(y), dtype=torch.int32, device=dev)
        j_key = torch.arange(40, 50, dtype=torch.int32, device=dev)
        nfeat, nsplit, nleft, ncnt0, ncnt1, j_node_off, _ = ops.forest_fit(
            codes_d, y_d, j_row_off, j_n, j_key, 16, 4, True, False, 0)
        phi = ops.treeshap(codes_d[:50].contiguous(), j_node_off, nfeat,
                           nsplit, nleft, ncnt0, ncnt1)
        got = phi.cpu().numpy() / 10
        np.testing.assert_allclose(got[:, :16], ref, atol=1e-9)

    def test_shap_stage_hip_vs_ref(self, ops):
        from flake16_framework_amd.configgrid import SHAP_CONFIGS
        from flake16_framework_amd.dataset.synthetic import (
            make_synthetic_tests,
        )
        from flake16_framework_amd.engine.shap_stage import (
            compute_shap_hip, compute_shap_ref,
        )
        tests = make_synthetic_tests(n_tests=250, seed=4)
        keys = SHAP_CONFIGS[0]
        ref = compute_shap_ref(keys, 0, tests=tests)
        got = compute_shap_hip(keys, 0, tests=tests)
        assert got.shape == ref.shape
        # scaling runs on device (fp64 reduction order differs) -> tolerance
        np.testing.assert_allclose(got, ref, atol=5e-4)


class TestStreamDeterminism:
    def test_stream_count_does_not_change_results(self, ops, monkeypatch):
        """Per-cell results are bitwise identical whether cells run on one
        stream or four (randomness is keyed on data identities, never on
        scheduling)."""
        from flake16_framework_amd.dataset.synthetic import (
            make_synthetic_tests,
        )
        from flake16_framework_amd.engine.scores import run_scores

        tests = make_synthetic_tests(n_tests=400, seed=2)
        cells = [0, 1, 107]
        monkeypatch.setenv("FLAKE16_STREAMS", "1")
        r1 = run_scores(tests=tests, backend="hip", cells=cells)
        monkeypatch.setenv("FLAKE16_STREAMS", "4")
        r4 = run_scores(tests=tests, backend="hip", cells=cells)
        for k in r1:
            assert r1[k][2] == r4[k][2]
            assert r1[k][3] == r4[k][3]


class TestWideForest:
    def test_ge_64k_samples_bitwise(self, ops, dev):
        """Nodes >= 2^16 samples take the WIDE (unpacked) histogram path;
        trees must still match the numpy reference bitwise."""
        from flake16_framework_amd.models.binning import (
            bin_codes, compute_bin_cuts,
        )
        from flake16_framework_amd.models.forest_ref import (
            ForestParams, fit_forest,
        )
        X, y = _data(70000, seed=11, sep=1.0)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        params = ForestParams(2, True, "best", "sqrt", 0)
        ref = fit_forest(codes, y, params, job_base=5, cuts=cuts)

        codes_d = torch.from_numpy(np.ascontiguousarray(codes)).to(dev)
        y_d = torch.from_numpy(y).to(dev)
        out = ops.forest_fit(
            codes_d, y_d,
            torch.zeros(2, dtype=torch.int32, device=dev),
            torch.full((2,), len(y), dtype=torch.int32, device=dev),
            torch.arange(5, 7, dtype=torch.int32, device=dev),
            16, 4, True, False, 0)
        node_alloc = out[6].cpu().numpy()
        assert node_alloc[0] == ref.trees[0].n_nodes
        assert node_alloc[1] == ref.trees[1].n_nodes


class TestTreeShapPaths:
    def test_leafpath_kernel_matches_ref(self, ops, dev):
        from flake16_framework_amd.models.binning import (
            bin_codes, compute_bin_cuts,
        )
        from flake16_framework_amd.models.forest_ref import (
            ForestParams, fit_forest,
        )
        from flake16_framework_amd.models.leafpaths import build_leaf_paths
        from flake16_framework_amd.models.treeshap_ref import forest_shap

        X, y = _data(300, f=16, seed=9)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        params = ForestParams(10, True, "best", "sqrt", 0)
        forest = fit_forest(codes, y, params, job_base=40, cuts=cuts)
        ref = forest_shap(forest, codes[:50], 16)

        codes_d = torch.from_numpy(np.ascontiguousarray(codes)).to(dev)
        y_d = torch.from_numpy(y).to(dev)
        nfeat, nsplit, nleft, ncnt0, ncnt1, j_node_off, node_alloc = \
            ops.forest_fit(
                codes_d, y_d,
                torch.zeros(10, dtype=torch.int32, device=dev),
                torch.full((10,), len(y), dtype=torch.int32, device=dev),
                torch.arange(40, 50, dtype=torch.int32, device=dev),
                16, 4, True, False, 0)
        leaf_tree, leaf_off, path_nodes, _ = build_leaf_paths(
            nfeat.cpu().numpy(), nleft.cpu().numpy(),
            j_node_off.cpu().numpy(), node_alloc.cpu().numpy())
        phi = ops.treeshap_paths(
            codes_d[:50].contiguous(),
            torch.from_numpy(leaf_tree).to(dev),
            torch.from_numpy(leaf_off).to(dev),
            torch.from_numpy(path_nodes).to(dev),
            j_node_off, nfeat, nsplit, nleft, ncnt0, ncnt1)
        np.testing.assert_allclose(phi.cpu().numpy() / 10, ref, atol=1e-9)
