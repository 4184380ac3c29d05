"""Golden tests: folds vs sklearn (exact), forest vs sklearn (metric-level),
preprocessing vs sklearn, balancing sanity."""

import numpy as np
import pytest

from flake16_framework_amd.balance import (
    apply_balancing, enn_mask, smote, tomek_links_mask,
)
from flake16_framework_amd.engine.folds import (
    stratified_kfold_assignments, stratified_kfold_split,
)
from flake16_framework_amd.models.binning import (
    bin_codes, compute_bin_cuts, n_bins_per_feature,
)
from flake16_framework_amd.models.forest_ref import (
    ForestParams, fit_forest, params_for_model, predict_forest,
)
from flake16_framework_amd.preprocess import (
    pca_fit_transform, scaler_fit_transform,
)

sklearn = pytest.importorskip("sklearn")


def _blob_data(n=400, f=16, seed=7, sep=1.6):
    rng = np.random.RandomState(seed)
    y = (rng.rand(n) < 0.3).astype(np.uint8)
    X = rng.randn(n, f).astype(np.float32)
    X[y == 1, : f // 2] += sep
    return X, y


class TestStratifiedKFold:
    @pytest.mark.parametrize("seed", [0, 1, 42])
    @pytest.mark.parametrize("n,frac", [(200, 0.3), (997, 0.08)])
    def test_matches_sklearn_exactly(self, seed, n, frac):
        from sklearn.model_selection import StratifiedKFold
        rng = np.random.RandomState(seed + 100)
        y = rng.rand(n) < frac
        X = np.zeros((n, 1))

        skf = StratifiedKFold(n_splits=10, shuffle=True, random_state=seed)
        ours = list(stratified_kfold_split(y, 10, True, seed))
        theirs = list(skf.split(X, y))
        for (otr, ote), (str_, ste) in zip(ours, theirs):
            np.testing.assert_array_equal(otr, str_)
            np.testing.assert_array_equal(ote, ste)

    def test_assignment_is_deterministic(self):
        y = np.array([0, 1] * 50)
        a = stratified_kfold_assignments(y, 10, True, 0)
        b = stratified_kfold_assignments(y, 10, True, 0)
        np.testing.assert_array_equal(a, b)


class TestPreprocessing:
    def test_scaler_matches_sklearn(self):
        from sklearn.preprocessing import StandardScaler
        X, _ = _blob_data(300)
        X[:, 5] = 3.25  # constant column
        ours = scaler_fit_transform(X)
        theirs = StandardScaler().fit_transform(X)
        np.testing.assert_allclose(ours, theirs, atol=1e-6)

    def test_pca_matches_sklearn_up_to_sign(self):
        from sklearn.decomposition import PCA
        X, _ = _blob_data(300)
        Xs = scaler_fit_transform(X)
        ours = pca_fit_transform(Xs)
        theirs = PCA(random_state=0).fit_transform(Xs)
        assert ours.shape == theirs.shape
        np.testing.assert_allclose(np.abs(ours), np.abs(theirs), atol=1e-6)


class TestBinning:
    def test_codes_reversible_thresholds(self):
        X, _ = _blob_data(500)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        assert codes.dtype == np.uint8
        f = 0
        b = 100
        # split "code <= b" must equal "x < cuts[f][b]"
        np.testing.assert_array_equal(
            codes[:, f] <= b, X[:, f].astype(np.float32) < cuts[f][b])

    def test_few_distinct_values_get_exact_cuts(self):
        X = np.array([[0.0], [1.0], [1.0], [3.0]], dtype=np.float32)
        cuts = compute_bin_cuts(X)
        np.testing.assert_allclose(cuts[0], [0.5, 2.0])
        assert n_bins_per_feature(cuts)[0] == 3


def _f1(y_true, y_pred):
    tp = int(((y_true == 1) & (y_pred == 1)).sum())
    fp = int(((y_true == 0) & (y_pred == 1)).sum())
    fn = int(((y_true == 1) & (y_pred == 0)).sum())
    if tp == 0:
        return 0.0
    p = tp / (tp + fp)
    r = tp / (tp + fn)
    return 2 * p * r / (p + r)


class TestForestGolden:
    """Metric-level parity with sklearn on held-out F1 (same folds)."""

    @pytest.mark.parametrize("kind,skl_cls", [
        ("decision_tree", "DecisionTreeClassifier"),
        ("random_forest", "RandomForestClassifier"),
        ("extra_trees", "ExtraTreesClassifier"),
    ])
    def test_f1_close_to_sklearn(self, kind, skl_cls):
        from sklearn.ensemble import (
            ExtraTreesClassifier, RandomForestClassifier,
        )
        from sklearn.tree import DecisionTreeClassifier
        cls = {"DecisionTreeClassifier": DecisionTreeClassifier,
               "RandomForestClassifier": RandomForestClassifier,
               "ExtraTreesClassifier": ExtraTreesClassifier}[skl_cls]

        X, y = _blob_data(600, sep=1.2)
        tr = np.arange(0, 450)
        te = np.arange(450, 600)

        n_est = 1 if kind == "decision_tree" else 50
        spec = {"kind": kind, "n_estimators": n_est}
        params = params_for_model(spec, seed=0)
        params.n_trees = n_est

        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        forest = fit_forest(codes[tr], y[tr], params, job_base=0, cuts=cuts)
        ours = predict_forest(forest, codes[te])

        kwargs = {"random_state": 0}
        if kind != "decision_tree":
            kwargs["n_estimators"] = n_est
        skl = cls(**kwargs).fit(X[tr], y[tr])
        theirs = skl.predict(X[te])

        f_ours, f_theirs = _f1(y[te], ours), _f1(y[te], theirs)
        assert f_ours > 0.6
        assert abs(f_ours - f_theirs) < 0.1, (f_ours, f_theirs)

    def test_pure_node_is_leaf_and_deterministic(self):
        X, y = _blob_data(200)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        params = ForestParams(5, True, "best", "sqrt", 0)
        f1 = fit_forest(codes, y, params, job_base=7, cuts=cuts)
        f2 = fit_forest(codes, y, params, job_base=7, cuts=cuts)
        for t1, t2 in zip(f1.trees, f2.trees):
            np.testing.assert_array_equal(t1.feature, t2.feature)
            np.testing.assert_array_equal(t1.split_bin, t2.split_bin)
        # training-set prediction of a fully-grown single tree is perfect
        dt = fit_forest(codes, y, ForestParams(1, False, "best", "all", 0),
                        job_base=0, cuts=cuts)
        np.testing.assert_array_equal(predict_forest(dt, codes), y)

    def test_job_base_changes_bootstrap(self):
        X, y = _blob_data(200)
        cuts = compute_bin_cuts(X)
        codes = bin_codes(X, cuts)
        params = ForestParams(1, True, "best", "sqrt", 0)
        fa = fit_forest(codes, y, params, job_base=0, cuts=cuts)
        fb = fit_forest(codes, y, params, job_base=1000, cuts=cuts)
        assert (fa.trees[0].n_nodes != fb.trees[0].n_nodes
                or not np.array_equal(fa.trees[0].feature, fb.trees[0].feature))


class TestBalancing:
    def test_smote_balances_to_parity(self):
        X, y = _blob_data(300)
        Xb, yb = smote(X, y, 0, 1)
        assert int(yb.sum()) == int(len(yb) - yb.sum())
        n_new = len(yb) - len(y)
        # synthetic rows are convex combinations: inside minority bbox
        mins = X[y == 1].min(axis=0) - 1e-5
        maxs = X[y == 1].max(axis=0) + 1e-5
        new = Xb[len(y):]
        assert ((new >= mins) & (new <= maxs)).all()
        assert n_new > 0 and (yb[len(y):] == 1).all()

    def test_smote_deterministic_per_key(self):
        X, y = _blob_data(300)
        Xa, _ = smote(X, y, 0, 5)
        Xb, _ = smote(X, y, 0, 5)
        Xc, _ = smote(X, y, 0, 6)
        np.testing.assert_array_equal(Xa, Xb)
        assert not np.array_equal(Xa, Xc)

    def test_tomek_removes_majority_of_links_only(self):
        # two interleaved points form a tomek link
        X = np.array([[0.0], [0.1], [5.0], [6.0], [7.0]], dtype=np.float32)
        y = np.array([1, 0, 0, 0, 0], dtype=np.uint8)
        keep = tomek_links_mask(X, y, "auto")
        assert not keep[1] and keep[0]
        keep_all = tomek_links_mask(X, y, "all")
        assert not keep_all[0] and not keep_all[1]

    def test_enn_removes_noisy_majority(self):
        rng = np.random.RandomState(0)
        X0 = rng.randn(100, 2).astype(np.float32)
        X1 = rng.randn(40, 2).astype(np.float32) + 8
        noisy = np.array([[8.0, 8.0]], dtype=np.float32)  # class-0 inside 1s
        X = np.vstack([X0, noisy, X1])
        y = np.array([0] * 101 + [1] * 40, dtype=np.uint8)
        keep = enn_mask(X, y, "auto")
        assert not keep[100]          # the intruder goes
        assert keep[:100].mean() > 0.9

    def test_apply_balancing_specs_run(self):
        X, y = _blob_data(250)
        for spec in [None, "tomek", "smote", "enn", "smote+enn",
                     "smote+tomek"]:
            Xb, yb = apply_balancing(X, y, spec, 0, 3)
            assert len(Xb) == len(yb) and Xb.dtype == np.float32
