"""Device-resident grid-cell evaluation (the MI355X execution path).

Mirrors engine/scores.evaluate_cell_ref step for step, but every stage runs
on the GPU through the hand-written HIP kernels (ops/hip/):
  preprocessing  -> scaler/PCA kernels (fp64)
  binning        -> bin_codes kernel (bitwise-exact vs numpy searchsorted)
  balancing      -> knn kernel + smote/enn/tomek kernels
  model fit      -> batched forest work-queue kernels (ALL 10 folds x
                    n_estimators trees of the cell in one forest_fit call)
  predict+score  -> traversal + confusion kernel
t_train / t_test are measured with HIP events around the fit and predict
calls and reported as the per-fold mean, preserving the scores.pkl contract
(reference experiment.py:455-489).

Determinism: trees are bit-identical to the numpy reference given identical
input bits (Philox keyed on node sample ranges; fp64 split scores with
-ffp-contract=off).  Cells with preprocessing run the linear algebra on
device in fp64, which matches the CPU reference only to fp tolerance — the
GPU-vs-CPU e2e test asserts exact confusion equality for 'None'-preprocessing
cells and metric tolerance for the rest.
"""

import numpy as np
import torch

from ..configgrid import MODEL_AXIS, resolve
from ..dataset.tests_io import load_feat_lab_proj
from ..models.binning import compute_bin_cuts
from ..ops.backend import get_ops
from .folds import stratified_kfold_split
from .metrics import finalize_scores

FPAD = 16
N_FOLDS = 10


def _pad16(X):
    """[n, F] -> contiguous [n, 16] (zero pad)."""
    n, f = X.shape
    if f == FPAD:
        return X.contiguous()
    out = X.new_zeros((n, FPAD))
    out[:, :f] = X
    return out.contiguous()


def _cuts_tensors(cuts, device):
    flat = np.concatenate(cuts) if len(cuts) else np.zeros(0, np.float32)
    off = np.zeros(len(cuts) + 1, dtype=np.int32)
    off[1:] = np.cumsum([len(c) for c in cuts])
    return (torch.from_numpy(flat.astype(np.float32)).to(device),
            torch.from_numpy(off).to(device))


def _balance_dev(ops, X32, y_np, spec, k0, k1, device):
    """Device balancing.  X32: [n,16] cuda fp32; y_np: host uint8.
    Returns (X_bal [m,16] cuda fp32, y_bal host uint8)."""
    n = len(y_np)
    y_dev = torch.from_numpy(y_np).to(device)

    if spec is None:
        return X32, y_np

    def _smote(X32, y_np, y_dev):
        n1 = int(y_np.sum())
        n0 = len(y_np) - n1
        if n0 == n1:
            return X32, y_np, y_dev
        min_label = 1 if n1 < n0 else 0
        n_new = abs(n0 - n1)
        min_rows = np.flatnonzero(y_np == min_label).astype(np.int32)
        k = min(5, len(min_rows) - 1)
        if k < 1:
            return X32, y_np, y_dev
        min_rows_dev = torch.from_numpy(min_rows).to(device)
        X_min = X32.index_select(0, min_rows_dev.long()).contiguous()
        nn = ops.knn(X_min, k, True)
        X_new = ops.smote_interpolate(X32, min_rows_dev, nn, n_new, k0, k1)
        Xb = torch.cat([X32, X_new], dim=0).contiguous()
        yb = np.concatenate(
            [y_np, np.full(n_new, min_label, dtype=np.uint8)])
        return Xb, yb, torch.from_numpy(yb).to(device)

    def _maj(y_np):
        n1 = int(y_np.sum())
        return 1 if n1 > len(y_np) - n1 else 0

    def _apply_keep(X32, y_np, keep_dev):
        keep_np = keep_dev.cpu().numpy().astype(bool)
        idx = torch.from_numpy(
            np.flatnonzero(keep_np).astype(np.int64)).to(device)
        return X32.index_select(0, idx).contiguous(), y_np[keep_np]

    if spec == "smote":
        Xb, yb, _ = _smote(X32, y_np, y_dev)
        return Xb, yb

    if spec == "tomek":
        if len(np.unique(y_np)) < 2:
            return X32, y_np
        nn1 = ops.knn(X32, 1, True)
        keep = ops.tomek_keep(y_dev, nn1[:, 0].contiguous(), _maj(y_np),
                              False)
        return _apply_keep(X32, y_np, keep)

    if spec == "enn":
        if len(np.unique(y_np)) < 2 or len(y_np) <= 3:
            return X32, y_np
        nn = ops.knn(X32, 3, True)
        keep = ops.enn_keep(y_dev, nn, 3, _maj(y_np), False)
        return _apply_keep(X32, y_np, keep)

    if spec in ("smote+enn", "smote+tomek"):
        Xs, ys, ys_dev = _smote(X32, y_np, y_dev)
        if spec == "smote+enn":
            if len(np.unique(ys)) < 2 or len(ys) <= 3:
                return Xs, ys
            nn = ops.knn(Xs, 3, True)
            keep = ops.enn_keep(ys_dev, nn, 3, _maj(ys), True)
        else:
            if len(np.unique(ys)) < 2:
                return Xs, ys
            nn1 = ops.knn(Xs, 1, True)
            keep = ops.tomek_keep(ys_dev, nn1[:, 0].contiguous(), _maj(ys),
                                  True)
        return _apply_keep(Xs, ys, keep)

    raise ValueError(spec)


def evaluate_cell_hip(config_keys, cell_idx, tests=None, tests_file=None,
                      seed=0, device=None):
    """Evaluate one grid cell on the GPU.  Returns the scores.pkl value
    [t_train, t_test, scores, scores_total]."""
    from .scores import job_ids_for

    ops = get_ops()
    device = device or torch.device("cuda")

    flaky_label, feature_set, preproc, balancing, model = resolve(config_keys)
    kwargs = {"tests": tests} if tests is not None else {"tests_file": tests_file}
    features, labels_b, projects = load_feat_lab_proj(flaky_label,
                                                      feature_set, **kwargs)
    labels = labels_b.astype(np.uint8)
    F = features.shape[1]
    n = len(labels)

    # --- preprocessing on device (fp64), then fp32 view ------------------
    X64 = _pad16(torch.from_numpy(np.ascontiguousarray(features)).to(device))
    if preproc == "scale":
        X64 = ops.scaler_fit_transform(X64)
    elif preproc == "scale+pca":
        X64 = ops.pca_fit_transform(ops.scaler_fit_transform(X64), F)
    X32 = X64.float().contiguous()

    # --- full-dataset bin cuts (host: tiny sort; values from device) -----
    X32_host = X32.cpu().numpy()[:, :F]
    cuts = compute_bin_cuts(X32_host)
    cuts_dev, cut_off_dev = _cuts_tensors(cuts, device)
    codes_all = ops.bin_codes(X32, cuts_dev, cut_off_dev, F)

    # --- folds (host) -----------------------------------------------------
    folds = list(stratified_kfold_split(labels, n_splits=N_FOLDS,
                                        random_state=seed))

    spec = MODEL_AXIS[config_keys[4]]
    n_trees = spec["n_estimators"]
    bootstrap = spec["bootstrap"]
    splitter_random = spec["kind"] == "extra_trees"
    max_features = F if spec["kind"] == "decision_tree" else max(
        1, int(np.sqrt(F)))

    # --- balancing + binning per fold, batched into one training buffer --
    ev = [torch.cuda.Event(enable_timing=True) for _ in range(4)]
    fold_codes, fold_labels = [], []
    j_row_off, j_n, j_key = [], [], []
    row_base = 0
    for i, (train, _) in enumerate(folds):
        bal_k1, job_base = job_ids_for(cell_idx, i)
        tr_idx = torch.from_numpy(train.astype(np.int64)).to(device)
        Xtr = X32.index_select(0, tr_idx).contiguous()
        Xb, yb = _balance_dev(ops, Xtr, labels[train], balancing,
                              seed, bal_k1, device)
        codes_b = ops.bin_codes(Xb, cuts_dev, cut_off_dev, F)
        fold_codes.append(codes_b)
        fold_labels.append(torch.from_numpy(yb).to(device))
        for t in range(n_trees):
            j_row_off.append(row_base)
            j_n.append(len(yb))
            j_key.append(job_base + t)
        row_base += len(yb)

    codes_train = torch.cat(fold_codes, dim=0).contiguous()
    labels_train = torch.cat(fold_labels, dim=0).contiguous()
    j_row_off = torch.tensor(j_row_off, dtype=torch.int32, device=device)
    j_n = torch.tensor(j_n, dtype=torch.int32, device=device)
    j_key = torch.tensor(j_key, dtype=torch.int32, device=device)

    # --- fit (timed) ------------------------------------------------------
    ev[0].record()
    nfeat, nsplit, nleft, ncnt0, ncnt1, j_node_off, node_alloc = \
        ops.forest_fit(codes_train, labels_train, j_row_off, j_n, j_key,
                       F, max_features, bootstrap, splitter_random, seed)
    ev[1].record()

    # --- predict + confusion (timed) -------------------------------------
    uniq_projects = list(dict.fromkeys(projects))
    proj_index = {p: i for i, p in enumerate(uniq_projects)}
    proj_id = torch.tensor([proj_index[p] for p in projects],
                           dtype=torch.int32, device=device)
    y_dev = torch.from_numpy(labels).to(device)

    pair_row = np.concatenate([test for _, test in folds]).astype(np.int32)
    pair_fold = np.concatenate(
        [np.full(len(test), i, np.int32) for i, (_, test) in
         enumerate(folds)])
    pair_row_d = torch.from_numpy(pair_row).to(device)
    pair_fold_d = torch.from_numpy(pair_fold).to(device)

    ev[2].record()
    pred, confusion = ops.forest_predict_confusion(
        codes_all, y_dev, proj_id, pair_row_d, pair_fold_d, j_node_off,
        nfeat, nsplit, nleft, ncnt0, ncnt1, n_trees, len(uniq_projects))
    ev[3].record()
    torch.cuda.synchronize(device)

    t_train = ev[0].elapsed_time(ev[1]) / 1000.0
    t_test = ev[2].elapsed_time(ev[3]) / 1000.0

    conf = confusion.cpu().numpy()
    scores = {p: [int(conf[i, 0]), int(conf[i, 1]), int(conf[i, 2])]
              for i, p in enumerate(uniq_projects)}
    scores_total = [int(v) for v in conf[len(uniq_projects)]]
    finalize_scores(scores, scores_total)
    return [t_train / N_FOLDS, t_test / N_FOLDS, scores, scores_total]
