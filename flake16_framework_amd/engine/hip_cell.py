"""Device-resident grid-sweep evaluation (the MI355X execution path).

Mirrors engine/scores.evaluate_cell_ref semantics, but every stage runs on
the GPU through the hand-written HIP kernels (ops/hip/), and redundant work
is shared ACROSS grid cells (the big win over the reference's
one-process-per-cell design — outputs are unchanged):

  - preprocessed view + bin cuts + full-dataset codes: one per
    (feature-set, preprocessing) pair — 6 variants, not 216 cells;
  - CV folds: one per flaky-type (they depend only on the labels);
  - balanced fold datasets: one per BALANCE GROUP (flaky x feature-set x
    preproc x balancing = 72 groups, each shared by its 3 model cells) —
    balancing RNG is keyed on the group (engine/scores.job_ids_for);
  - per cell, only the model work remains: one batched forest_fit over the
    cell's 10 folds x n_estimators trees, then predict + confusion.

t_train / t_test are HIP-event times around fit and predict, reported as
the per-fold mean (scores.pkl contract, reference experiment.py:455-489).

Determinism: trees are bit-identical to the numpy reference given identical
input bits; preprocessing (fp64 parallel reductions) matches the CPU
reference to fp tolerance — GPU tests assert exact confusion equality for
'None'-preprocessing cells and metric tolerance for the rest.
"""

import os

import numpy as np
import torch

from ..configgrid import MODEL_AXIS, resolve
from ..dataset.tests_io import load_feat_lab_proj, load_tests
from ..ops.backend import get_ops
from .folds import stratified_kfold_split
from .metrics import finalize_scores

FPAD = 16
N_FOLDS = 10


def _pad16(X):
    n, f = X.shape
    if f == FPAD:
        return X.contiguous()
    out = X.new_zeros((n, FPAD))
    out[:, :f] = X
    return out.contiguous()


def _device_cuts(X32, F, max_bins=256):
    """Bin cuts computed on device — bitwise-identical to
    models.binning.compute_bin_cuts (midpoints of uniques when <= max_bins
    distinct values, order-statistic positions otherwise: pure sorting and
    indexing, no float interpolation).  Returns (flat cuts, offsets)."""
    n = X32.shape[0]
    pos = torch.from_numpy(
        (np.ceil(np.arange(1, max_bins) * (n / max_bins))
         .astype(np.int64) - 1).clip(0, n - 1)).to(X32.device)
    parts = []
    lens = [0] * (FPAD + 1)
    for f in range(F):
        col, _ = torch.sort(X32[:, f].contiguous())
        vals = torch.unique_consecutive(col)
        if vals.numel() <= max_bins:
            c = ((vals[1:].double() + vals[:-1]) * 0.5).float()
        else:
            c = torch.unique_consecutive(col.index_select(0, pos))
        parts.append(c)
        lens[f + 1] = c.numel()
    off = np.cumsum(lens).astype(np.int32)
    flat = torch.cat(parts) if parts else X32.new_zeros(0)
    return flat.contiguous(), torch.from_numpy(off).to(X32.device)


class SweepContext:
    """Caches dataset views shared across cells of one scores sweep."""

    def __init__(self, tests=None, tests_file=None, seed=0, device=None):
        import threading

        self.ops = get_ops()
        self.device = device or torch.device("cuda")
        self.tests = tests if tests is not None else load_tests(tests_file)
        self.seed = seed
        self._views = {}      # (fset_key) raw features; (fset_key, prep) view
        self._labels = {}     # flaky_key -> (labels u8, folds, pair tensors)
        self._balanced = {}   # (flaky, fset, prep, bal) -> per-fold data
        self._projects = None
        # Per-key build locks: worker threads build missing cache entries
        # concurrently (on their own HIP streams); the builder synchronizes
        # its stream before publishing so consumers on other streams see
        # completed tensors.
        self._meta_lock = threading.Lock()
        self._key_locks = {}

    def _build_lock(self, key):
        with self._meta_lock:
            lock = self._key_locks.get(key)
            if lock is None:
                lock = self._key_locks[key] = __import__("threading").Lock()
        return lock

    def _publish_barrier(self):
        if self.device.type == "cuda":
            torch.cuda.current_stream(self.device).synchronize()

    # -- labels / folds / prediction index tensors -------------------------
    def labels_for(self, keys):
        flaky_key = keys[0]
        if flaky_key in self._labels:
            return self._labels[flaky_key]
        with self._build_lock(("labels", flaky_key)):
            if flaky_key in self._labels:
                return self._labels[flaky_key]
            return self._labels_build(keys, flaky_key)

    def _labels_build(self, keys, flaky_key):
        flaky_label, feature_set, *_ = resolve(keys)
        features, labels_b, projects = load_feat_lab_proj(
            flaky_label, feature_set, tests=self.tests)
        labels = labels_b.astype(np.uint8)
        if self._projects is None:
            self._projects = projects
            uniq = list(dict.fromkeys(projects))
            self._proj_uniq = uniq
            pidx = {p: i for i, p in enumerate(uniq)}
            self._proj_id = torch.tensor(
                [pidx[p] for p in projects], dtype=torch.int32,
                device=self.device)
        folds = list(stratified_kfold_split(labels, n_splits=N_FOLDS,
                                            random_state=self.seed))
        pair_row = np.concatenate([t for _, t in folds]).astype(np.int32)
        pair_fold = np.concatenate(
            [np.full(len(t), i, np.int32) for i, (_, t) in enumerate(folds)])
        entry = {
            "labels": labels,
            "labels_dev": torch.from_numpy(labels).to(self.device),
            "folds": folds,
            "pair_row": torch.from_numpy(pair_row).to(self.device),
            "pair_fold": torch.from_numpy(pair_fold).to(self.device),
        }
        self._publish_barrier()
        self._labels[flaky_key] = entry
        return entry

    # -- preprocessed views ------------------------------------------------
    def view_for(self, keys):
        fset_key, prep_key = keys[1], keys[2]
        ck = (fset_key, prep_key)
        if ck in self._views:
            return self._views[ck]
        with self._build_lock(("view", ck)):
            if ck in self._views:
                return self._views[ck]
            return self._view_build(keys, ck)

    def _view_build(self, keys, ck):
        ops = self.ops
        flaky_label, feature_set, preproc, *_ = resolve(keys)
        features, _, _ = load_feat_lab_proj(flaky_label, feature_set,
                                            tests=self.tests)
        F = features.shape[1]
        X64 = _pad16(torch.from_numpy(
            np.ascontiguousarray(features)).to(self.device))
        if preproc == "scale":
            X64 = ops.scaler_fit_transform(X64)
        elif preproc == "scale+pca":
            X64 = ops.pca_fit_transform(ops.scaler_fit_transform(X64), F)
        X32 = X64.float().contiguous()
        cuts_dev, cut_off_dev = _device_cuts(X32[:, :F], F)
        codes_all = ops.bin_codes(X32, cuts_dev, cut_off_dev, F)
        view = {"X32": X32, "cuts_dev": cuts_dev, "cut_off": cut_off_dev,
                "codes_all": codes_all, "F": F}
        self._publish_barrier()
        self._views[ck] = view
        return view

    # -- balancing ---------------------------------------------------------
    def _balance_dev(self, X32, y_np, spec, k1):
        ops, device, k0 = self.ops, self.device, self.seed

        def _smote(X32, y_np):
            n1 = int(y_np.sum())
            n0 = len(y_np) - n1
            if n0 == n1:
                return X32, y_np
            min_label = 1 if n1 < n0 else 0
            n_new = abs(n0 - n1)
            min_rows = np.flatnonzero(y_np == min_label).astype(np.int32)
            k = min(5, len(min_rows) - 1)
            if k < 1:
                return X32, y_np
            min_rows_dev = torch.from_numpy(min_rows).to(device)
            X_min = X32.index_select(0, min_rows_dev.long()).contiguous()
            nn = ops.knn(X_min, k, True)
            X_new = ops.smote_interpolate(X32, min_rows_dev, nn, n_new,
                                          k0, k1)
            Xb = torch.cat([X32, X_new], dim=0).contiguous()
            yb = np.concatenate(
                [y_np, np.full(n_new, min_label, dtype=np.uint8)])
            return Xb, yb

        def _maj(y_np):
            n1 = int(y_np.sum())
            return 1 if n1 > len(y_np) - n1 else 0

        def _apply_keep(X32, y_np, keep_dev):
            keep_np = keep_dev.cpu().numpy().astype(bool)
            idx = torch.from_numpy(
                np.flatnonzero(keep_np).astype(np.int64)).to(device)
            return X32.index_select(0, idx).contiguous(), y_np[keep_np]

        if spec is None:
            return X32, y_np
        if spec == "smote":
            return _smote(X32, y_np)
        if spec == "tomek":
            if len(np.unique(y_np)) < 2:
                return X32, y_np
            nn1 = ops.knn(X32, 1, True)
            keep = ops.tomek_keep(torch.from_numpy(y_np).to(device),
                                  nn1[:, 0].contiguous(), _maj(y_np), False)
            return _apply_keep(X32, y_np, keep)
        if spec == "enn":
            if len(np.unique(y_np)) < 2 or len(y_np) <= 3:
                return X32, y_np
            nn = ops.knn(X32, 3, True)
            keep = ops.enn_keep(torch.from_numpy(y_np).to(device), nn, 3,
                                _maj(y_np), False)
            return _apply_keep(X32, y_np, keep)
        if spec in ("smote+enn", "smote+tomek"):
            Xs, ys = _smote(X32, y_np)
            ys_dev = torch.from_numpy(ys).to(device)
            if spec == "smote+enn":
                if len(np.unique(ys)) < 2 or len(ys) <= 3:
                    return Xs, ys
                nn = ops.knn(Xs, 3, True)
                keep = ops.enn_keep(ys_dev, nn, 3, _maj(ys), True)
            else:
                if len(np.unique(ys)) < 2:
                    return Xs, ys
                nn1 = ops.knn(Xs, 1, True)
                keep = ops.tomek_keep(ys_dev, nn1[:, 0].contiguous(),
                                      _maj(ys), True)
            return _apply_keep(Xs, ys, keep)
        raise ValueError(spec)

    # -- fold-batched balancing (segmented k-NN fills the chip) ------------
    def _smote_folds(self, Xs, ys, bal_keys):
        """SMOTE across all folds with one segmented k-NN call; the
        synthesized rows are bit-identical to per-fold balance.smote."""
        ops, device, k0 = self.ops, self.device, self.seed
        metas = []
        for y in ys:
            n1 = int(y.sum())
            n0 = len(y) - n1
            if n0 == n1:
                metas.append(None)
                continue
            min_label = 1 if n1 < n0 else 0
            min_rows = np.flatnonzero(y == min_label).astype(np.int32)
            k = min(5, len(min_rows) - 1)
            if k < 1:
                metas.append(None)
                continue
            metas.append((min_label, abs(n0 - n1), min_rows, k))

        for kk in sorted({m[3] for m in metas if m}):
            idxs = [i for i, m in enumerate(metas) if m and m[3] == kk]
            mins = []
            seg = [0]
            for i in idxs:
                mr = torch.from_numpy(metas[i][2]).to(device)
                mins.append(Xs[i].index_select(0, mr.long()).contiguous())
                seg.append(seg[-1] + len(metas[i][2]))
            Xmin_cat = torch.cat(mins, dim=0).contiguous()
            seg_off = torch.tensor(seg, dtype=torch.int32, device=device)
            nn = ops.knn_segmented(Xmin_cat, seg_off, kk, True)
            for j, i in enumerate(idxs):
                min_label, n_new, min_rows, k = metas[i]
                nn_f = nn[seg[j]:seg[j + 1]].contiguous()
                mr_dev = torch.from_numpy(min_rows).to(device)
                X_new = ops.smote_interpolate(Xs[i], mr_dev, nn_f, n_new,
                                              k0, bal_keys[i])
                Xs[i] = torch.cat([Xs[i], X_new], dim=0).contiguous()
                ys[i] = np.concatenate(
                    [ys[i], np.full(n_new, min_label, dtype=np.uint8)])
        return Xs, ys

    def _clean_folds(self, Xs, ys, kind, clean_all):
        """ENN / Tomek keep-masks across folds with one segmented k-NN."""
        ops, device = self.ops, self.device
        k = 1 if kind == "tomek" else 3
        active = []
        for i, y in enumerate(ys):
            if len(np.unique(y)) < 2 or (kind == "enn" and len(y) <= 3):
                continue
            active.append(i)
        if not active:
            return Xs, ys

        seg = [0]
        for i in active:
            seg.append(seg[-1] + len(ys[i]))
        X_cat = torch.cat([Xs[i] for i in active], dim=0).contiguous()
        seg_off = torch.tensor(seg, dtype=torch.int32, device=device)
        nn = ops.knn_segmented(X_cat, seg_off, k, True)

        for j, i in enumerate(active):
            y = ys[i]
            n1 = int(y.sum())
            maj = 1 if n1 > len(y) - n1 else 0
            y_dev = torch.from_numpy(y).to(device)
            nn_f = nn[seg[j]:seg[j + 1]].contiguous()
            if kind == "tomek":
                keep = ops.tomek_keep(y_dev, nn_f[:, 0].contiguous(), maj,
                                      clean_all)
            else:
                keep = ops.enn_keep(y_dev, nn_f, 3, maj, clean_all)
            keep_np = keep.cpu().numpy().astype(bool)
            idx = torch.from_numpy(
                np.flatnonzero(keep_np).astype(np.int64)).to(device)
            Xs[i] = Xs[i].index_select(0, idx).contiguous()
            ys[i] = y[keep_np]
        return Xs, ys

    def balanced_for(self, keys, cell_idx):
        """Per-fold balanced training codes for the cell's balance group:
        [(codes_b, yb)] x 10.  k-NN work is batched across folds."""
        bk = tuple(keys[:4])
        if bk in self._balanced:
            return self._balanced[bk]
        with self._build_lock(("balanced", bk)):
            if bk in self._balanced:
                return self._balanced[bk]
            return self._balanced_build(keys, cell_idx, bk)

    def _balanced_build(self, keys, cell_idx, bk):
        from .scores import job_ids_for

        ops = self.ops
        _, _, _, balancing, _ = resolve(keys)
        lab = self.labels_for(keys)
        view = self.view_for(keys)
        X32, F = view["X32"], view["F"]

        Xs, ys, bal_keys = [], [], []
        for i, (train, _) in enumerate(lab["folds"]):
            bal_k1, _ = job_ids_for(keys, cell_idx, i)
            bal_keys.append(bal_k1)
            tr_idx = torch.from_numpy(train.astype(np.int64)).to(self.device)
            Xs.append(X32.index_select(0, tr_idx).contiguous())
            ys.append(lab["labels"][train])

        if balancing in ("smote", "smote+enn", "smote+tomek"):
            Xs, ys = self._smote_folds(Xs, ys, bal_keys)
        if balancing == "tomek":
            Xs, ys = self._clean_folds(Xs, ys, "tomek", False)
        elif balancing == "enn":
            Xs, ys = self._clean_folds(Xs, ys, "enn", False)
        elif balancing == "smote+tomek":
            Xs, ys = self._clean_folds(Xs, ys, "tomek", True)
        elif balancing == "smote+enn":
            Xs, ys = self._clean_folds(Xs, ys, "enn", True)

        out = []
        for i in range(N_FOLDS):
            codes_b = ops.bin_codes(Xs[i], view["cuts_dev"], view["cut_off"],
                                    F)
            out.append((codes_b, torch.from_numpy(ys[i]).to(self.device)))
        self._publish_barrier()
        self._balanced[bk] = out
        return out

    # -- evaluation -------------------------------------------------------
    def evaluate_cell(self, config_keys, cell_idx):
        return self.evaluate_group([(config_keys, cell_idx)])[config_keys]

    def evaluate_group(self, group_cells):
        """Evaluate 1-3 cells of ONE balance group (same keys[:4]) with a
        single fused forest_fit_multi over all their jobs — the DT + RF +
        ET trees of a group share one level pipeline, which cuts kernel
        launches ~3x and keeps the work queues full.

        Trees are bit-identical to per-cell fits: every job keeps the
        same Philox key (job_ids_for), and all kernels read the model
        spec per job.  The fused t_train is attributed to cells in
        proportion to their tree counts (documented deviation: the
        reference records per-process wall time, experiment.py:455).
        Returns {config_keys: [t_train, t_test, scores, scores_total]}.
        """
        from .scores import job_ids_for

        ops, device = self.ops, self.device
        keys0, cell0 = group_cells[0]
        view = self.view_for(keys0)
        lab = self.labels_for(keys0)
        balanced = self.balanced_for(keys0, cell0)
        F = view["F"]

        fold_codes = [c for c, _ in balanced]
        fold_labels = [y for _, y in balanced]
        n_per_fold = np.array([int(y.shape[0]) for y in fold_labels],
                              dtype=np.int64)
        fold_base = np.concatenate(([0], np.cumsum(n_per_fold)[:-1]))

        # Fusion pays while the batch is launch-bound; at large sample
        # totals the single long level pipeline loses to per-cell fits
        # overlapping across streams (same-box at N=40k: fused 11.4 s vs
        # per-cell 8.4 s; at N=10k fused 140.6 vs 128.4 configs/s), so
        # big groups fall back to per-cell evaluation.
        if len(group_cells) > 1:
            s_total = int(n_per_fold.sum()) * sum(
                MODEL_AXIS[keys[4]]["n_estimators"]
                for keys, _ in group_cells)
            fuse_max = int(os.environ.get("FLAKE16_FUSE_MAX_S",
                                          str(1 << 26)))
            if s_total > fuse_max:
                out = {}
                for cell in group_cells:
                    out.update(self.evaluate_group([cell]))
                return out

        # vectorized per-cell job blocks, concatenated cell-major
        parts = {"row": [], "n": [], "key": [], "mf": [], "rand": [],
                 "boot": []}
        cell_meta = []   # (keys, n_trees, job_off)
        job_off = 0
        for keys, cell_idx in group_cells:
            spec = MODEL_AXIS[keys[4]]
            n_trees = spec["n_estimators"]
            mf = F if spec["kind"] == "decision_tree" else max(
                1, int(np.sqrt(F)))
            job_bases = np.array(
                [job_ids_for(keys, cell_idx, i)[1]
                 for i in range(N_FOLDS)], dtype=np.int64)
            n_jobs = N_FOLDS * n_trees
            parts["row"].append(np.repeat(fold_base, n_trees))
            parts["n"].append(np.repeat(n_per_fold, n_trees))
            parts["key"].append(np.repeat(job_bases, n_trees)
                                + np.tile(np.arange(n_trees), N_FOLDS))
            parts["mf"].append(np.full(n_jobs, mf))
            parts["rand"].append(np.full(
                n_jobs, spec["kind"] == "extra_trees"))
            parts["boot"].append(np.full(n_jobs, spec["bootstrap"]))
            cell_meta.append((keys, n_trees, job_off))
            job_off += n_jobs

        codes_train = torch.cat(fold_codes, dim=0).contiguous()
        labels_train = torch.cat(fold_labels, dim=0).contiguous()
        cat32 = lambda k: np.concatenate(parts[k]).astype(np.int32)
        j_row_off = torch.from_numpy(cat32("row")).to(device)
        # j_n / j_mf / j_rand / j_boot stay CPU-resident: forest_fit_multi
        # needs their values host-side (sizing + kernel selection)
        j_n = torch.from_numpy(cat32("n"))
        j_key = torch.from_numpy(cat32("key")).to(device)
        j_mf = torch.from_numpy(cat32("mf"))
        j_rand = torch.from_numpy(
            np.concatenate(parts["rand"]).astype(np.uint8))
        j_boot = torch.from_numpy(
            np.concatenate(parts["boot"]).astype(np.uint8))

        ev_fit = [torch.cuda.Event(enable_timing=True) for _ in range(2)]
        ev_fit[0].record()
        nfeat, nsplit, nleft, ncnt0, ncnt1, j_node_off, node_alloc = \
            ops.forest_fit_multi(codes_train, labels_train, j_row_off, j_n,
                                 j_key, j_mf, j_rand, j_boot, F, self.seed)
        ev_fit[1].record()

        out = {}
        pred_events = []
        for keys, n_trees, off in cell_meta:
            e0 = torch.cuda.Event(enable_timing=True)
            e1 = torch.cuda.Event(enable_timing=True)
            e0.record()
            _, confusion = ops.forest_predict_confusion(
                view["codes_all"], lab["labels_dev"], self._proj_id,
                lab["pair_row"], lab["pair_fold"],
                j_node_off.narrow(0, off, N_FOLDS * n_trees),
                nfeat, nsplit, nleft, ncnt0, ncnt1, n_trees,
                len(self._proj_uniq))
            e1.record()
            pred_events.append((keys, confusion, e0, e1))

        pred_events[-1][3].synchronize()
        t_fit = ev_fit[0].elapsed_time(ev_fit[1]) / 1000.0
        total_trees = sum(nt for _, nt, _ in cell_meta)

        for (keys, n_trees, _), (_, confusion, e0, e1) in zip(cell_meta,
                                                              pred_events):
            t_train = t_fit * n_trees / total_trees
            t_test = e0.elapsed_time(e1) / 1000.0
            conf = confusion.cpu().numpy()
            scores = {p: [int(conf[i, 0]), int(conf[i, 1]), int(conf[i, 2])]
                      for i, p in enumerate(self._proj_uniq)}
            scores_total = [int(v) for v in conf[len(self._proj_uniq)]]
            finalize_scores(scores, scores_total)
            out[keys] = [t_train / N_FOLDS, t_test / N_FOLDS, scores,
                         scores_total]
        return out


def evaluate_cell_hip(config_keys, cell_idx, tests=None, tests_file=None,
                      seed=0, device=None, context=None):
    """One-cell convenience wrapper (tests); run_scores uses a shared
    SweepContext for the whole sweep."""
    ctx = context or SweepContext(tests=tests, tests_file=tests_file,
                                  seed=seed, device=device)
    return ctx.evaluate_cell(config_keys, cell_idx)
