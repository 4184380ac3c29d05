"""Class balancing: SMOTE, Tomek links, ENN and the SMOTE+cleaning combos.

Reference implementations (numpy) of the imbalanced-learn 0.9.0 estimators
used by the reference grid (experiment.py:89-93).  imblearn is not
installable in this environment, so these are re-implemented from the
published algorithms; sampling strategies follow imblearn defaults:

  SMOTE(random_state=0, k_neighbors=5), strategy 'auto':
      oversample the minority class to majority parity; each synthetic
      sample is x_i + gap * (x_nn - x_i) for a uniformly chosen minority
      row i, one of its 5 minority nearest neighbors (self excluded), and
      gap ~ U[0,1).  New rows are appended after the original data.
  TomekLinks, strategy 'auto':
      remove the MAJORITY member of each cross-class mutual-1-NN pair.
      (strategy 'all' removes both members — used inside SMOTE Tomek.)
  EditedNearestNeighbours(n_neighbors=3, kind_sel='all'), strategy 'auto':
      remove majority samples whose 3 nearest neighbors (self excluded,
      searched over ALL samples) are not all of the majority class.
      (strategy 'all' cleans every class — used inside SMOTE ENN.)
  SMOTEENN  = SMOTE then ENN(strategy 'all')
  SMOTETomek = SMOTE then TomekLinks(strategy 'all')

Determinism: neighbor searches use fp64 distances accumulated in fixed
feature order with ties broken by lower index; SMOTE draws come from the
framework Philox stream (tags TAG_SMOTE_PICK / TAG_SMOTE_GAP), keyed per
(cell, fold) job id.  The HIP kernels implement the identical contract.
"""

import numpy as np

from ..utils.philox import (
    TAG_SMOTE_GAP, TAG_SMOTE_PICK, bounded_int, draws_u32, u32_to_unit,
)


def knn_indices(X_query, X_cand, k, skip_identity=False):
    """k nearest candidate indices per query row (fp64 Euclidean; ties by
    lower candidate index).  skip_identity: candidate j == query row id j is
    excluded (use only when X_query IS X_cand)."""
    Xq = np.asarray(X_query, dtype=np.float64)
    Xc = np.asarray(X_cand, dtype=np.float64)
    # Feature-sequential fp64 accumulation — the exact op order of the HIP
    # knn kernel, so device and reference agree bitwise (tie behavior
    # included).
    d2 = np.zeros((Xq.shape[0], Xc.shape[0]))
    for f in range(Xq.shape[1]):
        diff = Xq[:, f, None] - Xc[None, :, f]
        d2 = d2 + diff * diff
    if skip_identity:
        np.fill_diagonal(d2, np.inf)
    # stable argsort => ties broken by lower index
    order = np.argsort(d2, axis=1, kind="stable")
    return order[:, :k]


def smote(X, y, k0, k1, k_neighbors=5):
    """Returns (X_res, y_res) float32/uint8 with synthetic minority rows
    appended.  If classes are already balanced, returns inputs unchanged."""
    X = np.asarray(X, dtype=np.float32)
    y = np.asarray(y, dtype=np.uint8)
    n1 = int(y.sum())
    n0 = len(y) - n1
    if n0 == n1:
        return X, y
    min_label = 1 if n1 < n0 else 0
    n_new = abs(n0 - n1)

    min_rows = np.flatnonzero(y == min_label)
    X_min = X[min_rows]
    k = min(k_neighbors, len(min_rows) - 1)
    if k < 1:
        return X, y
    nn = knn_indices(X_min, X_min, k, skip_identity=True)   # (n_min, k)

    u_pick = draws_u32(TAG_SMOTE_PICK, 0, 0, n_new, k0, k1)
    picks = bounded_int(u_pick, len(min_rows) * k)
    rows = picks // k
    cols = picks % k

    u_gap = draws_u32(TAG_SMOTE_GAP, 0, 0, n_new, k0, k1)
    gaps = u32_to_unit(u_gap)[:, None]

    base = X_min[rows]
    neigh = X_min[nn[rows, cols]]
    X_new = base + gaps * (neigh - base)

    X_res = np.vstack([X, X_new.astype(np.float32)])
    y_res = np.concatenate([y, np.full(n_new, min_label, dtype=np.uint8)])
    return X_res, y_res


def tomek_links_mask(X, y, strategy="auto"):
    """Boolean keep-mask after Tomek-link removal."""
    X = np.asarray(X, dtype=np.float32)
    y = np.asarray(y, dtype=np.uint8)
    if len(np.unique(y)) < 2:
        return np.ones(len(y), dtype=bool)
    nn1 = knn_indices(X, X, 1, skip_identity=True)[:, 0]
    n1 = int(y.sum())
    maj_label = 1 if n1 > len(y) - n1 else 0

    keep = np.ones(len(y), dtype=bool)
    for i in range(len(y)):
        j = nn1[i]
        if y[i] != y[j] and nn1[j] == i:
            if strategy == "all":
                keep[i] = keep[j] = False
            else:
                # remove the majority member only
                if y[i] == maj_label:
                    keep[i] = False
                else:
                    keep[j] = False
    return keep


def enn_mask(X, y, strategy="auto", n_neighbors=3):
    """Boolean keep-mask after Edited-Nearest-Neighbours cleaning
    (kind_sel='all': a cleaned sample is kept only if ALL its neighbors
    share its label)."""
    X = np.asarray(X, dtype=np.float32)
    y = np.asarray(y, dtype=np.uint8)
    if len(np.unique(y)) < 2 or len(y) <= n_neighbors:
        return np.ones(len(y), dtype=bool)
    nn = knn_indices(X, X, n_neighbors, skip_identity=True)
    n1 = int(y.sum())
    maj_label = 1 if n1 > len(y) - n1 else 0

    agree_all = (y[nn] == y[:, None]).all(axis=1)
    keep = np.ones(len(y), dtype=bool)
    if strategy == "all":
        targets = np.ones(len(y), dtype=bool)
    else:
        targets = y == maj_label
    keep[targets & ~agree_all] = False
    return keep


def apply_balancing(X, y, spec, k0, k1):
    """spec: None|'tomek'|'smote'|'enn'|'smote+enn'|'smote+tomek'
    (configgrid.BALANCING_AXIS).  Returns (X_res, y_res)."""
    X = np.asarray(X, dtype=np.float32)
    y = np.asarray(y, dtype=np.uint8)
    if spec is None:
        return X, y
    if spec == "tomek":
        keep = tomek_links_mask(X, y, "auto")
        return X[keep], y[keep]
    if spec == "smote":
        return smote(X, y, k0, k1)
    if spec == "enn":
        keep = enn_mask(X, y, "auto")
        return X[keep], y[keep]
    if spec == "smote+enn":
        Xs, ys = smote(X, y, k0, k1)
        keep = enn_mask(Xs, ys, "all")
        return Xs[keep], ys[keep]
    if spec == "smote+tomek":
        Xs, ys = smote(X, y, k0, k1)
        keep = tomek_links_mask(Xs, ys, "all")
        return Xs[keep], ys[keep]
    raise ValueError(spec)
