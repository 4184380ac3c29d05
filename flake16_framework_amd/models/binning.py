"""Feature binning for histogram-based tree building.

sklearn 1.0.2 trees use exact sorted splits; this framework (like every GPU
gradient-boosting/forest engine) discretizes each feature into <=256 quantile
bins and finds splits over bin boundaries.  SURVEY.md §7 "hard parts" calls
out this deliberate deviation: parity with the reference is METRIC-level (F1
on identical folds), validated by the golden tests against sklearn.

Bin semantics:
  cuts[f]  : ascending cut values (len <= 255)
  code(x)  = #{c in cuts[f] : c <= x}            (np.searchsorted side='right')
  split "code <= b"  <=>  x < cuts[f][b]
so a split bin b corresponds to the raw-space strict threshold cuts[f][b],
which is what gets exported for TreeSHAP / model export.
"""

import numpy as np

MAX_BINS = 256


def compute_bin_cuts(X, max_bins=MAX_BINS):
    """Per-feature cut points from the full matrix (float32 in, float32 cuts).

    Features with <= max_bins distinct values get exact midpoint cuts (the
    split set is then identical to sklearn's candidate set); others get
    quantile cuts.
    """
    X = np.asarray(X)
    n = X.shape[0]
    cuts = []
    for f in range(X.shape[1]):
        vals = np.unique(X[:, f].astype(np.float32))
        if len(vals) <= max_bins:
            c = ((vals[1:].astype(np.float64) + vals[:-1]) * 0.5).astype(np.float32)
        else:
            # order-statistic cuts: pure indexing into the sorted column —
            # no float interpolation, so the GPU path (torch.sort + the
            # same positions, engine/hip_cell._device_cuts) produces
            # BITWISE-identical cut arrays
            xs = np.sort(X[:, f].astype(np.float32))
            pos = (np.ceil(np.arange(1, max_bins) * (n / max_bins))
                   .astype(np.int64) - 1).clip(0, n - 1)
            c = np.unique(xs[pos])
        cuts.append(np.ascontiguousarray(c, dtype=np.float32))
    return cuts


def bin_codes(X, cuts):
    """float matrix -> uint8 bin codes with the given cuts."""
    X = np.asarray(X)
    codes = np.empty(X.shape, dtype=np.uint8)
    for f, c in enumerate(cuts):
        codes[:, f] = np.searchsorted(c, X[:, f].astype(np.float32),
                                      side="right").astype(np.uint8)
    return codes


def n_bins_per_feature(cuts):
    return np.array([len(c) + 1 for c in cuts], dtype=np.int32)
