"""Preprocessing: StandardScaler and PCA (full), reference implementations.

Semantics follow the reference grid (experiment.py:84-85):
  "Scaling" = StandardScaler().fit_transform
  "PCA"     = Pipeline(StandardScaler -> PCA(n_components=None, random_state=0))
and the reference's (leakage-style, deliberately preserved) protocol of
fitting on the FULL dataset before the CV split (experiment.py:452-453).

sklearn parity notes:
  - StandardScaler: population std (ddof=0); zero-variance columns divide
    by 1.0 (sklearn's _handle_zeros_in_scale).
  - PCA on an N x F (F<=16) matrix: full SVD; components are eigenvectors of
    the covariance; sklearn's deterministic svd_flip sign convention is
    applied (sign of the largest-|u| entry per component).  Sign choice
    cannot change tree splits (bins mirror with the sign), so metric parity
    is sign-independent; golden tests compare |values|.

The device path implements the same math as HIP kernels (ops/hip/pca.hip);
those are validated against this module within fp tolerance.
"""

import numpy as np


def scaler_fit_transform(X):
    X = np.asarray(X, dtype=np.float64)
    mean = X.mean(axis=0)
    var = X.var(axis=0)
    scale = np.sqrt(var)
    scale[scale == 0.0] = 1.0
    return (X - mean) / scale


def pca_fit_transform(X):
    """Full PCA via SVD of the centered matrix (input is pre-scaled)."""
    X = np.asarray(X, dtype=np.float64)
    Xc = X - X.mean(axis=0)
    u, s, vt = np.linalg.svd(Xc, full_matrices=False)
    # svd_flip: largest-|u| entry of each component made positive.
    max_rows = np.argmax(np.abs(u), axis=0)
    signs = np.sign(u[max_rows, range(u.shape[1])])
    signs[signs == 0.0] = 1.0
    return (u * s) * signs


def apply_preprocessing(X, spec):
    """spec: None | 'scale' | 'scale+pca' (configgrid.PREPROCESSING_AXIS)."""
    if spec is None:
        return np.asarray(X, dtype=np.float64)
    if spec == "scale":
        return scaler_fit_transform(X)
    if spec == "scale+pca":
        return pca_fit_transform(scaler_fit_transform(X))
    raise ValueError(spec)
