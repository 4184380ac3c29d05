"""The 216-cell configuration grid, declaratively.

The reference builds its grid out of live sklearn/imblearn estimator objects
(reference experiment.py:73-100).  Here the grid is pure data: each axis maps a
display key to a spec understood by the engine.  The AXIS ORDER and KEY ORDER
are a compatibility contract — `scores.pkl` is keyed by the tuple of display
keys in this order, and downstream consumers (top-10 tables, comparison
tables, shap configs) index it that way.

Axes (2 x 2 x 3 x 6 x 3 = 216 cells):
  0. flaky type   : NOD -> label FLAKY(2), OD -> label OD_FLAKY(1)
  1. feature set  : Flake16 (all 16 columns) | FlakeFlagger (7 columns)
  2. preprocessing: None | Scaling (StandardScaler) | PCA (Scaler -> full PCA)
  3. balancing    : None | Tomek Links | SMOTE | ENN | SMOTE ENN | SMOTE Tomek
  4. model        : Extra Trees | Random Forest | Decision Tree  (100/100/1
                    estimators, sklearn 1.0.2 defaults, random_state=0)
"""

import itertools

from .constants import FEATURE_NAMES, FLAKEFLAGGER_COLUMNS, FLAKY, OD_FLAKY

FLAKY_TYPE_AXIS = {
    "NOD": FLAKY,
    "OD": OD_FLAKY,
}

FEATURE_SET_AXIS = {
    "Flake16": tuple(range(len(FEATURE_NAMES))),
    "FlakeFlagger": FLAKEFLAGGER_COLUMNS,
}

PREPROCESSING_AXIS = {
    "None": None,
    "Scaling": "scale",
    "PCA": "scale+pca",       # StandardScaler -> PCA(all components), like the
                              # reference's Pipeline (experiment.py:85)
}

BALANCING_AXIS = {
    "None": None,
    "Tomek Links": "tomek",
    "SMOTE": "smote",
    "ENN": "enn",
    "SMOTE ENN": "smote+enn",
    "SMOTE Tomek": "smote+tomek",
}

MODEL_AXIS = {
    "Extra Trees": {"kind": "extra_trees", "n_estimators": 100, "bootstrap": False},
    "Random Forest": {"kind": "random_forest", "n_estimators": 100, "bootstrap": True},
    "Decision Tree": {"kind": "decision_tree", "n_estimators": 1, "bootstrap": False},
}

CONFIG_GRID = (
    FLAKY_TYPE_AXIS,
    FEATURE_SET_AXIS,
    PREPROCESSING_AXIS,
    BALANCING_AXIS,
    MODEL_AXIS,
)

# The two fixed shap-stage configs (reference experiment.py:524-525).
SHAP_CONFIGS = (
    ("NOD", "Flake16", "Scaling", "SMOTE Tomek", "Extra Trees"),
    ("OD", "Flake16", "Scaling", "SMOTE", "Random Forest"),
)


def iter_config_keys():
    """All 216 config-key tuples in itertools.product order — the same
    enumeration order as the reference's write_scores (experiment.py:494)."""
    return itertools.product(*[d.keys() for d in CONFIG_GRID])


def resolve(config_keys):
    """Config-key tuple -> (flaky_label, feature_set, preproc, balancing, model)."""
    return [CONFIG_GRID[i][k] for i, k in enumerate(config_keys)]


def balance_group_index(config_keys):
    """Index of the cell's (flaky-type, feature-set, preprocessing,
    balancing) combination in product order — 72 groups.

    The balanced training folds are identical for the 3 cells of a balance
    group (the model axis does not affect balancing), so balancing RNG is
    keyed on the GROUP id and the engine computes each group's balanced
    folds once (reference recomputes per cell; sharing changes no output).
    """
    import itertools as _it
    combos = list(_it.product(*[d.keys() for d in CONFIG_GRID[:4]]))
    return combos.index(tuple(config_keys[:4]))


# Relative cost factors calibrated from a measured 72-group trace on one
# MI355X at the headline N=10,000 (profiles/r02_kernel_notes.md): mean
# group duration by axis value / overall mean.  Preprocessing and flaky
# type measured within ~3% of neutral and are omitted.
_BALANCE_COST = {"None": 0.76, "Tomek Links": 0.72, "SMOTE": 1.16,
                 "ENN": 0.70, "SMOTE ENN": 1.22, "SMOTE Tomek": 1.45}
_FSET_COST = {"Flake16": 1.11, "FlakeFlagger": 0.89}


def cell_cost_estimate(config_keys):
    """Relative cost of a cell, for load balancing across ranks.

    Forest cells (100 trees) dominate; the balancing/feature-set factors
    come from measured group timings (see _BALANCE_COST)."""
    _, fset, _, balancing, model = config_keys
    n_trees = MODEL_AXIS[model]["n_estimators"]
    return (n_trees + 1.0) * _BALANCE_COST[balancing] * _FSET_COST[fset]
