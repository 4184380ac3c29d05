#!/usr/bin/env python3
"""CPU baseline for the scores sweep, for comparison with bench.py.

Two baselines on the same synthetic dataset and folds:
  1. sklearn: DecisionTree / RandomForest / ExtraTrees fit+predict per fold
     (the reference's model layer; imbalanced-learn is unavailable in this
     image, so balancing-free cells only — these bound the model cost that
     dominates the reference's grid).
  2. this framework's numpy reference backend on the same cells.

Usage: python scripts/cpu_baseline.py [--n-tests 10000] [--cells 6]
"""

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n-tests", type=int, default=10000)
    ap.add_argument("--cells", type=int, default=6,
                    help="number of balancing-free cells to time")
    args = ap.parse_args()

    from flake16_framework_amd.configgrid import iter_config_keys
    from flake16_framework_amd.constants import FLAKY
    from flake16_framework_amd.dataset.synthetic import make_synthetic_tests
    from flake16_framework_amd.dataset.tests_io import load_feat_lab_proj
    from flake16_framework_amd.engine.folds import stratified_kfold_split
    from flake16_framework_amd.engine.scores import run_scores

    tests = make_synthetic_tests(n_tests=args.n_tests, seed=0)
    all_keys = list(iter_config_keys())
    cells = [i for i, k in enumerate(all_keys)
             if k[2] == "None" and k[3] == "None"][:args.cells]
    names = [all_keys[i] for i in cells]

    # --- sklearn -----------------------------------------------------------
    try:
        from sklearn.ensemble import (
            ExtraTreesClassifier, RandomForestClassifier,
        )
        from sklearn.tree import DecisionTreeClassifier
        skl = {"Decision Tree": DecisionTreeClassifier,
               "Random Forest": RandomForestClassifier,
               "Extra Trees": ExtraTreesClassifier}

        X, y, _ = load_feat_lab_proj(FLAKY, tuple(range(16)), tests=tests)
        folds = list(stratified_kfold_split(y.astype(np.uint8)))
        t0 = time.time()
        for keys in names:
            model_cls = skl[keys[4]]
            for train, test in folds:
                m = model_cls(random_state=0)
                m.fit(X[train], y[train])
                m.predict(X[test])
        t_skl = time.time() - t0
        print(f"sklearn ({len(names)} balancing-free cells x 10 folds): "
              f"{t_skl:.1f}s  -> {len(names) / t_skl:.3f} configs/s")
        print(f"  extrapolated 216-cell grid (same mix): "
              f"{216 * t_skl / len(names):.0f}s")
    except ImportError:
        print("sklearn not available")

    # --- framework CPU reference ------------------------------------------
    t0 = time.time()
    run_scores(tests=tests, backend="ref", cells=cells)
    t_ref = time.time() - t0
    print(f"framework ref backend, same cells: {t_ref:.1f}s "
          f"-> {len(cells) / t_ref:.3f} configs/s")


if __name__ == "__main__":
    main()
