"""StratifiedKFold reproduction (n_splits=10, shuffle=True, random_state=0).

The reference evaluates every grid cell with sklearn's StratifiedKFold
(reference experiment.py:450, 458).  Fold membership must be reproduced
EXACTLY — F1 parity is defined on identical folds — so this reimplements
sklearn's allocation algorithm (sklearn/model_selection/_split.py,
StratifiedKFold._make_test_folds, stable since 0.22):

  1. encode classes by order of first appearance sorted by class value
  2. allocation[i, k] = count of class k in y_sorted[i::n_splits]
  3. per class: folds_for_class = arange(n_splits).repeat(allocation[:, k]),
     shuffled with the SAME np.random.RandomState stream, assigned to that
     class's samples in their original order.

np.random.RandomState (MT19937 + Fisher-Yates shuffle) is part of numpy's
frozen legacy API, so this matches sklearn bit-for-bit; a golden test against
the installed sklearn asserts it.
"""

import numpy as np


def stratified_kfold_assignments(y, n_splits=10, shuffle=True, random_state=0):
    """Return test_folds: int array, test_folds[i] = fold index of sample i."""
    y = np.asarray(y)
    rng = np.random.RandomState(random_state)

    _, y_idx, y_inv = np.unique(y, return_index=True, return_inverse=True)
    _, class_perm = np.unique(y_idx, return_inverse=True)
    y_encoded = class_perm[y_inv]

    n_classes = len(y_idx)
    y_counts = np.bincount(y_encoded)
    if np.min(y_counts) < n_splits:
        raise ValueError(
            f"n_splits={n_splits} greater than the number of members in the "
            f"least populated class ({np.min(y_counts)}).")

    y_order = np.sort(y_encoded)
    allocation = np.asarray(
        [np.bincount(y_order[i::n_splits], minlength=n_classes)
         for i in range(n_splits)])

    test_folds = np.empty(len(y), dtype="i")
    for k in range(n_classes):
        folds_for_class = np.arange(n_splits).repeat(allocation[:, k])
        if shuffle:
            rng.shuffle(folds_for_class)
        test_folds[y_encoded == k] = folds_for_class

    return test_folds


def stratified_kfold_split(y, n_splits=10, shuffle=True, random_state=0):
    """Yield (train_indices, test_indices) per fold, like sklearn's split()."""
    test_folds = stratified_kfold_assignments(y, n_splits, shuffle, random_state)
    indices = np.arange(len(test_folds))
    for i in range(n_splits):
        mask = test_folds == i
        yield indices[~mask], indices[mask]
