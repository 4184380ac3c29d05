"""Known-answer fixtures for the balancing semantics.

imbalanced-learn 0.9.0 is not installable here, so balance/__init__.py is
validated against HAND-DERIVED answers computed from the published
algorithm definitions (imblearn 0.9.0 docs: TomekLinks,
EditedNearestNeighbours(kind_sel='all'), SMOTE(k_neighbors=5), and the
SMOTEENN / SMOTETomek compositions with sampling_strategy='all' for the
cleaning step).  Every expected mask below was worked out by hand on
1-D/2-D point sets whose neighbor structure is unambiguous — these tests
fail if our reading of the imblearn semantics is wrong, independently of
the HIP kernels (reference experiment.py:89-93, requirements.txt:28).
"""

import numpy as np

from flake16_framework_amd.balance import (
    apply_balancing, enn_mask, knn_indices, smote, tomek_links_mask,
)


def col(v):
    """1-D feature column."""
    return np.asarray(v, dtype=np.float32)[:, None]


class TestKnnContract:
    def test_ties_break_to_lower_index(self):
        # query 1.0: candidates 0.0 (idx 1) and 2.0 (idx 2) equidistant
        X = col([1.0, 0.0, 2.0])
        nn = knn_indices(X, X, 1, skip_identity=True)
        assert nn[0, 0] == 1

    def test_ordering(self):
        X = col([0.0, 10.0, 3.0, 1.0])
        nn = knn_indices(X, X, 3, skip_identity=True)
        # neighbors of 0.0 in distance order: 1.0 (3), 3.0 (2), 10.0 (1)
        assert nn[0].tolist() == [3, 2, 1]


class TestTomekLinks:
    # X = [0, 1, 3, 10, 10.5, 20]; y = [maj, MIN, maj, maj, maj, maj]
    # (0.0, 1.0) is the only cross-class mutual-1-NN pair:
    #   nn(0.0)=1.0, nn(1.0)=0.0 -> Tomek link
    #   nn(3.0)=1.0 but nn(1.0)!=3.0 -> not mutual
    #   (10.0, 10.5) mutual but same class
    X = col([0.0, 1.0, 3.0, 10.0, 10.5, 20.0])
    y = np.array([0, 1, 0, 0, 0, 0], dtype=np.uint8)

    def test_auto_removes_majority_member(self):
        keep = tomek_links_mask(self.X, self.y, "auto")
        assert keep.tolist() == [False, True, True, True, True, True]

    def test_all_removes_both_members(self):
        keep = tomek_links_mask(self.X, self.y, "all")
        assert keep.tolist() == [False, False, True, True, True, True]

    def test_single_class_noop(self):
        keep = tomek_links_mask(self.X, np.zeros(6, np.uint8), "auto")
        assert keep.all()


class TestEnn:
    # X = [0, 1, 2, 2.5, 10, 11, 12, 13]; y = [1,1,1, 0, 0,0,0,0]
    # majority = 0 (5 vs 3).  3-NN (self excluded, all samples):
    #   2.5 -> {2, 1, 0}: all minority -> 2.5 removed under 'auto'
    #   10/11/12/13 -> each other: all majority -> kept
    X = col([0.0, 1.0, 2.0, 2.5, 10.0, 11.0, 12.0, 13.0])
    y = np.array([1, 1, 1, 0, 0, 0, 0, 0], dtype=np.uint8)

    def test_auto_targets_majority_only(self):
        keep = enn_mask(self.X, self.y, "auto")
        assert keep.tolist() == [True, True, True, False,
                                 True, True, True, True]

    def test_all_cleans_every_class(self):
        # additionally: 0 -> {1, 2, 2.5} has a majority neighbor -> removed
        #               1 -> {0, 2, 2.5} -> removed
        #               2 -> {2.5, 1, 0} -> removed
        keep = enn_mask(self.X, self.y, "all")
        assert keep.tolist() == [False, False, False, False,
                                 True, True, True, True]

    def test_too_small_noop(self):
        keep = enn_mask(col([0.0, 1.0, 2.0]),
                        np.array([0, 1, 0], np.uint8), "auto")
        assert keep.all()


class TestSmote:
    def test_synthetic_rows_on_minority_segments(self):
        # minority on the x==y diagonal: every synthetic point
        # base + gap*(neigh-base) must stay on it, coordinates bitwise
        # equal (identical fp32 ops on equal components)
        X = np.array([[0, 0], [1, 1], [2, 2],
                      [10, 0], [11, 0], [12, 0], [13, 0], [14, 0]],
                     dtype=np.float32)
        y = np.array([1, 1, 1, 0, 0, 0, 0, 0], dtype=np.uint8)
        Xr, yr = smote(X, y, k0=0, k1=7)
        assert len(yr) == 10 and yr[8:].tolist() == [1, 1]
        new = Xr[8:]
        assert (new[:, 0] == new[:, 1]).all()
        assert (new[:, 0] >= 0).all() and (new[:, 0] <= 2).all()

    def test_balanced_input_unchanged(self):
        X = col([0.0, 1.0, 10.0, 11.0])
        y = np.array([1, 1, 0, 0], dtype=np.uint8)
        Xr, yr = smote(X, y, 0, 1)
        assert len(yr) == 4 and np.array_equal(Xr, X.astype(np.float32))

    def test_deterministic_in_keys(self):
        X = np.random.RandomState(3).rand(40, 4).astype(np.float32)
        y = (np.arange(40) < 8).astype(np.uint8)
        a = smote(X, y, 0, 5)
        b = smote(X, y, 0, 5)
        c = smote(X, y, 0, 6)
        assert np.array_equal(a[0], b[0])
        assert not np.array_equal(a[0], c[0])

    def test_minority_neighbor_pool_only(self):
        # majority rows sit far away; synthetic rows must interpolate
        # minority rows only -> bounded by the minority bounding box
        X = col([0.0, 0.5, 1.0, 100.0, 101.0, 102.0, 103.0])
        y = np.array([1, 1, 1, 0, 0, 0, 0], dtype=np.uint8)
        Xr, yr = smote(X, y, 0, 2)
        assert (Xr[7:] <= 1.0).all() and (Xr[7:] >= 0.0).all()


class TestCombos:
    """On class-balanced inputs SMOTE is a no-op, so the combos reduce to
    the 'all'-strategy cleaners — giving exact hand-derived answers for
    the composition path."""

    def test_smote_tomek_all(self):
        # mutual-1-NN pairs: (0,1) cross -> both out; (10,11) same-class;
        # (20,21) same-class; (30,31) cross -> both out
        X = col([0.0, 1.0, 10.0, 11.0, 20.0, 21.0, 30.0, 31.0])
        y = np.array([0, 1, 0, 0, 1, 1, 0, 1], dtype=np.uint8)
        Xr, yr = apply_balancing(X, y, "smote+tomek", 0, 1)
        kept = [float(v) for v in Xr[:, 0]]
        assert kept == [10.0, 11.0, 20.0, 21.0]
        assert yr.tolist() == [0, 0, 1, 1]

    def test_smote_enn_all(self):
        # class1 = [0..4], class0 = [10..13, 4.4]; 3-NN (self excluded):
        #   0 -> {1,2,3}, 1 -> {0,2,3}, 2 -> {1,3,0}: pure -> kept
        #   3 -> {2, 4, 4.4}: mixed -> out
        #   4 -> {4.4, 3, 2}: mixed -> out
        #   4.4 -> {4, 3, 2}: all class1 -> out
        #   10..13 -> each other: pure -> kept
        X = col([0.0, 1.0, 2.0, 3.0, 4.0,
                 10.0, 11.0, 12.0, 13.0, 4.4])
        y = np.array([1, 1, 1, 1, 1, 0, 0, 0, 0, 0], dtype=np.uint8)
        Xr, yr = apply_balancing(X, y, "smote+enn", 0, 1)
        kept = [float(v) for v in Xr[:, 0]]
        assert kept == [0.0, 1.0, 2.0, 10.0, 11.0, 12.0, 13.0]
        assert yr.tolist() == [1, 1, 1, 0, 0, 0, 0]
