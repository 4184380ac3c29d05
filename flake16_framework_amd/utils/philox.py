"""Philox4x32-10 counter-based RNG (numpy, vectorized).

This is the framework's single source of randomness for model fitting and
balancing.  The HIP kernels implement the identical function
(flake16_framework_amd/ops/hip/philox.h); both sides are keyed on
DETERMINISTIC identities — (tag, node-range, draw index) for tree nodes,
(tag, row, draw) for SMOTE — so device work-queue scheduling order cannot
change any random draw, and the numpy reference reproduces device trees
bit-for-bit.

Counter layout convention used across the framework:
  c0 = tag | extra << 8     (domain separation tag, see TAG_*)
  c1, c2 = context          (e.g. node sample-range [start, end))
  c3 = draw index
  key = (k0, k1)            (k0 = global seed, k1 = job/stream id)
"""

import numpy as np

PHILOX_M0 = np.uint64(0xD2511F53)
PHILOX_M1 = np.uint64(0xCD9E8D57)
PHILOX_W0 = np.uint32(0x9E3779B9)
PHILOX_W1 = np.uint32(0xBB67AE85)

# Domain-separation tags (must match ops/hip/philox.h).
TAG_BOOTSTRAP = 1   # bootstrap sample draws (per tree)
TAG_FEATSEL = 2     # per-node feature-subset permutation
TAG_THRESH = 3      # Extra-Trees random threshold draws
TAG_SMOTE_PICK = 4  # SMOTE (row, neighbor) selection
TAG_SMOTE_GAP = 5   # SMOTE interpolation gap


def philox4x32(c0, c1, c2, c3, k0, k1):
    """10-round Philox4x32. All args uint32 scalars or arrays (broadcast).
    Returns (x0, x1, x2, x3) uint32 arrays."""
    c0 = np.asarray(c0, dtype=np.uint32)
    c1 = np.asarray(c1, dtype=np.uint32)
    c2 = np.asarray(c2, dtype=np.uint32)
    c3 = np.asarray(c3, dtype=np.uint32)
    k0 = np.uint32(k0)
    k1 = np.uint32(k1)

    for _ in range(10):
        p0 = PHILOX_M0 * c0.astype(np.uint64)
        p1 = PHILOX_M1 * c2.astype(np.uint64)
        hi0 = (p0 >> np.uint64(32)).astype(np.uint32)
        lo0 = p0.astype(np.uint32)
        hi1 = (p1 >> np.uint64(32)).astype(np.uint32)
        lo1 = p1.astype(np.uint32)

        c0, c1, c2, c3 = hi1 ^ c1 ^ k0, lo1, hi0 ^ c3 ^ k1, lo0
        k0 = np.uint32((int(k0) + int(PHILOX_W0)) & 0xFFFFFFFF)
        k1 = np.uint32((int(k1) + int(PHILOX_W1)) & 0xFFFFFFFF)

    return c0, c1, c2, c3


def u32_to_unit(u):
    """uint32 -> float32 in [0, 1): u * 2^-32 (matches the HIP side)."""
    return (np.asarray(u, dtype=np.uint64).astype(np.float64)
            * (1.0 / 4294967296.0)).astype(np.float32)


def bounded_int(u, n):
    """uint32 -> [0, n) by multiply-shift ((u * n) >> 32); slight bias is
    acceptable and identical on both sides."""
    return ((np.asarray(u, dtype=np.uint64) * np.uint64(n)) >> np.uint64(32)
            ).astype(np.int64)


def draws_u32(tag, c1, c2, n_draws, k0, k1):
    """n_draws uint32s at counters (tag, c1, c2, i) — one philox call per
    draw, first output word only (mirrors one-thread-one-draw on device)."""
    i = np.arange(n_draws, dtype=np.uint32)
    x0, _, _, _ = philox4x32(np.uint32(tag), np.uint32(c1), np.uint32(c2), i,
                             k0, k1)
    return x0
