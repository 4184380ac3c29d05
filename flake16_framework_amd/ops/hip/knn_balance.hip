// Brute-force k-NN and the SMOTE / ENN / Tomek balancing kernels (gfx950).
//
// Distances are fp64, accumulated in fixed feature order (file compiled
// -ffp-contract=off), ties broken by lower candidate index — the exact
// contract of balance/__init__.knn_indices, so device balancing reproduces
// the numpy reference bit-for-bit on identical input bits.
//
// The candidate set is staged through LDS in 256-row tiles (16 KiB fp32),
// each 256-thread workgroup owning 256 query rows.  imblearn 0.9.0
// semantics documented in balance/__init__.py.

#include <hip/hip_runtime.h>
#include <cstdint>

#include "philox.h"

#define FPAD 16
#define KMAX 8
#define KNN_BLK 256

// knn_kernel: for each query row, the k nearest candidate rows.
// X: [n, FPAD] fp32 (padded features are zero).  skip_identity: candidate
// j == query j excluded (query set IS candidate set).
__launch_bounds__(KNN_BLK)
__global__ void knn_kernel(const float* __restrict__ X,
                           int n, int k, int skip_identity,
                           int* __restrict__ out /* [n, k] */) {
    __shared__ float tile[KNN_BLK][FPAD];

    const int q = blockIdx.x * KNN_BLK + threadIdx.x;

    float qv[FPAD];
    if (q < n) {
        #pragma unroll
        for (int f = 0; f < FPAD; ++f) qv[f] = X[(size_t)q * FPAD + f];
    }

    double bd[KMAX];
    int bi[KMAX];
    for (int j = 0; j < KMAX; ++j) { bd[j] = 1.0e300; bi[j] = -1; }

    for (int base = 0; base < n; base += KNN_BLK) {
        const int c = base + threadIdx.x;
        if (c < n) {
            #pragma unroll
            for (int f = 0; f < FPAD; ++f)
                tile[threadIdx.x][f] = X[(size_t)c * FPAD + f];
        }
        __syncthreads();

        if (q < n) {
            const int tn = min(KNN_BLK, n - base);
            for (int t = 0; t < tn; ++t) {
                const int cand = base + t;
                if (skip_identity && cand == q) continue;
                double d = 0.0;
                #pragma unroll
                for (int f = 0; f < FPAD; ++f) {
                    double diff = (double)qv[f] - (double)tile[t][f];
                    d = d + diff * diff;
                }
                // insertion into the sorted top-k (ties: lower index first,
                // i.e. strictly-less replaces — candidates arrive in
                // ascending index order)
                if (d < bd[k - 1]) {
                    int j = k - 1;
                    while (j > 0 && d < bd[j - 1]) {
                        bd[j] = bd[j - 1];
                        bi[j] = bi[j - 1];
                        --j;
                    }
                    bd[j] = d;
                    bi[j] = cand;
                }
            }
        }
        __syncthreads();
    }

    if (q < n)
        for (int j = 0; j < k; ++j) out[(size_t)q * k + j] = bi[j];
}

// Segmented (fold-batched) k-NN: one call covers many independent
// candidate sets (e.g. the 10 CV folds of a balance group) so the chip is
// filled even when each segment alone is small.  seg_off[Nseg+1] bounds the
// segments in X; seg_blk[Nseg+1] is the prefix of per-segment block counts
// (ceil(n_seg / KNN_BLK)); outputs are SEGMENT-LOCAL indices — identical
// bits to a per-segment knn_kernel call.
__launch_bounds__(KNN_BLK)
__global__ void knn_segmented_kernel(const float* __restrict__ X,
                                     const int* __restrict__ seg_off,
                                     const int* __restrict__ seg_blk,
                                     int n_seg, int k, int skip_identity,
                                     int* __restrict__ out) {
    __shared__ float tile[KNN_BLK][FPAD];

    // locate this block's segment (binary search over seg_blk)
    int lo = 0, hi = n_seg;
    while (lo + 1 < hi) {
        int mid = (lo + hi) >> 1;
        if (seg_blk[mid] <= (int)blockIdx.x) lo = mid; else hi = mid;
    }
    const int seg = lo;
    const int chunk = blockIdx.x - seg_blk[seg];
    const int base_row = seg_off[seg];
    const int n = seg_off[seg + 1] - base_row;
    const int q = chunk * KNN_BLK + threadIdx.x;   // segment-local query

    float qv[FPAD];
    if (q < n) {
        #pragma unroll
        for (int f = 0; f < FPAD; ++f)
            qv[f] = X[(size_t)(base_row + q) * FPAD + f];
    }

    double bd[KMAX];
    int bi[KMAX];
    for (int j = 0; j < KMAX; ++j) { bd[j] = 1.0e300; bi[j] = -1; }

    for (int tb = 0; tb < n; tb += KNN_BLK) {
        const int c = tb + threadIdx.x;
        if (c < n) {
            #pragma unroll
            for (int f = 0; f < FPAD; ++f)
                tile[threadIdx.x][f] = X[(size_t)(base_row + c) * FPAD + f];
        }
        __syncthreads();

        if (q < n) {
            const int tn = min(KNN_BLK, n - tb);
            for (int t = 0; t < tn; ++t) {
                const int cand = tb + t;
                if (skip_identity && cand == q) continue;
                double d = 0.0;
                #pragma unroll
                for (int f = 0; f < FPAD; ++f) {
                    double diff = (double)qv[f] - (double)tile[t][f];
                    d = d + diff * diff;
                }
                if (d < bd[k - 1]) {
                    int j = k - 1;
                    while (j > 0 && d < bd[j - 1]) {
                        bd[j] = bd[j - 1];
                        bi[j] = bi[j - 1];
                        --j;
                    }
                    bd[j] = d;
                    bi[j] = cand;
                }
            }
        }
        __syncthreads();
    }

    if (q < n)
        for (int j = 0; j < k; ++j)
            out[(size_t)(base_row + q) * k + j] = bi[j];
}

// smote_kernel: one thread per synthetic sample.
// Draw i: pick = bounded(philox(TAG_SMOTE_PICK,0,0,i), n_min*k) ->
// (row = pick/k, col = pick%k); gap = unit(philox(TAG_SMOTE_GAP,0,0,i)).
// X_new = base + gap * (neigh - base), fp32, -ffp-contract=off: matches
// balance.smote exactly.
__global__ void smote_kernel(const float* __restrict__ X,   // [n, FPAD]
                             const int* __restrict__ min_rows,  // [n_min]
                             const int* __restrict__ nn,    // [n_min, k]
                             int n_min, int k, int n_new,
                             uint32_t k0, uint32_t k1,
                             float* __restrict__ X_new /* [n_new, FPAD] */) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n_new) return;

    uint32_t up = philox_draw(TAG_SMOTE_PICK, 0u, 0u, (uint32_t)i, k0, k1);
    uint32_t pick = philox_bounded(up, (uint32_t)(n_min * k));
    int row = (int)(pick / (uint32_t)k);
    int col = (int)(pick % (uint32_t)k);

    uint32_t ug = philox_draw(TAG_SMOTE_GAP, 0u, 0u, (uint32_t)i, k0, k1);
    float gap = philox_unit(ug);

    const float* base = X + (size_t)min_rows[row] * FPAD;
    const float* neigh = X + (size_t)min_rows[nn[(size_t)row * k + col]] * FPAD;
    float* outr = X_new + (size_t)i * FPAD;
    #pragma unroll
    for (int f = 0; f < FPAD; ++f) {
        float diff = neigh[f] - base[f];
        float step = gap * diff;
        outr[f] = base[f] + step;
    }
}

// enn_keep_kernel: kind_sel='all' — a targeted sample is kept only if ALL
// its n_neighbors nearest neighbors share its label.
// strategy 'auto': targets = majority class only; 'all': every class.
__global__ void enn_keep_kernel(const uint8_t* __restrict__ y,
                                const int* __restrict__ nn,  // [n, kq]
                                int n, int kq, int n_neighbors,
                                int maj_label, int clean_all,
                                uint8_t* __restrict__ keep) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    uint8_t yi = y[i];
    uint8_t kp = 1;
    if (clean_all || (int)yi == maj_label) {
        for (int j = 0; j < n_neighbors; ++j)
            if (y[nn[(size_t)i * kq + j]] != yi) { kp = 0; break; }
    }
    keep[i] = kp;
}

// tomek_keep_kernel: i is removed iff (i, nn1[i]) is a mutual cross-class
// 1-NN pair AND (strategy 'all', or i belongs to the majority class).
__global__ void tomek_keep_kernel(const uint8_t* __restrict__ y,
                                  const int* __restrict__ nn1,  // [n]
                                  int n, int maj_label, int remove_all,
                                  uint8_t* __restrict__ keep) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    int j = nn1[i];
    uint8_t kp = 1;
    if (j >= 0 && y[i] != y[j] && nn1[j] == i)
        if (remove_all || (int)y[i] == maj_label) kp = 0;
    keep[i] = kp;
}

// bin_codes_kernel: raw fp32 features -> uint8 bin codes.
// code = #{cut <= x} (searchsorted side='right'), binary search over the
// feature's cut array; bitwise-exact vs models/binning.bin_codes.
__global__ void bin_codes_kernel(const float* __restrict__ X,   // [n, FPAD]
                                 const float* __restrict__ cuts,
                                 const int* __restrict__ cut_off,  // [F+1]
                                 int n, int F,
                                 uint8_t* __restrict__ codes /* [n, FPAD] */) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    for (int f = 0; f < FPAD; ++f) {
        uint8_t code = 0;
        if (f < F) {
            const float x = X[(size_t)i * FPAD + f];
            const float* c = cuts + cut_off[f];
            int lo = 0, hi = cut_off[f + 1] - cut_off[f];
            while (lo < hi) {
                int mid = (lo + hi) >> 1;
                if (c[mid] <= x) lo = mid + 1; else hi = mid;
            }
            code = (uint8_t)lo;
        }
        codes[(size_t)i * FPAD + f] = code;
    }
}
