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
#define HWAVES (KNN_BLK / 64)

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

// ---------------------------------------------------------------------------
// MFMA k-NN (gfx950 matrix cores on the hot path).
//
// Exact-output two-phase design: phase 1 computes APPROXIMATE fp32
// distances d32 = |q|^2 + |c|^2 - 2 q.c for 16x16 (query, candidate)
// tiles on v_mfma_f32_16x16x4_f32 (K=16 in 4 chained steps; one candidate
// row is read once per 16 queries instead of once per pair, removing the
// LDS-bandwidth wall of the scalar kernel) and keeps a per-lane top-M
// candidate list; phase 2 re-evaluates the pooled survivors with the
// EXACT fp64 sequential-feature sum of balance/__init__.knn_indices and
// selects the final top-k with the same (distance, lower-index) order —
// so the output bits match the numpy reference exactly.  A query whose
// phase-1 list provably may have dropped a contender (discard_min within
// the error bound of the pooled kth distance) falls back to the scalar
// exact scan (knn_fallback_kernel); with continuous features this is
// rare-to-never.
//
// Error bound: |d32 - d| <= ~40 eps32 M (M = max squared row norm of the
// segment, ~20 roundings of magnitude <= 4M); the pool slack uses
// 4e-5 * M, a >15x margin.
// ---------------------------------------------------------------------------

typedef float v4f __attribute__((ext_vector_type(4)));

#define KNN_POOL_M 12          // per-lane phase-1 list length (>= k + 4)

__device__ __forceinline__ int knn_seg_of_row(
    const int* __restrict__ seg_off, int n_seg, int row) {
    int lo = 0, hi = n_seg;
    while (lo + 1 < hi) {
        int mid = (lo + hi) >> 1;
        if (seg_off[mid] <= row) lo = mid; else hi = mid;
    }
    return lo;
}

// Per-row squared norms (fp32).
__global__ void knn_norms_kernel(const float* __restrict__ X, int R,
                                 float* __restrict__ norms) {
    const int r = blockIdx.x * blockDim.x + threadIdx.x;
    if (r >= R) return;
    float s = 0.0f;
    #pragma unroll
    for (int f = 0; f < FPAD; ++f) {
        const float v = X[(size_t)r * FPAD + f];
        s = fmaf(v, v, s);
    }
    norms[r] = s;
}

// One 256-thread block = 4 waves = 64 queries of one segment; each wave
// owns 16 queries and sweeps all candidate tiles of the segment.
__launch_bounds__(KNN_BLK)
__global__ void knn_mfma_kernel(const float* __restrict__ X,
                                const int* __restrict__ seg_off,
                                const int* __restrict__ seg_blk,
                                int n_seg, int k, int skip_identity,
                                const float* __restrict__ norms,
                                int* __restrict__ out,
                                int* __restrict__ fb_list,
                                int* __restrict__ fb_count) {
    __shared__ float pd[HWAVES][64][KNN_POOL_M];
    __shared__ int pi[HWAVES][64][KNN_POOL_M];
    __shared__ float pdisc[HWAVES][64];

    // locate segment (binary search over per-segment block prefix)
    int lo = 0, hi = n_seg;
    while (lo + 1 < hi) {
        int mid = (lo + hi) >> 1;
        if (seg_blk[mid] <= (int)blockIdx.x) lo = mid; else hi = mid;
    }
    const int seg = lo;
    const int chunk = blockIdx.x - seg_blk[seg];
    const int base = seg_off[seg];
    const int n = seg_off[seg + 1] - base;

    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int q0 = chunk * 64 + wave * 16;      // segment-local query base
    if (q0 >= n) return;                         // wave-uniform

    const int jcol = lane & 15;                  // my query column
    const int kfrag = lane >> 4;                 // my K-slice (0..3)
    const int myq = q0 + jcol;
    const bool q_ok = myq < n;

    // B fragments: query jcol, features kfrag + 4*kk (zero-padded)
    float qf[4];
    #pragma unroll
    for (int kk = 0; kk < 4; ++kk)
        qf[kk] = q_ok ? X[(size_t)(base + myq) * FPAD + 4 * kk + kfrag]
                      : 0.0f;

    float td[KNN_POOL_M];
    int ti[KNN_POOL_M];
    #pragma unroll
    for (int j = 0; j < KNN_POOL_M; ++j) { td[j] = 1.0e30f; ti[j] = -1; }
    float discard_min = 1.0e30f;
    float worst = 1.0e30f;       // == td[KNN_POOL_M - 1], kept scalar
    const float qn = q_ok ? norms[base + myq] : 0.0f;

    // two 16-candidate tiles per iteration: independent accumulator
    // chains hide the MFMA dependent latency, loads amortize
    for (int c0 = 0; c0 < n; c0 += 32) {
        const int ca0 = c0 + jcol;
        const int ca1 = c0 + 16 + jcol;
        float cf0[4], cf1[4], nrm[8];
        #pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            cf0[kk] = ca0 < n
                ? X[(size_t)(base + ca0) * FPAD + 4 * kk + kfrag] : 0.0f;
            cf1[kk] = ca1 < n
                ? X[(size_t)(base + ca1) * FPAD + 4 * kk + kfrag] : 0.0f;
        }
        #pragma unroll
        for (int t = 0; t < 8; ++t) {
            const int c = c0 + (t >> 2) * 16 + kfrag * 4 + (t & 3);
            nrm[t] = c < n ? norms[base + c] : 0.0f;
        }

        v4f acc0 = {0.0f, 0.0f, 0.0f, 0.0f};
        v4f acc1 = {0.0f, 0.0f, 0.0f, 0.0f};
        #pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
            acc0 = __builtin_amdgcn_mfma_f32_16x16x4f32(cf0[kk], qf[kk],
                                                        acc0, 0, 0, 0);
            acc1 = __builtin_amdgcn_mfma_f32_16x16x4f32(cf1[kk], qf[kk],
                                                        acc1, 0, 0, 0);
        }

        // D rows this lane owns: candidates c0 (+16) + kfrag*4 + reg
        if (q_ok) {
            #pragma unroll
            for (int t = 0; t < 8; ++t) {
                const int c = c0 + (t >> 2) * 16 + kfrag * 4 + (t & 3);
                if (c >= n || (skip_identity && c == myq)) continue;
                const float g = (t < 4) ? acc0[t & 3] : acc1[t & 3];
                const float d32 = qn + nrm[t] - 2.0f * g;
                if (d32 < worst) {
                    // displaced entry becomes a discard
                    if (ti[KNN_POOL_M - 1] >= 0)
                        discard_min = fminf(discard_min,
                                            td[KNN_POOL_M - 1]);
                    int j = KNN_POOL_M - 1;
                    while (j > 0 && d32 < td[j - 1]) {
                        td[j] = td[j - 1];
                        ti[j] = ti[j - 1];
                        --j;
                    }
                    td[j] = d32;
                    ti[j] = c;
                    worst = td[KNN_POOL_M - 1];
                } else {
                    discard_min = fminf(discard_min, d32);
                }
            }
        }
    }

    // publish per-lane lists; wave lockstep makes them visible to the
    // merging lanes without a block barrier
    #pragma unroll
    for (int j = 0; j < KNN_POOL_M; ++j) {
        pd[wave][lane][j] = td[j];
        pi[wave][lane][j] = ti[j];
    }
    pdisc[wave][lane] = discard_min;
    __builtin_amdgcn_wave_barrier();

    // merge: lanes 0..15 finalize their query
    if (lane < 16 && q0 + lane < n) {
        const int q = q0 + lane;

        // tau = kth smallest pooled d32
        float topd[KMAX];
        #pragma unroll
        for (int j = 0; j < KMAX; ++j) topd[j] = 1.0e30f;
        float dmin = 1.0e30f;
        for (int g = 0; g < 4; ++g) {
            const int src = lane + g * 16;
            dmin = fminf(dmin, pdisc[wave][src]);
            for (int j = 0; j < KNN_POOL_M; ++j) {
                const float d = pd[wave][src][j];
                if (pi[wave][src][j] < 0) break;
                if (d < topd[k - 1]) {
                    int t = k - 1;
                    while (t > 0 && d < topd[t - 1]) {
                        topd[t] = topd[t - 1];
                        --t;
                    }
                    topd[t] = d;
                }
            }
        }
        // Per-query error bound: |d32 - d| <= ~24 eps32 B with
        // B = (|q| + |c|)^2 <= (2|q| + sqrt(tau))^2 for contenders
        // (24 roundings of magnitude <= B); slack carries an 8x margin.
        const float qnm = norms[base + q];
        const float rt = 2.0f * sqrtf(qnm) + sqrtf(fmaxf(topd[k - 1], 0.0f));
        const float slack = 1.2e-5f * rt * rt + 1.0e-30f;
        if (dmin <= topd[k - 1] + slack) {
            // phase-1 list may have dropped a contender: exact fallback
            atomicAdd(&fb_count[1 + seg], 1);
            fb_list[atomicAdd(fb_count, 1)] = base + q;
            return;
        }

        // exact phase: fp64 re-evaluation of the pooled survivors with
        // lexicographic (distance, index) selection — order-insensitive
        // restatement of "ascending index arrival + strictly-less
        // displaces", so the result bits match the reference
        double qv[FPAD];
        #pragma unroll
        for (int f = 0; f < FPAD; ++f)
            qv[f] = (double)X[(size_t)(base + q) * FPAD + f];

        double bd[KMAX];
        int bi[KMAX];
        #pragma unroll
        for (int j = 0; j < KMAX; ++j) { bd[j] = 1.0e300; bi[j] = -1; }

        for (int g = 0; g < 4; ++g) {
            const int src = lane + g * 16;
            for (int j = 0; j < KNN_POOL_M; ++j) {
                const int c = pi[wave][src][j];
                if (c < 0) break;
                double d = 0.0;
                #pragma unroll
                for (int f = 0; f < FPAD; ++f) {
                    const double diff =
                        qv[f] - (double)X[(size_t)(base + c) * FPAD + f];
                    d = d + diff * diff;
                }
                const bool lt = d < bd[k - 1] ||
                                (d == bd[k - 1] && bi[k - 1] >= 0 &&
                                 c < bi[k - 1]);
                if (lt) {
                    int t = k - 1;
                    while (t > 0 && (d < bd[t - 1] ||
                                     (d == bd[t - 1] && c < bi[t - 1]))) {
                        bd[t] = bd[t - 1];
                        bi[t] = bi[t - 1];
                        --t;
                    }
                    bd[t] = d;
                    bi[t] = c;
                }
            }
        }
        for (int j = 0; j < k; ++j)
            out[(size_t)(base + q) * k + j] = bi[j];
    }
}

// Fallback plumbing: flagged queries are regrouped by segment into
// 256-padded regions so the exact re-scan runs with the same LDS
// candidate tiling as the scalar kernel — the fallback then costs what
// the scalar path cost, but only for the flagged fraction.

// fb_count[0] = total flagged; fb_count[1 + s] = flagged in segment s.
// One thread: 256-padded exclusive offsets + cursor reset.
__global__ void knn_fb_scan_kernel(const int* __restrict__ fb_count,
                                   int n_seg, int* __restrict__ fb_off,
                                   int* __restrict__ fb_cursor) {
    if (threadIdx.x != 0 || blockIdx.x != 0) return;
    int acc = 0;
    for (int s = 0; s < n_seg; ++s) {
        fb_off[s] = acc;
        fb_cursor[s] = 0;
        acc += (fb_count[1 + s] + KNN_BLK - 1) / KNN_BLK * KNN_BLK;
    }
    fb_off[n_seg] = acc;
}

// -1-fill the used region of fb_sorted (device-bounded: costs nothing
// when no query was flagged).
__global__ void knn_fb_fill_kernel(const int* __restrict__ fb_off,
                                   int n_seg, int* __restrict__ fb_sorted) {
    const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < fb_off[n_seg]) fb_sorted[i] = -1;
}

// Scatter flagged rows into their segment's region (fb_sorted pre-filled
// with -1; every KNN_BLK-aligned block of it is single-segment).
__global__ void knn_fb_scatter_kernel(const int* __restrict__ fb_list,
                                      const int* __restrict__ fb_count,
                                      const int* __restrict__ seg_off,
                                      int n_seg,
                                      const int* __restrict__ fb_off,
                                      int* __restrict__ fb_cursor,
                                      int* __restrict__ fb_sorted) {
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= fb_count[0]) return;
    const int row = fb_list[i];
    const int seg = knn_seg_of_row(seg_off, n_seg, row);
    const int pos = atomicAdd(&fb_cursor[seg], 1);
    fb_sorted[fb_off[seg] + pos] = row;
}

// Exact scan for the flagged queries, LDS-tiled over the segment's
// candidates (numerics identical to knn_segmented_kernel).
__launch_bounds__(KNN_BLK)
__global__ void knn_fallback_kernel(const float* __restrict__ X,
                                    const int* __restrict__ seg_off,
                                    int n_seg, int k, int skip_identity,
                                    const int* __restrict__ fb_sorted,
                                    const int* __restrict__ fb_off,
                                    int* __restrict__ out) {
    __shared__ float tile[KNN_BLK][FPAD];
    __shared__ int sh_seg;

    const long slot = (long)blockIdx.x * KNN_BLK + threadIdx.x;
    if (threadIdx.x == 0) sh_seg = -1;
    __syncthreads();
    const int row = slot < fb_off[n_seg] ? fb_sorted[slot] : -1;
    if (row >= 0)
        sh_seg = knn_seg_of_row(seg_off, n_seg, row);  // benign same-value race
    __syncthreads();
    const int seg = sh_seg;
    if (seg < 0) return;   // all-padding block
    const int base = seg_off[seg];
    const int n = seg_off[seg + 1] - base;
    const int q = row - base;

    float qv[FPAD];
    if (row >= 0) {
        #pragma unroll
        for (int f = 0; f < FPAD; ++f)
            qv[f] = X[(size_t)row * FPAD + f];
    }

    double bd[KMAX];
    int bi[KMAX];
    for (int j = 0; j < KMAX; ++j) { bd[j] = 1.0e300; bi[j] = -1; }

    for (int tb = 0; tb < n; tb += KNN_BLK) {
        const int c = tb + threadIdx.x;
        if (c < n) {
            #pragma unroll
            for (int f = 0; f < FPAD; ++f)
                tile[threadIdx.x][f] = X[(size_t)(base + c) * FPAD + f];
        }
        __syncthreads();

        if (row >= 0) {
            const int tn = min(KNN_BLK, n - tb);
            for (int t = 0; t < tn; ++t) {
                const int cand = tb + t;
                if (skip_identity && cand == q) continue;
                double d = 0.0;
                #pragma unroll
                for (int f = 0; f < FPAD; ++f) {
                    const double diff = (double)qv[f] - (double)tile[t][f];
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

    if (row >= 0)
        for (int j = 0; j < k; ++j) out[(size_t)row * k + j] = bi[j];
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
