// StandardScaler and full PCA on MI355X (gfx950).
//
// Matches the semantics of preprocess/__init__.py (sklearn parity notes
// there; reference experiment.py:84-85): population std, zero-variance
// columns scale by 1.0; PCA = eigendecomposition of the centered Gram
// matrix (16x16) with sklearn's svd_flip sign convention (sign of the
// largest-|u| entry, ties to the lowest row index).  All statistics in
// fp64; validated against the numpy reference within fp tolerance (parallel
// reduction order differs — exactness is neither possible nor needed here,
// see SURVEY.md §7 reproducibility note).

#include <hip/hip_runtime.h>
#include <cstdint>

#define FPAD 16
#define RBLK 256

// Pass 1: per-column sum -> mean.  One block per feature.
__global__ void col_mean_kernel(const double* __restrict__ X, int n,
                                double* __restrict__ mean /* [FPAD] */) {
    __shared__ double red[RBLK];
    const int f = blockIdx.x;
    double s = 0.0;
    for (int i = threadIdx.x; i < n; i += RBLK)
        s += X[(size_t)i * FPAD + f];
    red[threadIdx.x] = s;
    __syncthreads();
    for (int d = RBLK / 2; d > 0; d >>= 1) {
        if (threadIdx.x < d) red[threadIdx.x] += red[threadIdx.x + d];
        __syncthreads();
    }
    if (threadIdx.x == 0) mean[f] = red[0] / (double)n;
}

// Pass 2: per-column variance (mean of squared deviations) -> scale.
__global__ void col_scale_kernel(const double* __restrict__ X, int n,
                                 const double* __restrict__ mean,
                                 double* __restrict__ scale /* [FPAD] */) {
    __shared__ double red[RBLK];
    const int f = blockIdx.x;
    const double m = mean[f];
    double s = 0.0;
    for (int i = threadIdx.x; i < n; i += RBLK) {
        double d = X[(size_t)i * FPAD + f] - m;
        s += d * d;
    }
    red[threadIdx.x] = s;
    __syncthreads();
    for (int d = RBLK / 2; d > 0; d >>= 1) {
        if (threadIdx.x < d) red[threadIdx.x] += red[threadIdx.x + d];
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        double sc = sqrt(red[0] / (double)n);
        scale[f] = (sc == 0.0) ? 1.0 : sc;
    }
}

// z-score transform, fp64 out (the PCA stage consumes fp64).
__global__ void scale_transform_kernel(const double* __restrict__ X, int n,
                                       const double* __restrict__ mean,
                                       const double* __restrict__ scale,
                                       double* __restrict__ out) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    #pragma unroll
    for (int f = 0; f < FPAD; ++f)
        out[(size_t)i * FPAD + f] =
            (X[(size_t)i * FPAD + f] - mean[f]) / scale[f];
}

// Centered Gram matrix C = Xc^T Xc.  One block per (a, b) pair, a <= b.
__global__ void gram_kernel(const double* __restrict__ X, int n,
                            const double* __restrict__ mean,
                            double* __restrict__ C /* [FPAD, FPAD] */) {
    __shared__ double red[RBLK];
    const int a = blockIdx.x / FPAD;
    const int b = blockIdx.x % FPAD;
    if (a > b) return;
    const double ma = mean[a], mb = mean[b];
    double s = 0.0;
    for (int i = threadIdx.x; i < n; i += RBLK) {
        double xa = X[(size_t)i * FPAD + a] - ma;
        double xb = X[(size_t)i * FPAD + b] - mb;
        s += xa * xb;
    }
    red[threadIdx.x] = s;
    __syncthreads();
    for (int d = RBLK / 2; d > 0; d >>= 1) {
        if (threadIdx.x < d) red[threadIdx.x] += red[threadIdx.x + d];
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        C[a * FPAD + b] = red[0];
        C[b * FPAD + a] = red[0];
    }
}

// Cyclic Jacobi eigendecomposition of the symmetric FPADxFPAD Gram matrix.
// Single thread — the matrix is 16x16, microseconds of work; eigenpairs are
// sorted by descending eigenvalue (stable: ties keep lower original index).
__global__ void jacobi_eigen_kernel(double* __restrict__ C,   // in: Gram
                                    double* __restrict__ V,   // out: [F,F] cols
                                    double* __restrict__ evals,
                                    int F) {
    if (blockIdx.x != 0 || threadIdx.x != 0) return;

    double A[FPAD][FPAD], Vv[FPAD][FPAD];
    for (int i = 0; i < F; ++i)
        for (int j = 0; j < F; ++j) {
            A[i][j] = C[i * FPAD + j];
            Vv[i][j] = (i == j) ? 1.0 : 0.0;
        }

    for (int sweep = 0; sweep < 64; ++sweep) {
        double off = 0.0;
        for (int p = 0; p < F; ++p)
            for (int q = p + 1; q < F; ++q) off += A[p][q] * A[p][q];
        if (off < 1e-24) break;

        for (int p = 0; p < F; ++p)
            for (int q = p + 1; q < F; ++q) {
                if (A[p][q] == 0.0) continue;
                double theta = (A[q][q] - A[p][p]) / (2.0 * A[p][q]);
                double t = (theta >= 0.0 ? 1.0 : -1.0)
                    / (fabs(theta) + sqrt(theta * theta + 1.0));
                double c = 1.0 / sqrt(t * t + 1.0);
                double s = t * c;
                for (int i = 0; i < F; ++i) {
                    double aip = A[i][p], aiq = A[i][q];
                    A[i][p] = c * aip - s * aiq;
                    A[i][q] = s * aip + c * aiq;
                }
                for (int i = 0; i < F; ++i) {
                    double api = A[p][i], aqi = A[q][i];
                    A[p][i] = c * api - s * aqi;
                    A[q][i] = s * api + c * aqi;
                }
                for (int i = 0; i < F; ++i) {
                    double vip = Vv[i][p], viq = Vv[i][q];
                    Vv[i][p] = c * vip - s * viq;
                    Vv[i][q] = s * vip + c * viq;
                }
            }
    }

    // Sort by descending eigenvalue (selection sort, stable).
    int order[FPAD];
    for (int i = 0; i < F; ++i) order[i] = i;
    for (int i = 0; i < F; ++i) {
        int best = i;
        for (int j = i + 1; j < F; ++j)
            if (A[order[j]][order[j]] > A[order[best]][order[best]]) best = j;
        int t = order[i]; order[i] = order[best]; order[best] = t;
    }
    for (int k = 0; k < F; ++k) {
        evals[k] = A[order[k]][order[k]];
        for (int i = 0; i < F; ++i) V[i * FPAD + k] = Vv[i][order[k]];
    }
}

// Project: T = Xc V.  One thread per row.
__global__ void pca_project_kernel(const double* __restrict__ X, int n,
                                   const double* __restrict__ mean,
                                   const double* __restrict__ V,
                                   int F,
                                   double* __restrict__ T) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    double xc[FPAD];
    for (int f = 0; f < FPAD; ++f)
        xc[f] = X[(size_t)i * FPAD + f] - mean[f];
    for (int k = 0; k < FPAD; ++k) {
        double s = 0.0;
        if (k < F)
            for (int f = 0; f < F; ++f) s += xc[f] * V[f * FPAD + k];
        T[(size_t)i * FPAD + k] = s;
    }
}

// svd_flip sign: per component, sign of the entry with the largest |T|,
// ties to the lowest row (numpy argmax).  One block per component.
__global__ void pca_signflip_kernel(double* __restrict__ T, int n, int F) {
    __shared__ double red_v[RBLK];
    __shared__ int red_i[RBLK];
    const int k = blockIdx.x;
    if (k >= F) return;

    double bv = -1.0;
    int bi = 0;
    for (int i = threadIdx.x; i < n; i += RBLK) {
        double v = fabs(T[(size_t)i * FPAD + k]);
        if (v > bv) { bv = v; bi = i; }
    }
    red_v[threadIdx.x] = bv;
    red_i[threadIdx.x] = bi;
    __syncthreads();
    for (int d = RBLK / 2; d > 0; d >>= 1) {
        if (threadIdx.x < d) {
            double ov = red_v[threadIdx.x + d];
            int oi = red_i[threadIdx.x + d];
            if (ov > red_v[threadIdx.x] ||
                (ov == red_v[threadIdx.x] && oi < red_i[threadIdx.x])) {
                red_v[threadIdx.x] = ov;
                red_i[threadIdx.x] = oi;
            }
        }
        __syncthreads();
    }
    const double sgn = (T[(size_t)red_i[0] * FPAD + k] < 0.0) ? -1.0 : 1.0;
    __syncthreads();
    for (int i = threadIdx.x; i < n; i += RBLK)
        T[(size_t)i * FPAD + k] *= sgn;
}

// fp64 -> fp32 matrix cast (the engine bins fp32 values).
__global__ void cast_f64_f32_kernel(const double* __restrict__ in,
                                    float* __restrict__ out, long count) {
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < count) out[i] = (float)in[i];
}
