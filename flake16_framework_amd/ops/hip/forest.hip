// Batched histogram-forest construction for MI355X (gfx950).
//
// One forest_fit call builds ALL trees of one grid cell (10 folds x
// n_estimators jobs) device-resident.  Work is tiered by node size:
//   > 2048 samples   level-synchronous work queue, one 256-thread
//                    workgroup per node, packed 16-bit LDS histograms
//                    (hist_split_kernel; a WIDE unpacked instantiation
//                    covers nodes >= 2^16) with parent-minus-smaller-child
//                    histogram-subtraction pools; Extra-Trees jobs use the
//                    histogram-free two-pass et_split_kernel instead
//   65..2048         mid_subtree_kernel: the block stages the node's code
//                    rows into LDS once and finishes the whole subtree
//                    internally (LDS histograms + LDS index partitions)
//   <= 64            small_subtree_kernel: one 64-lane wave finishes the
//                    subtree from register-resident samples with
//                    ballot/popcount counting
// Split selection is fp64 (file compiled -ffp-contract=off) and all
// randomness is Philox keyed on (tag, node sample-range, draw), so trees
// are bit-identical to the numpy reference in models/forest_ref.py no
// matter how the device schedules the work.
//
// Reference semantics being implemented: sklearn 1.0.2 defaults for
// DecisionTree/RandomForest/ExtraTrees (see models/forest_ref.py docstring;
// reference experiment.py:96-98, 469, 473).

#include <hip/hip_runtime.h>
#include <cstdint>

#include "philox.h"

#define HBLK 256
#define FPAD 16          // codes row stride (bytes); F <= 16
#define LEAF_SENTINEL (-1)

struct WorkItem {
    int job;
    int node;       // job-local node id
    int start;      // job-local sidx range [start, end)
    int end;
    int depth;
    int hist_slot;  // precomputed-histogram slot in pool[depth&1], or -1
};

struct ForestDev {
    const uint8_t* __restrict__ codes;    // [R, FPAD]
    const uint8_t* __restrict__ labels;   // [R]
    const int* __restrict__ j_row_off;    // [J] fold data base row
    const int* __restrict__ j_n;          // [J] samples per job
    const long* __restrict__ j_sidx_off;  // [J] base into sidx buffers
    const long* __restrict__ j_node_off;  // [J] base into node arrays
    const int* __restrict__ j_key;        // [J] philox k1 per job
    int* __restrict__ node_alloc;         // [J]
    int* __restrict__ nfeat;              // per node (LEAF_SENTINEL = leaf)
    int* __restrict__ nsplit;
    int* __restrict__ nleft;              // left child id; right = left + 1
    float* __restrict__ ncnt0;
    float* __restrict__ ncnt1;
    const int* __restrict__ sidx_cur;
    int* __restrict__ sidx_nxt;
    const WorkItem* __restrict__ cur;
    const int* __restrict__ cur_count;
    WorkItem* __restrict__ nxt;
    int* __restrict__ nxt_count;
    int* __restrict__ err_flag;
    int F;
    // Per-job model spec (a fused fit batches a balance group's three
    // model cells: DT + RF + ET jobs in one build).
    const int* __restrict__ j_mf;        // [J] max_features
    const uint8_t* __restrict__ j_rand;  // [J] splitter: 1 = random (ET)
    uint32_t seed;
    int work_cap;
    // Histogram-subtraction pools: a splitting node >= hist_save_min
    // samples accumulates its SMALLER child's histogram itself and derives
    // the larger child's by subtraction (exact integer counts); both are
    // stored in pool[(depth+1)&1] and the children skip accumulation.
    uint32_t* __restrict__ hist_pool0;
    uint32_t* __restrict__ hist_pool1;
    int* __restrict__ pool_count;   // strided level-state slots (see host)
    int pool_cap;
    int hist_save_min;
    // Small-subtree routing: children with n <= SMALL_N go to this queue
    // and are finished wholesale by small_subtree_kernel (one wave builds
    // the whole subtree from register-resident samples).
    WorkItem* __restrict__ small;
    int* __restrict__ small_count;
    int small_cap;
    // Mid-subtree routing: children with SMALL_N < n <= MID_N are staged
    // into LDS by mid_subtree_kernel, which recurses internally (no level
    // round-trips) and farms its <=SMALL_N descendants to the small queue.
    WorkItem* __restrict__ mid;
    int* __restrict__ mid_count;
    int mid_cap;
    // mid_subtree_kernel mode: wave-parallel node processing (4 nodes in
    // flight per block) when max_features <= WAVE_CANDS; 0 = block-serial
    // DFS (DT, or FLAKE16_NO_WAVE_MID=1 for A/B runs).
    int wave_mid;
};

#define SMALL_N 64
#define MID_N 2048

// Three-way child routing: <=SMALL_N -> wave-subtree queue, <=MID_N ->
// LDS-staged mid-subtree queue, else next level.  Called by one thread.
__device__ __forceinline__ void route_child(const ForestDev& a,
                                            const WorkItem& w) {
    const int n = w.end - w.start;
    if (n <= SMALL_N) {
        const int i = atomicAdd(a.small_count, 1);
        if (i < a.small_cap) a.small[i] = w;
        else atomicExch(a.err_flag, 1);
    } else if (n <= MID_N && w.hist_slot < 0) {
        const int i = atomicAdd(a.mid_count, 1);
        if (i < a.mid_cap) a.mid[i] = w;
        else atomicExch(a.err_flag, 1);
    } else {
        const int i = atomicAdd(a.nxt_count, 1);
        if (i < a.work_cap) a.nxt[i] = w;
        else atomicExch(a.err_flag, 1);
    }
}

// ---------------------------------------------------------------------------
// Init: fill per-job sample indices (bootstrap or identity) and root items.
// Bootstrap draw i: bounded(philox(TAG_BOOTSTRAP, 0, 0, i), n)  — matches
// forest_ref.fit_forest.
// ---------------------------------------------------------------------------
__global__ void forest_init_kernel(
    const int* __restrict__ j_row_off, const int* __restrict__ j_n,
    const long* __restrict__ j_sidx_off, const int* __restrict__ j_key,
    const long* __restrict__ j_node_off, int* __restrict__ nfeat,
    int* __restrict__ node_alloc, int* __restrict__ sidx,
    WorkItem* __restrict__ work, const uint8_t* __restrict__ j_boot,
    uint32_t seed) {
    int job = blockIdx.x;
    int n = j_n[job];
    long off = j_sidx_off[job];
    int row0 = j_row_off[job];
    uint32_t key = (uint32_t)j_key[job];
    const int bootstrap = j_boot[job];

    for (int i = threadIdx.x; i < n; i += blockDim.x) {
        int s;
        if (bootstrap) {
            uint32_t u = philox_draw(TAG_BOOTSTRAP, 0u, 0u, (uint32_t)i,
                                     seed, key);
            s = (int)philox_bounded(u, (uint32_t)n);
        } else {
            s = i;
        }
        sidx[off + i] = row0 + s;
    }
    if (threadIdx.x == 0) {
        node_alloc[job] = 1;
        nfeat[j_node_off[job]] = LEAF_SENTINEL;   // root starts as a leaf
        work[job] = {job, 0, 0, n, 0, -1};
    }
}

// ---------------------------------------------------------------------------
// The level kernel: histogram + split + partition, one workgroup per item.
// ---------------------------------------------------------------------------
// Histogram packing (WIDE = false): one uint32 per (feature, bin) with
// the total count in bits 0-15 and the class-1 count in bits 16-31 — one
// LDS atomic per (sample, feature) and 16 KiB LDS (double the resident
// blocks per CU).  Nodes with n >= 2^16 overflow the packing; the WIDE
// instantiation uses two unpacked count planes (32 KiB LDS, two atomics)
// and handles exactly those nodes — the two instantiations filter the
// same work queue by node size, so arbitrarily large training sets work
// (top levels wide, everything below packed).
template <bool WIDE>
__launch_bounds__(HBLK)
__global__ void hist_split_kernel(ForestDev a) {
    __shared__ uint32_t hist[FPAD * 256 * (WIDE ? 2 : 1)];
    __shared__ int sh_scan[HBLK];
    __shared__ int sh_bmin[FPAD], sh_bmax[FPAD];
    __shared__ int sh_cand[FPAD], sh_ncand;
    __shared__ double sh_score[FPAD];
    __shared__ int sh_bin[FPAD], sh_nL[FPAD];
    __shared__ int sh_bestf, sh_bestbin, sh_bestnL;
    __shared__ int sh_loff, sh_roff;
    __shared__ int sh_lid;                 // allocated left-child node id
    __shared__ uint32_t sh_draws[FPAD];
    __shared__ int sh_accum_small;         // phase-8 plan flags
    __shared__ int sh_slot_small, sh_slot_large, sh_small_is_left;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int n_items = *a.cur_count;

    for (int wi = blockIdx.x; wi < n_items; wi += gridDim.x) {
        WorkItem it = a.cur[wi];
        const int n = it.end - it.start;
        if ((n >= 65536) != WIDE) continue;   // size-class filter (uniform)
        if (a.j_rand[it.job]) continue;       // random-splitter: et_split's
        const int max_features = a.j_mf[it.job];
        const long sbase = a.j_sidx_off[it.job];
        const long nbase = a.j_node_off[it.job];
        const uint32_t key = (uint32_t)a.j_key[it.job];
        const int F = a.F;

        // helpers: per-(feature, bin) total / class-1 counts
        auto h_n = [&](int f, int b) -> uint32_t {
            return WIDE ? hist[f * 256 + b] : (hist[f * 256 + b] & 0xFFFFu);
        };
        auto h_1 = [&](int f, int b) -> uint32_t {
            return WIDE ? hist[FPAD * 256 + f * 256 + b]
                        : (hist[f * 256 + b] >> 16);
        };

        // Phase 0/1: obtain the node histogram — either load the slot the
        // parent precomputed (subtraction scheme), or zero + accumulate.
        if (!WIDE && it.hist_slot >= 0) {
            const uint32_t* src =
                ((it.depth & 1) ? a.hist_pool1 : a.hist_pool0)
                + (size_t)it.hist_slot * (FPAD * 256);
            for (int i = tid; i < F * 64; i += HBLK)
                reinterpret_cast<uint4*>(hist)[i] =
                    reinterpret_cast<const uint4*>(src)[i];
        } else {
            const int zwords = F * 64 * (WIDE ? 2 : 1);
            for (int i = tid; i < zwords; i += HBLK)
                reinterpret_cast<uint4*>(hist)[i] = uint4{0, 0, 0, 0};
            __syncthreads();
            // One uint4 = the sample's 16 packed bin codes.
            for (int i = it.start + tid; i < it.end; i += HBLK) {
                int row = a.sidx_cur[sbase + i];
                uint4 cw = *reinterpret_cast<const uint4*>(
                    a.codes + (size_t)row * FPAD);
                uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
                const uint32_t lab = a.labels[row];
                const uint32_t inc = WIDE ? 1u : (1u | (lab << 16));
                for (int f = 0; f < F; ++f) {
                    uint32_t b = (w[f >> 2] >> ((f & 3) * 8)) & 0xFFu;
                    atomicAdd(&hist[f * 256 + b], inc);
                    if (WIDE && lab)
                        atomicAdd(&hist[FPAD * 256 + f * 256 + b], 1u);
                }
            }
        }
        __syncthreads();

        // Phase 2: class counts from feature-0 histogram (wave reduce +
        // one cross-wave combine: 1 barrier instead of 8).
        {
            int v = (int)h_1(0, tid);
            for (int d = 32; d > 0; d >>= 1) v += __shfl_down(v, d);
            if (lane == 0) sh_scan[wave] = v;
        }
        __syncthreads();
        const int c1 = sh_scan[0] + sh_scan[1] + sh_scan[2] + sh_scan[3];
        const int c0 = n - c1;
        __syncthreads();

        if (tid == 0) {
            a.ncnt0[nbase + it.node] = (float)c0;
            a.ncnt1[nbase + it.node] = (float)c1;
        }

        if (n < 2 || c0 == 0 || c1 == 0) {
            __syncthreads();
            continue;  // leaf (nfeat stays LEAF_SENTINEL)
        }

        // Phase 3: occupied-bin range per feature (parallel: each of the
        // 256 threads scans one 16-bin segment of one feature).
        if (tid < F) {
            sh_bmin[tid] = 256;
            sh_bmax[tid] = -1;
        }
        __syncthreads();
        {
            const int f3 = tid >> 4;
            const int seg = (tid & 15) * 16;
            if (f3 < F) {
                int lo = 256, hi = -1;
                for (int b = seg; b < seg + 16; ++b)
                    if (h_n(f3, b)) {
                        if (lo == 256) lo = b;
                        hi = b;
                    }
                if (hi >= 0) {
                    atomicMin(&sh_bmin[f3], lo);
                    atomicMax(&sh_bmax[f3], hi);
                }
            }
        }
        __syncthreads();

        // Phase 4: feature permutation (partial Fisher-Yates, counters
        // (TAG_FEATSEL|depth<<8, start, end, i)) — draws computed by the
        // first F-1 threads in parallel, swap walk on thread 0 — and the
        // candidate list: walk perm, non-constant until max_features.
        if (tid < FPAD)
            sh_draws[tid] = philox_draw(
                TAG_FEATSEL | ((uint32_t)(it.depth & 0xFF) << 8),
                (uint32_t)it.start, (uint32_t)it.end, (uint32_t)tid,
                a.seed, key);
        __syncthreads();
        if (tid == 0) {
            int perm[FPAD];
            for (int f = 0; f < F; ++f) perm[f] = f;
            for (int i = 0; i < F - 1; ++i) {
                int j = i + (int)philox_bounded(sh_draws[i],
                                                (uint32_t)(F - i));
                int t = perm[i]; perm[i] = perm[j]; perm[j] = t;
            }
            int nc = 0;
            for (int i = 0; i < F && nc < max_features; ++i) {
                int f = perm[i];
                if (sh_bmin[f] != sh_bmax[f]) sh_cand[nc++] = f;
            }
            sh_ncand = nc;
        }
        __syncthreads();

        const int ncand = sh_ncand;
        if (ncand == 0) {
            __syncthreads();
            continue;  // all features constant: leaf
        }

        // Phase 5: evaluate candidates; wave w handles candidates w, w+4, ...
        // Scores in fp64, expression order matching the reference
        // (-ffp-contract=off).
        for (int ci = wave; ci < ncand; ci += HBLK / 64) {
            const int f = sh_cand[ci];
            const int bmin = sh_bmin[f], bmax = sh_bmax[f];

            // Per-lane 4-bin partial sums, then wave-inclusive scan.
            int ln[4], l1[4];
            int tn = 0, t1 = 0;
            for (int k = 0; k < 4; ++k) {
                int b = lane * 4 + k;
                ln[k] = (int)h_n(f, b);
                l1[k] = (int)h_1(f, b);
                tn += ln[k];
                t1 += l1[k];
            }
            int scn = tn, sc1 = t1;
            for (int d = 1; d < 64; d <<= 1) {
                int un = __shfl_up(scn, d);
                int u1 = __shfl_up(sc1, d);
                if (lane >= d) { scn += un; sc1 += u1; }
            }
            const int excl_n = scn - tn, excl_1 = sc1 - t1;

            double best_s = -1.0;  // all real scores are > 0
            int best_b = -1, best_nl = 0;

            if (a.j_rand[it.job]) {
                uint32_t tag = TAG_THRESH | ((uint32_t)(it.depth & 0xFF) << 8);
                uint32_t u = philox_draw(tag, (uint32_t)it.start,
                                         (uint32_t)it.end, (uint32_t)f,
                                         a.seed, key);
                int b = bmin + (int)philox_bounded(u, (uint32_t)(bmax - bmin));
                // the lane owning bin b evaluates it
                if (b >= lane * 4 && b < lane * 4 + 4) {
                    int cn = excl_n, c1f = excl_1;
                    for (int k = 0; k <= b - lane * 4; ++k) {
                        cn += ln[k];
                        c1f += l1[k];
                    }
                    long nL = cn, n1L = c1f;
                    long n0L = nL - n1L, nR = n - nL;
                    long n1R = c1 - n1L, n0R = c0 - n0L;
                    best_s = (double)(n0L * n0L + n1L * n1L) / (double)nL
                           + (double)(n0R * n0R + n1R * n1R) / (double)nR;
                    best_b = b;
                    best_nl = (int)nL;
                }
            } else {
                int cn = excl_n, c1f = excl_1;
                for (int k = 0; k < 4; ++k) {
                    int b = lane * 4 + k;
                    cn += ln[k];
                    c1f += l1[k];
                    // split scores only change at occupied bins, and the
                    // first bin of an equal-score run is occupied, so
                    // skipping empty bins preserves the argmax-first result
                    if (ln[k] == 0 || b < bmin || b >= bmax) continue;
                    long nL = cn, n1L = c1f;
                    long n0L = nL - n1L, nR = n - nL;
                    long n1R = c1 - n1L, n0R = c0 - n0L;
                    double s = (double)(n0L * n0L + n1L * n1L) / (double)nL
                             + (double)(n0R * n0R + n1R * n1R) / (double)nR;
                    if (s > best_s) { best_s = s; best_b = b; best_nl = (int)nL; }
                }
            }

            // Wave arg-max with first-of-ties (lowest bin) tie-break.
            for (int d = 32; d > 0; d >>= 1) {
                double os = __shfl_down(best_s, d);
                int ob = __shfl_down(best_b, d);
                int onl = __shfl_down(best_nl, d);
                if (os > best_s || (os == best_s && ob != -1 &&
                                    (best_b == -1 || ob < best_b))) {
                    best_s = os; best_b = ob; best_nl = onl;
                }
            }
            if (lane == 0) {
                sh_score[ci] = best_s;
                sh_bin[ci] = best_b;
                sh_nL[ci] = best_nl;
            }
        }
        __syncthreads();

        // Phase 6 (thread 0): sequential select in perm order, strict >.
        if (tid == 0) {
            double best_s = -1.0e300;
            int bf = -1, bb = -1, bnl = 0;
            for (int ci = 0; ci < ncand; ++ci) {
                if (sh_bin[ci] >= 0 && sh_score[ci] > best_s) {
                    best_s = sh_score[ci];
                    bf = sh_cand[ci];
                    bb = sh_bin[ci];
                    bnl = sh_nL[ci];
                }
            }
            sh_bestf = bf;
            sh_bestbin = bb;
            sh_bestnL = bnl;
            if (bf >= 0) {
                int l = atomicAdd(&a.node_alloc[it.job], 2);
                a.nfeat[nbase + l] = LEAF_SENTINEL;       // children start
                a.nfeat[nbase + l + 1] = LEAF_SENTINEL;   // as leaves
                a.nfeat[nbase + it.node] = bf;
                a.nsplit[nbase + it.node] = bb;
                a.nleft[nbase + it.node] = l;
                sh_lid = l;
            }
            sh_loff = 0;
            sh_roff = 0;
        }
        __syncthreads();

        const int bf = sh_bestf;
        if (bf < 0) {
            __syncthreads();
            continue;  // no valid split: leaf
        }
        const int bb = sh_bestbin;
        const int nL = sh_bestnL;

        // Phase 7: stable partition into sidx_nxt, 256-wide tiles.
        // Ranks come from wave ballots (valid lanes are a prefix of the
        // tile, so cross-wave offsets are 4 LDS words): 2 barriers per
        // tile instead of 16.
        for (int base = it.start; base < it.end; base += HBLK) {
            const int i = base + tid;
            const bool valid = i < it.end;
            int row = 0, flag = 0;
            if (valid) {
                row = a.sidx_cur[sbase + i];
                uint32_t b = a.codes[(size_t)row * FPAD + bf];
                flag = (int)(b <= (uint32_t)bb);
            }
            const unsigned long long lm = __ballot(valid && flag);
            const unsigned long long below = (1ULL << lane) - 1ULL;
            const int left_rank = __popcll(lm & below);
            const int wave_left = __popcll(lm);
            if (lane == 0) sh_scan[wave] = wave_left;
            __syncthreads();
            int wave_left_excl = 0;
            for (int ww = 0; ww < wave; ++ww)
                wave_left_excl += sh_scan[ww];
            const int tile_left = sh_scan[0] + sh_scan[1] + sh_scan[2]
                                  + sh_scan[3];
            const int tile_n = min(HBLK, it.end - base);
            if (valid) {
                if (flag) {
                    a.sidx_nxt[sbase + it.start + sh_loff + wave_left_excl
                               + left_rank] = row;
                } else {
                    // rights before me = my tile index - lefts before me
                    const int rights_before =
                        (i - base) - (wave_left_excl + left_rank);
                    a.sidx_nxt[sbase + it.start + nL + sh_roff
                               + rights_before] = row;
                }
            }
            __syncthreads();
            if (tid == 0) {
                sh_loff += tile_left;
                sh_roff += tile_n - tile_left;
            }
            __syncthreads();
        }

        // Phase 8: histogram-subtraction bookkeeping + child item push.
        // A big splitting node accumulates its SMALLER child's histogram
        // here (the children's samples are now contiguous in sidx_nxt) and
        // stores smaller + (parent - smaller) so both children skip their
        // accumulation.  Exact integer counts — results are unchanged.
        if (tid == 0) {
            // class-1 count of the left child (prefix over chosen feature)
            int n1L = 0;
            for (int b = 0; b <= bb; ++b)
                n1L += (int)h_1(bf, b);
            const int nR = n - nL;
            const int n1R = c1 - n1L;
            // only children that will run the histogram path need a slot
            const bool l_needs = nL > SMALL_N && n1L > 0 && n1L < nL;
            const bool r_needs = nR > SMALL_N && n1R > 0 && n1R < nR;
            sh_small_is_left = nL <= nR;
            const bool small_needs = sh_small_is_left ? l_needs : r_needs;
            const bool large_needs = sh_small_is_left ? r_needs : l_needs;

            sh_slot_small = sh_slot_large = -1;
            sh_accum_small = 0;
            // no pools on the WIDE path: packed 16-bit slots cannot hold
            // its counts (its children > 2048 re-accumulate)
            if (!WIDE && n >= a.hist_save_min &&
                (small_needs || large_needs)) {
                const int wp = (it.depth + 1) & 1;
                const int want = (small_needs ? 1 : 0) +
                                 (large_needs ? 1 : 0);
                int s = atomicAdd(&a.pool_count[wp * 4], want);
                if (s + want <= a.pool_cap) {
                    sh_accum_small = 1;
                    if (small_needs) sh_slot_small = s++;
                    if (large_needs) sh_slot_large = s;
                }
            }

            const int sl = sh_small_is_left ? sh_slot_small : sh_slot_large;
            const int sr = sh_small_is_left ? sh_slot_large : sh_slot_small;
            route_child(a, {it.job, sh_lid, it.start, it.start + nL,
                            it.depth + 1, sh_accum_small ? sl : -1});
            route_child(a, {it.job, sh_lid + 1, it.start + nL, it.end,
                            it.depth + 1, sh_accum_small ? sr : -1});
        }
        __syncthreads();

        if (sh_accum_small) {
            // stash the parent histogram in registers (16 words/thread)
            uint32_t stash[FPAD];
            #pragma unroll
            for (int r = 0; r < FPAD; ++r)
                stash[r] = (tid + r * HBLK < F * 256)
                               ? hist[tid + r * HBLK] : 0u;
            __syncthreads();

            for (int i = tid; i < F * 64; i += HBLK)
                reinterpret_cast<uint4*>(hist)[i] = uint4{0, 0, 0, 0};
            __syncthreads();

            const int s0 = sh_small_is_left ? it.start : it.start + nL;
            const int s1 = sh_small_is_left ? it.start + nL : it.end;
            for (int i = s0 + tid; i < s1; i += HBLK) {
                int row = a.sidx_nxt[sbase + i];
                uint4 cw = *reinterpret_cast<const uint4*>(
                    a.codes + (size_t)row * FPAD);
                uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
                const uint32_t inc = 1u | ((uint32_t)a.labels[row] << 16);
                for (int f = 0; f < F; ++f) {
                    uint32_t b = (w[f >> 2] >> ((f & 3) * 8)) & 0xFFu;
                    atomicAdd(&hist[f * 256 + b], inc);
                }
            }
            __syncthreads();

            uint32_t* wpool = ((it.depth + 1) & 1) ? a.hist_pool1
                                                   : a.hist_pool0;
            if (sh_slot_small >= 0) {
                uint32_t* dst = wpool + (size_t)sh_slot_small * (FPAD * 256);
                for (int i = tid; i < F * 256; i += HBLK)
                    dst[i] = hist[i];
            }
            if (sh_slot_large >= 0) {
                uint32_t* dst = wpool + (size_t)sh_slot_large * (FPAD * 256);
                #pragma unroll
                for (int r = 0; r < FPAD; ++r) {
                    const int i = tid + r * HBLK;
                    if (i < F * 256) dst[i] = stash[r] - hist[i];
                }
            }
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Extra-Trees level kernel: histogram-free.
//
// The random splitter needs only (a) per-feature occupied min/max codes to
// know constancy and the draw range, and (b) left counts at ONE drawn bin
// per candidate feature — two atomic-free passes over the node's samples
// instead of a 16-way-atomic histogram pass.  Scores, draws and
// tie-breaking are identical to the histogram path (same Philox counters,
// same fp64 expression), so trees are unchanged.  Used for all
// splitter_random jobs; the wave-subtree kernel handles their <=64 tails.
// ---------------------------------------------------------------------------
__launch_bounds__(HBLK)
__global__ void et_split_kernel(ForestDev a) {
    __shared__ int sh_scan[HBLK];
    __shared__ int sh_min[FPAD], sh_max[FPAD];
    __shared__ int sh_cand[FPAD], sh_cbin[FPAD], sh_ncand;
    __shared__ uint32_t sh_draws[FPAD], sh_tdraws[FPAD];
    __shared__ int sh_cnt[4][FPAD], sh_cnt1[4][FPAD];
    __shared__ int sh_bestf, sh_bestbin, sh_bestnL;
    __shared__ int sh_loff, sh_roff, sh_lid;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int n_items = *a.cur_count;

    for (int wi = blockIdx.x; wi < n_items; wi += gridDim.x) {
        WorkItem it = a.cur[wi];
        const int n = it.end - it.start;
        if (!a.j_rand[it.job]) continue;     // best-splitter: hist_split's
        const int max_features = a.j_mf[it.job];
        const long sbase = a.j_sidx_off[it.job];
        const long nbase = a.j_node_off[it.job];
        const uint32_t key = (uint32_t)a.j_key[it.job];
        const int F = a.F;

        if (tid < F) {
            sh_min[tid] = 256;
            sh_max[tid] = -1;
        }
        __syncthreads();

        // Pass 1: per-feature min/max + class-1 count.
        int lmin[FPAD], lmax[FPAD];
        #pragma unroll
        for (int f = 0; f < FPAD; ++f) { lmin[f] = 256; lmax[f] = -1; }
        int lc1 = 0;
        for (int i = it.start + tid; i < it.end; i += HBLK) {
            int row = a.sidx_cur[sbase + i];
            uint4 cw = *reinterpret_cast<const uint4*>(
                a.codes + (size_t)row * FPAD);
            uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
            lc1 += a.labels[row];
            for (int f = 0; f < F; ++f) {
                const int b = (int)((w[f >> 2] >> ((f & 3) * 8)) & 0xFFu);
                lmin[f] = min(lmin[f], b);
                lmax[f] = max(lmax[f], b);
            }
        }
        for (int d = 32; d > 0; d >>= 1) lc1 += __shfl_down(lc1, d);
        if (lane == 0) sh_scan[wave] = lc1;
        for (int f = 0; f < F; ++f) {
            int mn = lmin[f], mx = lmax[f];
            for (int d = 32; d > 0; d >>= 1) {
                mn = min(mn, __shfl_xor(mn, d));
                mx = max(mx, __shfl_xor(mx, d));
            }
            if (lane == 0) {
                atomicMin(&sh_min[f], mn);
                atomicMax(&sh_max[f], mx);
            }
        }
        __syncthreads();
        const int c1 = sh_scan[0] + sh_scan[1] + sh_scan[2] + sh_scan[3];
        const int c0 = n - c1;
        __syncthreads();

        if (tid == 0) {
            a.ncnt0[nbase + it.node] = (float)c0;
            a.ncnt1[nbase + it.node] = (float)c1;
        }
        if (n < 2 || c0 == 0 || c1 == 0) {
            __syncthreads();
            continue;
        }

        // Permutation walk + threshold draws: all Philox draws computed
        // by the first F threads in parallel, walk on thread 0.
        if (tid < FPAD) {
            sh_draws[tid] = philox_draw(
                TAG_FEATSEL | ((uint32_t)(it.depth & 0xFF) << 8),
                (uint32_t)it.start, (uint32_t)it.end, (uint32_t)tid,
                a.seed, key);
            sh_tdraws[tid] = philox_draw(
                TAG_THRESH | ((uint32_t)(it.depth & 0xFF) << 8),
                (uint32_t)it.start, (uint32_t)it.end, (uint32_t)tid,
                a.seed, key);
        }
        __syncthreads();
        if (tid == 0) {
            int perm[FPAD];
            for (int f = 0; f < F; ++f) perm[f] = f;
            for (int i = 0; i < F - 1; ++i) {
                int j = i + (int)philox_bounded(sh_draws[i],
                                                (uint32_t)(F - i));
                int t = perm[i]; perm[i] = perm[j]; perm[j] = t;
            }
            int nc = 0;
            for (int i = 0; i < F && nc < max_features; ++i) {
                int f = perm[i];
                if (sh_min[f] == sh_max[f]) continue;
                sh_cand[nc] = f;
                sh_cbin[nc] = sh_min[f] + (int)philox_bounded(
                    sh_tdraws[f], (uint32_t)(sh_max[f] - sh_min[f]));
                ++nc;
            }
            sh_ncand = nc;
        }
        __syncthreads();

        const int ncand = sh_ncand;
        if (ncand == 0) {
            __syncthreads();
            continue;
        }

        // Pass 2: left counts at each candidate's drawn bin.
        int cnt[FPAD], cnt1[FPAD];
        for (int ci = 0; ci < ncand; ++ci) { cnt[ci] = 0; cnt1[ci] = 0; }
        for (int i = it.start + tid; i < it.end; i += HBLK) {
            int row = a.sidx_cur[sbase + i];
            uint4 cw = *reinterpret_cast<const uint4*>(
                a.codes + (size_t)row * FPAD);
            uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
            const int lab = a.labels[row];
            for (int ci = 0; ci < ncand; ++ci) {
                const int f = sh_cand[ci];
                const int b = (int)((w[f >> 2] >> ((f & 3) * 8)) & 0xFFu);
                const int le = b <= sh_cbin[ci];
                cnt[ci] += le;
                cnt1[ci] += le & lab;
            }
        }
        for (int ci = 0; ci < ncand; ++ci) {
            int c = cnt[ci], c1l = cnt1[ci];
            for (int d = 32; d > 0; d >>= 1) {
                c += __shfl_down(c, d);
                c1l += __shfl_down(c1l, d);
            }
            if (lane == 0) {
                sh_cnt[wave][ci] = c;
                sh_cnt1[wave][ci] = c1l;
            }
        }
        __syncthreads();

        // Select (thread 0): candidates in perm order, strict >.
        if (tid == 0) {
            double best_s = -1.0e300;
            int bf = -1, bb = -1, bnl = 0;
            for (int ci = 0; ci < ncand; ++ci) {
                const long nL = sh_cnt[0][ci] + sh_cnt[1][ci]
                                + sh_cnt[2][ci] + sh_cnt[3][ci];
                const long n1L = sh_cnt1[0][ci] + sh_cnt1[1][ci]
                                 + sh_cnt1[2][ci] + sh_cnt1[3][ci];
                const long n0L = nL - n1L, nR = n - nL;
                const long n1R = c1 - n1L, n0R = c0 - n0L;
                if (nL == 0 || nR == 0) continue;
                double sc = (double)(n0L * n0L + n1L * n1L) / (double)nL
                            + (double)(n0R * n0R + n1R * n1R) / (double)nR;
                if (sc > best_s) {
                    best_s = sc;
                    bf = sh_cand[ci];
                    bb = sh_cbin[ci];
                    bnl = (int)nL;
                }
            }
            sh_bestf = bf;
            sh_bestbin = bb;
            sh_bestnL = bnl;
            if (bf >= 0) {
                int l = atomicAdd(&a.node_alloc[it.job], 2);
                a.nfeat[nbase + l] = LEAF_SENTINEL;       // children start
                a.nfeat[nbase + l + 1] = LEAF_SENTINEL;   // as leaves
                a.nfeat[nbase + it.node] = bf;
                a.nsplit[nbase + it.node] = bb;
                a.nleft[nbase + it.node] = l;
                sh_lid = l;
            }
            sh_loff = 0;
            sh_roff = 0;
        }
        __syncthreads();

        const int bf = sh_bestf;
        if (bf < 0) {
            __syncthreads();
            continue;
        }
        const int bb = sh_bestbin;
        const int nL = sh_bestnL;

        // Partition (same scheme as hist_split_kernel phase 7).
        for (int base = it.start; base < it.end; base += HBLK) {
            const int i = base + tid;
            const bool valid = i < it.end;
            int row = 0, flag = 0;
            if (valid) {
                row = a.sidx_cur[sbase + i];
                uint32_t b = a.codes[(size_t)row * FPAD + bf];
                flag = (int)(b <= (uint32_t)bb);
            }
            const unsigned long long lm = __ballot(valid && flag);
            const unsigned long long below = (1ULL << lane) - 1ULL;
            const int left_rank = __popcll(lm & below);
            if (lane == 0) sh_scan[wave] = __popcll(lm);
            __syncthreads();
            int wave_left_excl = 0;
            for (int ww = 0; ww < wave; ++ww)
                wave_left_excl += sh_scan[ww];
            const int tile_left = sh_scan[0] + sh_scan[1] + sh_scan[2]
                                  + sh_scan[3];
            const int tile_n = min(HBLK, it.end - base);
            if (valid) {
                if (flag)
                    a.sidx_nxt[sbase + it.start + sh_loff + wave_left_excl
                               + left_rank] = row;
                else
                    a.sidx_nxt[sbase + it.start + nL + sh_roff
                               + (i - base) - (wave_left_excl + left_rank)]
                        = row;
            }
            __syncthreads();
            if (tid == 0) {
                sh_loff += tile_left;
                sh_roff += tile_n - tile_left;
            }
            __syncthreads();
        }

        // Push children (no histogram slots on the ET path).
        if (tid == 0) {
            route_child(a, {it.job, sh_lid, it.start, it.start + nL,
                            it.depth + 1, -1});
            route_child(a, {it.job, sh_lid + 1, it.start + nL, it.end,
                            it.depth + 1, -1});
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Candidate-only Random-Forest level kernel.
//
// RF evaluates max_features (= sqrt(F) ~ 4) candidate features per node,
// but the generic histogram kernel accumulates all 16.  Here pass 1
// computes per-feature min/max (constancy) + class counts; the Fisher-
// Yates walk then fixes the <= max_features candidates, and pass 2
// histograms ONLY those (~4 LDS atomics per sample instead of 16).
// Split scores, tie-breaks and partition are identical to
// hist_split_kernel, so trees are unchanged.  No histogram-subtraction
// pools on this path (children of other candidates are useless to them).
// ---------------------------------------------------------------------------
__launch_bounds__(HBLK)
__global__ void rf_cand_split_kernel(ForestDev a) {
    __shared__ uint32_t hist[FPAD * 256];   // rows = candidate index
    __shared__ int sh_scan[HBLK];
    __shared__ int sh_min[FPAD], sh_max[FPAD];
    __shared__ int sh_cand[FPAD], sh_ncand;
    __shared__ uint32_t sh_draws[FPAD];
    __shared__ double sh_score[FPAD];
    __shared__ int sh_bin[FPAD], sh_nL[FPAD];
    __shared__ int sh_bestf, sh_bestbin, sh_bestnL;
    __shared__ int sh_loff, sh_roff, sh_lid;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int n_items = *a.cur_count;

    for (int wi = blockIdx.x; wi < n_items; wi += gridDim.x) {
        WorkItem it = a.cur[wi];
        const int n = it.end - it.start;
        if (a.j_rand[it.job]) continue;      // random-splitter: et_split's
        const int max_features = a.j_mf[it.job];
        const long sbase = a.j_sidx_off[it.job];
        const long nbase = a.j_node_off[it.job];
        const uint32_t key = (uint32_t)a.j_key[it.job];
        const int F = a.F;

        if (tid < F) {
            sh_min[tid] = 256;
            sh_max[tid] = -1;
        }
        __syncthreads();

        // Pass 1: per-feature min/max + class-1 count.
        int lmin[FPAD], lmax[FPAD];
        #pragma unroll
        for (int f = 0; f < FPAD; ++f) { lmin[f] = 256; lmax[f] = -1; }
        int lc1 = 0;
        for (int i = it.start + tid; i < it.end; i += HBLK) {
            int row = a.sidx_cur[sbase + i];
            uint4 cw = *reinterpret_cast<const uint4*>(
                a.codes + (size_t)row * FPAD);
            uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
            lc1 += a.labels[row];
            for (int f = 0; f < F; ++f) {
                const int b = (int)((w[f >> 2] >> ((f & 3) * 8)) & 0xFFu);
                lmin[f] = min(lmin[f], b);
                lmax[f] = max(lmax[f], b);
            }
        }
        for (int d = 32; d > 0; d >>= 1) lc1 += __shfl_down(lc1, d);
        if (lane == 0) sh_scan[wave] = lc1;
        for (int f = 0; f < F; ++f) {
            int mn = lmin[f], mx = lmax[f];
            for (int d = 32; d > 0; d >>= 1) {
                mn = min(mn, __shfl_xor(mn, d));
                mx = max(mx, __shfl_xor(mx, d));
            }
            if (lane == 0) {
                atomicMin(&sh_min[f], mn);
                atomicMax(&sh_max[f], mx);
            }
        }
        __syncthreads();
        const int c1 = sh_scan[0] + sh_scan[1] + sh_scan[2] + sh_scan[3];
        const int c0 = n - c1;
        __syncthreads();

        if (tid == 0) {
            a.ncnt0[nbase + it.node] = (float)c0;
            a.ncnt1[nbase + it.node] = (float)c1;
        }
        if (n < 2 || c0 == 0 || c1 == 0) {
            __syncthreads();
            continue;
        }

        // Candidate walk: parallel draws, thread-0 swap walk (same
        // permutation/constancy semantics as the histogram path).
        if (tid < FPAD)
            sh_draws[tid] = philox_draw(
                TAG_FEATSEL | ((uint32_t)(it.depth & 0xFF) << 8),
                (uint32_t)it.start, (uint32_t)it.end, (uint32_t)tid,
                a.seed, key);
        __syncthreads();
        if (tid == 0) {
            int perm[FPAD];
            for (int f = 0; f < F; ++f) perm[f] = f;
            for (int i = 0; i < F - 1; ++i) {
                int j = i + (int)philox_bounded(sh_draws[i],
                                                (uint32_t)(F - i));
                int t = perm[i]; perm[i] = perm[j]; perm[j] = t;
            }
            int nc = 0;
            for (int i = 0; i < F && nc < max_features; ++i) {
                int f = perm[i];
                if (sh_min[f] != sh_max[f]) sh_cand[nc++] = f;
            }
            sh_ncand = nc;
        }
        __syncthreads();

        const int ncand = sh_ncand;
        if (ncand == 0) {
            __syncthreads();
            continue;
        }

        // Pass 2: packed histograms of the candidate features only.
        for (int i = tid; i < ncand * 256; i += HBLK) hist[i] = 0;
        __syncthreads();
        for (int i = it.start + tid; i < it.end; i += HBLK) {
            int row = a.sidx_cur[sbase + i];
            uint4 cw = *reinterpret_cast<const uint4*>(
                a.codes + (size_t)row * FPAD);
            uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
            const uint32_t inc = 1u | ((uint32_t)a.labels[row] << 16);
            for (int ci = 0; ci < ncand; ++ci) {
                const int f = sh_cand[ci];
                uint32_t b = (w[f >> 2] >> ((f & 3) * 8)) & 0xFFu;
                atomicAdd(&hist[ci * 256 + b], inc);
            }
        }
        __syncthreads();

        // Evaluate: wave w handles candidates w, w+4, ... (identical fp64
        // scores and tie-breaks to hist_split_kernel phase 5).
        for (int ci = wave; ci < ncand; ci += HBLK / 64) {
            const int f = sh_cand[ci];
            const int bmin = sh_min[f], bmax = sh_max[f];
            int ln[4], l1[4];
            int tn = 0, t1 = 0;
            for (int k = 0; k < 4; ++k) {
                uint32_t v = hist[ci * 256 + lane * 4 + k];
                ln[k] = (int)(v & 0xFFFFu);
                l1[k] = (int)(v >> 16);
                tn += ln[k];
                t1 += l1[k];
            }
            int scn = tn, sc1 = t1;
            for (int d = 1; d < 64; d <<= 1) {
                int un = __shfl_up(scn, d);
                int u1 = __shfl_up(sc1, d);
                if (lane >= d) { scn += un; sc1 += u1; }
            }
            const int excl_n = scn - tn, excl_1 = sc1 - t1;

            double best_s = -1.0;
            int best_b = -1, best_nl = 0;
            int cn = excl_n, c1f = excl_1;
            for (int k = 0; k < 4; ++k) {
                int b = lane * 4 + k;
                cn += ln[k];
                c1f += l1[k];
                if (ln[k] == 0 || b < bmin || b >= bmax) continue;
                long nL = cn, n1L = c1f;
                long n0L = nL - n1L, nR = n - nL;
                long n1R = c1 - n1L, n0R = c0 - n0L;
                double s = (double)(n0L * n0L + n1L * n1L) / (double)nL
                         + (double)(n0R * n0R + n1R * n1R) / (double)nR;
                if (s > best_s) { best_s = s; best_b = b; best_nl = (int)nL; }
            }
            for (int d = 32; d > 0; d >>= 1) {
                double os = __shfl_down(best_s, d);
                int ob = __shfl_down(best_b, d);
                int onl = __shfl_down(best_nl, d);
                if (os > best_s || (os == best_s && ob != -1 &&
                                    (best_b == -1 || ob < best_b))) {
                    best_s = os; best_b = ob; best_nl = onl;
                }
            }
            if (lane == 0) {
                sh_score[ci] = best_s;
                sh_bin[ci] = best_b;
                sh_nL[ci] = best_nl;
            }
        }
        __syncthreads();

        if (tid == 0) {
            double best_s = -1.0e300;
            int bf = -1, bb = -1, bnl = 0;
            for (int ci = 0; ci < ncand; ++ci) {
                if (sh_bin[ci] >= 0 && sh_score[ci] > best_s) {
                    best_s = sh_score[ci];
                    bf = sh_cand[ci];
                    bb = sh_bin[ci];
                    bnl = sh_nL[ci];
                }
            }
            sh_bestf = bf;
            sh_bestbin = bb;
            sh_bestnL = bnl;
            if (bf >= 0) {
                int l = atomicAdd(&a.node_alloc[it.job], 2);
                a.nfeat[nbase + l] = LEAF_SENTINEL;       // children start
                a.nfeat[nbase + l + 1] = LEAF_SENTINEL;   // as leaves
                a.nfeat[nbase + it.node] = bf;
                a.nsplit[nbase + it.node] = bb;
                a.nleft[nbase + it.node] = l;
                sh_lid = l;
            }
            sh_loff = 0;
            sh_roff = 0;
        }
        __syncthreads();

        const int bf = sh_bestf;
        if (bf < 0) {
            __syncthreads();
            continue;
        }
        const int bb = sh_bestbin;
        const int nL = sh_bestnL;

        // Partition (identical to hist_split_kernel phase 7).
        for (int base = it.start; base < it.end; base += HBLK) {
            const int i = base + tid;
            const bool valid = i < it.end;
            int row = 0, flag = 0;
            if (valid) {
                row = a.sidx_cur[sbase + i];
                uint32_t b = a.codes[(size_t)row * FPAD + bf];
                flag = (int)(b <= (uint32_t)bb);
            }
            const unsigned long long lm = __ballot(valid && flag);
            const unsigned long long below = (1ULL << lane) - 1ULL;
            const int left_rank = __popcll(lm & below);
            if (lane == 0) sh_scan[wave] = __popcll(lm);
            __syncthreads();
            int wave_left_excl = 0;
            for (int ww = 0; ww < wave; ++ww)
                wave_left_excl += sh_scan[ww];
            const int tile_left = sh_scan[0] + sh_scan[1] + sh_scan[2]
                                  + sh_scan[3];
            const int tile_n = min(HBLK, it.end - base);
            if (valid) {
                if (flag)
                    a.sidx_nxt[sbase + it.start + sh_loff + wave_left_excl
                               + left_rank] = row;
                else
                    a.sidx_nxt[sbase + it.start + nL + sh_roff
                               + (i - base) - (wave_left_excl + left_rank)]
                        = row;
            }
            __syncthreads();
            if (tid == 0) {
                sh_loff += tile_left;
                sh_roff += tile_n - tile_left;
            }
            __syncthreads();
        }

        if (tid == 0) {
            route_child(a, {it.job, sh_lid, it.start, it.start + nL,
                            it.depth + 1, -1});
            route_child(a, {it.job, sh_lid + 1, it.start + nL, it.end,
                            it.depth + 1, -1});
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Mid-subtree builder: one BLOCK finishes the whole subtree of a node with
// SMALL_N < n <= MID_N samples from LDS-resident data.
//
// The node's code rows + labels are staged into LDS ONCE; the block then
// recurses (smaller-child-first DFS, stack in LDS) over its > SMALL_N
// descendants with LDS histograms and LDS index partitions — no level
// round-trips, no global gathers per level (PMC showed the level kernels
// 40-69% wait-bound on exactly those).  <= SMALL_N descendants are written
// back (final index arrangement -> sidx buffer) and pushed to the small
// queue for the wave kernel.  Split semantics, RNG counters (global
// job-relative (start,end) ranges) and fp64 scores are identical to
// hist_split_kernel — trees are bit-identical.
// Handles both splitters (best: prefix scan over bins; random: drawn bin).
// ---------------------------------------------------------------------------
struct MidFrame {
    short s, e;     // LOCAL sample range within the staged window
    int depth;
    int node;
};

// Block-serial DFS (DT / fallback) needs <= log2(MID_N)+2 entries; the
// wave-parallel mode (4 nodes in flight, breadth-ish order) can hold the
// whole >SMALL_N frontier of a MID_N subtree (~ MID_N/SMALL_N * 2, plus
// slack); overflow falls back to the next-level global work queue.
#define MID_STACK 96
#define WAVE_CANDS 4   // wave-parallel path: max_features <= 4 (RF / ET)

// One 64-lane wave splits one node [ls, le) of the staged window and
// routes its children (shared LDS stack for > SMALL_N, global small queue
// otherwise).  Control flow is wave-uniform throughout; split semantics,
// Philox counters (global job-relative ranges) and fp64 scores are
// identical to the block-wide path — trees are bit-identical.  Requires
// max_features <= WAVE_CANDS (RF / ET; DT keeps the block path).
// whist: this wave's WAVE_CANDS*256-word LDS histogram region (RF only).
template <bool STAGED>
__device__ __forceinline__ void mid_wave_node(
    const ForestDev& a, const WorkItem& it, int ls, int le, int depth,
    int node, const uint4* m_codes, const uint8_t* m_lab,
    const int* m_rows, uint16_t* m_idx,
    uint16_t* m_idx2, uint32_t* whist, int* wperm, uint32_t* wthr,
    int* wcand, int* wcbin, MidFrame* stack, int* sh_count) {
    // STAGED reads the LDS-staged code rows / labels; the unstaged
    // variant reads them from L1/L2 through the m_rows map (half the
    // LDS per block -> double the resident blocks).
    auto code_at = [&](int o) -> uint4 {
        if (STAGED) return m_codes[o];
        return *reinterpret_cast<const uint4*>(
            a.codes + (size_t)m_rows[o] * FPAD);
    };
    auto lab_at = [&](int o) -> int {
        return STAGED ? (int)m_lab[o] : (int)a.labels[m_rows[o]];
    };
    const int lane = threadIdx.x & 63;
    const int n = le - ls;
    const long nbase = a.j_node_off[it.job];
    const uint32_t key = (uint32_t)a.j_key[it.job];
    const int F = a.F;
    const int max_features = a.j_mf[it.job];
    const int splitter_random = a.j_rand[it.job];
    const int gs = it.start + ls;   // global job-relative range (RNG id)
    const int ge = it.start + le;

    // Pass 1: per-feature occupied min/max + class-1 count.
    int lmin[FPAD], lmax[FPAD];
    #pragma unroll
    for (int f = 0; f < FPAD; ++f) { lmin[f] = 256; lmax[f] = -1; }
    int lc1 = 0;
    for (int i = ls + lane; i < le; i += 64) {
        const int o = m_idx[i];
        const uint4 cw = code_at(o);
        const uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
        lc1 += lab_at(o);
        #pragma unroll
        for (int f = 0; f < FPAD; ++f) {
            const int b = (int)((w[f >> 2] >> ((f & 3) * 8)) & 0xFFu);
            lmin[f] = min(lmin[f], b);
            lmax[f] = max(lmax[f], b);
        }
    }
    for (int d = 32; d > 0; d >>= 1) lc1 += __shfl_down(lc1, d);
    const int c1 = __shfl(lc1, 0);
    const int c0 = n - c1;
    #pragma unroll
    for (int f = 0; f < FPAD; ++f) {
        int mn = lmin[f], mx = lmax[f];
        for (int d = 32; d > 0; d >>= 1) {
            mn = min(mn, __shfl_xor(mn, d));
            mx = max(mx, __shfl_xor(mx, d));
        }
        lmin[f] = mn;   // wave-uniform from here on
        lmax[f] = mx;
    }

    if (lane == 0) {
        a.ncnt0[nbase + node] = (float)c0;
        a.ncnt1[nbase + node] = (float)c1;
    }
    if (n < 2 || c0 == 0 || c1 == 0) return;   // leaf

    // Feature permutation + candidate walk (draws parallel across lanes,
    // walk on lane 0 through this wave's LDS scratch).
    {
        const uint32_t uf = philox_draw(
            TAG_FEATSEL | ((uint32_t)(depth & 0xFF) << 8),
            (uint32_t)gs, (uint32_t)ge,
            (uint32_t)(lane < FPAD ? lane : 0), a.seed, key);
        if (splitter_random && lane < FPAD)
            wthr[lane] = philox_draw(
                TAG_THRESH | ((uint32_t)(depth & 0xFF) << 8),
                (uint32_t)gs, (uint32_t)ge, (uint32_t)lane, a.seed, key);
        if (lane == 0)
            for (int f = 0; f < F; ++f) wperm[f] = f;
        for (int i = 0; i < F - 1; ++i) {
            const uint32_t u = __shfl(uf, i);
            if (lane == 0) {
                int j = i + (int)philox_bounded(u, (uint32_t)(F - i));
                int t = wperm[i]; wperm[i] = wperm[j]; wperm[j] = t;
            }
        }
    }
    int ncand = 0;
    if (lane == 0) {
        for (int i = 0; i < F && ncand < max_features; ++i) {
            const int f = wperm[i];
            if (lmin[f] == lmax[f]) continue;
            wcand[ncand] = f;
            if (splitter_random)
                wcbin[ncand] = lmin[f] + (int)philox_bounded(
                    wthr[f], (uint32_t)(lmax[f] - lmin[f]));
            ++ncand;
        }
    }
    ncand = __shfl(ncand, 0);
    if (ncand == 0) return;   // all candidates constant: leaf

    double best_s = -1.0e300;
    int bf = -1, bb = -1, bnl = 0;

    if (splitter_random) {
        // ET: counts at each candidate's drawn bin — histogram-free.
        int cnt[WAVE_CANDS], cnt1[WAVE_CANDS];
        #pragma unroll
        for (int ci = 0; ci < WAVE_CANDS; ++ci) { cnt[ci] = 0; cnt1[ci] = 0; }
        for (int i = ls + lane; i < le; i += 64) {
            const int o = m_idx[i];
            const uint4 cw = code_at(o);
            const uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
            const int lab = lab_at(o);
            #pragma unroll
            for (int ci = 0; ci < WAVE_CANDS; ++ci) {
                if (ci >= ncand) break;
                const int f = wcand[ci];
                const int b = (int)((w[f >> 2] >> ((f & 3) * 8)) & 0xFFu);
                const int le_ = b <= wcbin[ci];
                cnt[ci] += le_;
                cnt1[ci] += le_ & lab;
            }
        }
        #pragma unroll
        for (int ci = 0; ci < WAVE_CANDS; ++ci) {
            if (ci >= ncand) break;
            int c = cnt[ci], cl = cnt1[ci];
            for (int d = 32; d > 0; d >>= 1) {
                c += __shfl_down(c, d);
                cl += __shfl_down(cl, d);
            }
            c = __shfl(c, 0);
            cl = __shfl(cl, 0);
            const long nL = c, n1L = cl;
            const long n0L = nL - n1L, nR = n - nL;
            const long n1R = c1 - n1L, n0R = c0 - n0L;
            if (nL == 0 || nR == 0) continue;
            const double sc = (double)(n0L * n0L + n1L * n1L) / (double)nL
                            + (double)(n0R * n0R + n1R * n1R) / (double)nR;
            if (sc > best_s) {
                best_s = sc;
                bf = wcand[ci];
                bb = wcbin[ci];
                bnl = (int)nL;
            }
        }
    } else {
        // RF: candidate-only packed LDS histograms.
        for (int i = lane; i < ncand * 256; i += 64) whist[i] = 0;
        for (int i = ls + lane; i < le; i += 64) {
            const int o = m_idx[i];
            const uint4 cw = code_at(o);
            const uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
            const uint32_t inc = 1u | ((uint32_t)lab_at(o) << 16);
            #pragma unroll
            for (int ci = 0; ci < WAVE_CANDS; ++ci) {
                if (ci >= ncand) break;
                const int f = wcand[ci];
                const uint32_t b = (w[f >> 2] >> ((f & 3) * 8)) & 0xFFu;
                atomicAdd(&whist[ci * 256 + b], inc);
            }
        }
        for (int ci = 0; ci < ncand; ++ci) {
            const int f = wcand[ci];
            const int bmin = lmin[f], bmax = lmax[f];
            int ln[4], l1[4];
            int tn = 0, t1 = 0;
            #pragma unroll
            for (int k = 0; k < 4; ++k) {
                const uint32_t v = whist[ci * 256 + lane * 4 + k];
                ln[k] = (int)(v & 0xFFFFu);
                l1[k] = (int)(v >> 16);
                tn += ln[k];
                t1 += l1[k];
            }
            int scn = tn, sc1 = t1;
            for (int d = 1; d < 64; d <<= 1) {
                const int un = __shfl_up(scn, d);
                const int u1 = __shfl_up(sc1, d);
                if (lane >= d) { scn += un; sc1 += u1; }
            }
            int cn = scn - tn, c1f = sc1 - t1;
            double cand_s = -1.0;
            int cand_b = -1, cand_nl = 0;
            #pragma unroll
            for (int k = 0; k < 4; ++k) {
                const int b = lane * 4 + k;
                cn += ln[k];
                c1f += l1[k];
                if (ln[k] == 0 || b < bmin || b >= bmax) continue;
                const long nL = cn, n1L = c1f;
                const long n0L = nL - n1L, nR = n - nL;
                const long n1R = c1 - n1L, n0R = c0 - n0L;
                const double s =
                    (double)(n0L * n0L + n1L * n1L) / (double)nL
                    + (double)(n0R * n0R + n1R * n1R) / (double)nR;
                if (s > cand_s) { cand_s = s; cand_b = b; cand_nl = (int)nL; }
            }
            for (int d = 32; d > 0; d >>= 1) {
                const double os = __shfl_down(cand_s, d);
                const int ob = __shfl_down(cand_b, d);
                const int onl = __shfl_down(cand_nl, d);
                if (os > cand_s || (os == cand_s && ob != -1 &&
                                    (cand_b == -1 || ob < cand_b))) {
                    cand_s = os; cand_b = ob; cand_nl = onl;
                }
            }
            cand_s = __shfl(cand_s, 0);
            cand_b = __shfl(cand_b, 0);
            cand_nl = __shfl(cand_nl, 0);
            // sequential select in perm order, strict > (phase-6 contract)
            if (cand_b >= 0 && cand_s > best_s) {
                best_s = cand_s;
                bf = wcand[ci];
                bb = cand_b;
                bnl = cand_nl;
            }
        }
    }

    if (bf < 0) return;   // no valid split: leaf
    const int nL = bnl;

    int lid = 0;
    if (lane == 0) {
        lid = atomicAdd(&a.node_alloc[it.job], 2);
        a.nfeat[nbase + lid] = LEAF_SENTINEL;
        a.nfeat[nbase + lid + 1] = LEAF_SENTINEL;
        a.nfeat[nbase + node] = bf;
        a.nsplit[nbase + node] = bb;
        a.nleft[nbase + node] = lid;
    }
    lid = __shfl(lid, 0);

    // Stable partition of m_idx[ls, le) via this node's private m_idx2
    // range (sibling ranges are disjoint — no cross-wave aliasing).
    int lo = 0, ro = 0;
    for (int base = ls; base < le; base += 64) {
        const int i = base + lane;
        const bool valid = i < le;
        int o = 0, flag = 0;
        if (valid) {
            o = m_idx[i];
            const uint4 cw = code_at(o);
            const uint32_t wsel = (&cw.x)[bf >> 2];
            flag = (int)(((wsel >> ((bf & 3) * 8)) & 0xFFu)
                         <= (uint32_t)bb);
        }
        const unsigned long long lm = __ballot(valid && flag);
        const unsigned long long below = (1ULL << lane) - 1ULL;
        const int rank = __popcll(lm & below);
        const int chunk_left = __popcll(lm);
        const int chunk_n = min(64, le - base);
        if (valid) {
            const int dst = flag
                ? ls + lo + rank
                : ls + nL + ro + (i - base) - rank;
            m_idx2[dst] = (uint16_t)o;
        }
        lo += chunk_left;
        ro += chunk_n - chunk_left;
    }
    for (int i = ls + lane; i < le; i += 64) m_idx[i] = m_idx2[i];

    // Route children: > SMALL_N to the shared stack (overflow: next-level
    // global work queue — correct, just slower); <= SMALL_N to the global
    // small queue with final global ranges.
    if (lane == 0) {
        const MidFrame kids[2] = {
            {(short)ls, (short)(ls + nL), depth + 1, lid},
            {(short)(ls + nL), (short)le, depth + 1, lid + 1},
        };
        for (int x = 0; x < 2; ++x) {
            const MidFrame& fr = kids[x];
            const int fn = fr.e - fr.s;
            if (fn > SMALL_N) {
                const int idx = atomicAdd(sh_count, 1);
                if (idx < MID_STACK) {
                    stack[idx] = fr;
                } else {
                    atomicSub(sh_count, 1);
                    const int i2 = atomicAdd(a.nxt_count, 1);
                    if (i2 < a.work_cap)
                        a.nxt[i2] = {it.job, fr.node, it.start + fr.s,
                                     it.start + fr.e, fr.depth, -1};
                    else
                        atomicExch(a.err_flag, 1);
                }
            } else {
                const int i2 = atomicAdd(a.small_count, 1);
                if (i2 < a.small_cap)
                    a.small[i2] = {it.job, fr.node, it.start + fr.s,
                                   it.start + fr.e, fr.depth, -1};
                else
                    atomicExch(a.err_flag, 1);
            }
        }
    }
}

template <bool STAGED>
__launch_bounds__(HBLK)
__global__ void mid_subtree_kernel(ForestDev a,
                                   const int* __restrict__ sidx_level) {
    __shared__ uint4 m_codes[STAGED ? MID_N : 1];   // 32 KiB staged rows
    __shared__ int m_rows[MID_N];             // 4 KiB original global rows
    __shared__ uint8_t m_lab[STAGED ? MID_N : 1];   // 2 KiB labels
    __shared__ uint16_t m_idx[MID_N];         // 2 KiB sample ordering
    __shared__ uint16_t m_idx2[MID_N];        // 2 KiB partition scratch
    __shared__ uint32_t hist[FPAD * 256];     // 16 KiB packed histogram
    __shared__ int sh_scan[HBLK];
    __shared__ int sh_bmin[FPAD], sh_bmax[FPAD];
    __shared__ int sh_cand[FPAD], sh_ncand;
    __shared__ uint32_t sh_draws[FPAD];
    __shared__ double sh_score[FPAD];
    __shared__ int sh_bin[FPAD], sh_nL[FPAD];
    __shared__ int sh_bestf, sh_bestbin, sh_bestnL, sh_lid;
    __shared__ MidFrame stack[MID_STACK];
    __shared__ int sh_sp;
    __shared__ int sh_lo, sh_ro;
    // wave-parallel mode scratch (one slice per wave)
    __shared__ MidFrame wframes[HBLK / 64];
    __shared__ int wperm_ws[HBLK / 64][FPAD];
    __shared__ uint32_t wthr_ws[HBLK / 64][FPAD];
    __shared__ int wcand_ws[HBLK / 64][WAVE_CANDS];
    __shared__ int wcbin_ws[HBLK / 64][WAVE_CANDS];
    __shared__ int sh_count, sh_take;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int n_items = *a.mid_count;

    for (int wi = blockIdx.x; wi < n_items; wi += gridDim.x) {
        const WorkItem it = a.mid[wi];
        const int n0 = it.end - it.start;
        const long sbase = a.j_sidx_off[it.job];
        const long nbase = a.j_node_off[it.job];
        const uint32_t key = (uint32_t)a.j_key[it.job];
        const int F = a.F;

        // Stage the subtree's samples into LDS (one gather, reused by
        // every internal level).
        for (int i = tid; i < n0; i += HBLK) {
            const int row = sidx_level[sbase + it.start + i];
            m_rows[i] = row;
            if (STAGED) {
                m_codes[i] = *reinterpret_cast<const uint4*>(
                    a.codes + (size_t)row * FPAD);
                m_lab[i] = a.labels[row];
            }
            m_idx[i] = (uint16_t)i;
        }

        auto code_at = [&](int o) -> uint4 {
            if (STAGED) return m_codes[o];
            return *reinterpret_cast<const uint4*>(
                a.codes + (size_t)m_rows[o] * FPAD);
        };
        auto lab_at = [&](int o) -> int {
            return STAGED ? (int)m_lab[o] : (int)a.labels[m_rows[o]];
        };

        // Per-node block-wide builder (histogram, split, partition over
        // the staged window) shared by the DFS path and the wave mode's
        // single-frontier rounds.  dfs_mode selects the child routing:
        // the DFS stack (smaller child on top) or the wave stack.
        auto block_node = [&](int ls, int le, int depth, int node,
                              bool dfs_mode) {
            const int n = le - ls;
            const int max_features = a.j_mf[it.job];
            const int splitter_random = a.j_rand[it.job];
            // global job-relative range (the RNG identity)
            const int gs = it.start + ls;
            const int ge = it.start + le;

            // histogram from LDS-resident samples
            for (int i = tid; i < F * 64; i += HBLK)
                reinterpret_cast<uint4*>(hist)[i] = uint4{0, 0, 0, 0};
            __syncthreads();
            for (int i = ls + tid; i < le; i += HBLK) {
                const int o = m_idx[i];
                const uint4 cw = code_at(o);
                const uint32_t w[4] = {cw.x, cw.y, cw.z, cw.w};
                const uint32_t inc = 1u | ((uint32_t)lab_at(o) << 16);
                for (int f = 0; f < F; ++f) {
                    uint32_t b = (w[f >> 2] >> ((f & 3) * 8)) & 0xFFu;
                    atomicAdd(&hist[f * 256 + b], inc);
                }
            }
            __syncthreads();

            // class counts (wave reduce over feature-0 histogram)
            {
                int v = (int)(hist[tid] >> 16);
                for (int d = 32; d > 0; d >>= 1) v += __shfl_down(v, d);
                if (lane == 0) sh_scan[wave] = v;
            }
            __syncthreads();
            const int c1 = sh_scan[0] + sh_scan[1] + sh_scan[2] + sh_scan[3];
            const int c0 = n - c1;
            __syncthreads();

            if (tid == 0) {
                a.ncnt0[nbase + node] = (float)c0;
                a.ncnt1[nbase + node] = (float)c1;
            }
            if (n < 2 || c0 == 0 || c1 == 0) {
                __syncthreads();
                return;
            }

            // occupied-bin range per feature
            if (tid < F) {
                sh_bmin[tid] = 256;
                sh_bmax[tid] = -1;
            }
            __syncthreads();
            {
                const int f3 = tid >> 4;
                const int seg = (tid & 15) * 16;
                if (f3 < F) {
                    int lo = 256, hi = -1;
                    for (int b = seg; b < seg + 16; ++b)
                        if (hist[f3 * 256 + b] & 0xFFFFu) {
                            if (lo == 256) lo = b;
                            hi = b;
                        }
                    if (hi >= 0) {
                        atomicMin(&sh_bmin[f3], lo);
                        atomicMax(&sh_bmax[f3], hi);
                    }
                }
            }
            __syncthreads();

            // permutation + candidate walk (parallel draws)
            if (tid < FPAD)
                sh_draws[tid] = philox_draw(
                    TAG_FEATSEL | ((uint32_t)(depth & 0xFF) << 8),
                    (uint32_t)gs, (uint32_t)ge, (uint32_t)tid,
                    a.seed, key);
            __syncthreads();
            if (tid == 0) {
                int perm[FPAD];
                for (int f = 0; f < F; ++f) perm[f] = f;
                for (int i = 0; i < F - 1; ++i) {
                    int j = i + (int)philox_bounded(sh_draws[i],
                                                    (uint32_t)(F - i));
                    int t = perm[i]; perm[i] = perm[j]; perm[j] = t;
                }
                int nc = 0;
                for (int i = 0; i < F && nc < max_features; ++i) {
                    int f = perm[i];
                    if (sh_bmin[f] != sh_bmax[f]) sh_cand[nc++] = f;
                }
                sh_ncand = nc;
            }
            __syncthreads();

            const int ncand = sh_ncand;
            if (ncand == 0) {
                __syncthreads();
                return;
            }

            // evaluate candidates (wave per candidate; identical math and
            // tie-breaks to hist_split_kernel phase 5)
            for (int ci = wave; ci < ncand; ci += HBLK / 64) {
                const int f = sh_cand[ci];
                const int bmin = sh_bmin[f], bmax = sh_bmax[f];
                int ln[4], l1[4];
                int tn = 0, t1 = 0;
                for (int k = 0; k < 4; ++k) {
                    uint32_t v = hist[f * 256 + lane * 4 + k];
                    ln[k] = (int)(v & 0xFFFFu);
                    l1[k] = (int)(v >> 16);
                    tn += ln[k];
                    t1 += l1[k];
                }
                int scn = tn, sc1 = t1;
                for (int d = 1; d < 64; d <<= 1) {
                    int un = __shfl_up(scn, d);
                    int u1 = __shfl_up(sc1, d);
                    if (lane >= d) { scn += un; sc1 += u1; }
                }
                const int excl_n = scn - tn, excl_1 = sc1 - t1;

                double best_s = -1.0;
                int best_b = -1, best_nl = 0;

                if (splitter_random) {
                    uint32_t tag = TAG_THRESH |
                                   ((uint32_t)(depth & 0xFF) << 8);
                    uint32_t u = philox_draw(tag, (uint32_t)gs,
                                             (uint32_t)ge, (uint32_t)f,
                                             a.seed, key);
                    int b = bmin + (int)philox_bounded(
                        u, (uint32_t)(bmax - bmin));
                    if (b >= lane * 4 && b < lane * 4 + 4) {
                        int cn = excl_n, c1f = excl_1;
                        for (int k = 0; k <= b - lane * 4; ++k) {
                            cn += ln[k];
                            c1f += l1[k];
                        }
                        long nL = cn, n1L = c1f;
                        long n0L = nL - n1L, nR = n - nL;
                        long n1R = c1 - n1L, n0R = c0 - n0L;
                        best_s = (double)(n0L * n0L + n1L * n1L)
                                     / (double)nL
                               + (double)(n0R * n0R + n1R * n1R)
                                     / (double)nR;
                        best_b = b;
                        best_nl = (int)nL;
                    }
                } else {
                    int cn = excl_n, c1f = excl_1;
                    for (int k = 0; k < 4; ++k) {
                        int b = lane * 4 + k;
                        cn += ln[k];
                        c1f += l1[k];
                        if (ln[k] == 0 || b < bmin || b >= bmax) continue;
                        long nL = cn, n1L = c1f;
                        long n0L = nL - n1L, nR = n - nL;
                        long n1R = c1 - n1L, n0R = c0 - n0L;
                        double sc2 = (double)(n0L * n0L + n1L * n1L)
                                         / (double)nL
                                   + (double)(n0R * n0R + n1R * n1R)
                                         / (double)nR;
                        if (sc2 > best_s) {
                            best_s = sc2; best_b = b; best_nl = (int)nL;
                        }
                    }
                }
                for (int d = 32; d > 0; d >>= 1) {
                    double os = __shfl_down(best_s, d);
                    int ob = __shfl_down(best_b, d);
                    int onl = __shfl_down(best_nl, d);
                    if (os > best_s || (os == best_s && ob != -1 &&
                                        (best_b == -1 || ob < best_b))) {
                        best_s = os; best_b = ob; best_nl = onl;
                    }
                }
                if (lane == 0) {
                    sh_score[ci] = best_s;
                    sh_bin[ci] = best_b;
                    sh_nL[ci] = best_nl;
                }
            }
            __syncthreads();

            if (tid == 0) {
                double best_s = -1.0e300;
                int bf = -1, bb = -1, bnl = 0;
                for (int ci = 0; ci < ncand; ++ci) {
                    if (sh_bin[ci] >= 0 && sh_score[ci] > best_s) {
                        best_s = sh_score[ci];
                        bf = sh_cand[ci];
                        bb = sh_bin[ci];
                        bnl = sh_nL[ci];
                    }
                }
                sh_bestf = bf;
                sh_bestbin = bb;
                sh_bestnL = bnl;
                if (bf >= 0) {
                    int l = atomicAdd(&a.node_alloc[it.job], 2);
                    a.nfeat[nbase + l] = LEAF_SENTINEL;
                    a.nfeat[nbase + l + 1] = LEAF_SENTINEL;
                    a.nfeat[nbase + node] = bf;
                    a.nsplit[nbase + node] = bb;
                    a.nleft[nbase + node] = l;
                    sh_lid = l;
                }
            }
            __syncthreads();

            const int bf = sh_bestf;
            if (bf < 0) {
                __syncthreads();
                return;
            }
            const int bb = sh_bestbin;
            const int nL = sh_bestnL;

            // stable partition of m_idx[ls,le) via m_idx2 (LDS)
            if (tid == 0) { sh_lo = 0; sh_ro = 0; }
            __syncthreads();
            for (int base = ls; base < le; base += HBLK) {
                const int i = base + tid;
                const bool valid = i < le;
                int o = 0, flag = 0;
                if (valid) {
                    o = m_idx[i];
                    const uint4 cw = code_at(o);
                    const uint32_t wsel = (&cw.x)[bf >> 2];
                    flag = (int)(((wsel >> ((bf & 3) * 8)) & 0xFFu)
                                 <= (uint32_t)bb);
                }
                const unsigned long long lm = __ballot(valid && flag);
                const unsigned long long below = (1ULL << lane) - 1ULL;
                const int left_rank = __popcll(lm & below);
                if (lane == 0) sh_scan[wave] = __popcll(lm);
                __syncthreads();
                int wave_left_excl = 0;
                for (int ww = 0; ww < wave; ++ww)
                    wave_left_excl += sh_scan[ww];
                const int tile_left = sh_scan[0] + sh_scan[1] + sh_scan[2]
                                      + sh_scan[3];
                const int tile_n = min(HBLK, le - base);
                if (valid) {
                    const int dst = flag
                        ? ls + sh_lo + wave_left_excl + left_rank
                        : ls + nL + sh_ro + (i - base)
                          - (wave_left_excl + left_rank);
                    m_idx2[dst] = (uint16_t)o;
                }
                __syncthreads();
                if (tid == 0) {
                    sh_lo += tile_left;
                    sh_ro += tile_n - tile_left;
                }
                __syncthreads();
            }
            for (int i = ls + tid; i < le; i += HBLK) m_idx[i] = m_idx2[i];
            __syncthreads();

            if (tid == 0) {
                const int nR = n - nL;
                const MidFrame left = {(short)ls, (short)(ls + nL),
                                       depth + 1, sh_lid};
                const MidFrame right = {(short)(ls + nL), (short)le,
                                        depth + 1, sh_lid + 1};
                const bool left_small = nL <= nR;
                const MidFrame* ordered[2] = {
                    left_small ? &right : &left,   // larger pushed first
                    left_small ? &left : &right,
                };
                for (int x = 0; x < 2; ++x) {
                    const MidFrame& fr = *ordered[x];
                    const int fn = fr.e - fr.s;
                    if (fn <= SMALL_N) {
                        const int i = atomicAdd(a.small_count, 1);
                        if (i < a.small_cap)
                            a.small[i] = {it.job, fr.node,
                                          it.start + fr.s, it.start + fr.e,
                                          fr.depth, -1};
                        else
                            atomicExch(a.err_flag, 1);
                    } else if (dfs_mode) {
                        stack[++sh_sp] = fr;
                    } else if (sh_count < MID_STACK) {
                        stack[sh_count++] = fr;
                    } else {
                        const int i2 = atomicAdd(a.nxt_count, 1);
                        if (i2 < a.work_cap)
                            a.nxt[i2] = {it.job, fr.node, it.start + fr.s,
                                         it.start + fr.e, fr.depth, -1};
                        else
                            atomicExch(a.err_flag, 1);
                    }
                }
            }
            __syncthreads();
        };

        if (a.wave_mid && a.j_mf[it.job] <= WAVE_CANDS) {
            // Wave-parallel rounds: up to HBLK/64 nodes in flight, each
            // built by one wave (no block barriers inside a node); a
            // single-node frontier — the top of every staged subtree —
            // is built block-wide instead, so no wave idles there.  The
            // block-wide 16 KiB `hist` array is re-sliced as one
            // WAVE_CANDS*256-word histogram region per wave.
            if (tid == 0) {
                sh_count = 1;
                stack[0] = {0, (short)n0, it.depth, it.node};
            }
            while (true) {
                __syncthreads();
                if (tid == 0) {
                    const int take = min(HBLK / 64, sh_count);
                    for (int w = 0; w < take; ++w)
                        wframes[w] = stack[sh_count - 1 - w];
                    sh_count -= take;
                    sh_take = take;
                }
                __syncthreads();
                if (sh_take == 0) break;
                if (sh_take == 1) {
                    const MidFrame fr = wframes[0];
                    block_node(fr.s, fr.e, fr.depth, fr.node, false);
                } else if (sh_take == 2 &&
                           (wframes[0].e - wframes[0].s)
                               + (wframes[1].e - wframes[1].s) > 1024) {
                    // two big frontier nodes: sequential block-wide
                    // processing beats two waves with two idle
                    for (int x = 0; x < 2; ++x) {
                        const MidFrame fr = wframes[x];
                        block_node(fr.s, fr.e, fr.depth, fr.node, false);
                    }
                } else if (wave < sh_take) {
                    const MidFrame fr = wframes[wave];
                    mid_wave_node<STAGED>(
                        a, it, fr.s, fr.e, fr.depth, fr.node,
                        m_codes, m_lab, m_rows, m_idx, m_idx2,
                        hist + wave * (WAVE_CANDS * 256),
                        wperm_ws[wave], wthr_ws[wave],
                        wcand_ws[wave], wcbin_ws[wave],
                        stack, &sh_count);
                }
            }
            __syncthreads();
            for (int i = tid; i < n0; i += HBLK)
                a.sidx_nxt[sbase + it.start + i] = m_rows[m_idx[i]];
            __syncthreads();
            continue;
        }

        if (tid == 0) {
            sh_sp = 0;
            stack[0] = {0, (short)n0, it.depth, it.node};
        }
        __syncthreads();

        while (true) {
            const int sp = sh_sp;
            if (sp < 0) break;
            const MidFrame fr = stack[sp];
            __syncthreads();
            if (tid == 0) --sh_sp;
            block_node(fr.s, fr.e, fr.depth, fr.node, true);
        }

        // write the final arrangement back so the small kernel (launched
        // after this one) reads each farmed subtree's samples
        __syncthreads();
        for (int i = tid; i < n0; i += HBLK)
            a.sidx_nxt[sbase + it.start + i] = m_rows[m_idx[i]];
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Wave-per-subtree builder for nodes with n <= SMALL_N (=64) samples.
//
// One 64-lane wave finishes the WHOLE subtree: each lane holds one
// sample's 16 packed codes + label in registers; node membership is a
// 64-bit lane mask; class/prefix counts come from __ballot+popcount; the
// DFS stack lives in LDS.  No histograms, no sample-index traffic, no
// further level round-trips.  All randomness stays keyed on the node's
// (start, end) range (child ranges follow the stable-partition arithmetic
// [s, s+nL) / [s+nL, e)), and split scores use the identical fp64
// expression — trees are bit-identical to the histogram path / numpy
// reference.
// ---------------------------------------------------------------------------
struct SmallFrame {
    unsigned long long mask;
    int s, e, depth, node;
};

#define SMALL_STACK 68

__launch_bounds__(HBLK)
__global__ void small_subtree_kernel(ForestDev a,
                                     const int* __restrict__ sidx_level) {
    __shared__ SmallFrame stack_ws[HBLK / 64][SMALL_STACK];
    __shared__ int perm_ws[HBLK / 64][FPAD];

    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int waves_total = gridDim.x * (HBLK / 64);
    const int wave_id = blockIdx.x * (HBLK / 64) + wave;
    const int n_items = *a.small_count;
    SmallFrame* stack = stack_ws[wave];
    int* perm = perm_ws[wave];

    for (int wi = wave_id; wi < n_items; wi += waves_total) {
        const WorkItem it = a.small[min(wi, a.small_cap - 1)];
        const int n0 = it.end - it.start;
        const long sbase = a.j_sidx_off[it.job];
        const long nbase = a.j_node_off[it.job];
        const uint32_t key = (uint32_t)a.j_key[it.job];
        const int F = a.F;
        const int max_features = a.j_mf[it.job];
        const int splitter_random = a.j_rand[it.job];

        // lane -> sample (register-resident)
        uint32_t w[4] = {0, 0, 0, 0};
        int label = 0;
        if (lane < n0) {
            const int row = sidx_level[sbase + it.start + lane];
            uint4 cw = *reinterpret_cast<const uint4*>(
                a.codes + (size_t)row * FPAD);
            w[0] = cw.x; w[1] = cw.y; w[2] = cw.z; w[3] = cw.w;
            label = a.labels[row];
        }
        const unsigned long long lab_mask = __ballot(label != 0);

        int sp = 0;
        if (lane == 0)
            stack[0] = {n0 >= 64 ? ~0ULL : ((1ULL << n0) - 1ULL),
                        it.start, it.end, it.depth, it.node};

        while (sp >= 0) {
            // pop (lane-uniform: all lanes read the same LDS entry)
            const unsigned long long mask = stack[sp].mask;
            const int s = stack[sp].s;
            const int e = stack[sp].e;
            const int depth = stack[sp].depth;
            const int node = stack[sp].node;
            --sp;

            const int n = __popcll(mask);
            const int c1 = __popcll(mask & lab_mask);
            const int c0 = n - c1;
            if (lane == 0) {
                a.ncnt0[nbase + node] = (float)c0;
                a.ncnt1[nbase + node] = (float)c1;
            }
            if (n < 2 || c0 == 0 || c1 == 0) continue;

            // feature permutation: each of the first F-1 lanes computes
            // its Philox draw in parallel; lane 0 runs the swap loop with
            // the draws broadcast by shuffle
            {
                uint32_t tag = TAG_FEATSEL |
                               ((uint32_t)(depth & 0xFF) << 8);
                const uint32_t u_mine = philox_draw(
                    tag, (uint32_t)s, (uint32_t)e,
                    (uint32_t)(lane < FPAD ? lane : 0), a.seed, key);
                if (lane == 0)
                    for (int f = 0; f < F; ++f) perm[f] = f;
                for (int i = 0; i < F - 1; ++i) {
                    const uint32_t u = __shfl(u_mine, i);
                    if (lane == 0) {
                        int j = i + (int)philox_bounded(
                            u, (uint32_t)(F - i));
                        int t = perm[i]; perm[i] = perm[j]; perm[j] = t;
                    }
                }
            }
            // lane 0's LDS writes are visible to the wave in program order

            uint32_t u_thresh = 0;
            if (splitter_random) {
                uint32_t ttag = TAG_THRESH |
                                ((uint32_t)(depth & 0xFF) << 8);
                u_thresh = philox_draw(ttag, (uint32_t)s, (uint32_t)e,
                                       (uint32_t)(lane < FPAD ? lane : 0),
                                       a.seed, key);
            }

            double best_s = -1.0e300;
            int best_f = -1, best_b = -1, best_nl = 0;
            int n_eval = 0;

            for (int pi = 0; pi < F && n_eval < max_features; ++pi) {
                const int f = perm[pi];
                const int my_code = (int)((w[f >> 2] >> ((f & 3) * 8))
                                          & 0xFFu);
                const bool in = (mask >> lane) & 1ULL;
                // occupied-bin range via wave min/max (inactive lanes
                // neutralized)
                int cmin = in ? my_code : 256;
                int cmax = in ? my_code : -1;
                for (int d = 32; d > 0; d >>= 1) {
                    cmin = min(cmin, __shfl_xor(cmin, d));
                    cmax = max(cmax, __shfl_xor(cmax, d));
                }
                if (cmin == cmax) continue;   // constant: not counted
                ++n_eval;

                if (splitter_random) {
                    const int b = cmin + (int)philox_bounded(
                        __shfl(u_thresh, f), (uint32_t)(cmax - cmin));
                    const unsigned long long lm =
                        mask & __ballot(in && my_code <= b);
                    const long nL = __popcll(lm);
                    const long n1L = __popcll(lm & lab_mask);
                    const long n0L = nL - n1L, nR = n - nL;
                    const long n1R = c1 - n1L, n0R = c0 - n0L;
                    if (nL > 0 && nR > 0) {
                        double sc = (double)(n0L * n0L + n1L * n1L)
                                        / (double)nL
                                    + (double)(n0R * n0R + n1R * n1R)
                                        / (double)nR;
                        if (sc > best_s) {
                            best_s = sc; best_f = f; best_b = b;
                            best_nl = (int)nL;
                        }
                    }
                } else {
                    // parallel-rank: every in-mask lane scores its OWN
                    // code as the threshold candidate (duplicates give
                    // identical (score, bin) pairs; the argmax tie-break
                    // to the lowest bin matches the ascending sequential
                    // scan exactly).  The rank (lanes with code <= mine)
                    // comes from 8 bit-plane ballots instead of a 64-step
                    // shuffle scan: after the loop E = lanes whose code
                    // equals mine, L = lanes whose code is strictly less.
                    unsigned long long E = ~0ULL, L = 0ULL;
                    #pragma unroll
                    for (int b = 7; b >= 0; --b) {
                        const unsigned long long B =
                            __ballot((my_code >> b) & 1);
                        if ((my_code >> b) & 1) {
                            L |= E & ~B;
                            E &= B;
                        } else {
                            E &= ~B;
                        }
                    }
                    const unsigned long long le_mask = (L | E) & mask;
                    const int cnt = __popcll(le_mask);
                    const int cnt1 = __popcll(le_mask & lab_mask);
                    double sc = -1.0e300;
                    int sb = 0x7FFFFFFF, snl = 0;
                    if (in && my_code < cmax) {
                        const long nL = cnt, n1L = cnt1;
                        const long n0L = nL - n1L, nR = n - nL;
                        const long n1R = c1 - n1L, n0R = c0 - n0L;
                        sc = (double)(n0L * n0L + n1L * n1L) / (double)nL
                           + (double)(n0R * n0R + n1R * n1R) / (double)nR;
                        sb = my_code;
                        snl = (int)nL;
                    }
                    for (int d = 32; d > 0; d >>= 1) {
                        const double os = __shfl_down(sc, d);
                        const int ob = __shfl_down(sb, d);
                        const int onl = __shfl_down(snl, d);
                        if (os > sc || (os == sc && ob < sb)) {
                            sc = os; sb = ob; snl = onl;
                        }
                    }
                    sc = __shfl(sc, 0);
                    sb = __shfl(sb, 0);
                    snl = __shfl(snl, 0);
                    if (sc > best_s) {
                        best_s = sc; best_f = f; best_b = sb;
                        best_nl = snl;
                    }
                }
            }

            if (best_f < 0) continue;   // no valid split: leaf

            int lid = 0;
            if (lane == 0) {
                lid = atomicAdd(&a.node_alloc[it.job], 2);
                a.nfeat[nbase + lid] = LEAF_SENTINEL;
                a.nfeat[nbase + lid + 1] = LEAF_SENTINEL;
                a.nfeat[nbase + node] = best_f;
                a.nsplit[nbase + node] = best_b;
                a.nleft[nbase + node] = lid;
            }
            lid = __shfl(lid, 0);

            const int bc = (int)((w[best_f >> 2] >> ((best_f & 3) * 8))
                                 & 0xFFu);
            const unsigned long long lmask =
                mask & __ballot(((mask >> lane) & 1ULL) && bc <= best_b);
            const int nL = best_nl;

            if (lane == 0) {
                stack[sp + 1] = {mask & ~lmask, s + nL, e, depth + 1,
                                 lid + 1};
                stack[sp + 2] = {lmask, s, s + nL, depth + 1, lid};
            }
            sp += 2;   // left child on top: DFS order (order immaterial)
        }
    }
}

// ---------------------------------------------------------------------------
// Ensemble prediction + confusion, two stages for parallelism:
//   stage 1: one thread per (pair, tree-block of PREDICT_TREE_BLOCK) —
//            partial fp64 probability sums (sequential within the block);
//   stage 2: one thread per pair — blocks summed in ASCENDING order
//            (deterministic; forest_ref.predict_forest uses the same
//            association), then the confusion atomics
//            (k = 2*y + pred - 1, true negatives skipped; reference
//            experiment.py:476-483).
// ---------------------------------------------------------------------------
#define PREDICT_TREE_BLOCK 10

__global__ void predict_partial_kernel(
    const uint8_t* __restrict__ codes_test,   // [M, FPAD]
    const int* __restrict__ pair_row,         // [P]
    const int* __restrict__ pair_fold,        // [P]
    int n_pairs,
    const long* __restrict__ j_node_off,      // [J]
    const int* __restrict__ nfeat, const int* __restrict__ nsplit,
    const int* __restrict__ nleft,
    const float* __restrict__ ncnt0, const float* __restrict__ ncnt1,
    int trees_per_fold, int n_blocks,
    double* __restrict__ partial /* [P, n_blocks, 2] */) {
    const long total = (long)n_pairs * n_blocks;
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;
    const int p = (int)(idx % n_pairs);
    const int tb = (int)(idx / n_pairs);

    const int row = pair_row[p];
    const int fold = pair_fold[p];
    const uint8_t* cr = codes_test + (size_t)row * FPAD;

    const int t0 = tb * PREDICT_TREE_BLOCK;
    const int t1 = min(trees_per_fold, t0 + PREDICT_TREE_BLOCK);
    double acc0 = 0.0, acc1 = 0.0;
    for (int t = t0; t < t1; ++t) {
        const long nbase = j_node_off[fold * trees_per_fold + t];
        int node = 0;
        int f = nfeat[nbase];
        while (f != LEAF_SENTINEL) {
            int go_right = (int)cr[f] > nsplit[nbase + node];
            node = nleft[nbase + node] + go_right;
            f = nfeat[nbase + node];
        }
        const double c0 = (double)ncnt0[nbase + node];
        const double c1 = (double)ncnt1[nbase + node];
        const double tot = c0 + c1;
        acc0 += c0 / tot;
        acc1 += c1 / tot;
    }
    partial[((size_t)p * n_blocks + tb) * 2 + 0] = acc0;
    partial[((size_t)p * n_blocks + tb) * 2 + 1] = acc1;
}

__global__ void predict_combine_kernel(
    const uint8_t* __restrict__ y_test,       // [M]
    const int* __restrict__ proj_id,          // [M]
    const int* __restrict__ pair_row, int n_pairs,
    const double* __restrict__ partial, int n_blocks,
    uint8_t* __restrict__ pred_out,
    int* __restrict__ confusion, int n_proj) {
    const int p = blockIdx.x * blockDim.x + threadIdx.x;
    if (p >= n_pairs) return;
    double acc0 = 0.0, acc1 = 0.0;
    for (int tb = 0; tb < n_blocks; ++tb) {
        acc0 += partial[((size_t)p * n_blocks + tb) * 2 + 0];
        acc1 += partial[((size_t)p * n_blocks + tb) * 2 + 1];
    }
    const int pred = acc1 > acc0;
    pred_out[p] = (uint8_t)pred;
    const int row = pair_row[p];
    const int k = 2 * (int)y_test[row] + pred - 1;
    if (k >= 0) {
        atomicAdd(&confusion[proj_id[row] * 3 + k], 1);
        atomicAdd(&confusion[n_proj * 3 + k], 1);
    }
}

// (single-stage original, no longer launched — kept for reference builds)
__global__ void predict_confusion_kernel(
    const uint8_t* __restrict__ codes_test,   // [M, FPAD] full-dataset codes
    const uint8_t* __restrict__ y_test,       // [M]
    const int* __restrict__ proj_id,          // [M] project index per row
    const int* __restrict__ pair_row,         // [P] dataset row of pair
    const int* __restrict__ pair_fold,        // [P] fold of pair
    int n_pairs,
    const long* __restrict__ j_node_off,      // [J]
    const int* __restrict__ nfeat, const int* __restrict__ nsplit,
    const int* __restrict__ nleft,
    const float* __restrict__ ncnt0, const float* __restrict__ ncnt1,
    int trees_per_fold,
    uint8_t* __restrict__ pred_out,           // [P]
    int* __restrict__ confusion,              // [n_proj+1, 3]
    int n_proj) {
    int p = blockIdx.x * blockDim.x + threadIdx.x;
    if (p >= n_pairs) return;

    const int row = pair_row[p];
    const int fold = pair_fold[p];
    const uint8_t* cr = codes_test + (size_t)row * FPAD;

    double acc0 = 0.0, acc1 = 0.0;
    for (int t = 0; t < trees_per_fold; ++t) {
        const long nbase = j_node_off[fold * trees_per_fold + t];
        int node = 0;
        int f = nfeat[nbase];
        while (f != LEAF_SENTINEL) {
            int go_right = (int)cr[f] > nsplit[nbase + node];
            node = nleft[nbase + node] + go_right;
            f = nfeat[nbase + node];
        }
        double c0 = (double)ncnt0[nbase + node];
        double c1 = (double)ncnt1[nbase + node];
        double tot = c0 + c1;
        acc0 += c0 / tot;
        acc1 += c1 / tot;
    }
    const int pred = acc1 > acc0;
    pred_out[p] = (uint8_t)pred;

    const int k = 2 * (int)y_test[row] + pred - 1;
    if (k >= 0) {
        atomicAdd(&confusion[proj_id[row] * 3 + k], 1);
        atomicAdd(&confusion[n_proj * 3 + k], 1);
    }
}
