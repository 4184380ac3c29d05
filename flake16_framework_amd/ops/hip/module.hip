// Torch-extension host layer for the flake16 MI355X kernels.
//
// Single translation unit: includes the kernel files and exposes
// tensor-based entry points.  Compiled by hipcc for gfx950 with
// -ffp-contract=off (bit-parity of fp64 split scores / fp32 SMOTE
// interpolation with the numpy reference).

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDAGuard.h>

#include "forest.hip"
#include "knn_balance.hip"
#include "scaler_pca.hip"
#include "treeshap.hip"

#include <vector>
#include <algorithm>
#include <cstdlib>

#define CHECK_HIP(expr)                                                     \
    do {                                                                    \
        hipError_t _e = (expr);                                             \
        TORCH_CHECK(_e == hipSuccess, "HIP error: ",                        \
                    hipGetErrorString(_e));                                 \
    } while (0)

static hipStream_t current_stream() {
    return (hipStream_t)at::cuda::getCurrentCUDAStream().stream();
}

// ---------------------------------------------------------------------------
// forest_fit
// ---------------------------------------------------------------------------
// Core fit over a batch of jobs with PER-JOB model spec (max_features,
// splitter, bootstrap) — one call can build a whole balance group's
// DT + RF + ET forests in a single level pipeline.
static std::vector<at::Tensor> forest_fit_impl(
    at::Tensor codes, at::Tensor labels, at::Tensor j_row_off,
    at::Tensor j_n, at::Tensor j_key, at::Tensor j_mf, at::Tensor j_rand,
    at::Tensor j_boot, int64_t F, int64_t seed) {
    TORCH_CHECK(codes.is_cuda() && codes.dtype() == at::kByte &&
                codes.size(1) == FPAD && codes.is_contiguous());
    TORCH_CHECK(labels.is_cuda() && labels.dtype() == at::kByte);
    const at::cuda::OptionalCUDAGuard guard(codes.device());

    const int J = j_n.size(0);
    auto opts_i32 = codes.options().dtype(at::kInt);
    auto opts_i64 = codes.options().dtype(at::kLong);
    auto opts_f32 = codes.options().dtype(at::kFloat);

    // Job arrays may arrive CPU-resident (preferred: the host needs
    // j_n / j_rand values for sizing and kernel selection, and device
    // tensors would force blocking D2H copies here)
    auto j_n_cpu = j_n.is_cuda() ? j_n.to(at::kCPU) : j_n;
    auto j_n_dev = j_n.is_cuda() ? j_n : j_n.to(codes.device());
    auto j_rand_cpu = j_rand.is_cuda() ? j_rand.to(at::kCPU) : j_rand;
    auto j_rand_dev = j_rand.is_cuda() ? j_rand : j_rand.to(codes.device());
    auto j_mf_dev = j_mf.is_cuda() ? j_mf : j_mf.to(codes.device());
    auto j_boot_dev = j_boot.is_cuda() ? j_boot : j_boot.to(codes.device());
    const int* jn = j_n_cpu.data_ptr<int>();
    const uint8_t* jr = j_rand_cpu.data_ptr<uint8_t>();
    bool any_rand = false, any_best = false;
    for (int j = 0; j < J; ++j)
        (jr[j] ? any_rand : any_best) = true;
    std::vector<long> sidx_off(J), node_off(J);
    long S = 0, Ntot = 0;
    bool has_wide = false;   // any node >= 2^16 samples needs the WIDE path
    for (int j = 0; j < J; ++j) {
        has_wide |= jn[j] >= 65536;
        sidx_off[j] = S;
        node_off[j] = Ntot;
        S += jn[j];
        Ntot += 2L * jn[j] + 1;
    }
    TORCH_CHECK(S > 0, "empty forest_fit batch");

    auto j_sidx_off = at::from_blob(sidx_off.data(), {J}, at::kLong)
                          .to(codes.device());
    auto j_node_off = at::from_blob(node_off.data(), {J}, at::kLong)
                          .to(codes.device());

    // Workspace sizes are rounded to a coarse granularity so cells of
    // similar size hit the same caching-allocator buckets: per-cell
    // workspaces here reach GBs at large N, and exact (per-balance-group)
    // sizes would force hipMalloc/hipFree churn — whose device syncs
    // serialize the concurrent cell streams (measured: the N=40k sweep
    // ran at ~79% single-stream occupancy before this).
    const long GRAN = 1L << 23;
    const long S_alloc = (S + GRAN - 1) / GRAN * GRAN;
    const long Ntot_alloc = (Ntot + GRAN - 1) / GRAN * GRAN;

    // No global sentinel fill (it was 1 GB/group at large N): every
    // allocated node is written LEAF_SENTINEL at allocation time (root in
    // forest_init_kernel, children at their parent's split), and the
    // other node fields are written for every reached node before any
    // read — so plain at::empty suffices for all five arrays.
    auto nfeat = at::empty({Ntot_alloc}, opts_i32);
    auto nsplit = at::empty({Ntot_alloc}, opts_i32);
    auto nleft = at::empty({Ntot_alloc}, opts_i32);
    auto ncnt0 = at::empty({Ntot_alloc}, opts_f32);
    auto ncnt1 = at::empty({Ntot_alloc}, opts_f32);
    auto node_alloc = at::zeros({J}, opts_i32);

    auto sidx_a = at::empty({S_alloc}, opts_i32);
    auto sidx_b = at::empty({S_alloc}, opts_i32);

    // Histogram-subtraction pools (see forest.hip): sized for the worst
    // per-level allocation, 2 slots per splitting node >= HIST_SAVE_MIN.
    // Larger batches raise the subtraction threshold: with the pool
    // clamped at 131072 slots, small thresholds exhaust it mid-band and
    // the 16 KiB/slot store+load traffic overtakes the saved
    // re-accumulation (measured at N=40k fused: 23.0 -> 15.3 s with
    // 8192).
    int HIST_SAVE_MIN = S < (1L << 24) ? 2048 : 8192;
    if (const char* e = getenv("FLAKE16_HIST_SAVE_MIN"))
        HIST_SAVE_MIN = atoi(e);
    // cap bounds pool memory at ~2 GB per parity (16 KiB per slot);
    // exhaustion is correct but silently degrades to full accumulation
    const long pool_cap =
        std::min<long>(2 * (S_alloc / HIST_SAVE_MIN) + 8, 131072);

    // Level work queue: only nodes > MID_N samples (disjoint: <= S/MID_N)
    // plus <= MID_N children carrying a subtraction slot (<= 2 per pool
    // slot) and the J roots; overflow trips err_flag loudly.
    const long work_cap = S_alloc / MID_N + 2 * pool_cap + J + 64;
    auto work_a = at::empty({work_cap * (long)sizeof(WorkItem)},
                            codes.options().dtype(at::kByte));
    auto work_b = at::empty({work_cap * (long)sizeof(WorkItem)},
                            codes.options().dtype(at::kByte));
    // per-level counter state, one 16 B memset per level:
    // state[parity*4 + 0] = work count, +1 = pool count, +2 = small,
    // +3 = mid
    auto state = at::zeros({8}, opts_i32);
    auto err = at::zeros({1}, opts_i32);

    // Small queue: disjoint <= SMALL_N subtrees; S/4 covers any
    // non-adversarial level (err_flag guards the theoretical S bound).
    const long small_cap = S_alloc / 4 + 1024;
    auto small_q = at::empty({small_cap * (long)sizeof(WorkItem)},
                             codes.options().dtype(at::kByte));

    // mid-subtree queue: disjoint ranges of > SMALL_N samples each
    const long mid_cap = S_alloc / (SMALL_N + 1) + 64;
    auto mid_q = at::empty({mid_cap * (long)sizeof(WorkItem)},
                           codes.options().dtype(at::kByte));
    auto hist_pool0 = at::empty({pool_cap * FPAD * 256},
                                codes.options().dtype(at::kInt));
    auto hist_pool1 = at::empty({pool_cap * FPAD * 256},
                                codes.options().dtype(at::kInt));

    hipStream_t stream = current_stream();

    forest_init_kernel<<<J, HBLK, 0, stream>>>(
        j_row_off.data_ptr<int>(), j_n_dev.data_ptr<int>(),
        j_sidx_off.data_ptr<long>(), j_key.data_ptr<int>(),
        j_node_off.data_ptr<long>(), nfeat.data_ptr<int>(),
        node_alloc.data_ptr<int>(), sidx_a.data_ptr<int>(),
        (WorkItem*)work_a.data_ptr(), j_boot_dev.data_ptr<uint8_t>(),
        (uint32_t)seed);
    state.narrow(0, 0, 1).fill_((int)J);

    const int PINSZ = 64;
    auto pinned = at::empty({PINSZ}, at::TensorOptions()
                                         .dtype(at::kInt)
                                         .pinned_memory(true));
    int* pinned_p = pinned.data_ptr<int>();

    ForestDev a{};
    a.codes = codes.data_ptr<uint8_t>();
    a.labels = labels.data_ptr<uint8_t>();
    a.j_row_off = j_row_off.data_ptr<int>();
    a.j_n = j_n_dev.data_ptr<int>();
    a.j_sidx_off = j_sidx_off.data_ptr<long>();
    a.j_node_off = j_node_off.data_ptr<long>();
    a.j_key = j_key.data_ptr<int>();
    a.node_alloc = node_alloc.data_ptr<int>();
    a.nfeat = nfeat.data_ptr<int>();
    a.nsplit = nsplit.data_ptr<int>();
    a.nleft = nleft.data_ptr<int>();
    a.ncnt0 = ncnt0.data_ptr<float>();
    a.ncnt1 = ncnt1.data_ptr<float>();
    a.err_flag = err.data_ptr<int>();
    a.F = (int)F;
    a.j_mf = j_mf_dev.data_ptr<int>();
    a.j_rand = j_rand_dev.data_ptr<uint8_t>();
    a.seed = (uint32_t)seed;
    a.work_cap = (int)work_cap;
    a.hist_pool0 = (uint32_t*)hist_pool0.data_ptr<int>();
    a.hist_pool1 = (uint32_t*)hist_pool1.data_ptr<int>();
    a.pool_cap = (int)pool_cap;
    a.hist_save_min = HIST_SAVE_MIN;
    a.small = (WorkItem*)small_q.data_ptr();
    a.small_cap = (int)small_cap;
    a.mid = (WorkItem*)mid_q.data_ptr();
    a.mid_cap = (int)mid_cap;
    int* st = state.data_ptr<int>();
    a.pool_count = st + 1;   // kernel indexes [wp * 4]
    // wave-parallel mid-subtree mode (per-item: jobs with
    // max_features <= WAVE_CANDS); the env var forces the block-serial
    // DFS for same-box A/B runs
    a.wave_mid = getenv("FLAKE16_NO_WAVE_MID") ? 0 : 1;

    int GRID = 4096;
    if (const char* e = getenv("FLAKE16_FIT_GRID")) GRID = atoi(e);
    // STAGED keeps code rows in LDS (2 blocks/CU); the unstaged variant
    // reads them through L1/L2 and doubles the resident blocks — A/B via
    // FLAKE16_MID_UNSTAGED=1.
    const bool mid_staged = !getenv("FLAKE16_MID_UNSTAGED");

    // Adaptive level grids: the deep tail of the band (skewed-split
    // chains) has a handful of items per level, where scheduling the
    // full 4096/2048 empty-block grids is the dominant cost; once the
    // synced queue count is tiny the next chunk launches small grids
    // (grid-stride loops keep any count correct regardless).
    int lv_grid = GRID, sub_grid = 2048;

    // One level's dispatches: clear next-parity counters, split kernel(s),
    // mid- and small-subtree drains, next-level count -> pinned slot.
    auto level_ops = [&](int cur_par, int pinned_slot) {
        const int nx = cur_par ^ 1;
        // one 16 B memset clears next-level work/pool/small/mid counts
        CHECK_HIP(hipMemsetAsync(st + nx * 4, 0, 16, stream));
        a.sidx_cur = (cur_par == 0 ? sidx_a : sidx_b).data_ptr<int>();
        a.sidx_nxt = (cur_par == 0 ? sidx_b : sidx_a).data_ptr<int>();
        a.cur = (const WorkItem*)(cur_par == 0 ? work_a : work_b).data_ptr();
        a.nxt = (WorkItem*)(cur_par == 0 ? work_b : work_a).data_ptr();
        a.cur_count = st + cur_par * 4;
        a.nxt_count = st + nx * 4;
        a.small_count = st + nx * 4 + 2;
        a.mid_count = st + nx * 4 + 3;
        if (any_best) {
            if (!has_wide && getenv("FLAKE16_RF_CANDONLY")) {
                // ablation variant: candidate-only RF histograms measured
                // ~3% SLOWER than full histograms + subtraction pools
                rf_cand_split_kernel<<<lv_grid, HBLK, 0, stream>>>(a);
            } else {
                hist_split_kernel<false><<<lv_grid, HBLK, 0, stream>>>(a);
                if (has_wide)
                    hist_split_kernel<true><<<lv_grid, HBLK, 0, stream>>>(a);
            }
        }
        if (any_rand)
            et_split_kernel<<<lv_grid, HBLK, 0, stream>>>(a);
        if (mid_staged)
            mid_subtree_kernel<true><<<sub_grid, HBLK, 0, stream>>>(
                a, a.sidx_nxt);
        else
            mid_subtree_kernel<false><<<sub_grid, HBLK, 0, stream>>>(
                a, a.sidx_nxt);
        small_subtree_kernel<<<sub_grid, HBLK, 0, stream>>>(a, a.sidx_nxt);
        CHECK_HIP(hipMemcpyAsync(pinned_p + pinned_slot, st + nx * 4, 4,
                                 hipMemcpyDeviceToHost, stream));
    };

    if (getenv("FLAKE16_HIPGRAPH") && stream != nullptr) {
        // MEASURED AND REJECTED (kept for A/B): the dispatch sequence is
        // parity-periodic with every pointer fixed, so one even+odd level
        // pair is captured into a hipGraph and replayed — but the
        // per-fit capture+instantiate costs more than the ~12 dispatches
        // per pair it saves (same-box sweep 123.9 vs 127.1 configs/s),
        // so the plain loop below is the default.  Requires a non-default
        // stream (capture on the legacy stream is not permitted).
        // Replaying past exhaustion is harmless — all kernels early-exit
        // on zero counts.
        hipGraph_t graph = nullptr;
        hipGraphExec_t gexec = nullptr;
        CHECK_HIP(hipStreamBeginCapture(stream,
                                        hipStreamCaptureModeThreadLocal));
        level_ops(0, 0);
        level_ops(1, 1);
        CHECK_HIP(hipStreamEndCapture(stream, &graph));
        CHECK_HIP(hipGraphInstantiate(&gexec, graph, nullptr, nullptr, 0));

        const int PAIRS_PER_SYNC = 4;
        long pairs = 0;
        bool done = false;
        while (!done) {
            for (int p = 0; p < PAIRS_PER_SYNC; ++p)
                CHECK_HIP(hipGraphLaunch(gexec, stream));
            CHECK_HIP(hipStreamSynchronize(stream));
            pairs += PAIRS_PER_SYNC;
            done = pinned_p[0] == 0 || pinned_p[1] == 0;
            TORCH_CHECK(pairs < 4096, "forest_fit: depth limit exceeded");
        }
        CHECK_HIP(hipGraphExecDestroy(gexec));
        CHECK_HIP(hipGraphDestroy(graph));
    } else {
        const int CHUNK = 8;
        int cur = 0;
        long lev = 0;
        bool done = false;
        while (!done) {
            for (int c = 0; c < CHUNK; ++c, ++lev) {
                level_ops(cur, c % PINSZ);
                cur ^= 1;
            }
            CHECK_HIP(hipStreamSynchronize(stream));
            for (int c = 0; c < CHUNK; ++c)
                if (pinned_p[c] == 0) { done = true; break; }
            // shrink the next chunk's grids when the band has thinned
            // (items can at most double per level: 16 items now bound
            // the whole next chunk well under a 512-block grid)
            const int last_cnt = pinned_p[CHUNK - 1];
            lv_grid = last_cnt <= 16 ? 512 : GRID;
            sub_grid = last_cnt <= 16 ? 512 : 2048;
            TORCH_CHECK(lev < 8192, "forest_fit: depth limit exceeded");
        }
    }

    TORCH_CHECK(err.to(at::kCPU).item<int>() == 0,
                "forest_fit: work-queue capacity exceeded");

    return {nfeat, nsplit, nleft, ncnt0, ncnt1, j_node_off, node_alloc};
}

// Uniform-spec compatibility surface (single-model batches).
std::vector<at::Tensor> forest_fit(
    at::Tensor codes, at::Tensor labels, at::Tensor j_row_off,
    at::Tensor j_n, at::Tensor j_key, int64_t F, int64_t max_features,
    bool bootstrap, bool splitter_random, int64_t seed) {
    const int J = j_n.size(0);
    auto mf = at::full({J}, (int)max_features, at::kInt);
    auto rnd = at::full({J}, splitter_random ? 1 : 0, at::kByte);
    auto boot = at::full({J}, bootstrap ? 1 : 0, at::kByte);
    return forest_fit_impl(codes, labels, j_row_off, j_n, j_key, mf, rnd,
                           boot, F, seed);
}

// Mixed-spec batches (fused balance-group fits).
std::vector<at::Tensor> forest_fit_multi(
    at::Tensor codes, at::Tensor labels, at::Tensor j_row_off,
    at::Tensor j_n, at::Tensor j_key, at::Tensor j_mf, at::Tensor j_rand,
    at::Tensor j_boot, int64_t F, int64_t seed) {
    return forest_fit_impl(codes, labels, j_row_off, j_n, j_key, j_mf,
                           j_rand, j_boot, F, seed);
}

// ---------------------------------------------------------------------------
// forest_predict_confusion
// ---------------------------------------------------------------------------
std::vector<at::Tensor> forest_predict_confusion(
    at::Tensor codes_test, at::Tensor y_test, at::Tensor proj_id,
    at::Tensor pair_row, at::Tensor pair_fold, at::Tensor j_node_off,
    at::Tensor nfeat, at::Tensor nsplit, at::Tensor nleft,
    at::Tensor ncnt0, at::Tensor ncnt1, int64_t trees_per_fold,
    int64_t n_proj) {
    const at::cuda::OptionalCUDAGuard guard(codes_test.device());
    const int P = pair_row.size(0);
    const int n_blocks =
        ((int)trees_per_fold + PREDICT_TREE_BLOCK - 1) / PREDICT_TREE_BLOCK;
    auto pred = at::zeros({P}, codes_test.options().dtype(at::kByte));
    auto confusion = at::zeros({n_proj + 1, 3},
                               codes_test.options().dtype(at::kInt));
    auto partial = at::empty({(long)P * n_blocks * 2},
                             codes_test.options().dtype(at::kDouble));
    hipStream_t s = current_stream();
    const long total = (long)P * n_blocks;
    predict_partial_kernel<<<(int)((total + 255) / 256), 256, 0, s>>>(
        codes_test.data_ptr<uint8_t>(), pair_row.data_ptr<int>(),
        pair_fold.data_ptr<int>(), P, j_node_off.data_ptr<long>(),
        nfeat.data_ptr<int>(), nsplit.data_ptr<int>(),
        nleft.data_ptr<int>(), ncnt0.data_ptr<float>(),
        ncnt1.data_ptr<float>(), (int)trees_per_fold, n_blocks,
        partial.data_ptr<double>());
    predict_combine_kernel<<<(P + 255) / 256, 256, 0, s>>>(
        y_test.data_ptr<uint8_t>(), proj_id.data_ptr<int>(),
        pair_row.data_ptr<int>(), P, partial.data_ptr<double>(), n_blocks,
        pred.data_ptr<uint8_t>(), confusion.data_ptr<int>(), (int)n_proj);
    return {pred, confusion};
}

// ---------------------------------------------------------------------------
// knn / balancing / binning
// ---------------------------------------------------------------------------
// Shared implementation: MFMA two-phase kernel by default (matrix cores
// compute the fp32 distance filter; survivors are re-checked in exact
// fp64 — output bits identical to the scalar kernels), scalar tile
// kernels under FLAKE16_NO_MFMA_KNN=1 for A/B runs.
static at::Tensor knn_run(at::Tensor X, at::Tensor seg_off, const int* so,
                          int n_seg, int64_t k, bool skip_identity) {
    const int R = X.size(0);
    hipStream_t stream = current_stream();
    auto out = at::empty({R, k}, X.options().dtype(at::kInt));

    if (getenv("FLAKE16_NO_MFMA_KNN")) {
        std::vector<int> blk(n_seg + 1);
        blk[0] = 0;
        for (int s = 0; s < n_seg; ++s) {
            int n = so[s + 1] - so[s];
            blk[s + 1] = blk[s] + (n + KNN_BLK - 1) / KNN_BLK;
        }
        auto seg_blk = at::from_blob(blk.data(), {n_seg + 1}, at::kInt)
                           .to(X.device());
        if (blk[n_seg] > 0)
            knn_segmented_kernel<<<blk[n_seg], KNN_BLK, 0, stream>>>(
                X.data_ptr<float>(), seg_off.data_ptr<int>(),
                seg_blk.data_ptr<int>(), n_seg, (int)k,
                skip_identity ? 1 : 0, out.data_ptr<int>());
        return out;
    }

    std::vector<int> blk(n_seg + 1);
    blk[0] = 0;
    for (int s = 0; s < n_seg; ++s) {
        int n = so[s + 1] - so[s];
        blk[s + 1] = blk[s] + (n + 63) / 64;
    }
    auto seg_blk = at::from_blob(blk.data(), {n_seg + 1}, at::kInt)
                       .to(X.device());
    auto norms = at::empty({R}, X.options());
    auto fb_list = at::empty({R}, X.options().dtype(at::kInt));
    auto fb_count = at::zeros({1 + n_seg}, X.options().dtype(at::kInt));
    auto fb_off = at::empty({n_seg + 1}, X.options().dtype(at::kInt));
    auto fb_cursor = at::empty({n_seg}, X.options().dtype(at::kInt));
    const long fb_cap = (long)R + (long)KNN_BLK * n_seg;
    auto fb_sorted = at::empty({fb_cap}, X.options().dtype(at::kInt));

    knn_norms_kernel<<<(R + 255) / 256, 256, 0, stream>>>(
        X.data_ptr<float>(), R, norms.data_ptr<float>());
    if (blk[n_seg] > 0) {
        knn_mfma_kernel<<<blk[n_seg], KNN_BLK, 0, stream>>>(
            X.data_ptr<float>(), seg_off.data_ptr<int>(),
            seg_blk.data_ptr<int>(), n_seg, (int)k, skip_identity ? 1 : 0,
            norms.data_ptr<float>(), out.data_ptr<int>(),
            fb_list.data_ptr<int>(), fb_count.data_ptr<int>());
        // regroup flagged queries by segment (256-padded) and re-scan
        // them exactly with LDS candidate tiling; all plumbing is
        // device-bounded — when nothing was flagged these kernels are
        // a few empty dispatches
        knn_fb_scan_kernel<<<1, 1, 0, stream>>>(
            fb_count.data_ptr<int>(), n_seg, fb_off.data_ptr<int>(),
            fb_cursor.data_ptr<int>());
        knn_fb_fill_kernel<<<(int)((fb_cap + 255) / 256), 256, 0, stream>>>(
            fb_off.data_ptr<int>(), n_seg, fb_sorted.data_ptr<int>());
        knn_fb_scatter_kernel<<<(R + 255) / 256, 256, 0, stream>>>(
            fb_list.data_ptr<int>(), fb_count.data_ptr<int>(),
            seg_off.data_ptr<int>(), n_seg, fb_off.data_ptr<int>(),
            fb_cursor.data_ptr<int>(), fb_sorted.data_ptr<int>());
        knn_fallback_kernel<<<(int)((fb_cap + KNN_BLK - 1) / KNN_BLK),
                              KNN_BLK, 0, stream>>>(
            X.data_ptr<float>(), seg_off.data_ptr<int>(), n_seg, (int)k,
            skip_identity ? 1 : 0, fb_sorted.data_ptr<int>(),
            fb_off.data_ptr<int>(), out.data_ptr<int>());
    }
    if (getenv("FLAKE16_KNN_DEBUG")) {
        const int fb = fb_count.narrow(0, 0, 1).to(at::kCPU).item<int>();
        fprintf(stderr, "[knn] R=%d n_seg=%d k=%ld fallback=%d\n",
                R, n_seg, (long)k, fb);
    }
    return out;
}

at::Tensor knn(at::Tensor X, int64_t k, bool skip_identity) {
    const at::cuda::OptionalCUDAGuard guard(X.device());
    TORCH_CHECK(X.is_cuda() && X.dtype() == at::kFloat &&
                X.size(1) == FPAD && X.is_contiguous());
    TORCH_CHECK(k >= 1 && k <= KMAX);
    const int n = X.size(0);
    int so[2] = {0, n};
    auto seg_off = at::from_blob(so, {2}, at::kInt).to(X.device());
    return knn_run(X, seg_off, so, 1, k, skip_identity);
}

at::Tensor knn_segmented(at::Tensor X, at::Tensor seg_off, int64_t k,
                         bool skip_identity) {
    const at::cuda::OptionalCUDAGuard guard(X.device());
    TORCH_CHECK(X.is_cuda() && X.dtype() == at::kFloat &&
                X.size(1) == FPAD && X.is_contiguous());
    TORCH_CHECK(k >= 1 && k <= KMAX);
    const int n_seg = seg_off.size(0) - 1;
    auto seg_off_cpu = seg_off.to(at::kCPU);
    return knn_run(X, seg_off, seg_off_cpu.data_ptr<int>(), n_seg, k,
                   skip_identity);
}

at::Tensor smote_interpolate(at::Tensor X, at::Tensor min_rows,
                             at::Tensor nn, int64_t n_new, int64_t k0,
                             int64_t k1) {
    const at::cuda::OptionalCUDAGuard guard(X.device());
    const int n_min = min_rows.size(0);
    const int k = nn.size(1);
    auto out = at::empty({n_new, FPAD}, X.options());
    const int grid = (n_new + 255) / 256;
    smote_kernel<<<grid, 256, 0, current_stream()>>>(
        X.data_ptr<float>(), min_rows.data_ptr<int>(), nn.data_ptr<int>(),
        n_min, k, (int)n_new, (uint32_t)k0, (uint32_t)k1,
        out.data_ptr<float>());
    return out;
}

at::Tensor enn_keep(at::Tensor y, at::Tensor nn, int64_t n_neighbors,
                    int64_t maj_label, bool clean_all) {
    const at::cuda::OptionalCUDAGuard guard(y.device());
    const int n = y.size(0);
    auto keep = at::empty({n}, y.options());
    const int grid = (n + 255) / 256;
    enn_keep_kernel<<<grid, 256, 0, current_stream()>>>(
        y.data_ptr<uint8_t>(), nn.data_ptr<int>(), n, (int)nn.size(1),
        (int)n_neighbors, (int)maj_label, clean_all ? 1 : 0,
        keep.data_ptr<uint8_t>());
    return keep;
}

at::Tensor tomek_keep(at::Tensor y, at::Tensor nn1, int64_t maj_label,
                      bool remove_all) {
    const at::cuda::OptionalCUDAGuard guard(y.device());
    const int n = y.size(0);
    auto keep = at::empty({n}, y.options());
    const int grid = (n + 255) / 256;
    tomek_keep_kernel<<<grid, 256, 0, current_stream()>>>(
        y.data_ptr<uint8_t>(), nn1.data_ptr<int>(), n, (int)maj_label,
        remove_all ? 1 : 0, keep.data_ptr<uint8_t>());
    return keep;
}

at::Tensor bin_codes_dev(at::Tensor X, at::Tensor cuts, at::Tensor cut_off,
                         int64_t F) {
    const at::cuda::OptionalCUDAGuard guard(X.device());
    const int n = X.size(0);
    auto codes = at::zeros({n, FPAD}, X.options().dtype(at::kByte));
    const int grid = (n + 255) / 256;
    bin_codes_kernel<<<grid, 256, 0, current_stream()>>>(
        X.data_ptr<float>(), cuts.data_ptr<float>(), cut_off.data_ptr<int>(),
        n, (int)F, codes.data_ptr<uint8_t>());
    return codes;
}

// ---------------------------------------------------------------------------
// scaler / pca  (fp64 in, fp64 out)
// ---------------------------------------------------------------------------
at::Tensor scaler_fit_transform_dev(at::Tensor X) {
    const at::cuda::OptionalCUDAGuard guard(X.device());
    TORCH_CHECK(X.dtype() == at::kDouble && X.size(1) == FPAD);
    const int n = X.size(0);
    auto mean = at::empty({FPAD}, X.options());
    auto scale = at::empty({FPAD}, X.options());
    auto out = at::empty_like(X);
    hipStream_t s = current_stream();
    col_mean_kernel<<<FPAD, RBLK, 0, s>>>(X.data_ptr<double>(), n,
                                          mean.data_ptr<double>());
    col_scale_kernel<<<FPAD, RBLK, 0, s>>>(X.data_ptr<double>(), n,
                                           mean.data_ptr<double>(),
                                           scale.data_ptr<double>());
    scale_transform_kernel<<<(n + 255) / 256, 256, 0, s>>>(
        X.data_ptr<double>(), n, mean.data_ptr<double>(),
        scale.data_ptr<double>(), out.data_ptr<double>());
    return out;
}

at::Tensor pca_fit_transform_dev(at::Tensor X, int64_t F) {
    const at::cuda::OptionalCUDAGuard guard(X.device());
    TORCH_CHECK(X.dtype() == at::kDouble && X.size(1) == FPAD);
    const int n = X.size(0);
    auto mean = at::empty({FPAD}, X.options());
    auto C = at::zeros({FPAD, FPAD}, X.options());
    auto V = at::zeros({FPAD, FPAD}, X.options());
    auto evals = at::zeros({FPAD}, X.options());
    auto T = at::empty_like(X);
    hipStream_t s = current_stream();
    col_mean_kernel<<<FPAD, RBLK, 0, s>>>(X.data_ptr<double>(), n,
                                          mean.data_ptr<double>());
    gram_kernel<<<FPAD * FPAD, RBLK, 0, s>>>(X.data_ptr<double>(), n,
                                             mean.data_ptr<double>(),
                                             C.data_ptr<double>());
    jacobi_eigen_kernel<<<1, 1, 0, s>>>(C.data_ptr<double>(),
                                        V.data_ptr<double>(),
                                        evals.data_ptr<double>(), (int)F);
    pca_project_kernel<<<(n + 255) / 256, 256, 0, s>>>(
        X.data_ptr<double>(), n, mean.data_ptr<double>(),
        V.data_ptr<double>(), (int)F, T.data_ptr<double>());
    pca_signflip_kernel<<<FPAD, RBLK, 0, s>>>(T.data_ptr<double>(), n,
                                              (int)F);
    return T;
}

// ---------------------------------------------------------------------------
// treeshap
// ---------------------------------------------------------------------------
at::Tensor treeshap(at::Tensor codes, at::Tensor j_node_off,
                    at::Tensor nfeat, at::Tensor nsplit, at::Tensor nleft,
                    at::Tensor ncnt0, at::Tensor ncnt1) {
    const at::cuda::OptionalCUDAGuard guard(codes.device());
    const int n_samples = codes.size(0);
    const int n_trees = j_node_off.size(0);
    hipStream_t s = current_stream();

    auto depth = at::zeros({n_trees}, codes.options().dtype(at::kInt));
    tree_depth_kernel<<<(n_trees + 63) / 64, 64, 0, s>>>(
        j_node_off.data_ptr<long>(), nfeat.data_ptr<int>(),
        nleft.data_ptr<int>(), n_trees, depth.data_ptr<int>());
    const int d_max = depth.max().to(at::kCPU).item<int>() + 1;

    int GRID = 2048;   // same-box sweep: 512/1024/2048/4096 -> 21.7/15.1/11.3/12.2 s
    if (const char* e = getenv("FLAKE16_SHAP_GRID")) GRID = atoi(e);
    const long n_threads = (long)GRID * SHAP_BLK;
    const long tri = (long)(d_max + 1) * (d_max + 2) / 2;
    auto path_ws = at::empty({n_threads * tri * (long)sizeof(PathElem)},
                             codes.options().dtype(at::kByte));
    auto frame_ws = at::empty(
        {n_threads * (d_max + 2) * (long)sizeof(ShapFrame)},
        codes.options().dtype(at::kByte));
    auto phi = at::zeros({n_samples, 16},
                         codes.options().dtype(at::kDouble));

    treeshap_kernel<<<GRID, SHAP_BLK, 0, s>>>(
        codes.data_ptr<uint8_t>(), n_samples, j_node_off.data_ptr<long>(),
        nfeat.data_ptr<int>(), nsplit.data_ptr<int>(),
        nleft.data_ptr<int>(), ncnt0.data_ptr<float>(),
        ncnt1.data_ptr<float>(), n_trees, d_max,
        (PathElem*)path_ws.data_ptr(), (ShapFrame*)frame_ws.data_ptr(),
        phi.data_ptr<double>());
    return phi;
}

at::Tensor treeshap_paths(at::Tensor codes, at::Tensor leaf_tree,
                          at::Tensor leaf_off, at::Tensor path_nodes,
                          at::Tensor j_node_off, at::Tensor nfeat,
                          at::Tensor nsplit, at::Tensor nleft,
                          at::Tensor ncnt0, at::Tensor ncnt1) {
    const at::cuda::OptionalCUDAGuard guard(codes.device());
    const int n_samples = codes.size(0);
    const int n_leaves = leaf_tree.size(0);
    auto phi = at::zeros({n_samples, 16},
                         codes.options().dtype(at::kDouble));
    treeshap_paths_kernel<<<2048, SHAP_BLK, 0, current_stream()>>>(
        codes.data_ptr<uint8_t>(), n_samples, leaf_tree.data_ptr<int>(),
        leaf_off.data_ptr<int>(), path_nodes.data_ptr<int>(),
        j_node_off.data_ptr<long>(), nfeat.data_ptr<int>(),
        nsplit.data_ptr<int>(), nleft.data_ptr<int>(),
        ncnt0.data_ptr<float>(), ncnt1.data_ptr<float>(), n_leaves,
        phi.data_ptr<double>());
    return phi;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("treeshap", &treeshap, py::call_guard<py::gil_scoped_release>(),
          "Path-dependent TreeSHAP (class-0), summed over trees");
    m.def("treeshap_paths", &treeshap_paths,
          py::call_guard<py::gil_scoped_release>(),
          "Leaf-path TreeSHAP (class-0), summed over trees");
    // gil_scoped_release: the forest_fit level loop blocks on stream syncs;
    // releasing the GIL lets other Python threads drive their own streams.
    m.def("forest_fit", &forest_fit,
          py::call_guard<py::gil_scoped_release>(),
          "Batched histogram-forest fit (gfx950)");
    m.def("forest_fit_multi", &forest_fit_multi,
          py::call_guard<py::gil_scoped_release>(),
          "Mixed-model batched forest fit (per-job spec)");
    m.def("forest_predict_confusion", &forest_predict_confusion,
          py::call_guard<py::gil_scoped_release>(),
          "Ensemble predict + confusion accumulation");
    m.def("knn", &knn, py::call_guard<py::gil_scoped_release>(),
          "Brute-force k-NN (fp64 distances)");
    m.def("knn_segmented", &knn_segmented,
          py::call_guard<py::gil_scoped_release>(),
          "Fold-batched segmented k-NN (segment-local indices)");
    m.def("smote_interpolate", &smote_interpolate,
          py::call_guard<py::gil_scoped_release>(), "SMOTE synthesis");
    m.def("enn_keep", &enn_keep, py::call_guard<py::gil_scoped_release>(),
          "ENN keep-mask");
    m.def("tomek_keep", &tomek_keep,
          py::call_guard<py::gil_scoped_release>(), "Tomek-link keep-mask");
    m.def("bin_codes", &bin_codes_dev,
          py::call_guard<py::gil_scoped_release>(), "Quantile-bin codes");
    m.def("scaler_fit_transform", &scaler_fit_transform_dev,
          py::call_guard<py::gil_scoped_release>(),
          "StandardScaler fit_transform");
    m.def("pca_fit_transform", &pca_fit_transform_dev,
          py::call_guard<py::gil_scoped_release>(),
          "Full PCA fit_transform");
}
