"""The `scores` stage: evaluate the 216-cell grid, write scores.pkl.

Output contract (reference experiment.py:446-501):
  scores.pkl = { (k0..k4): [t_train, t_test, scores, scores_total] }
    scores       = {proj: [FP, FN, TP, P, R, F]}  (all projects in dataset)
    scores_total = [FP, FN, TP, P, R, F]
    t_train/t_test = MEAN seconds per fold over the 10 folds
  quirks preserved: preprocessing is fit on the FULL dataset before the CV
  split; true negatives are skipped; P/R/F are None on zero denominators.

Execution model (MI355X-first, nothing like the reference's process pool):
  cells are sharded across ranks (one rank per GPU); each rank evaluates its
  cells device-resident — fold-batched balancing, binning, forest fit,
  prediction and confusion — and the per-cell result blobs are combined with
  one RCCL all-reduce at the end (parallel/comm.py).  The 'ref' backend runs
  the identical algorithm in numpy on CPU, optionally over a process pool
  (the reference's execution model).
"""

import pickle
import time

import numpy as np

from ..balance import apply_balancing
from ..configgrid import iter_config_keys, resolve
from ..constants import SCORES_FILE
from ..dataset.tests_io import load_feat_lab_proj
from ..models.binning import bin_codes, compute_bin_cuts
from ..models.forest_ref import fit_forest, params_for_model, predict_forest
from ..preprocess import apply_preprocessing
from .folds import stratified_kfold_split
from .metrics import finalize_scores

GLOBAL_SEED = 0
N_FOLDS = 10


def job_ids_for(config_keys, cell_idx, fold):
    """Philox key-ids for one (cell, fold): (balance_k1, tree_job_base).

    Balancing is keyed on the cell's BALANCE GROUP (the 72 distinct
    (flaky, feature-set, preproc, balancing) combos) so the engine can share
    balanced folds across the 3 model-axis cells of a group; trees are keyed
    per cell.  Tree t of this fold uses k1 = tree_job_base + t (t < 128).
    Philox tags domain-separate the two key spaces.
    """
    from ..configgrid import balance_group_index
    bal_k1 = balance_group_index(config_keys) * N_FOLDS + fold
    ctx = cell_idx * N_FOLDS + fold
    return bal_k1, ctx * 128


def evaluate_cell_ref(config_keys, cell_idx, tests=None, tests_file=None,
                      seed=GLOBAL_SEED):
    """Evaluate one grid cell with the numpy reference backend.

    Returns [t_train, t_test, scores, scores_total] (the scores.pkl value).
    """
    flaky_label, feature_set, preproc, balancing, model = resolve(config_keys)
    kwargs = {"tests": tests} if tests is not None else {"tests_file": tests_file}
    features, labels, projects = load_feat_lab_proj(flaky_label, feature_set,
                                                    **kwargs)

    X = apply_preprocessing(features, preproc).astype(np.float32)
    labels = labels.astype(np.uint8)

    # Full-dataset bin cuts (mirrors the full-data preprocessing-fit quirk).
    cuts = compute_bin_cuts(X)
    codes_all = bin_codes(X, cuts)
    params = params_for_model(model, seed=seed)

    t_train = t_test = 0.0
    scores = {proj: [0] * 6 for proj in projects}
    scores_total = [0] * 6

    for i, (train, test) in enumerate(
            stratified_kfold_split(labels, n_splits=N_FOLDS,
                                   random_state=seed)):
        bal_k1, job_base = job_ids_for(config_keys, cell_idx, i)
        Xb, yb = apply_balancing(X[train], labels[train], balancing,
                                 seed, bal_k1)
        codes_tr = bin_codes(Xb, cuts)

        t0 = time.time()
        forest = fit_forest(codes_tr, yb, params, job_base=job_base, cuts=cuts)
        t_train += time.time() - t0

        t0 = time.time()
        preds = predict_forest(forest, codes_all[test])
        t_test += time.time() - t0

        y_test = labels[test]
        projects_test = projects[test]
        for j in range(len(test)):
            k = int(2 * y_test[j] + preds[j]) - 1
            if k == -1:
                continue
            scores[projects_test[j]][k] += 1
            scores_total[k] += 1

    finalize_scores(scores, scores_total)
    return [t_train / N_FOLDS, t_test / N_FOLDS, scores, scores_total]


def run_scores(tests_file=None, tests=None, backend="auto", cells=None,
               progress=None, seed=GLOBAL_SEED, on_result=None,
               processes=1):
    """Evaluate `cells` (an iterable of (cell_idx, config_keys); default all
    216) and return {config_keys: [t_train, t_test, scores, scores_total]}.
    on_result(config_keys, value): optional per-cell callback (used by the
    checkpoint writer).  processes > 1 runs the 'ref' backend cells in a
    multiprocessing pool (the reference's execution model,
    experiment.py:496-498); the hip backend uses streams instead.
    """
    from ..utils.trace import trace_span

    all_cells = list(enumerate(iter_config_keys()))
    if cells is not None:
        wanted = set(cells)
        all_cells = [(i, k) for i, k in all_cells if i in wanted]

    if backend == "auto":
        backend = _auto_backend()

    if backend == "hip":
        return _run_scores_hip(all_cells, tests, tests_file, seed, progress,
                               on_result=on_result)

    out = {}
    t_start = time.time()

    if processes > 1:
        from multiprocessing import Pool
        args = [(keys, ci, tests, tests_file, seed)
                for ci, keys in all_cells]
        with Pool(processes=processes) as pool:
            for n_done, (keys, value) in enumerate(
                    pool.imap_unordered(_eval_cell_ref_task, args)):
                out[keys] = value
                if on_result:
                    on_result(keys, value)
                if progress:
                    progress(n_done + 1, len(all_cells),
                             time.time() - t_start, ", ".join(keys))
        return out

    for n_done, (cell_idx, config_keys) in enumerate(all_cells):
        with trace_span("cell", cell=cell_idx, backend="ref"):
            out[config_keys] = evaluate_cell_ref(
                config_keys, cell_idx, tests=tests, tests_file=tests_file,
                seed=seed)
        if on_result:
            on_result(config_keys, out[config_keys])
        if progress:
            progress(n_done + 1, len(all_cells), time.time() - t_start,
                     ", ".join(config_keys))
    return out


def _eval_cell_ref_task(args):
    keys, cell_idx, tests, tests_file, seed = args
    return keys, evaluate_cell_ref(keys, cell_idx, tests=tests,
                                   tests_file=tests_file, seed=seed)


def _run_scores_hip(all_cells, tests, tests_file, seed, progress,
                    n_streams=None, on_result=None):
    """Device sweep: prebuild the shared caches (views, folds, balanced
    groups) on the default stream, then evaluate BALANCE GROUPS (the 1-3
    model cells sharing balanced folds, fused into one mixed-model
    forest_fit) concurrently on worker threads with one HIP stream each —
    host-side bookkeeping overlaps other groups' kernels (extension calls
    release the GIL)."""
    import os
    import threading
    from concurrent.futures import ThreadPoolExecutor

    import torch

    from ..configgrid import balance_group_index
    from ..utils.trace import trace_span
    from .hip_cell import SweepContext

    if n_streams is None:
        n_streams = int(os.environ.get("FLAKE16_STREAMS", "4"))

    # Shared caches (views, folds, balanced groups) build lazily under
    # per-key locks: the first worker to need an entry builds it on its own
    # stream while other workers proceed with other groups.
    context = SweepContext(tests=tests, tests_file=tests_file, seed=seed)

    if os.environ.get("FLAKE16_NO_GROUP_FUSE"):
        tasks = [[(config_keys, cell_idx)]
                 for cell_idx, config_keys in all_cells]
    else:
        groups = {}
        for cell_idx, config_keys in all_cells:
            g = balance_group_index(config_keys)
            groups.setdefault(g, []).append((config_keys, cell_idx))
        tasks = [groups[g] for g in sorted(groups)]

    out = {}
    lock = threading.Lock()
    t_start = time.time()
    n_done = [0]

    def eval_group(group_cells):
        stream = torch.cuda.Stream()
        with trace_span("group", cells=len(group_cells), backend="hip"):
            with torch.cuda.stream(stream):
                result = context.evaluate_group(group_cells)
        with lock:
            for config_keys, value in result.items():
                out[config_keys] = value
                n_done[0] += 1
                if on_result:
                    on_result(config_keys, value)
                if progress:
                    progress(n_done[0], len(all_cells),
                             time.time() - t_start, ", ".join(config_keys))

    if n_streams <= 1:
        for group_cells in tasks:
            eval_group(group_cells)
    else:
        with ThreadPoolExecutor(max_workers=n_streams) as pool:
            list(pool.map(eval_group, tasks))
    return out


def _auto_backend():
    try:
        import torch
        if torch.cuda.is_available():
            return "hip"
    except ImportError:
        pass
    return "ref"


def _load_checkpoint(path):
    """Sequence-of-pickles checkpoint -> {config_keys: value}."""
    import os
    done = {}
    if not os.path.exists(path):
        return done
    with open(path, "rb") as fd:
        while True:
            try:
                keys, value = pickle.load(fd)
                done[keys] = value
            except EOFError:
                break
            except Exception:
                break  # truncated trailing record from a crash: drop it
    return done


def write_scores(tests_file=None, scores_file=SCORES_FILE, backend="auto",
                 checkpoint=None, n_cells=None):
    """Full 216-cell sweep (sharded across ranks if distributed is
    initialized) -> scores.pkl on rank 0.

    checkpoint: optional path PREFIX for crash-restart resumability (the
    scores-stage analogue of the reference run stage's log.txt,
    experiment.py:222-237): each completed cell is appended to
    <prefix>.rank<r>; a re-invocation skips completed cells.
    """
    from ..parallel import comm
    from ..utils.trace import trace_span

    rank, world = comm.rank_world()
    my_cells = comm.shard_cells(world, rank, n_cells=n_cells)

    done = {}
    on_result = None
    if checkpoint:
        ckpt_path = f"{checkpoint}.rank{rank}"
        done = _load_checkpoint(ckpt_path)
        all_keys = list(iter_config_keys())
        # Keep only entries this rank OWNS under the current shard: a resume
        # with a different world size would otherwise merge a cell from the
        # old shard AND recompute it on its new owner, and the all-reduce SUM
        # in gather_scores would double its counts.
        my_keys = {all_keys[c] for c in my_cells}
        done = {k: v for k, v in done.items() if k in my_keys}
        my_cells = [c for c in my_cells if all_keys[c] not in done]
        ckpt_fd = open(ckpt_path, "ab")
        ckpt_lock = __import__("threading").Lock()

        def on_result(keys, value):
            with ckpt_lock:
                pickle.dump((keys, value), ckpt_fd)
                ckpt_fd.flush()

    def progress(done_n, total, elapsed, name):
        eta = elapsed / done_n * (total - done_n)
        print(f"[rank {rank}] {done_n}/{total} {name} "
              f"({elapsed:.0f}s elapsed, eta {eta:.0f}s)", flush=True)

    import os as _os
    n_proc = int(_os.environ.get("FLAKE16_REF_PROCS",
                                 str(_os.cpu_count() or 1)))
    with trace_span("scores_sweep", rank_cells=len(my_cells)):
        result = run_scores(tests_file=tests_file, backend=backend,
                            cells=my_cells, progress=progress,
                            on_result=on_result, processes=n_proc)
    result.update(done)
    result = comm.gather_scores(result)

    if checkpoint:
        ckpt_fd.close()
    if rank == 0:
        with open(scores_file, "wb") as fd:
            pickle.dump(result, fd)
    return result
