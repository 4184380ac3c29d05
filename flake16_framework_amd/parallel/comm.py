"""Distributed execution: one rank per GPU, RCCL (= the "nccl" backend of
torch.distributed on ROCm) over xGMI; gloo on CPU for tests.

The reference's only transport is a multiprocessing.Pool result queue
(experiment.py:191-211, 496).  Here the 216 grid cells are sharded across
ranks cost-balanced, each rank evaluates its cells device-resident, and the
per-cell result blobs are combined with ONE all-reduce at the end: payloads
are tiny (per-cell confusion counts, ~70 KB for the whole grid), so latency
dominates and a single fused collective beats any chatter.
"""

import os

import numpy as np

from ..configgrid import cell_cost_estimate, iter_config_keys
from ..engine.metrics import finalize_scores

_N_CONFUSION = 3  # FP, FN, TP


def dist():
    import torch.distributed as d
    return d


def is_initialized():
    try:
        return dist().is_initialized()
    except Exception:
        return False


def rank_world():
    if is_initialized():
        return dist().get_rank(), dist().get_world_size()
    return 0, 1


def init_from_env(backend=None):
    """Initialize the process group from torchrun env vars if present.

    Backend selection: RCCL ("nccl") when there is one GPU per rank — the
    production layout.  When ranks OVERSUBSCRIBE the visible GPUs (e.g. a
    2-rank smoke test on a 1-GPU box) RCCL refuses duplicate devices
    ("Duplicate GPU detected"), so collectives fall back to gloo on host
    memory while the compute path stays on CUDA; payloads here are ~KB, so
    the transport does not matter for validation runs."""
    if is_initialized() or "RANK" not in os.environ:
        return rank_world()
    import torch
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if backend is None:
        if torch.cuda.is_available() and world <= torch.cuda.device_count():
            backend = "nccl"
        else:
            backend = "gloo"
    if torch.cuda.is_available():
        torch.cuda.set_device(min(int(os.environ.get("LOCAL_RANK", "0")),
                                  torch.cuda.device_count() - 1))
    dist().init_process_group(backend=backend)
    return rank_world()


def collective_device():
    """Device collective tensors should live on for the current backend."""
    if is_initialized() and dist().get_backend() == "nccl":
        return "cuda"
    return "cpu"


def shard_cells(world, rank, n_cells=None):
    """Cost-balanced static shard of the grid cells, BALANCE-GROUP aware:
    the 3 model cells of a (flaky x feature-set x preproc x balancing)
    group share their balanced folds and one fused fit, so whole groups
    are assigned LPT-greedily (heaviest group to the lightest rank) with
    measured-calibrated costs (configgrid.cell_cost_estimate).
    Deterministic: ties break to the lower group index / lower rank.
    Returns the sorted list of cell indices owned by `rank`.

    (A view-affine contiguous fill was used before round 2's trace
    showed view builds cost only 5-20 ms per rank while its boundary
    effects left a 1.24 max/mean load imbalance at 8 ranks; LPT brings
    the measured-cost imbalance to ~1.06.)"""
    from ..configgrid import balance_group_index

    keys = list(iter_config_keys())
    if n_cells is not None:
        keys = keys[:n_cells]

    groups = {}
    for i, k in enumerate(keys):
        groups.setdefault(balance_group_index(k), []).append(i)
    gcost = {g: sum(cell_cost_estimate(keys[i]) for i in cells)
             for g, cells in groups.items()}

    loads = [0.0] * world
    mine = []
    for g in sorted(groups, key=lambda g: (-gcost[g], g)):
        r = min(range(world), key=lambda r: (loads[r], r))
        loads[r] += gcost[g]
        if r == rank:
            mine.extend(groups[g])

    return sorted(mine)


def _pack(result, cell_order, projects):
    """result {config_keys: [t_train, t_test, scores, scores_total]} ->
    float64 tensor [n_cells, 2 + (n_proj+1)*3 + 1]; the trailing column
    marks cells this rank actually evaluated (so a partial-grid run does
    not gather phantom zero-count cells)."""
    n_proj = len(projects)
    buf = np.zeros((len(cell_order), 2 + (n_proj + 1) * _N_CONFUSION + 1))
    for ci, keys in enumerate(cell_order):
        if keys not in result:
            continue
        t_train, t_test, scores, scores_total = result[keys]
        buf[ci, 0] = t_train
        buf[ci, 1] = t_test
        for pi, proj in enumerate(projects):
            buf[ci, 2 + pi * 3: 2 + pi * 3 + 3] = scores[proj][:3]
        buf[ci, 2 + n_proj * 3: 2 + (n_proj + 1) * 3] = scores_total[:3]
        buf[ci, -1] = 1.0
    return buf


def _unpack(buf, cell_order, projects):
    n_proj = len(projects)
    out = {}
    for ci, keys in enumerate(cell_order):
        if buf[ci, -1] == 0.0:
            continue   # no rank evaluated this cell
        scores = {}
        for pi, proj in enumerate(projects):
            scores[proj] = [int(v) for v in buf[ci, 2 + pi * 3: 2 + pi * 3 + 3]]
        scores_total = [int(v)
                        for v in buf[ci, 2 + n_proj * 3: 2 + (n_proj + 1) * 3]]
        finalize_scores(scores, scores_total)
        out[keys] = [float(buf[ci, 0]), float(buf[ci, 1]), scores, scores_total]
    return out


def gather_scores(result):
    """Combine per-rank partial results into the full scores dict on every
    rank via one all-reduce(SUM) over a flat tensor.  world==1: passthrough.
    """
    rank, world = rank_world()
    if world == 1:
        return result

    import torch
    d = dist()

    cell_order = list(iter_config_keys())
    # Every evaluated cell carries the full project list in dataset order.
    projects = None
    for v in result.values():
        projects = list(v[2].keys())
        break
    # All ranks must agree on the project list; broadcast from rank 0's copy
    # is unnecessary since all load the same tests.json, but a rank could in
    # principle own zero cells — exchange via all_gather_object.
    plists = [None] * world
    d.all_gather_object(plists, projects)
    projects = next((p for p in plists if p is not None), None)
    if projects is None:
        return result  # no rank evaluated anything

    buf = _pack(result, cell_order, projects)
    use_cuda = d.get_backend() == "nccl"
    t = torch.from_numpy(buf)
    if use_cuda:
        t = t.cuda()
    d.all_reduce(t, op=d.ReduceOp.SUM)
    buf = t.cpu().numpy()

    return _unpack(buf, cell_order, projects)
