"""The `shap` stage: TreeSHAP attributions for the two best configs.

Output contract (reference experiment.py:504-530): shap.pkl is a 2-list
[shap_nod, shap_od] of N x 16 arrays — class-0 SHAP values of the model
fitted on the FULL (preprocessed, balanced) dataset, evaluated over the
full preprocessed (UNbalanced) feature matrix.  Configs are fixed
(configgrid.SHAP_CONFIGS).  The reference's latent NameError on the
unbalanced path (experiment.py:515 uses `feature`) is fixed here: an
unbalanced config simply fits on the full data.
"""

import pickle

import numpy as np

from ..balance import apply_balancing
from ..configgrid import SHAP_CONFIGS, resolve
from ..constants import SHAP_FILE
from ..dataset.tests_io import load_feat_lab_proj
from ..models.binning import bin_codes, compute_bin_cuts
from ..models.forest_ref import fit_forest, params_for_model
from ..models.treeshap_ref import forest_shap
from ..preprocess import apply_preprocessing

# Philox key bases for the shap stage (disjoint from the scores stage
# keying in engine/scores.job_ids_for by construction: scores balance keys
# are < 720, tree job keys are cell-ctx*128 with ctx < 2160*10; shap uses a
# distinct high range).
SHAP_BAL_KEY = 1 << 20
SHAP_JOB_BASE = 1 << 21


def compute_shap_ref(config_keys, idx, tests=None, tests_file=None, seed=0):
    flaky_label, feature_set, preproc, balancing, model = resolve(config_keys)
    kwargs = {"tests": tests} if tests is not None else {"tests_file": tests_file}
    features, labels_b, _ = load_feat_lab_proj(flaky_label, feature_set,
                                               **kwargs)
    labels = labels_b.astype(np.uint8)
    X = apply_preprocessing(features, preproc).astype(np.float32)
    F = X.shape[1]

    cuts = compute_bin_cuts(X)
    codes_all = bin_codes(X, cuts)

    Xb, yb = apply_balancing(X, labels, balancing, seed, SHAP_BAL_KEY + idx)
    codes_b = bin_codes(Xb, cuts)

    params = params_for_model(model, seed=seed)
    forest = fit_forest(codes_b, yb, params,
                        job_base=SHAP_JOB_BASE + idx * 128, cuts=cuts)
    return forest_shap(forest, codes_all, F)


def compute_shap_hip(config_keys, idx, tests=None, tests_file=None, seed=0):
    import torch

    from ..configgrid import MODEL_AXIS
    from ..ops.backend import get_ops
    from .hip_cell import SweepContext

    ops = get_ops()
    ctx = SweepContext(tests=tests, tests_file=tests_file, seed=seed)
    device = ctx.device

    view = ctx.view_for(config_keys)
    lab = ctx.labels_for(config_keys)
    F = view["F"]
    _, _, _, balancing_spec, model_spec = resolve(config_keys)
    Xb, yb = ctx._balance_dev(view["X32"], lab["labels"], balancing_spec,
                              SHAP_BAL_KEY + idx)
    codes_b = ops.bin_codes(Xb, view["cuts_dev"], view["cut_off"], F)

    spec = MODEL_AXIS[config_keys[4]]
    n_trees = spec["n_estimators"]
    max_features = F if spec["kind"] == "decision_tree" else max(
        1, int(np.sqrt(F)))
    J = n_trees
    j_row_off = torch.zeros(J, dtype=torch.int32, device=device)
    j_n = torch.full((J,), len(yb), dtype=torch.int32, device=device)
    j_key = torch.arange(SHAP_JOB_BASE + idx * 128,
                         SHAP_JOB_BASE + idx * 128 + J,
                         dtype=torch.int32, device=device)
    labels_b_dev = torch.from_numpy(yb).to(device)
    nfeat, nsplit, nleft, ncnt0, ncnt1, j_node_off, node_alloc = \
        ops.forest_fit(codes_b, labels_b_dev, j_row_off, j_n, j_key, F,
                       max_features, spec["bootstrap"],
                       spec["kind"] == "extra_trees", seed)

    import os
    if os.environ.get("FLAKE16_SHAP_RECURSE"):
        # the original per-(sample, tree) recursion — kept for A/B
        phi = ops.treeshap(view["codes_all"], j_node_off, nfeat, nsplit,
                           nleft, ncnt0, ncnt1)
    else:
        # leaf-path formulation: wave-uniform walks, no path copies
        from ..models.leafpaths import build_leaf_paths
        leaf_tree, leaf_off, path_nodes, max_depth = build_leaf_paths(
            nfeat.cpu().numpy(), nleft.cpu().numpy(),
            j_node_off.cpu().numpy(), node_alloc.cpu().numpy())
        if max_depth > 128:
            raise RuntimeError(
                f"treeshap: tree depth {max_depth} exceeds SHAP_DMAX")
        phi = ops.treeshap_paths(
            view["codes_all"],
            torch.from_numpy(leaf_tree).to(device),
            torch.from_numpy(leaf_off).to(device),
            torch.from_numpy(path_nodes).to(device),
            j_node_off, nfeat, nsplit, nleft, ncnt0, ncnt1)
    return (phi.cpu().numpy() / n_trees)[:, :F]


def write_shap(tests_file=None, tests=None, shap_file=SHAP_FILE,
               backend="auto", seed=0):
    from ..parallel import comm
    from .scores import _auto_backend

    if backend == "auto":
        backend = _auto_backend()
    rank, world = comm.rank_world()

    results = [None, None]
    for i, keys in enumerate(SHAP_CONFIGS):
        if i % world != rank:
            continue
        if backend == "hip":
            results[i] = compute_shap_hip(keys, i, tests=tests,
                                          tests_file=tests_file, seed=seed)
        else:
            results[i] = compute_shap_ref(keys, i, tests=tests,
                                          tests_file=tests_file, seed=seed)

    if world > 1:
        gathered = [None] * world
        comm.dist().all_gather_object(gathered, results)
        for part in gathered:
            for i in range(2):
                if part[i] is not None:
                    results[i] = part[i]

    if rank == 0:
        with open(shap_file, "wb") as fd:
            pickle.dump(results, fd)
    return results
