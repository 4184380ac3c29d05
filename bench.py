#!/usr/bin/env python3
"""Flagship benchmark: the 216-cell grid sweep (the reference's `scores`
stage headline, BASELINE.json) on synthetic data.

One step = one full 216-cell sweep {flaky-type x feature-set x preprocessing
x balancing x model} with 10-fold stratified CV per cell, sharded across the
N GPUs (strong scaling: total work fixed), evaluated device-resident through
the HIP kernels and combined with one RCCL all-reduce.  The reported value
is grid-configs/sec for the WHOLE job.

Usage (driver contract):
  python bench.py --gpus N --steps K --warmup W
  # N>1 may be launched via torch.distributed.run (one rank per GPU, RCCL);
  # invoked directly with --gpus N>1 and no torchrun env, bench.py
  # SELF-LAUNCHES N ranks via torch.distributed.run on 127.0.0.1.

Secondary surface: --stage shap benchmarks the shap stage (the 2-config
TreeSHAP pass, reference experiment.py:520-530) instead of the scores sweep.
"""

import argparse
import json
import os
import socket
import subprocess
import sys
import time


def parse_args(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=2)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--n-tests", type=int, default=10000,
                   help="synthetic dataset size (tests.json rows)")
    p.add_argument("--cells", type=int, default=216,
                   help="number of grid cells (216 = full sweep)")
    p.add_argument("--stage", default="scores", choices=["scores", "shap"])
    p.add_argument("--backend", default="hip", choices=["hip", "ref"])
    return p.parse_args(argv)


def _free_port():
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def self_launch(args):
    """Re-exec as N ranks under torch.distributed.run (one rank per GPU).

    Entered only when --gpus > 1 and no torchrun rendezvous env is present
    (the driver usually provides its own torchrun; this makes a direct
    `python bench.py --gpus N` measure N real ranks instead of silently
    timing one)."""
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={args.gpus}",
        "--master-addr", "127.0.0.1",
        "--master-port", str(_free_port()),
        os.path.abspath(__file__),
    ] + sys.argv[1:]
    return subprocess.call(cmd)


def main():
    args = parse_args()

    if args.gpus > 1 and "RANK" not in os.environ:
        return self_launch(args)

    import torch

    from flake16_framework_amd.parallel import comm

    rank, world = comm.init_from_env()
    use_cuda = torch.cuda.is_available() and args.backend == "hip"
    if use_cuda:
        # clamp so oversubscribed smoke runs (2 ranks, 1 GPU) still work
        torch.cuda.set_device(min(int(os.environ.get("LOCAL_RANK", "0")),
                                  torch.cuda.device_count() - 1))

    def barrier_sync():
        if world > 1:
            import torch.distributed as dist
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    stage = bench_scores if args.stage == "scores" else bench_shap
    one_step, report = stage(args, rank, world, use_cuda)

    for _ in range(args.warmup):
        result = one_step()

    barrier_sync()
    t0 = time.time()
    for _ in range(args.steps):
        result = one_step()
    barrier_sync()
    elapsed = time.time() - t0

    # MAX elapsed over ranks
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=comm.collective_device())
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu().item())

    if rank == 0:
        print(json.dumps(report(result, elapsed)), flush=True)

    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()
    return 0


def bench_scores(args, rank, world, use_cuda):
    """One step = one full grid sweep, cells sharded over the world."""
    from flake16_framework_amd.configgrid import iter_config_keys
    from flake16_framework_amd.dataset.synthetic import make_synthetic_tests
    from flake16_framework_amd.engine.scores import run_scores
    from flake16_framework_amd.parallel import comm

    tests = make_synthetic_tests(n_tests=args.n_tests, seed=0)
    my_cells = comm.shard_cells(world, rank, n_cells=args.cells)
    backend = "hip" if use_cuda else "ref"

    def one_step():
        result = run_scores(tests=tests, backend=backend, cells=my_cells)
        return comm.gather_scores(result)

    def report(result, elapsed):
        configs_per_sec = args.cells * args.steps / elapsed

        # headline-quality check: best F1 among NOD/Flake16/Extra Trees cells
        f1_best = None
        for keys in iter_config_keys():
            if (keys[0], keys[1], keys[4]) == ("NOD", "Flake16",
                                               "Extra Trees") \
                    and keys in result:
                f = result[keys][3][5]
                if f is not None and (f1_best is None or f > f1_best):
                    f1_best = f

        return {
            "metric": "grid-configs/sec over 216-cell sweep",
            "value": configs_per_sec,
            "unit": "configs/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "flake16-scores-grid216",
                "grid_cells": args.cells,
                "n_tests": args.n_tests,
                "n_features": 16,
                "cv": "StratifiedKFold(10, shuffle, seed 0)",
                "parallelism": f"cell-sharded dp{world} + RCCL all-reduce",
                "f1_nod_flake16_extratrees": f1_best,
            },
        }

    return one_step, report


def bench_shap(args, rank, world, use_cuda):
    """One step = the full shap stage: TreeSHAP attributions for the two
    fixed best configs (fit on balanced full data, explain all N rows)."""
    from flake16_framework_amd.configgrid import SHAP_CONFIGS
    from flake16_framework_amd.dataset.synthetic import make_synthetic_tests
    from flake16_framework_amd.engine import shap_stage

    tests = make_synthetic_tests(n_tests=args.n_tests, seed=0)
    compute = (shap_stage.compute_shap_hip if use_cuda
               else shap_stage.compute_shap_ref)

    def one_step():
        results = [None, None]
        for i, keys in enumerate(SHAP_CONFIGS):
            if i % world != rank:
                continue
            results[i] = compute(keys, i, tests=tests)
        if world > 1:
            from flake16_framework_amd.parallel import comm
            gathered = [None] * world
            comm.dist().all_gather_object(gathered, results)
            for part in gathered:
                for i in range(2):
                    if part[i] is not None:
                        results[i] = part[i]
        return results

    def report(result, elapsed):
        n_configs = len(SHAP_CONFIGS)
        return {
            "metric": "shap-configs/sec (2-config TreeSHAP stage)",
            "value": n_configs * args.steps / elapsed,
            "unit": "configs/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "flake16-shap-stage",
                "shap_configs": [", ".join(k) for k in SHAP_CONFIGS],
                "n_tests": args.n_tests,
                "n_features": 16,
                "parallelism": f"config-sharded dp{world}",
            },
        }

    return one_step, report


if __name__ == "__main__":
    sys.exit(main())
