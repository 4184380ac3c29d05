"""Command-line interface — compatible with the reference's
`python experiment.py COMMAND` surface (experiment.py:693-714) plus
GPU/synthetic extensions:

  setup                      provision subject venvs (docker build time)
  container NAME CMD...      in-container runner
  run MODE...                data collection (baseline/shuffle/testinspect)
  tests                      collate data/ -> tests.json
  scores                     216-cell grid -> scores.pkl (GPU when present)
  shap                       TreeSHAP for the 2 best configs -> shap.pkl
  figures                    LaTeX tables/plots

  synthetic [--n-tests N] [--seed S]   write a synthetic tests.json
  scores/shap extras: --backend {auto,hip,ref}

Distributed: launch via `python -m torch.distributed.run --nproc-per-node N
experiment.py scores` — one rank per GPU over RCCL; rank 0 writes the
artifacts.
"""

import sys


def _pop_flag(args, name, default=None, boolean=False):
    if name in args:
        i = args.index(name)
        if boolean:
            args.pop(i)
            return True
        args.pop(i)
        return args.pop(i)
    return default


def main(argv=None):
    argv = list(sys.argv[1:] if argv is None else argv)
    if not argv:
        raise ValueError("No command given")

    command, *args = argv

    if command == "setup":
        from .orchestrate.runner import provision_all
        provision_all()
    elif command == "container":
        from .orchestrate.runner import exec_suite
        exec_suite(*args)
    elif command == "run":
        from .orchestrate.runner import drive_runs
        drive_runs(*args)
    elif command == "tests":
        from .dataset.tests_io import write_tests
        write_tests()
    elif command == "scores":
        backend = _pop_flag(args, "--backend", "auto")
        checkpoint = _pop_flag(args, "--checkpoint", None)
        n_cells = _pop_flag(args, "--cells", None)
        trace = _pop_flag(args, "--trace", None)
        if trace:
            from .utils.trace import set_trace_file
            set_trace_file(trace)
        from .engine.scores import write_scores
        from .parallel import comm
        comm.init_from_env()
        write_scores(backend=backend, checkpoint=checkpoint,
                     n_cells=int(n_cells) if n_cells else None)
    elif command == "shap":
        backend = _pop_flag(args, "--backend", "auto")
        from .engine.shap_stage import write_shap
        from .parallel import comm
        comm.init_from_env()
        write_shap(backend=backend)
    elif command == "figures":
        offline = _pop_flag(args, "--offline", False, boolean=True)
        from .report.figures import write_figures
        write_figures(offline=offline)
    elif command == "synthetic":
        n_tests = int(_pop_flag(args, "--n-tests", "10000"))
        seed = int(_pop_flag(args, "--seed", "0"))
        from .constants import TESTS_FILE
        from .dataset.synthetic import write_synthetic_tests
        write_synthetic_tests(TESTS_FILE, n_tests=n_tests, seed=seed)
        print(f"wrote {TESTS_FILE} ({n_tests} synthetic tests, seed {seed})")
    else:
        raise ValueError("Unrecognized command given")


if __name__ == "__main__":
    main()
