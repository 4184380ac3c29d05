"""Execution orchestration: the `run`, `container`, and `setup` commands.

Semantic port of the reference's data-collection layer
(experiment.py:110-239): 130,026 per-run Docker containers (26 subjects x
{2,500 baseline + 2,500 shuffle + 1 testinspect}), a process pool with a
progress meter, and crash-restart resumability through an append-only
log.txt of completed container names.

The pytest plugins installed into each subject venv are this package's
collect.showflakes / collect.testinspect equivalents (the reference's are
empty submodules).  Docker interaction is isolated behind run_container's
`runner` argument so the layer is testable without Docker.
"""

import os
import random
import shlex
import subprocess as sp
import sys
import time
from multiprocessing import Pool

from ..constants import (
    CONT_DATA_DIR, CONT_TIMEOUT, DATA_DIR, IMAGE_NAME, LOG_FILE,
    N_RUNS, PLUGIN_BLACKLIST, STDOUT_DIR, SUBJECTS_DIR, SUBJECTS_FILE,
)

N_PROC = os.cpu_count()
PIP_VERSION = "pip==21.2.1"
PIP_INSTALL = ["pip", "install", "-I", "--no-deps"]

# Plugin registration: our in-package collectors replace the reference's
# external showflakes/testinspect plugins.
COLLECT_PLUGINS = ("flake16_framework_amd.collect.showflakes",
                   "flake16_framework_amd.collect.testinspect")

# Root of the framework checkout (the directory holding the
# flake16_framework_amd package) — exposed to subject venvs by
# install_plugins so `pytest -p flake16_framework_amd.collect.*` resolves.
FRAMEWORK_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def iter_subjects(subjects_file=SUBJECTS_FILE):
    """subjects.txt rows: owner/repo,sha,package_dir[,pre-command...]."""
    with open(subjects_file, "r") as fd:
        for line in fd:
            line = line.strip()
            if not line:
                continue
            repo, *rest = line.split(",")
            yield repo.split("/", 1)[1], repo, *rest


def setup_project(proj, url, sha, package_dir, subjects_dir=SUBJECTS_DIR):
    """Create the subject's venv, clone at the pinned SHA, install pinned
    requirements and the project itself."""
    proj_dir = os.path.join(subjects_dir, proj, proj)
    venv_dir = os.path.join(subjects_dir, proj, "venv")
    requirements_file = os.path.join(subjects_dir, proj, "requirements.txt")

    env = os.environ.copy()
    env["PATH"] = os.path.join(venv_dir, "bin") + ":" + env["PATH"]

    sp.run(["virtualenv", venv_dir], check=True)
    sp.run(["git", "clone", url, proj_dir], check=True)
    sp.run(["git", "reset", "--hard", sha], cwd=proj_dir, check=True)

    package_dir = os.path.join(proj_dir, package_dir)
    sp.run([*PIP_INSTALL, PIP_VERSION], env=env, check=True)
    sp.run([*PIP_INSTALL, "-r", requirements_file], env=env, check=True)
    sp.run([*PIP_INSTALL, "-e", package_dir], env=env, check=True)
    install_plugins(venv_dir)


def install_plugins(venv_dir, framework_root=FRAMEWORK_ROOT):
    """Make the collection plugins importable from the subject venv (the
    reference pip-installs its showflakes/testinspect checkouts into each
    venv, experiment.py:125).  The framework's own setup.py builds the HIP
    extension and needs torch, which subject venvs don't ship, so the
    package is exposed with a .pth in the venv's site-packages instead of
    a pip install — equivalent for `pytest -p flake16_framework_amd...`."""
    import glob
    site_dirs = glob.glob(os.path.join(venv_dir, "lib", "python*",
                                       "site-packages"))
    if not site_dirs:
        raise RuntimeError(f"no site-packages under {venv_dir}")
    for site_dir in site_dirs:
        pth = os.path.join(site_dir, "flake16_framework_amd.pth")
        with open(pth, "w") as fd:
            fd.write(framework_root + "\n")


def setup_image():
    """Provision all subjects in parallel (runs inside `docker build`)."""
    os.makedirs(CONT_DATA_DIR, exist_ok=True)
    args = [(proj, f"https://github.com/{repo}", sha, package_dir)
            for proj, repo, sha, package_dir, *_ in iter_subjects()]
    with Pool(processes=N_PROC) as pool:
        pool.starmap(setup_project, args)


def mode_flags(mode, data_file):
    """Per-mode pytest flags (matches the showflakes/testinspect CLI)."""
    return {
        "testinspect": [f"--testinspect={data_file}"],
        "baseline": [f"--record-file={data_file}.tsv"],
        "shuffle": [f"--record-file={data_file}.tsv", "--shuffle"],
    }[mode]


def manage_container(cont_name, *commands, subjects_dir=None,
                     data_dir=None, run=sp.run):
    """Inside the container: run the subject's pre-commands, then pytest
    with the plugin blacklist, our collectors, and the mode flags.
    Directory overrides via FLAKE16_SUBJECTS_DIR / FLAKE16_DATA_DIR
    support the local (docker-less) mode."""
    subjects_dir = subjects_dir or os.environ.get("FLAKE16_SUBJECTS_DIR",
                                                  SUBJECTS_DIR)
    data_dir = data_dir or os.environ.get("FLAKE16_DATA_DIR", CONT_DATA_DIR)
    proj, mode, _ = cont_name.split("_", 2)
    proj_dir = os.path.join(subjects_dir, proj, proj)
    data_file = os.path.join(data_dir, cont_name)
    bin_dir = os.path.join(subjects_dir, proj, "venv", "bin")

    env = os.environ.copy()
    env["PATH"] = bin_dir + ":" + env["PATH"]

    for cmd in commands[:-1]:
        run(shlex.split(cmd), cwd=proj_dir, env=env, check=True)

    plugin_args = []
    for plug in COLLECT_PLUGINS:
        plugin_args += ["-p", plug]

    run(
        [*shlex.split(commands[-1]), *PLUGIN_BLACKLIST, *plugin_args,
         "--set-exitstatus", *mode_flags(mode, data_file)],
        timeout=CONT_TIMEOUT, cwd=proj_dir, check=True, env=env)


def docker_run_argv(cont_name, commands, host_data_dir):
    return [
        "docker", "run", "-it",
        f"-v={host_data_dir}:{CONT_DATA_DIR}:rw", "--rm", "--init",
        "--cpus=1", f"--name={cont_name}", IMAGE_NAME, "python3",
        "experiment.py", "container", cont_name, *commands,
    ]


def _local_run_argv(cont_name, commands):
    """Local (no-docker) mode: the per-run isolation is a fresh python
    process instead of a container.  Enabled by FLAKE16_LOCAL_RUN=1 with
    FLAKE16_SUBJECTS_DIR / FLAKE16_DATA_DIR pointing at the work tree —
    used by the integration tests and available for docker-less hosts."""
    import sys
    return [sys.executable, "-m", "flake16_framework_amd.cli", "container",
            cont_name, *commands]


def run_container(args, runner=None, stdout_dir=STDOUT_DIR):
    """Launch one container run; append its stdout; report success."""
    cont_name, commands = args
    host_data_dir = os.path.join(os.getcwd(), DATA_DIR)
    stdout_file = os.path.join(stdout_dir, cont_name)

    if os.environ.get("FLAKE16_LOCAL_RUN"):
        argv = _local_run_argv(cont_name, commands)
    else:
        argv = docker_run_argv(cont_name, commands, host_data_dir)

    if runner is None:
        def runner(argv, fd):
            return sp.run(argv, stdout=fd, stderr=sp.STDOUT).returncode

    with open(stdout_file, "a") as fd:
        returncode = runner(argv, fd)

    succeeded = returncode == 0
    message = "succeeded" if succeeded else "failed"
    return f"{message}: {cont_name}", (succeeded, cont_name)


def iter_containers(run_modes, subjects_file=SUBJECTS_FILE, n_runs=None):
    n_runs = n_runs or N_RUNS
    for proj, _, _, _, *commands in iter_subjects(subjects_file):
        for mode in set(run_modes):
            for run_n in range(n_runs[mode]):
                yield f"{proj}_{mode}_{run_n}", commands


def manage_pool(pool, fn, args, out=None):
    """Shuffle, imap_unordered, and print per-task progress lines with
    elapsed/ETA minutes — the reference's pool meter (experiment.py:191).
    `out` resolves to the CURRENT sys.stdout at call time (an import-time
    default would capture a stale stream under redirection)."""
    out = out or sys.stdout
    n_finish = 0
    t_start = time.time()

    random.shuffle(args)
    out.write(f"0/{len(args)} 0/?\r")

    for message, result in pool.imap_unordered(fn, args):
        n_finish += 1
        n_remain = len(args) - n_finish
        t_elapse = time.time() - t_start
        t_remain = t_elapse / n_finish * n_remain
        out.write(f"{message}\n\r")
        out.write(f"{n_finish}/{n_remain} "
                  f"{round(t_elapse / 60)}/{round(t_remain / 60)}\r")
        yield result


def read_log(log_file=LOG_FILE):
    if not os.path.exists(log_file):
        return []
    with open(log_file, "r") as fd:
        return [line.strip() for line in fd]


def run_experiment(*run_modes, subjects_file=SUBJECTS_FILE, n_runs=None,
                   runner=None, log_file=LOG_FILE):
    """The resumable run driver: skip completed runs (log.txt), launch the
    rest through a process pool, append completions, exit 1 on any
    failure."""
    os.makedirs(DATA_DIR, exist_ok=True)
    os.makedirs(STDOUT_DIR, exist_ok=True)

    log = set(read_log(log_file))
    args = [(cont_name, commands)
            for cont_name, commands in iter_containers(run_modes,
                                                       subjects_file, n_runs)
            if cont_name not in log]

    exitstatus = 0
    run_fn = run_container if runner is None else \
        (lambda a: run_container(a, runner=runner))

    with Pool(processes=N_PROC) as pool:
        for succeeded, cont_name in manage_pool(pool, run_fn, args):
            if succeeded:
                with open(log_file, "a") as fd:
                    fd.write(f"{cont_name}\n")
            else:
                exitstatus = 1

    sys.exit(exitstatus)
