"""Execution orchestration: the `run`, `container`, and `setup` commands.

Covers the reference's data-collection layer (the BEHAVIOR of
experiment.py:110-239 — 130,026 per-run containers over 26 subjects, a
process pool with a progress meter, crash-restart resumability through an
append-only completion log).  The structure here is this framework's own;
the compatibility contracts preserved byte/argv-exactly are:

  - subjects.txt rows: owner/repo,sha,package_dir[,pre-command...]
  - the docker invocation (docker run -it -v data:rw --rm --init --cpus=1
    --name=<cont> flake16framework python3 experiment.py container ...)
  - the pytest collector flags (--record-file/--shuffle/--testinspect/
    --set-exitstatus) and the plugin blacklist
  - the pool progress meter lines ("<message>\\n" + "done/remain
    elapsed/eta" carriage-return updates)
  - log.txt: one completed container name per line, appended on success

Docker interaction is isolated behind the `runner` callable so the layer
is testable without Docker; FLAKE16_LOCAL_RUN=1 substitutes a fresh local
python process for the container (used by the integration tests and
docker-less hosts).
"""

import os
import random
import shlex
import subprocess as sp
import sys
import time
from multiprocessing import Pool
from typing import NamedTuple

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


# ---------------------------------------------------------------------------
# Subjects
# ---------------------------------------------------------------------------

class Subject(NamedTuple):
    """One study subject: a pinned open-source project."""
    proj: str          # short name (repo without the owner)
    repo: str          # owner/repo
    sha: str           # pinned commit
    package_dir: str   # directory pip-installed -e (relative to checkout)
    commands: tuple    # pre-commands + the pytest command (last)

    @classmethod
    def parse(cls, line):
        repo, sha, package_dir, *commands = line.split(",")
        return cls(repo.split("/", 1)[1], repo, sha, package_dir,
                   tuple(commands))


def read_subjects(subjects_file=SUBJECTS_FILE):
    """subjects.txt -> [Subject]; blank lines ignored."""
    with open(subjects_file, "r") as fd:
        return [Subject.parse(line.strip()) for line in fd if line.strip()]


# ---------------------------------------------------------------------------
# Subject environment provisioning (the `setup` stage)
# ---------------------------------------------------------------------------

def create_venv(venv_dir, run=sp.run):
    """virtualenv when available (the reference's tool; the runner image
    installs it), `python -m venv` otherwise, degrading to --without-pip
    where ensurepip has no bundled wheels — every variant yields the
    bin/ + site-packages layout the rest relies on."""
    import shutil
    if shutil.which("virtualenv"):
        run(["virtualenv", venv_dir], check=True)
        return
    try:
        run([sys.executable, "-m", "venv", venv_dir], check=True)
    except sp.CalledProcessError:
        shutil.rmtree(venv_dir, ignore_errors=True)
        run([sys.executable, "-m", "venv", "--without-pip", venv_dir],
            check=True)


def provision_subject(subject, subjects_dir=SUBJECTS_DIR, run=sp.run,
                      pip_run=None):
    """Create the subject's venv, clone at the pinned SHA, install the
    pinned requirements + the project, and expose the collectors.

    run / pip_run: injectable process runners (pip_run defaults to run);
    FLAKE16_GIT_BASE overrides the clone base URL (local mirrors, tests).
    """
    pip_run = pip_run or run
    root = os.path.join(subjects_dir, subject.proj)
    checkout = os.path.join(root, subject.proj)
    venv_dir = os.path.join(root, "venv")
    pins = os.path.join(root, "requirements.txt")
    git_base = os.environ.get("FLAKE16_GIT_BASE", "https://github.com/")

    env = os.environ.copy()
    env["PATH"] = os.path.join(venv_dir, "bin") + ":" + env["PATH"]

    create_venv(venv_dir, run=run)
    run(["git", "clone", git_base + subject.repo, checkout], check=True)
    run(["git", "reset", "--hard", subject.sha], cwd=checkout, check=True)

    pip_run([*PIP_INSTALL, PIP_VERSION], env=env, check=True)
    pip_run([*PIP_INSTALL, "-r", pins], env=env, check=True)
    pip_run([*PIP_INSTALL, "-e",
             os.path.join(checkout, subject.package_dir)],
            env=env, check=True)
    install_plugins(venv_dir)


def install_plugins(venv_dir, framework_root=FRAMEWORK_ROOT):
    """Make the collection plugins importable from the subject venv (the
    reference pip-installs its showflakes/testinspect checkouts into each
    venv).  The framework's own setup.py builds the HIP extension and
    needs torch, which subject venvs don't ship, so the package is exposed
    with a .pth in the venv's site-packages instead of a pip install —
    equivalent for `pytest -p flake16_framework_amd...`."""
    import glob
    site_dirs = glob.glob(os.path.join(venv_dir, "lib", "python*",
                                       "site-packages"))
    if not site_dirs:
        raise RuntimeError(f"no site-packages under {venv_dir}")
    for site_dir in site_dirs:
        pth = os.path.join(site_dir, "flake16_framework_amd.pth")
        with open(pth, "w") as fd:
            fd.write(framework_root + "\n")


def provision_all(subjects_file=SUBJECTS_FILE):
    """Provision every subject in parallel (runs inside `docker build`)."""
    os.makedirs(CONT_DATA_DIR, exist_ok=True)
    with Pool(processes=N_PROC) as pool:
        pool.map(provision_subject, read_subjects(subjects_file))


# ---------------------------------------------------------------------------
# One suite execution (the `container` command, inside the container)
# ---------------------------------------------------------------------------

def collector_flags(mode, data_file):
    """Per-mode pytest flags (the showflakes/testinspect CLI contract)."""
    return {
        "testinspect": [f"--testinspect={data_file}"],
        "baseline": [f"--record-file={data_file}.tsv"],
        "shuffle": [f"--record-file={data_file}.tsv", "--shuffle"],
    }[mode]


def exec_suite(cont_name, *commands, subjects_dir=None, data_dir=None,
               run=sp.run):
    """Run the subject's pre-commands, then pytest with the plugin
    blacklist, our collectors, and the mode flags.  Directory overrides
    via FLAKE16_SUBJECTS_DIR / FLAKE16_DATA_DIR support the local
    (docker-less) mode."""
    subjects_dir = subjects_dir or os.environ.get("FLAKE16_SUBJECTS_DIR",
                                                  SUBJECTS_DIR)
    data_dir = data_dir or os.environ.get("FLAKE16_DATA_DIR", CONT_DATA_DIR)
    proj, mode, _ = cont_name.split("_", 2)
    checkout = os.path.join(subjects_dir, proj, proj)
    data_file = os.path.join(data_dir, cont_name)
    venv_bin = os.path.join(subjects_dir, proj, "venv", "bin")

    env = os.environ.copy()
    env["PATH"] = venv_bin + ":" + env["PATH"]

    for cmd in commands[:-1]:
        run(shlex.split(cmd), cwd=checkout, env=env, check=True)

    plugin_args = []
    for plug in COLLECT_PLUGINS:
        plugin_args += ["-p", plug]

    run(
        [*shlex.split(commands[-1]), *PLUGIN_BLACKLIST, *plugin_args,
         "--set-exitstatus", *collector_flags(mode, data_file)],
        timeout=CONT_TIMEOUT, cwd=checkout, check=True, env=env)


# ---------------------------------------------------------------------------
# Run launching (host side)
# ---------------------------------------------------------------------------

def docker_run_argv(cont_name, commands, host_data_dir):
    """The container invocation — argv-compatible with the reference."""
    return [
        "docker", "run", "-it",
        f"-v={host_data_dir}:{CONT_DATA_DIR}:rw", "--rm", "--init",
        "--cpus=1", f"--name={cont_name}", IMAGE_NAME, "python3",
        "experiment.py", "container", cont_name, *commands,
    ]


def local_run_argv(cont_name, commands):
    """Local (no-docker) mode: per-run isolation is a fresh python
    process instead of a container (FLAKE16_LOCAL_RUN=1)."""
    return [sys.executable, "-m", "flake16_framework_amd.cli", "container",
            cont_name, *commands]


def launch_run(args, runner=None, stdout_dir=STDOUT_DIR):
    """Launch one run; append its stdout; report (message, result)."""
    cont_name, commands = args
    host_data_dir = os.path.join(os.getcwd(), DATA_DIR)
    stdout_file = os.path.join(stdout_dir, cont_name)

    if os.environ.get("FLAKE16_LOCAL_RUN"):
        argv = local_run_argv(cont_name, commands)
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


def enumerate_runs(run_modes, subjects_file=SUBJECTS_FILE, n_runs=None):
    """All (container name, commands) pairs for the requested modes."""
    n_runs = n_runs or N_RUNS
    for subject in read_subjects(subjects_file):
        for mode in set(run_modes):
            for run_n in range(n_runs[mode]):
                yield f"{subject.proj}_{mode}_{run_n}", subject.commands


# ---------------------------------------------------------------------------
# Pool + progress + resumability
# ---------------------------------------------------------------------------

class ProgressMeter:
    """The reference pool meter's line format: a completion message per
    task, then "finished/remaining elapsed/eta" (minutes) with a trailing
    carriage return."""

    def __init__(self, total, out=None):
        # resolve the CURRENT sys.stdout at call time (an import-time
        # default would capture a stale stream under redirection)
        self.out = out or sys.stdout
        self.total = total
        self.finished = 0
        self.started = time.time()
        self.out.write(f"0/{total} 0/?\r")

    def task_done(self, message):
        self.finished += 1
        remain = self.total - self.finished
        elapsed = time.time() - self.started
        eta = elapsed / self.finished * remain
        self.out.write(f"{message}\n\r")
        self.out.write(f"{self.finished}/{remain} "
                       f"{round(elapsed / 60)}/{round(eta / 60)}\r")


def pooled_progress(pool, fn, args, out=None):
    """imap_unordered over shuffled args with the progress meter; yields
    each task's result."""
    random.shuffle(args)
    meter = ProgressMeter(len(args), out=out)
    for message, result in pool.imap_unordered(fn, args):
        meter.task_done(message)
        yield result


class CompletionLog:
    """Append-only completion log (log.txt): the crash-restart checkpoint
    of the run stage, one container name per line."""

    def __init__(self, path=LOG_FILE):
        self.path = path

    def completed(self):
        if not os.path.exists(self.path):
            return set()
        with open(self.path, "r") as fd:
            return {line.strip() for line in fd}

    def mark(self, cont_name):
        with open(self.path, "a") as fd:
            fd.write(f"{cont_name}\n")


def read_log(log_file=LOG_FILE):
    return sorted(CompletionLog(log_file).completed())


def drive_runs(*run_modes, subjects_file=SUBJECTS_FILE, n_runs=None,
               runner=None, log_file=LOG_FILE):
    """The resumable run driver: skip completed runs, launch the rest
    through a process pool, log completions, exit 1 on any failure."""
    os.makedirs(DATA_DIR, exist_ok=True)
    os.makedirs(STDOUT_DIR, exist_ok=True)

    log = CompletionLog(log_file)
    done = log.completed()
    args = [(cont_name, commands)
            for cont_name, commands in enumerate_runs(run_modes,
                                                      subjects_file, n_runs)
            if cont_name not in done]

    exitstatus = 0
    run_fn = launch_run if runner is None else \
        (lambda a: launch_run(a, runner=runner))

    with Pool(processes=N_PROC) as pool:
        for succeeded, cont_name in pooled_progress(pool, run_fn, args):
            if succeeded:
                log.mark(cont_name)
            else:
                exitstatus = 1

    sys.exit(exitstatus)
