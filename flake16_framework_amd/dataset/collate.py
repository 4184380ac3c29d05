"""Collation of raw per-run data files into the per-project accumulator.

Data-plane parity with the reference (experiment.py:242-336):
  data/<proj>_<mode>_<runN>.<ext> where
    mode baseline|shuffle, ext tsv : one "outcome\\tnodeid" line per test
    mode testinspect, ext sqlite3  : coverage.py DB with per-test contexts
    mode testinspect, ext tsv      : "6 rusage floats \\t nodeid" lines
    mode testinspect, ext pkl      : pickle (test_fn_ids, test_fn_data,
                                     test_files, churn)

The per-project accumulator is
  collated[proj] = [test_data, test_fn_data, test_files, churn]
  test_data[nid] = [runs, cov, rusage, fid]
    runs[mode]   = [n_runs, n_fails, min_failing_run, min_passing_run]
    cov          = {relpath: set(line_numbers)}
"""

import os
import pickle
import sqlite3

from ..constants import DATA_DIR, SUBJECTS_DIR


def iter_data_dir(data_dir=DATA_DIR):
    """Yield (path, proj, mode, run_n, ext) for each file in data/."""
    for file_name in os.listdir(data_dir):
        proj, mode, rest = file_name.split("_", 2)
        run_n, ext = rest.split(".", 1)
        yield os.path.join(data_dir, file_name), proj, mode, int(run_n), ext


def iter_tsv(fd, n_split):
    for line in fd:
        yield line.strip().split("\t", n_split)


def get_test_data_nid(collated_proj, nid):
    return collated_proj[0].setdefault(nid, [{}, {}, None, None])


def update_collated_runs(fd, mode, run_n, collated_proj):
    """Accumulate pass/fail statistics from one run's outcome TSV.

    Failure detection is substring-based ("failed" in outcome), matching the
    showflakes output contract (reference experiment.py:260-277).
    """
    for outcome, nid in iter_tsv(fd, 1):
        runs_nid = get_test_data_nid(collated_proj, nid)[0]
        runs_mode = runs_nid.setdefault(mode, [0, 0, None, None])
        runs_mode[0] += 1

        if "failed" in outcome:
            runs_mode[1] += 1
            runs_mode[2] = run_n if runs_mode[2] is None else min(runs_mode[2], run_n)
        else:
            runs_mode[3] = run_n if runs_mode[3] is None else min(runs_mode[3], run_n)


def _numbits_to_nums(numbits_blob):
    """Decode coverage.py's numbits packing: bit i of byte b set means line
    number b*8+i was executed (little-endian bit order within each byte)."""
    nums = []
    for byte_i, byte in enumerate(numbits_blob):
        for bit_i in range(8):
            if byte & (1 << bit_i):
                nums.append(byte_i * 8 + bit_i)
    return nums


def update_collated_cov(con, proj, collated_proj, subjects_dir=SUBJECTS_DIR):
    """Ingest a coverage.py sqlite3 DB with per-test dynamic contexts.

    Schema consumed (coverage 6.2): context(id, context), file(id, path),
    line_bits(context_id, file_id, numbits).  File paths are stored relative
    to the project checkout (reference experiment.py:280-299).
    """
    cur = con.cursor()

    nodeids = dict(cur.execute("SELECT id, context FROM context").fetchall())

    proj_dir = os.path.join(subjects_dir, proj, proj)
    files = {
        file_id: os.path.relpath(path, start=proj_dir)
        for file_id, path in cur.execute("SELECT id, path FROM file").fetchall()
    }

    rows = cur.execute("SELECT context_id, file_id, numbits FROM line_bits")
    for context_id, file_id, nb in rows.fetchall():
        cov_nid = get_test_data_nid(collated_proj, nodeids[context_id])[1]
        cov_nid[files[file_id]] = set(_numbits_to_nums(nb))


def update_collated_rusage(fd, collated_proj):
    """Rusage TSV: 6 floats (Execution Time, Read Count, Write Count,
    Context Switches, Max. Threads, Max. Memory) then the nodeid."""
    for *rusage, nid in iter_tsv(fd, 6):
        get_test_data_nid(collated_proj, nid)[2] = [float(x) for x in rusage]


def update_collated_static(fd, collated_proj):
    """Static-metrics pickle: (test_fn_ids {nid->fid}, test_fn_data
    {fid->7 static metrics}, test_files set, churn {file->{line->count}})."""
    test_fn_ids, test_fn_data, test_files, churn = pickle.load(fd)
    collated_proj[1] = test_fn_data
    collated_proj[2] = test_files
    collated_proj[3] = churn

    for nid, fid in test_fn_ids.items():
        get_test_data_nid(collated_proj, nid)[3] = fid


def get_collated(data_dir=DATA_DIR, subjects_dir=SUBJECTS_DIR):
    """Walk data/ and build the full collated structure."""
    collated = {}

    for file_name, proj, mode, run_n, ext in iter_data_dir(data_dir):
        collated_proj = collated.setdefault(proj, [{}, None, None, None])

        if mode in {"baseline", "shuffle"}:
            with open(file_name, "r") as fd:
                update_collated_runs(fd, mode, run_n, collated_proj)
        elif mode == "testinspect":
            if ext == "sqlite3":
                with sqlite3.connect(file_name) as con:
                    update_collated_cov(con, proj, collated_proj, subjects_dir)
            elif ext == "tsv":
                with open(file_name, "r") as fd:
                    update_collated_rusage(fd, collated_proj)
            elif ext == "pkl":
                with open(file_name, "rb") as fd:
                    update_collated_static(fd, collated_proj)

    return collated
