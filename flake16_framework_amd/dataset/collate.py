"""Collation of raw per-run data files into per-project records.

Data-plane parity with the reference (the FILE FORMATS of
experiment.py:242-336 are the contract; the in-memory model here is this
framework's own record classes):

  data/<proj>_<mode>_<runN>.<ext> where
    mode baseline|shuffle, ext tsv : one "outcome\\tnodeid" line per test
    mode testinspect, ext sqlite3  : coverage.py DB with per-test contexts
    mode testinspect, ext tsv      : "6 rusage floats \\t nodeid" lines
    mode testinspect, ext pkl      : pickle (test_fn_ids, test_fn_data,
                                     test_files, churn)

The model:
  ProjectData  one subject project's accumulated evidence
    .tests[nid] -> TestRecord
    .fn_metrics {fid: 7 static metrics}, .test_files set,
    .churn {file: {line: change_count}}
  TestRecord   one test's evidence across all runs
    .runs[mode] -> RunStats, .coverage {relpath: set(lines)},
    .rusage [6 floats], .static_id fid
  RunStats     pass/fail accounting for one mode
"""

import os
import pickle
import sqlite3

from ..constants import DATA_DIR, SUBJECTS_DIR


class RunStats:
    """Pass/fail accounting for one (test, mode) across repeated runs.

    Failure detection is substring-based ("failed" in outcome), matching
    the showflakes output contract."""

    __slots__ = ("total", "failures", "first_fail", "first_pass")

    def __init__(self):
        self.total = 0
        self.failures = 0
        self.first_fail = None
        self.first_pass = None

    def observe(self, outcome, run_n):
        self.total += 1
        if "failed" in outcome:
            self.failures += 1
            if self.first_fail is None or run_n < self.first_fail:
                self.first_fail = run_n
        else:
            if self.first_pass is None or run_n < self.first_pass:
                self.first_pass = run_n

    @property
    def never_failed(self):
        return self.failures == 0

    @property
    def always_failed(self):
        return self.failures == self.total


class TestRecord:
    """Everything collected about one test node."""

    __slots__ = ("runs", "coverage", "rusage", "static_id")

    def __init__(self):
        self.runs = {}            # mode -> RunStats
        self.coverage = None      # {relpath: set(line_numbers)}
        self.rusage = None        # [6 floats]
        self.static_id = None     # fid into ProjectData.fn_metrics

    def stats(self, mode):
        if mode not in self.runs:
            self.runs[mode] = RunStats()
        return self.runs[mode]

    @property
    def complete(self):
        """All four evidence parts present (reference drops others)."""
        return bool(self.runs) and bool(self.coverage) and \
            bool(self.rusage) and self.static_id is not None


def _tsv_rows(fd, n_split):
    for line in fd:
        yield line.strip().split("\t", n_split)


def _numbits_to_nums(numbits_blob):
    """Decode coverage.py's numbits packing: bit i of byte b set means line
    number b*8+i was executed (little-endian bit order within each byte)."""
    nums = []
    for byte_i, byte in enumerate(numbits_blob):
        for bit_i in range(8):
            if byte & (1 << bit_i):
                nums.append(byte_i * 8 + bit_i)
    return nums


class ProjectData:
    """One project's accumulated evidence + the per-file ingestors."""

    def __init__(self, proj, subjects_dir=SUBJECTS_DIR):
        self.proj = proj
        self.subjects_dir = subjects_dir
        self.tests = {}           # nid -> TestRecord
        self.fn_metrics = None    # {fid: [7 static metrics]}
        self.test_files = None    # set of test file paths
        self.churn = None         # {file: {line: change_count}}

    def record(self, nid):
        if nid not in self.tests:
            self.tests[nid] = TestRecord()
        return self.tests[nid]

    # -- ingestors (one per raw-file kind) --------------------------------
    def add_outcomes(self, fd, mode, run_n):
        """One run's showflakes outcome TSV."""
        for outcome, nid in _tsv_rows(fd, 1):
            self.record(nid).stats(mode).observe(outcome, run_n)

    def add_coverage_db(self, con):
        """coverage.py 6.2 sqlite3 DB with per-test dynamic contexts
        (tables context / file / line_bits); file paths become relative
        to the project checkout."""
        cur = con.cursor()
        contexts = dict(cur.execute("SELECT id, context FROM context"))
        checkout = os.path.join(self.subjects_dir, self.proj, self.proj)
        rel_paths = {
            file_id: os.path.relpath(path, start=checkout)
            for file_id, path in cur.execute("SELECT id, path FROM file")
        }
        for ctx_id, file_id, blob in cur.execute(
                "SELECT context_id, file_id, numbits FROM line_bits"):
            rec = self.record(contexts[ctx_id])
            if rec.coverage is None:
                rec.coverage = {}
            rec.coverage[rel_paths[file_id]] = set(_numbits_to_nums(blob))

    def add_rusage(self, fd):
        """Rusage TSV: 6 floats (Execution Time, Read Count, Write Count,
        Context Switches, Max. Threads, Max. Memory) then the nodeid."""
        for *values, nid in _tsv_rows(fd, 6):
            self.record(nid).rusage = [float(v) for v in values]

    def add_static(self, fd):
        """Static-metrics pickle: (test_fn_ids {nid->fid}, test_fn_data
        {fid->7 metrics}, test_files set, churn {file->{line->count}})."""
        fn_ids, self.fn_metrics, self.test_files, self.churn = \
            pickle.load(fd)
        for nid, fid in fn_ids.items():
            self.record(nid).static_id = fid

    @property
    def complete(self):
        """Static parts present (projects without them are dropped)."""
        return self.fn_metrics is not None and \
            bool(self.test_files) and bool(self.churn) and bool(self.tests)


def iter_data_dir(data_dir=DATA_DIR):
    """Yield (path, proj, mode, run_n, ext) for each file in data/."""
    for file_name in os.listdir(data_dir):
        proj, mode, rest = file_name.split("_", 2)
        run_n, ext = rest.split(".", 1)
        yield os.path.join(data_dir, file_name), proj, mode, int(run_n), ext


def collate(data_dir=DATA_DIR, subjects_dir=SUBJECTS_DIR):
    """Walk data/ and build {proj: ProjectData}."""
    projects = {}

    for path, proj, mode, run_n, ext in iter_data_dir(data_dir):
        if proj not in projects:
            projects[proj] = ProjectData(proj, subjects_dir)
        data = projects[proj]

        if mode in ("baseline", "shuffle"):
            with open(path, "r") as fd:
                data.add_outcomes(fd, mode, run_n)
        elif mode == "testinspect":
            if ext == "sqlite3":
                with sqlite3.connect(path) as con:
                    data.add_coverage_db(con)
            elif ext == "tsv":
                with open(path, "r") as fd:
                    data.add_rusage(fd)
            elif ext == "pkl":
                with open(path, "rb") as fd:
                    data.add_static(fd)

    return projects
