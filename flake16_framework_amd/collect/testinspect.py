"""testinspect-equivalent pytest plugin.

The reference's `testinspect` plugin (empty git submodule) produces three
files per instrumented run, whose formats are fixed by the collation layer
(reference experiment.py:280-313; our dataset/collate.py):

  PATH.sqlite3  coverage DB with per-test dynamic contexts: tables
                context(id, context), file(id, path),
                line_bits(context_id, file_id, numbits)
  PATH.tsv      "t_exec \\t read_count \\t write_count \\t ctx_switches \\t
                max_threads \\t max_memory \\t nodeid" per test
                (Flake16 features 3-8)
  PATH.pkl      pickle (test_fn_ids {nid->fid}, test_fn_data {fid->7
                static metrics}, test_files set, churn {file->{line->n}})
                (features 9-15: AST Depth, Assertions, External Modules,
                Halstead Volume, Cyclomatic Complexity, Test LoC,
                Maintainability)

This reimplementation is self-contained: coverage via sys.settrace (the
numbits blob is a plain little-endian bitmap, matching
coverage.numbits/our decoder), rusage via psutil, static metrics via ast
(the original used radon, unavailable here — Halstead/CC/MI follow the
standard formulas on the ast node stream), churn via `git log -p` hunk
line numbers (an approximation of per-line change counts).

Register with `-p flake16_framework_amd.collect.testinspect`.
"""

import ast
import math
import os
import pickle
import sqlite3
import subprocess
import sys
import time

from pytest import hookimpl as _pytest_hookimpl


def pytest_addoption(parser):
    group = parser.getgroup("testinspect")
    group.addoption("--testinspect", action="store", default=None,
                    help="output path prefix for the instrumented run")


def nums_to_numbits(nums):
    """Line numbers -> coverage.py numbits blob (bit i of byte b = line
    b*8+i)."""
    if not nums:
        return b""
    nbytes = max(nums) // 8 + 1
    blob = bytearray(nbytes)
    for n in nums:
        blob[n // 8] |= 1 << (n % 8)
    return bytes(blob)


class _LineTracer:
    """Per-test line coverage over project files via sys.settrace."""

    def __init__(self, root):
        self.root = os.path.abspath(root)
        self.lines = {}   # file -> set(lineno)

    def __call__(self, frame, event, arg):
        if event == "call":
            fn = frame.f_code.co_filename
            if not fn.startswith(self.root) or "site-packages" in fn:
                return None
            return self._local
        return None

    def _local(self, frame, event, arg):
        if event == "line":
            self.lines.setdefault(frame.f_code.co_filename,
                                  set()).add(frame.f_lineno)
        return self._local


# ---------------------------------------------------------------------------
# static metrics (ast-based, radon-rule implementations)
#
# The original study computed Halstead Volume / Cyclomatic Complexity /
# Maintainability via radon 5.1.0 (reference requirements.txt:25).  radon
# is not installable here; these follow radon's documented rules:
#   CC: +1 per if/elif/ternary/assert/with; +1 per loop plus +1 for a
#       loop else; per try: +1 per except handler plus +1 for try-else;
#       +1 per comprehension generator plus +1 per condition in it;
#       boolean op chains add (#values - 1).
#   Halstead: operators are the arithmetic/boolean/comparison operator
#       instances (BinOp/UnaryOp/BoolOp/AugAssign/Compare ops); operands
#       are their direct Name/Constant/Attribute operands.
#       Volume = (N1 + N2) * log2(n1 + n2).
#   MI: max(0, (171 - 5.2 ln V - 0.23 CC - 16.2 ln LOC) * 100 / 171)
#       (the comment term is 0: radon's sin-of-comment-ratio applies to
#       commented code only).
# Fixture tests with hand-computed values: tests/test_static_metrics.py.
# ---------------------------------------------------------------------------
_HAL_OPERAND_TYPES = (ast.Name, ast.Constant, ast.Attribute)


def _ast_depth(node, depth=0):
    children = list(ast.iter_child_nodes(node))
    if not children:
        return depth
    return max(_ast_depth(c, depth + 1) for c in children)


def _hal_operand_name(n):
    if isinstance(n, ast.Name):
        return n.id
    if isinstance(n, ast.Attribute):
        return n.attr
    return repr(n.value)


def _halstead_volume(fn_node):
    operators, operands = [], []

    def operand(n):
        if isinstance(n, _HAL_OPERAND_TYPES):
            operands.append(_hal_operand_name(n))

    for n in ast.walk(fn_node):
        if isinstance(n, ast.BinOp):
            operators.append(type(n.op).__name__)
            operand(n.left)
            operand(n.right)
        elif isinstance(n, ast.UnaryOp):
            operators.append(type(n.op).__name__)
            operand(n.operand)
        elif isinstance(n, ast.BoolOp):
            operators.extend([type(n.op).__name__] * (len(n.values) - 1))
            for v in n.values:
                operand(v)
        elif isinstance(n, ast.AugAssign):
            operators.append(type(n.op).__name__)
            operand(n.target)
            operand(n.value)
        elif isinstance(n, ast.Compare):
            operators.extend(type(op).__name__ for op in n.ops)
            operand(n.left)
            for c in n.comparators:
                operand(c)
    n1, n2 = len(set(operators)), len(set(operands))
    N1, N2 = len(operators), len(operands)
    vocab = n1 + n2
    length = N1 + N2
    return length * math.log2(vocab) if vocab > 0 and length > 0 else 0.0


def _cyclomatic(fn_node):
    cc = 1
    for n in ast.walk(fn_node):
        if isinstance(n, (ast.If, ast.IfExp, ast.Assert, ast.With,
                          ast.AsyncWith)):
            cc += 1
        elif isinstance(n, (ast.For, ast.AsyncFor, ast.While)):
            cc += 1 + bool(n.orelse)
        elif isinstance(n, ast.Try):
            cc += len(n.handlers) + bool(n.orelse)
        elif isinstance(n, ast.BoolOp):
            cc += len(n.values) - 1
        elif isinstance(n, ast.comprehension):
            cc += 1 + len(n.ifs)
    return cc


def static_metrics(fn_node, module_imports):
    """The 7 static Flake16 features of one test function."""
    depth = _ast_depth(fn_node)
    assertions = sum(isinstance(n, ast.Assert) for n in ast.walk(fn_node))
    names = {n.id for n in ast.walk(fn_node) if isinstance(n, ast.Name)}
    attrs = {n.value.id for n in ast.walk(fn_node)
             if isinstance(n, ast.Attribute) and
             isinstance(n.value, ast.Name)}
    ext_modules = len(module_imports & (names | attrs))
    hv = _halstead_volume(fn_node)
    cc = _cyclomatic(fn_node)
    loc = (fn_node.end_lineno or fn_node.lineno) - fn_node.lineno + 1
    # maintainability index (standard SEI formula, clamped to [0, 100])
    mi = 171.0 - 5.2 * math.log(max(hv, 1.0)) - 0.23 * cc \
        - 16.2 * math.log(max(loc, 1))
    mi = max(0.0, min(100.0, mi * 100.0 / 171.0))
    return (depth, assertions, ext_modules, hv, cc, loc, mi)


def collect_static(test_files):
    """Parse the given test files -> (fn_metrics {(file, name) -> metrics},
    per-file imports)."""
    out = {}
    for path in test_files:
        try:
            with open(path, "r") as fd:
                tree = ast.parse(fd.read())
        except (OSError, SyntaxError):
            continue
        imports = set()
        for n in ast.walk(tree):
            if isinstance(n, ast.Import):
                imports |= {a.asname or a.name.split(".")[0] for a in n.names}
            elif isinstance(n, ast.ImportFrom) and n.module:
                imports |= {a.asname or a.name for a in n.names}
        for n in ast.walk(tree):
            if isinstance(n, (ast.FunctionDef, ast.AsyncFunctionDef)):
                out[(path, n.name)] = static_metrics(n, imports)
    return out


def collect_churn(root, files):
    """{relpath: {line: change_count}} from `git log -p` hunk headers —
    counts how many commits touched each (current-ish) line number."""
    churn = {}
    try:
        log = subprocess.run(
            ["git", "log", "-p", "--unified=0", "--no-color"],
            cwd=root, capture_output=True, text=True, timeout=300).stdout
    except (OSError, subprocess.TimeoutExpired):
        return churn
    import re
    cur_file = None
    for line in log.splitlines():
        m = re.match(r"\+\+\+ b/(.*)", line)
        if m:
            cur_file = m.group(1)
            continue
        m = re.match(r"@@ -\d+(?:,\d+)? \+(\d+)(?:,(\d+))? @@", line)
        if m and cur_file and (not files or cur_file in files):
            start = int(m.group(1))
            count = int(m.group(2) or "1")
            cf = churn.setdefault(cur_file, {})
            for ln in range(start, start + count):
                cf[ln] = cf.get(ln, 0) + 1
    return churn


# ---------------------------------------------------------------------------
# the plugin
# ---------------------------------------------------------------------------
class TestInspectPlugin:
    def __init__(self, prefix, root):
        self.prefix = prefix
        self.root = os.path.abspath(root)
        self.cov = {}       # nid -> {abspath: set(lines)}
        self.rusage = {}    # nid -> [6 floats]
        self.test_items = {}   # nid -> (file, function name)

    def pytest_collection_modifyitems(self, config, items):
        for item in items:
            path = str(getattr(item, "path", item.fspath))
            name = item.name.split("[")[0]
            self.test_items[item.nodeid] = (path, name)

    # hookwrapper: the default pytest_runtest_call runs the test inside
    # our tracing (a plain hookimpl that called item.runtest() itself
    # would EXECUTE THE TEST TWICE — the default impl still runs).
    @_pytest_hookimpl(hookwrapper=True)
    def pytest_runtest_call(self, item):
        proc = None
        try:
            import psutil
            proc = psutil.Process()
            io0 = proc.io_counters() if hasattr(proc, "io_counters") else None
            cs0 = proc.num_ctx_switches()
        except Exception:
            io0 = cs0 = None

        tracer = _LineTracer(self.root)
        t0 = time.time()
        old = sys.gettrace()
        sys.settrace(tracer)
        try:
            yield
        finally:
            sys.settrace(old)
            t_exec = time.time() - t0
            reads = writes = switches = threads = mem = 0.0
            if proc is not None:
                try:
                    if io0 is not None:
                        io1 = proc.io_counters()
                        reads = io1.read_count - io0.read_count
                        writes = io1.write_count - io0.write_count
                    cs1 = proc.num_ctx_switches()
                    switches = ((cs1.voluntary + cs1.involuntary)
                                - (cs0.voluntary + cs0.involuntary))
                    threads = proc.num_threads()
                    mem = proc.memory_info().rss
                except Exception:
                    pass
            self.rusage[item.nodeid] = [t_exec, reads, writes, switches,
                                        threads, mem]
            self.cov[item.nodeid] = tracer.lines

    def write_outputs(self):
        # -- sqlite3 (coverage schema subset the collator reads) ----------
        db = f"{self.prefix}.sqlite3"
        if os.path.exists(db):
            os.remove(db)
        con = sqlite3.connect(db)
        cur = con.cursor()
        cur.execute("CREATE TABLE context (id INTEGER PRIMARY KEY, "
                    "context TEXT)")
        cur.execute("CREATE TABLE file (id INTEGER PRIMARY KEY, path TEXT)")
        cur.execute("CREATE TABLE line_bits (context_id INT, file_id INT, "
                    "numbits BLOB)")
        file_ids = {}
        for ci, (nid, cov) in enumerate(sorted(self.cov.items()), start=1):
            cur.execute("INSERT INTO context VALUES (?, ?)", (ci, nid))
            for path, lines in cov.items():
                if path not in file_ids:
                    file_ids[path] = len(file_ids) + 1
                    cur.execute("INSERT INTO file VALUES (?, ?)",
                                (file_ids[path], path))
                cur.execute(
                    "INSERT INTO line_bits VALUES (?, ?, ?)",
                    (ci, file_ids[path], nums_to_numbits(sorted(lines))))
        con.commit()
        con.close()

        # -- rusage tsv ----------------------------------------------------
        with open(f"{self.prefix}.tsv", "w") as fd:
            for nid, ru in sorted(self.rusage.items()):
                fd.write("\t".join(str(x) for x in ru) + f"\t{nid}\n")

        # -- static pickle ---------------------------------------------
        test_files = {path for path, _ in self.test_items.values()}
        fn_metrics = collect_static(test_files)
        test_fn_ids, test_fn_data = {}, {}
        fid_of = {}
        for nid, key in self.test_items.items():
            if key not in fn_metrics:
                continue
            if key not in fid_of:
                fid_of[key] = len(fid_of) + 1
                test_fn_data[fid_of[key]] = fn_metrics[key]
            test_fn_ids[nid] = fid_of[key]

        rel_test_files = {os.path.relpath(p, self.root) for p in test_files}
        churn = collect_churn(self.root, None)
        with open(f"{self.prefix}.pkl", "wb") as fd:
            pickle.dump((test_fn_ids, test_fn_data, rel_test_files, churn),
                        fd)


def pytest_configure(config):
    prefix = config.getoption("--testinspect", default=None)
    if prefix:
        plugin = TestInspectPlugin(prefix, os.getcwd())
        config._testinspect = plugin
        config.pluginmanager.register(plugin, "testinspect-collector")


def pytest_sessionfinish(session, exitstatus):
    plugin = getattr(session.config, "_testinspect", None)
    if plugin is not None:
        plugin.write_outputs()
