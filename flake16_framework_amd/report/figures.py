"""The `figures` stage: LaTeX tables and plots from the stage artifacts.

Emits the reference's eight .tex artifacts (reference experiment.py:533-690
defines the OUTPUT BYTES; the implementation here is this framework's own —
a block-structured ``TexTable`` renderer plus one builder per artifact):

  tests.tex     per-project stars/tests/NOD/OD counts + totals
  req-runs.tex  cumulative distribution of runs-to-detection (pgfplots)
  corr.tex      Spearman correlation matrix of the 16 features
  nod-top.tex / od-top.tex   top-10 configs by F1, FlakeFlagger vs Flake16
                side by side
  nod-comp.tex / od-comp.tex per-project comparison of the paper's baseline
                vs extended configs
  shap.tex      mean |SHAP| per feature, NOD and OD side by side

Output format contracts preserved byte-for-byte: cell rendering ("%.2f"
floats, "-" for zero ints, "%.3f" SHAP, gray \\cellcolor scaled by |corr|),
\\rowcolor{gray!20} zebra on odd rows, \\midrule between blocks, and the
25-bin normalized detection CDF.

The subject list (repo names for tests.tex row order and the stars column)
comes from subjects.txt when present — matching the reference, which always
reads it — and falls back to the tests.json project keys offline.  The
GitHub stars fetch (reference experiment.py:533-535) is optional: with no
network (or offline=True) the star count is -1.
"""

import json
import os
import pickle

import numpy as np
from scipy import stats

from ..constants import (
    FEATURE_NAMES, FLAKY, OD_FLAKY, SCORES_FILE, SHAP_FILE, SUBJECTS_FILE,
    TESTS_FILE,
)

COMPARISON_CONFIGS = {
    # (baseline config, extended config) per flaky type — the pairs the
    # paper compares (reference experiment.py:672-682).
    "nod": (("NOD", "FlakeFlagger", "None", "Tomek Links", "Extra Trees"),
            ("NOD", "Flake16", "PCA", "SMOTE", "Extra Trees")),
    "od": (("OD", "FlakeFlagger", "None", "SMOTE Tomek", "Extra Trees"),
           ("OD", "Flake16", "Scaling", "SMOTE", "Random Forest")),
}

# Detection-CDF sampling grid: 25 bins at 100-run steps, normalized by the
# final bin (the labeling budget is 2,500 runs per mode).
CDF_STEP = 100
CDF_BINS = 25


# ---------------------------------------------------------------------------
# Cell formatting + table rendering
# ---------------------------------------------------------------------------

class CellFormat:
    """Renders one table cell to LaTeX source.

    float_fmt: printf format for floats; zero_int_dash: render integer 0 as
    "-"; shade: prefix floats with a gray \\cellcolor scaled by |value|.
    """

    def __init__(self, float_fmt="%.2f", zero_int_dash=True, shade=False):
        self.float_fmt = float_fmt
        self.zero_int_dash = zero_int_dash
        self.shade = shade

    def __call__(self, cell):
        if isinstance(cell, str):
            return cell
        if isinstance(cell, float):
            text = self.float_fmt % cell
            if self.shade:
                return "\\cellcolor{gray!%d} %s" % (int(50 * abs(cell)), text)
            return text
        if isinstance(cell, (int, np.integer)):
            if self.zero_int_dash and cell == 0:
                return "-"
            return str(cell)
        return str(cell)


#: the three cell styles the artifacts use
cellfn_default = CellFormat()
cellfn_corr = CellFormat(shade=True, zero_int_dash=False)
cellfn_shap = CellFormat(float_fmt="%.3f")


class TexTable:
    """Block-structured LaTeX table body.

    Blocks are separated by \\midrule; rows within a block are optionally
    zebra-shaded (\\rowcolor{gray!20} before every odd row).
    """

    def __init__(self, zebra=True, fmt=cellfn_default):
        self.blocks = [[]]
        self.zebra = zebra
        self.fmt = fmt

    def row(self, *cells):
        self.blocks[-1].append(list(cells))
        return self

    def rule(self):
        self.blocks.append([])
        return self

    def render(self):
        pieces = []
        for bi, block in enumerate(self.blocks):
            if bi:
                pieces.append("\\midrule\n")
            for ri, cells in enumerate(block):
                if self.zebra and ri % 2:
                    pieces.append("\\rowcolor{gray!20}\n")
                pieces.append(" & ".join(self.fmt(c) for c in cells)
                              + " \\\\\n")
        return "".join(pieces)

    def write(self, path):
        with open(path, "w") as fd:
            fd.write(self.render())


def write_table(table_file, tab, rowcol=True, cellfn=cellfn_default):
    """Nested-list compatibility wrapper: tab = [block][row][cell]."""
    t = TexTable(zebra=rowcol, fmt=cellfn)
    for bi, block in enumerate(tab):
        if bi:
            t.rule()
        for cells in block:
            t.row(*cells)
    t.write(table_file)


# ---------------------------------------------------------------------------
# Subjects
# ---------------------------------------------------------------------------

def load_subjects(tests, subjects_file=None):
    """[(proj, repo)] — from subjects.txt when it describes THIS dataset
    (reference row order and 'owner/repo' first column), else the
    tests.json keys.  A synthetic tests.json has proj00..proj25 keys, so
    the real study subject list must not be applied to it."""
    path = subjects_file or SUBJECTS_FILE
    if os.path.exists(path):
        from ..orchestrate.runner import read_subjects
        subjects = [(s.proj, s.repo) for s in read_subjects(path)]
        if all(proj in tests for proj, _ in subjects):
            return subjects
    return [(proj, proj) for proj in tests.keys()]


def get_n_stars(repo, offline=False):
    if offline:
        return -1
    try:
        import requests
        info = requests.get(f"https://api.github.com/repos/{repo}",
                            timeout=10).json()
        return info.get("stargazers_count", -1)
    except Exception:
        return -1


# ---------------------------------------------------------------------------
# Artifact builders
# ---------------------------------------------------------------------------

class DatasetSummary:
    """Single pass over tests.json: per-project test/NOD/OD counts, the
    runs-to-detection histograms, and the feature matrix."""

    def __init__(self, tests, subjects):
        self.counts = {proj: [0, 0, 0] for proj, _ in subjects}  # n, nod, od
        self.req_runs = {FLAKY: {}, OD_FLAKY: {}}
        self.features = []
        for proj, _ in subjects:
            for req_runs, label, *feats in tests[proj].values():
                self.counts[proj][0] += 1
                if label in (FLAKY, OD_FLAKY):
                    self.counts[proj][1 if label == FLAKY else 2] += 1
                    hist = self.req_runs[label]
                    hist[req_runs] = hist.get(req_runs, 0) + 1
                self.features.append(feats)

    def tests_table(self, subjects, offline):
        t = TexTable()
        totals = [0, 0, 0, 0]
        for proj, repo in subjects:
            n, nod, od = self.counts[proj]
            stars = get_n_stars(repo, offline)
            t.row(repo, stars, n, nod, od)
            for i, v in enumerate((stars, n, nod, od)):
                totals[i] += v
        t.rule()
        t.row("{\\bf Total}", *totals)
        return t

    def corr_table(self):
        corr = stats.spearmanr(self.features).correlation
        t = TexTable(zebra=False, fmt=cellfn_corr)
        for i, name in enumerate(FEATURE_NAMES):
            t.row(name, *corr[i])
        return t


def get_req_runs_plot_coords(req_runs):
    """Normalized cumulative detection counts at 100-run steps.

    Counts stay integers until the final division so the rendered floats
    match the reference byte-for-byte."""
    edges = [CDF_STEP * (i + 1) for i in range(CDF_BINS)]
    cum = [sum(freq for runs, freq in req_runs.items() if runs <= edge)
           for edge in edges]
    denom = cum[-1] or 1
    return " ".join(f"({edge},{count / denom})"
                    for edge, count in zip(edges, cum))


def render_req_runs_plot(req_runs_nod, req_runs_od):
    marks = (("x", "NOD", req_runs_nod), ("o", "OD", req_runs_od))
    lines = []
    for mark, legend, hist in marks:
        coords = get_req_runs_plot_coords(hist)
        lines.append(
            f"\\addplot[mark={mark},only marks] coordinates {{{coords}}};\n"
            f"\\addlegendentry{{{legend}}}")
    return "\n".join(lines)


def write_req_runs_plot(req_runs_nod, req_runs_od, path="req-runs.tex"):
    with open(path, "w") as fd:
        fd.write(render_req_runs_plot(req_runs_nod, req_runs_od))


def get_top_tables(scores, top_n=10):
    """Per flaky type: the top-N configs of each feature set by F1 (rows
    where F1 is defined), FlakeFlagger and Flake16 columns side by side.

    Returns the [block][row][cell] nesting write_table consumes."""
    ranked = {}   # (flaky_type, feature_set) -> [(axes..., tt, tp, f1)]
    for keys, (t_train, t_test, _, total) in scores.items():
        f1 = total[-1]
        if f1 is None:
            continue
        bucket = ranked.setdefault((keys[0], keys[1]), [])
        bucket.append((*keys[2:], t_train, t_test, f1))

    tables = []
    for flaky in ("NOD", "OD"):
        left = sorted(ranked.get((flaky, "FlakeFlagger"), []),
                      key=lambda row: -row[-1])
        right = sorted(ranked.get((flaky, "Flake16"), []),
                       key=lambda row: -row[-1])
        n = min(top_n, len(left), len(right))
        tables.append([[left[i] + right[i] for i in range(n)]])
    return tuple(tables)


def comparison_table(scores, orig_keys, ext_keys):
    """Per-project confusion+PRF of the baseline config next to the
    extended config, rows where every metric is defined, plus Total."""
    _, _, orig, orig_total = scores[orig_keys]
    _, _, ext, ext_total = scores[ext_keys]
    t = TexTable()
    for proj, orig_row in orig.items():
        cells = orig_row + ext[proj]
        if any(v is None for v in cells):
            continue
        t.row(proj, *cells)
    t.rule()
    t.row("{\\bf Total}", *orig_total, *ext_total)
    return t


def shap_table(shap_nod, shap_od):
    """Mean |SHAP| per feature, each flaky type sorted descending."""
    def ranking(phi):
        means = np.abs(phi).mean(axis=0)
        return sorted(zip(FEATURE_NAMES, means), key=lambda nv: -nv[1])

    t = TexTable(fmt=cellfn_shap)
    for (n_name, n_val), (o_name, o_val) in zip(ranking(shap_nod),
                                                ranking(shap_od)):
        t.row(n_name, float(n_val), o_name, float(o_val))
    return t


# ---------------------------------------------------------------------------
# Driver
# ---------------------------------------------------------------------------

def write_figures(tests_file=TESTS_FILE, scores_file=SCORES_FILE,
                  shap_file=SHAP_FILE, subjects=None, offline=False,
                  out_dir="."):
    """Emit all eight artifacts into out_dir.

    subjects: optional [(proj, repo), ...] override; default resolves via
    load_subjects (subjects.txt if present, else tests.json keys)."""
    with open(tests_file, "r") as fd:
        tests = json.load(fd)
    if subjects is None:
        subjects = load_subjects(tests)

    dest = lambda name: os.path.join(out_dir, name)

    summary = DatasetSummary(tests, subjects)
    summary.tests_table(subjects, offline).write(dest("tests.tex"))
    write_req_runs_plot(summary.req_runs[FLAKY], summary.req_runs[OD_FLAKY],
                        dest("req-runs.tex"))
    summary.corr_table().write(dest("corr.tex"))

    with open(scores_file, "rb") as fd:
        scores = pickle.load(fd)

    tab_nod_top, tab_od_top = get_top_tables(scores)
    write_table(dest("nod-top.tex"), tab_nod_top)
    write_table(dest("od-top.tex"), tab_od_top)

    for name, (orig_keys, ext_keys) in COMPARISON_CONFIGS.items():
        if orig_keys not in scores or ext_keys not in scores:
            # partial sweep (e.g. --cells subsets): emit an empty table
            # instead of failing the whole stage
            TexTable().write(dest(f"{name}-comp.tex"))
            continue
        comparison_table(scores, orig_keys, ext_keys).write(
            dest(f"{name}-comp.tex"))

    with open(shap_file, "rb") as fd:
        shap_nod, shap_od = pickle.load(fd)
    shap_table(shap_nod, shap_od).write(dest("shap.tex"))
