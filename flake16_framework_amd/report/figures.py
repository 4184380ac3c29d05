"""The `figures` stage: LaTeX tables and plots from the stage artifacts.

Semantic port of the reference's reporting layer (experiment.py:533-690):
  tests.tex     per-project stars/tests/NOD/OD counts + totals
  req-runs.tex  cumulative distribution of runs-to-detection (pgfplots)
  corr.tex      Spearman correlation matrix of the 16 features
  nod-top.tex / od-top.tex   top-10 configs by F1, FlakeFlagger vs Flake16
                side by side
  nod-comp.tex / od-comp.tex per-project comparison of the paper's baseline
                vs extended configs
  shap.tex      mean |SHAP| per feature, NOD and OD side by side

The GitHub stars fetch (reference experiment.py:533-535) is optional here:
with no network (or offline=True) the star count is -1.
"""

import json
import pickle

import numpy as np
from scipy import stats

from ..constants import (
    FEATURE_NAMES, FLAKY, OD_FLAKY, SCORES_FILE, SHAP_FILE, TESTS_FILE,
)

COMPARISON_CONFIGS = {
    # (baseline config, extended config) per flaky type — the pairs the
    # paper compares (reference experiment.py:672-682).
    "nod": (("NOD", "FlakeFlagger", "None", "Tomek Links", "Extra Trees"),
            ("NOD", "Flake16", "PCA", "SMOTE", "Extra Trees")),
    "od": (("OD", "FlakeFlagger", "None", "SMOTE Tomek", "Extra Trees"),
           ("OD", "Flake16", "Scaling", "SMOTE", "Random Forest")),
}


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


def get_req_runs_plot_coords(req_runs):
    coords = [[100 * (i + 1), 0] for i in range(25)]
    for c in coords:
        for runs, freq in req_runs.items():
            c[1] += (runs <= c[0]) * freq
    denom = coords[24][1] or 1
    return " ".join(f"({x},{y / denom})" for x, y in coords)


def write_req_runs_plot(req_runs_nod, req_runs_od, path="req-runs.tex"):
    with open(path, "w") as fd:
        coords = get_req_runs_plot_coords(req_runs_nod)
        fd.write(f"\\addplot[mark=x,only marks] coordinates {{{coords}}};\n")
        fd.write("\\addlegendentry{NOD}\n")
        coords = get_req_runs_plot_coords(req_runs_od)
        fd.write(f"\\addplot[mark=o,only marks] coordinates {{{coords}}};\n")
        fd.write("\\addlegendentry{OD}")


def get_top_tables(scores, top_n=10):
    """Bucket configs by (flaky-type, feature-set), drop F=None, sort by F1
    desc, pair FlakeFlagger/Flake16 rows side by side."""
    configs = [[] for _ in range(4)]
    for config_keys in scores:
        flaky_type, feature_set, *rest = config_keys
        t_train, t_test, _, (*_, f) = scores[config_keys]
        i = 2 * (flaky_type == "OD") + (feature_set == "Flake16")
        configs[i].append((*rest, t_train, t_test, f))

    for i in range(4):
        configs[i] = [c for c in configs[i] if c[-1] is not None]
        configs[i] = sorted(configs[i], key=lambda c: -c[-1])

    n_nod = min(top_n, len(configs[0]), len(configs[1]))
    n_od = min(top_n, len(configs[2]), len(configs[3]))
    tab_nod = [[configs[0][i] + configs[1][i] for i in range(n_nod)]]
    tab_od = [[configs[2][i] + configs[3][i] for i in range(n_od)]]
    return tab_nod, tab_od


def get_comparison_table(scores_orig, scores_ext):
    orig, orig_total = scores_orig[2:]
    ext, ext_total = scores_ext[2:]
    tab = []
    for proj, orig_proj in orig.items():
        if all(all(x is not None for x in y) for y in (orig_proj, ext[proj])):
            tab.append([proj, *orig_proj, *ext[proj]])
    return [tab, [["{\\bf Total}", *orig_total, *ext_total]]]


def get_shap_table(shap_nod, shap_od):
    shap_nod = sorted(zip(FEATURE_NAMES, abs(shap_nod).mean(axis=0)),
                      key=lambda x: -x[1])
    shap_od = sorted(zip(FEATURE_NAMES, abs(shap_od).mean(axis=0)),
                     key=lambda x: -x[1])
    return [[shap_nod[i] + shap_od[i] for i in range(len(FEATURE_NAMES))]]


def cellfn_default(cell):
    if isinstance(cell, str):
        return cell
    if isinstance(cell, float):
        return "%.2f" % cell
    if isinstance(cell, (int, np.integer)):
        return "-" if cell == 0 else str(cell)
    return str(cell)


def cellfn_corr(cell):
    if isinstance(cell, str):
        return cell
    if isinstance(cell, float):
        return "\\cellcolor{gray!%d} %.2f" % (int(50 * abs(cell)), cell)
    return str(cell)


def cellfn_shap(cell):
    if isinstance(cell, str):
        return cell
    if isinstance(cell, float):
        return "%.3f" % cell
    return str(cell)


def write_table(table_file, tab, rowcol=True, cellfn=cellfn_default):
    with open(table_file, "w") as fd:
        for i, tab_i in enumerate(tab):
            if i:
                fd.write("\\midrule\n")
            for j, tab_j in enumerate(tab_i):
                if rowcol and j % 2:
                    fd.write("\\rowcolor{gray!20}\n")
                fd.write(" & ".join([cellfn(c) for c in tab_j]) + " \\\\\n")


def write_figures(tests_file=TESTS_FILE, scores_file=SCORES_FILE,
                  shap_file=SHAP_FILE, subjects=None, offline=False,
                  out_dir="."):
    """subjects: optional [(proj, repo), ...]; default derives proj names
    from tests.json with repo == proj (the real study reads subjects.txt)."""
    import os

    with open(tests_file, "r") as fd:
        tests = json.load(fd)

    if subjects is None:
        subjects = [(proj, proj) for proj in tests.keys()]

    tab_tests = [[], [["{\\bf Total}", *[0] * 4]]]
    req_runs_nod, req_runs_od = {}, {}
    features = []

    for i, (proj, repo) in enumerate(subjects):
        tab_tests[0].append(
            [repo, get_n_stars(repo, offline), len(tests[proj]), 0, 0])
        for (req_runs, label_nid, *features_nid) in tests[proj].values():
            if label_nid == FLAKY:
                tab_tests[0][i][3] += 1
                req_runs_nod[req_runs] = req_runs_nod.get(req_runs, 0) + 1
            elif label_nid == OD_FLAKY:
                tab_tests[0][i][4] += 1
                req_runs_od[req_runs] = req_runs_od.get(req_runs, 0) + 1
            features.append(features_nid)
        for j in range(1, 5):
            tab_tests[1][0][j] += tab_tests[0][i][j]

    p = lambda name: os.path.join(out_dir, name)
    write_table(p("tests.tex"), tab_tests)
    write_req_runs_plot(req_runs_nod, req_runs_od, p("req-runs.tex"))

    corr = stats.spearmanr(features).correlation
    tab_corr = [[[f_i, *corr[i]] for i, f_i in enumerate(FEATURE_NAMES)]]
    write_table(p("corr.tex"), tab_corr, rowcol=False, cellfn=cellfn_corr)

    with open(scores_file, "rb") as fd:
        scores = pickle.load(fd)

    tab_nod_top, tab_od_top = get_top_tables(scores)
    write_table(p("nod-top.tex"), tab_nod_top)
    write_table(p("od-top.tex"), tab_od_top)

    for name, (orig_keys, ext_keys) in COMPARISON_CONFIGS.items():
        tab = get_comparison_table(scores[orig_keys], scores[ext_keys])
        write_table(p(f"{name}-comp.tex"), tab)

    with open(shap_file, "rb") as fd:
        shap_nod, shap_od = pickle.load(fd)
    tab_shap = get_shap_table(shap_nod, shap_od)
    write_table(p("shap.tex"), tab_shap, cellfn=cellfn_shap)
