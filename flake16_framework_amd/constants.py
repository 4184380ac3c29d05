"""Module-level constants: artifact names, label encoding, feature schema.

Semantics-parity notes (reference: /root/reference/experiment.py:32-71):
  - artifact file names and directory layout match the reference so that the
    staged pipeline (run -> tests -> scores -> shap -> figures) is drop-in
    compatible.
  - label encoding follows the CODE, not the README: NON_FLAKY=0, OD_FLAKY=1,
    FLAKY=2 where FLAKY means NOD-flaky (reference experiment.py:50; the
    README's "1=NOD, 2=OD" contradicts the code and is ignored).
"""

import os

LOG_FILE = "log.txt"
SHAP_FILE = "shap.pkl"
TESTS_FILE = "tests.json"
SCORES_FILE = "scores.pkl"
SUBJECTS_FILE = "subjects.txt"
REQUIREMENTS_FILE = "requirements.txt"

DATA_DIR = "data"
STDOUT_DIR = "stdout"
WORK_DIR = os.path.join("/", "home", "user")
SUBJECTS_DIR = os.path.join(WORK_DIR, "subjects")
CONT_DATA_DIR = os.path.join(WORK_DIR, DATA_DIR)

CONT_TIMEOUT = 7200
IMAGE_NAME = "flake16framework"

NON_FLAKY, OD_FLAKY, FLAKY = 0, 1, 2

# Runs per mode used by the labeling rule (reference experiment.py:52).
N_RUNS = {"baseline": 2500, "shuffle": 2500, "testinspect": 1}

# pytest plugins that interfere with deterministic suite execution
# (reference experiment.py:54-59).
PLUGIN_BLACKLIST = (
    "-p", "no:cov", "-p", "no:flaky", "-p", "no:xdist", "-p", "no:sugar",
    "-p", "no:replay", "-p", "no:forked", "-p", "no:ordering",
    "-p", "no:randomly", "-p", "no:flakefinder", "-p", "no:random_order",
    "-p", "no:rerunfailures",
)

# The 16 Flake16 features, canonical column order (reference experiment.py:65-71).
FEATURE_NAMES = (
    "Covered Lines", "Covered Changes", "Source Covered Lines",
    "Execution Time", "Read Count", "Write Count", "Context Switches",
    "Max. Threads", "Max. Memory", "AST Depth", "Assertions",
    "External Modules", "Halstead Volume", "Cyclomatic Complexity",
    "Test Lines of Code", "Maintainability",
)

# Column subset used by the FlakeFlagger feature set (reference experiment.py:81).
FLAKEFLAGGER_COLUMNS = (0, 1, 2, 3, 10, 11, 14)
