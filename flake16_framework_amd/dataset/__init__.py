from .collate import ProjectData, RunStats, TestRecord, collate
from .features import get_features_cov
from .labeling import classify
from .synthetic import make_synthetic_tests, write_synthetic_tests
from .tests_io import build_tests, load_feat_lab_proj, load_tests, write_tests
