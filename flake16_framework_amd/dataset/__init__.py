from .labeling import get_req_runs_label
from .features import get_features_cov
from .tests_io import build_tests, load_feat_lab_proj, load_tests, write_tests
from .synthetic import make_synthetic_tests, write_synthetic_tests
