"""Coverage-derived features (Flake16 features 0-2).

Semantics of the reference's get_features_nid_cov (experiment.py:362-373):
  Covered Lines        = total executed lines over all covered files
  Covered Changes      = sum of churn change-counts over the covered lines
  Source Covered Lines = covered lines in files that are NOT test files
"""


def get_features_cov(cov_nid, test_files, churn):
    n_lines = n_changes = n_src_lines = 0

    for file_name, cov_file in cov_nid.items():
        n_lines += len(cov_file)
        churn_file = churn.get(file_name, {})
        n_changes += sum(churn_file.get(l_no, 0) for l_no in cov_file)

        if file_name not in test_files:
            n_src_lines += len(cov_file)

    return n_lines, n_changes, n_src_lines
