"""Confusion accounting and precision/recall/F1.

Quirks preserved from the reference (experiment.py:430-443, 476-483):
  - the per-sample confusion key is k = 2*y_true + y_pred - 1: true negatives
    (k == -1) are SKIPPED entirely; 0 = FP, 1 = FN, 2 = TP.
  - precision/recall/F1 are None when their denominator is zero.
"""


def div_none(a, b):
    return a / b if b else None


def get_prf(fp, fn, tp):
    p = div_none(tp, tp + fp)
    r = div_none(tp, tp + fn)

    if p is None or r is None:
        f = None
    else:
        f = div_none(2 * p * r, p + r)

    return p, r, f


def finalize_scores(scores, scores_total):
    """Append [P, R, F] to each per-project [FP, FN, TP] triple, in place."""
    for scores_proj in [*scores.values(), scores_total]:
        scores_proj[3:] = get_prf(*scores_proj[:3])
    return scores, scores_total
