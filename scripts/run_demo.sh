#!/bin/bash
# End-to-end demo of the GPU pipeline on synthetic data:
#   synthetic tests.json -> scores.pkl (216 cells) -> shap.pkl -> figures
# Run from the repo root on an MI355X box.  Artifacts land in demo_out/.
set -e
cd "$(dirname "$0")/.."
mkdir -p demo_out && cd demo_out
export PYTHONPATH=..

python ../experiment.py synthetic --n-tests 10000 --seed 0
time python ../experiment.py scores --checkpoint scores_ckpt
time python ../experiment.py shap
python ../experiment.py figures --offline
ls -la *.tex scores.pkl shap.pkl tests.json
python - <<'PY'
import pickle
scores = pickle.load(open("scores.pkl", "rb"))
best = max((v[3][5], k) for k, v in scores.items() if v[3][5] is not None)
print(f"{len(scores)} cells; best F1 {best[0]:.3f} at {best[1]}")
PY
