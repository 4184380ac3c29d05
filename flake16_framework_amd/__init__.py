"""flake16_framework_amd — MI355X-native Flake16 training/evaluation engine.

A from-scratch rebuild of the capabilities of flake-it/flake16-framework
(the ICST'22 Flake16 flaky-test study pipeline) designed for AMD Instinct
MI355X (gfx950): the `scores` and `shap` stages run device-resident with
hand-written CDNA4 HIP kernels (histogram split-finding, k-NN for
SMOTE/ENN/Tomek, PCA, ensemble prediction, TreeSHAP), sharded across GPUs
with RCCL collectives over xGMI, while keeping the reference's CLI surface
(`experiment.py COMMAND`) and artifact formats (tests.json / scores.pkl /
shap.pkl) compatible.
"""

__version__ = "0.1.0"
