"""cilfw — MI355X-native class-incremental learning framework.

A from-scratch rebuild of the capabilities of the WA (Weight Aligning) class-incremental
learning baseline (reference: G-U-N/a-PyTorch-Tutorial-to-Class-Incremental-Learning,
see SURVEY.md) as an AMD MI355X-first framework:

- PyTorch-ROCm frontend, hand-written CDNA4 HIP kernels for the compute path
  (MFMA implicit-GEMM convolutions, fused BN+ReLU, fused CE / KD losses, fused SGD,
  on-device herding) — see ``cilfw/csrc`` and ``cilfw/ops``;
- a first-party data-parallel engine doing bucketed, backward-overlapped gradient
  all-reduce on RCCL over xGMI — see ``cilfw/distributed``;
- a continuum-style class-incremental scenario engine, rehearsal memory with herding
  exemplar selection, KD teacher management, weight-norm alignment and per-task
  checkpointing — see ``cilfw/data`` and ``cilfw/cil``.
"""

__version__ = "0.1.0"
