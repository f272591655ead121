"""fast_autoaugment_amd — an MI355X-native Fast AutoAugment framework.

A from-scratch re-design of the capabilities of kakaobrain/fast-autoaugment
(reference layout documented in SURVEY.md) for AMD Instinct MI355X (gfx950):
PyTorch-ROCm orchestration, hand-written CDNA4 HIP kernels for the hot ops
(GPU-resident augmentation pipeline, fused optimizer/EMA/loss kernels,
stochastic-regularizer kernels), and RCCL over xGMI for data parallelism.

Top-level namespaces:
  config    — YAML+CLI configuration singleton (replaces `theconf`)
  policies  — augmentation-policy archives, encoder/decoder (reference archive.py)
  aug       — augmentation op registry; CPU reference + GPU pipeline
  models    — model registry (WideResNet, ResNet, Shake-Shake, PyramidNet,
              EfficientNet) (reference networks/)
  engine    — trainer / eval loops (reference train.py)
  parallel  — RCCL data-parallel engine + local multi-GPU scheduler
  search    — TPE policy search + density-matching evaluator (reference search.py)
  ops       — HIP kernel extension loader + autograd wrappers
"""

__version__ = "0.1.0"
