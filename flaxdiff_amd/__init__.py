"""flaxdiff_amd — MI355X-native diffusion training + sampling framework.

A from-scratch re-design of FlaxDiff's capabilities for AMD Instinct MI355X
(gfx950, CDNA4): PyTorch-ROCm framework layer, hand-written HIP kernels for
the hot path (fused GroupNorm+SiLU, LDS-tiled implicit-GEMM 3x3 convs, MFMA
attention, fused Adam+EMA), RCCL-over-xGMI data parallelism.

Reference for behavior/API parity: AshishKumar4/FlaxDiff.
"""

__version__ = "0.1.0"

from . import ops, schedulers, predictors, samplers, models, utils  # noqa: F401
