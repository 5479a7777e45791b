"""tensor2robot_amd: MI355X-native robot-learning framework.

A from-scratch re-design of google-research/tensor2robot's capabilities for
AMD Instinct MI355X (gfx950): PyTorch-ROCm framework layer, hand-written
CDNA4 HIP kernels for the vision hot path, RCCL-over-xGMI data parallelism.
"""

from tensor2robot_amd import ginconfig as gin

__version__ = "0.1.0"
__all__ = ["gin"]
