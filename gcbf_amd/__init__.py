"""gcbf_amd — MI355X-native Graph Control Barrier Function framework.

A from-scratch rebuild of the capabilities of MIT-REALM/gcbf-pytorch for AMD
Instinct MI355X (gfx950): PyTorch-ROCm for autograd/optimizers, hand-written
HIP/CDNA4 kernels for the hot-path graph ops, and RCCL-over-xGMI data
parallelism across the 8 GPUs of a node.
"""
__version__ = "0.1.0"

from .graph import GraphBatch

__all__ = ["GraphBatch", "__version__"]
