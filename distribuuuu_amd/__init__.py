"""distribuuuu_amd: MI355X-native distributed image-classification training framework.

A from-scratch rebuild of BIGBALLON/distribuuuu's capabilities for AMD Instinct
MI355X (gfx950, CDNA4): PyTorch-ROCm supplies autograd and process-group plumbing;
every hot op is a hand-written HIP kernel (MFMA implicit-GEMM convs, fused
BN+ReLU epilogues, fused attention, multi-tensor SGD); collectives run on RCCL
over xGMI through our own bucketed DDP and SyncBN.
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
from .config import cfg  # noqa: F401
