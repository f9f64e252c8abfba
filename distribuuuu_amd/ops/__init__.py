"""MI355X-native op layer.

Every hot op of the reference's model zoo dispatches here:
on a GPU the hand-written gfx950 HIP kernels in ``distribuuuu_amd/csrc`` run via the
in-tree ``_hip_ops`` extension; on CPU a plain PyTorch fp32 composition of the same
math runs (this is also what kernel numerics tests compare against).

The extension is REQUIRED on a GPU machine: if a CUDA/ROCm device is visible and the
extension is missing, op calls raise instead of silently falling back to ATen.
"""

from .dispatch import ext, hip_op_available, require_ext  # noqa: F401
from .modules import (  # noqa: F401
    AdaptiveAvgPool2d,
    AvgPool2d,
    BatchNorm2d,
    Conv2d,
    Dropout,
    Linear,
    MaxPool2d,
    ReLU,
    SiLU,
    Sigmoid,
)
