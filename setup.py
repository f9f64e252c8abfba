"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built distribuuuu_amd/_hip_ops*.so travels with the repo snapshot (it is
git-ignored but NOT gpurun-ignored), so GPU boxes load the in-tree binary.
"""

import glob
import os

from setuptools import setup

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "distribuuuu_amd", "csrc")

# *_hip.hip are torch-hipify build fallout (byte-copies of the hand-written
# sources with one #include swapped) — regenerated every build, never sources
sources = sorted(glob.glob(os.path.join(CSRC, "*.cpp"))) + sorted(
    f for f in glob.glob(os.path.join(CSRC, "*.hip"))
    if not f.endswith("_hip.hip")
)

setup(
    name="distribuuuu_amd_hip_ops",
    ext_modules=[
        CUDAExtension(
            name="distribuuuu_amd._hip_ops",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
