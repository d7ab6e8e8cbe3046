"""In-tree build of the gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces deeplearning_cfn_amd/ops/_hip_ops*.so (loaded via
torch.ops.load_library).  The .so is git-ignored but travels to the GPU
box with the gpurun snapshot.
"""

import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROOT = Path(__file__).parent
CSRC = ROOT / "deeplearning_cfn_amd" / "ops" / "csrc"

sources = sorted(str(p) for p in CSRC.glob("*.hip"))

ext = cpp_extension.CUDAExtension(
    name="deeplearning_cfn_amd.ops._hip_ops",
    sources=sources,
    extra_compile_args={
        "cxx": ["-O3", "-std=c++17"],
        "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
    },
)

setup(
    name="deeplearning_cfn_amd",
    version="0.1.0",
    packages=["deeplearning_cfn_amd"],
    ext_modules=[ext],
    cmdclass={"build_ext": cpp_extension.BuildExtension.with_options(
        no_python_abi_suffix=False)},
)
