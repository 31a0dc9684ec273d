"""In-tree build of the adanet_amd HIP kernel extension for gfx950.

Usage:  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace
The resulting adanet_amd/_adanet_hip*.so travels with the repo snapshot to
GPU boxes (it is git-ignored but NOT gpurun-ignored).
"""

import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

SRC = [
    "adanet_amd/csrc/binding.cpp",
    "adanet_amd/csrc/gemm.hip",
    "adanet_amd/csrc/gemm_8ph.hip",
    "adanet_amd/csrc/gemm_tn.hip",
    "adanet_amd/csrc/transpose.hip",
    "adanet_amd/csrc/softmax_xent.hip",
    "adanet_amd/csrc/mixer.hip",
    "adanet_amd/csrc/optim.hip",
    "adanet_amd/csrc/layernorm.hip",
    "adanet_amd/csrc/batchnorm.hip",
    "adanet_amd/csrc/pool.hip",
    "adanet_amd/csrc/im2col.hip",
    "adanet_amd/csrc/elementwise.hip",
    "adanet_amd/csrc/reduce.hip",
    "adanet_amd/csrc/depthwise.hip",
]

def _version():
    ns = {}
    with open("adanet_amd/version.py") as f:
        exec(f.read(), ns)
    return ns["__version__"]


setup(
    name="adanet-amd",
    version=_version(),
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="adanet_amd._adanet_hip",
            sources=SRC,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension},
)
