"""In-tree build of the MI355X native extension (gfx950 only).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Produces mi355x_ddp/_C.*.so next to the package so the artifact travels
with the source tree (no JIT cache dependency).
"""

import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ext = CUDAExtension(
    name="mi355x_ddp._C",
    sources=[
        "mi355x_ddp/ops/csrc/kernels.hip",
        "mi355x_ddp/ops/csrc/rccl_comm.hip",
        "mi355x_ddp/ops/csrc/p2p_mesh.hip",
        "mi355x_ddp/ops/csrc/reducer_core.hip",
        "mi355x_ddp/ops/csrc/autograd_ops.hip",
        "mi355x_ddp/ops/csrc/bindings.hip",
    ],
    libraries=["rccl"],
    extra_compile_args={
        "cxx": ["-O3"],
        "nvcc": ["-O3", "-std=c++17"],
    },
)

setup(
    name="mi355x_ddp",
    version="0.1.0",
    packages=["mi355x_ddp", "mi355x_ddp.models", "mi355x_ddp.ops",
              "mi355x_ddp.parallel", "mi355x_ddp.utils"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension},
)
