"""Build the in-tree gfx950 HIP extension:

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The resulting moco_amd/_C*.so is git-ignored but ships with the repo
snapshot to the GPU box (it is NOT gpurun-ignored).
"""

import os

from setuptools import setup

from torch.utils.cpp_extension import BuildExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))

sources = [
    "moco_amd/ops/bindings.cpp",
    "moco_amd/ops/hip/l2norm.hip",
    "moco_amd/ops/hip/contrastive.hip",
    "moco_amd/ops/hip/softmax_ce.hip",
    "moco_amd/ops/hip/multi_tensor.hip",
    "moco_amd/ops/hip/queue.hip",
    "moco_amd/ops/hip/bn.hip",
    "moco_amd/ops/hip/pool.hip",
    "moco_amd/ops/hip/conv1x1.hip",
    "moco_amd/ops/hip/conv3x3.hip",
    "moco_amd/ops/hip/augment.hip",
]

setup(
    name="moco_amd_C",
    ext_modules=[
        CUDAExtension(
            name="moco_amd._C",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
