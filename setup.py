"""Build the in-tree gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built _hipcore*.so lands next to bodywork_mlops_demo_amd/ops/ so it
travels with the repo snapshot to GPU boxes (no JIT cache dependency).
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

import torch  # noqa: E402
from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join("bodywork_mlops_demo_amd", "ops", "hip")
TORCH_LIB = os.path.join(os.path.dirname(torch.__file__), "lib")

sources = [
    os.path.join(HIP_DIR, "bindings.cpp"),
    os.path.join(HIP_DIR, "datagen.hip"),
    os.path.join(HIP_DIR, "linreg.hip"),
    os.path.join(HIP_DIR, "mlp_small.hip"),
    os.path.join(HIP_DIR, "gemm.hip"),
    os.path.join(HIP_DIR, "gemm8.hip"),
    os.path.join(HIP_DIR, "gemm_mx8.hip"),
    os.path.join(HIP_DIR, "optim.hip"),
]

setup(
    name="bodywork_hipcore",
    ext_modules=[
        CUDAExtension(
            name="bodywork_mlops_demo_amd.ops._hipcore",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
            # rpath so the in-tree .so resolves torch/ROCm libs on any box
            # with this image (no LD_LIBRARY_PATH needed)
            extra_link_args=[
                f"-Wl,-rpath,{TORCH_LIB}",
                "-Wl,-rpath,/opt/rocm/lib",
            ],
        )
    ],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
