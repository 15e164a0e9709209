"""In-tree build of the pertgnn HIP extension for gfx950 (MI355X).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

produces pertgnn/_C.cpython-*.so next to the package sources so the binary
travels with the repo snapshot to GPU boxes.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="pertgnn",
    version="0.1.0",
    packages=["pertgnn"],
    ext_modules=[
        CUDAExtension(
            name="pertgnn._C",
            sources=[
                "csrc/bindings.cpp",
                "csrc/collate.cpp",
                "csrc/hip/edge_attn.hip",
                "csrc/hip/edge_attn_fused.hip",
                "csrc/hip/segops.hip",
                "csrc/hip/gemm.hip",
                "csrc/hip/gemm_bf16.hip",
            ],
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
