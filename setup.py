"""In-tree build of the flink_ms_amd HIP extension for MI355X (gfx950).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built .so lands next to the package (flink_ms_amd/_hip_ops*.so) so it
travels with the source tree to GPU boxes.
"""
import os

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension  # noqa: E402

setup(
    name="flink_ms_amd",
    version="0.1.0",
    packages=["flink_ms_amd"],
    ext_modules=[
        CUDAExtension(
            name="flink_ms_amd._hip_ops",
            sources=[
                "flink_ms_amd/ops/csrc/bindings.cpp",
                "flink_ms_amd/ops/csrc/als_kernels.hip",
                "flink_ms_amd/ops/csrc/svm_kernels.hip",
                "flink_ms_amd/ops/csrc/serve_kernels.hip",
            ],
            extra_compile_args={
                "cxx": ["-O2"],
                "nvcc": ["-O3", "-std=c++17"],
            },
        )
    ],
    cmdclass={"build_ext": BuildExtension},
)
