"""Package setup.  The HIP extension is built by a direct hipcc driver
(`python -m flink_ms_amd.ops.build`), NOT by torch's CUDAExtension pipeline
(whose CUDA->HIP rewriter would mangle the native HIP sources); the custom
build_ext below just delegates so `python setup.py build_ext --inplace`
keeps working.
"""

from setuptools import setup
from setuptools.command.build_ext import build_ext as _build_ext


class HipccBuildExt(_build_ext):
    def run(self):
        from flink_ms_amd.ops.build import build
        build(verbose=True)


setup(
    name="flink_ms_amd",
    version="0.2.0",
    packages=["flink_ms_amd", "flink_ms_amd.cli", "flink_ms_amd.data",
              "flink_ms_amd.models", "flink_ms_amd.ops",
              "flink_ms_amd.parallel", "flink_ms_amd.serving",
              "flink_ms_amd.utils"],
    cmdclass={"build_ext": HipccBuildExt},
)
