"""setuptools shim around build.py (the hipcc/gfx950 build driver).

`python setup.py build_ext --inplace` (or `pip install -e .`) compiles the
native engine in-tree as sharedtensor_amd/_core.so.
"""
import os
import sys

from setuptools import setup
from setuptools.command.build_ext import build_ext as _build_ext

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


class HipccBuild(_build_ext):
    def run(self):
        import build as build_mod
        build_mod.build()


setup(
    name="sharedtensor_amd",
    version="0.1.0",
    description="MI355X-native distributed shared-tensor engine "
                "(CDNA4 HIP kernels + RCCL/xGMI, reference-compatible API)",
    packages=["sharedtensor_amd", "sharedtensor_amd.ops",
              "sharedtensor_amd.models", "sharedtensor_amd.parallel",
              "sharedtensor_amd.utils"],
    package_data={"sharedtensor_amd": ["_core.so"]},
    cmdclass={"build_ext": HipccBuild},
    ext_modules=[],  # built by build.py/hipcc, not by setuptools compilers
    python_requires=">=3.9",
)
