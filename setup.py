"""Packaging + in-tree extension build.

`python setup.py build_ext --inplace` (or __graft_entry__.build()) drives
torched_impala_amd/build_ext.py: g++ for the host ring extension and
hipcc --offload-arch=gfx950 for the HIP kernel extension. The .so files
land inside the package so the repo snapshot is self-contained.
"""
import os
import sys

from setuptools import setup, find_packages
from setuptools.command.build_ext import build_ext as _build_ext


class BuildExt(_build_ext):
    def run(self):
        sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
        from torched_impala_amd import build_ext as bx
        bx.build_all()


setup(
    name="torched_impala_amd",
    version="0.1.0",
    description="MI355X-native IMPALA actor-learner RL engine",
    packages=find_packages(include=["torched_impala_amd*"]),
    package_data={"torched_impala_amd": ["*.so"]},
    cmdclass={"build_ext": BuildExt},
    python_requires=">=3.10",
)
