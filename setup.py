"""Builds the in-tree HIP extension (gfx950) then installs the package.

The native build is driven by tools/build_ext.py (explicit hipcc, no
hipify); `python setup.py build_ext --inplace` produces
mpi4torch_amd/_C.so in-tree, which is how the repo is meant to be used
(the .so travels with the source tree).
"""
import os
import sys

from setuptools import setup
from setuptools.command.build_ext import build_ext as _build_ext

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "tools"))


class HipccBuildExt(_build_ext):
    def run(self):
        from build_ext import build

        build()


setup(cmdclass={"build_ext": HipccBuildExt}, ext_modules=[])
