"""setup.py shim: builds the in-tree gfx950 engine (_core.so) via hipcc on
`pip install .` / `python setup.py build_ext --inplace` (reference analogue:
Makefile + packaging/ deb/rpm templates)."""

from setuptools import setup
from setuptools.command.build_ext import build_ext as _build_ext


class HipccBuildExt(_build_ext):
    def run(self):
        from elbencho_amd import build as eb_build

        eb_build.build()


setup(cmdclass={"build_ext": HipccBuildExt})
