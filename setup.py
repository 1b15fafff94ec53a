"""setup.py shim: `python setup.py build_ext --inplace` compiles the HIP
engine extension in-tree for gfx950 (the .so stays inside the package so it
travels with repo snapshots)."""

import sys

from setuptools import setup
from setuptools.command.build_ext import build_ext


class BuildHip(build_ext):
    def run(self):
        sys.path.insert(0, ".")
        from crowdllama_amd.ops import build as b
        b.build(verbose=True)


setup(cmdclass={"build_ext": BuildHip})
