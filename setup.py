"""In-tree extension build: python setup.py build_ext --inplace.

Delegates to active_learning_amd.ops.build (direct hipcc for gfx950; the
.so lands inside the package so repo snapshots carry it)."""

import sys

from setuptools import Command, find_packages, setup


class BuildExt(Command):
    user_options = [("inplace", "i", "build in place (always true here)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        from active_learning_amd.ops.build import build
        build()


setup(
    name="active_learning_amd",
    version="0.1.0",
    packages=find_packages(include=["active_learning_amd*"]),
    cmdclass={"build_ext": BuildExt},
)
