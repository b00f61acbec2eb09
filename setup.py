"""Build entry: ``python setup.py build_ext --inplace`` compiles the
in-tree gfx950 HIP extension (byzpy_amd/_hip_ops.so) with explicit hipcc.
"""
import sys

from setuptools import Command, find_packages, setup


class BuildHip(Command):
    description = "build the gfx950 HIP extension in-tree"
    user_options = [("inplace", "i", "build in-tree (always on)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        from byzpy_amd.hip.build import build

        build(force="--force" in sys.argv)


setup(
    name="byzpy_amd",
    version="0.1.0",
    packages=find_packages(include=["byzpy_amd*"]),
    cmdclass={"build_ext": BuildHip},
    entry_points={"console_scripts": ["byzpy-amd=byzpy_amd.cli:main"]},
)
