"""Packaging for edl_amd. `python setup.py build_ext --inplace` drives the
direct-hipcc build (build_hip.py) — no hipify, gfx950 only."""
import os
import sys

from setuptools import Command, find_packages, setup

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


class BuildHip(Command):
    description = "build the gfx950 HIP extension in-tree"
    user_options = [("inplace", "i", "build in-tree (always on)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        import build_hip

        build_hip.build()


setup(
    name="edl_amd",
    version="0.1.0",
    description="MI355X-native elastic deep learning framework",
    packages=find_packages(include=["edl_amd*"]),
    cmdclass={"build_ext": BuildHip},
    entry_points={"console_scripts": ["edlrun=edl_amd.launch:main"]},
    python_requires=">=3.8",
)
