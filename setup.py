"""Package setup. `python setup.py build_ext --inplace` compiles the
in-tree gfx950 HIP extension (delegates to build_hip.py)."""

import os
import sys

from setuptools import Command, find_packages, setup

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


class BuildHipExt(Command):
    user_options = [("inplace", "i", "build in-tree (always true)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        import build_hip

        build_hip.build()


setup(
    name="deepfake_detection_amd",
    version="0.1.0",
    description="MI355X-native distributed deepfake-detection training/inference stack",
    packages=find_packages(include=["deepfake_detection_amd", "deepfake_detection_amd.*"]),
    package_data={"deepfake_detection_amd": ["_hip_ops.so", "ops/hip/*.hip", "ops/hip/*.h"]},
    python_requires=">=3.8",
    cmdclass={"build_ext": BuildHipExt},
)
