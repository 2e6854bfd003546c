"""hetu-amd packaging.

`python setup.py build_ext --inplace` (or `pip install -e .`) builds the
gfx950 HIP extension in-tree via hipcc (no GPU needed to compile); the
.so lives at hetu_amd/ops/hip/_hetu_hip.so so source checkouts and
editable installs both find it.
"""
import os
import sys

from setuptools import Command, find_packages, setup

HERE = os.path.dirname(os.path.abspath(__file__))


class BuildHip(Command):
    description = "compile the gfx950 HIP extension in-tree"
    user_options = [("inplace", "i", "build in-tree (always true)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        sys.path.insert(0, HERE)
        from hetu_amd.ops.hip.build import build
        build()


setup(
    name="hetu-amd",
    version="0.1.0",
    description="MI355X-native distributed training framework "
                "(PKU-DAIR/Hetu capabilities, CDNA4-first design)",
    packages=find_packages(include=["hetu_amd", "hetu_amd.*"]),
    package_data={"hetu_amd.ops.hip": ["*.so", "*.hip", "*.cpp", "*.h"]},
    python_requires=">=3.10",
    install_requires=["torch", "numpy", "safetensors", "pyyaml"],
    cmdclass={"build_ext": BuildHip},
)
