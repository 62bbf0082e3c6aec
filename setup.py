"""In-tree build of dpo_amd (the HIP extension is compiled by hipcc for
gfx950; see dpo_amd/ops/build.py). `python setup.py build_ext --inplace`
or just `python -m dpo_amd.ops.build`."""
import subprocess
import sys

from setuptools import Command, find_packages, setup


class BuildHip(Command):
    user_options = []

    def initialize_options(self):
        pass

    def finalize_options(self):
        pass

    def run(self):
        from dpo_amd.ops.build import build
        build(force=True)


setup(
    name="dpo_amd",
    version="0.1.0",
    description="MI355X-native distributed pose-graph optimization",
    packages=find_packages(include=["dpo_amd", "dpo_amd.*"]),
    package_data={"dpo_amd.ops": ["hip/*.hip", "hip/*.cpp", "hip/*.so"]},
    cmdclass={"build_hip": BuildHip},
    python_requires=">=3.9",
)
