"""setup.py — `python setup.py build_ext --inplace` builds the gfx950 HIP
extension in-tree via build_hip.py (hipcc directly; no hipify, no torch C++
ABI)."""

import sys

from setuptools import Command, find_packages, setup


class BuildHip(Command):
    user_options = [("force", "f", "rebuild even if up to date"),
                    ("inplace", "i", "compat no-op (always in-place)")]

    def initialize_options(self):
        self.force = False
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        import build_hip
        build_hip.build(force=bool(self.force))


setup(
    name="asyncframework_amd",
    version="0.1.0",
    description="MI355X-native asynchronous optimization engine "
                "(ASGD/ASAGA parameter server, HIP/CDNA4 kernels, RCCL)",
    packages=find_packages(include=["asyncframework_amd*"]),
    cmdclass={"build_ext": BuildHip},
    python_requires=">=3.9",
    entry_points={
        "console_scripts": [
            "async-asgd-thread=asyncframework_amd.cli.drivers:asgd_thread",
            "async-asgd-sync=asyncframework_amd.cli.drivers:asgd_sync",
            "async-asaga-thread=asyncframework_amd.cli.drivers:asaga_thread",
            "async-asaga-sync=asyncframework_amd.cli.drivers:asaga_sync",
            "async-sgd-mllib=asyncframework_amd.cli.drivers:sgd_mllib",
        ]
    },
)
