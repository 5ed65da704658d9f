"""Packaging for maggy_amd.

``python setup.py build_ext --inplace`` builds the HIP extension in-tree
via ops/build.py (hipcc --offload-arch=gfx950); plain ``pip install -e .``
installs the python package and builds lazily on first GPU use.
"""
from setuptools import Command, find_packages, setup


class BuildHip(Command):
    description = "build the gfx950 HIP extension in-tree"
    user_options = []

    def initialize_options(self):
        pass

    def finalize_options(self):
        pass

    def run(self):
        from maggy_amd.ops.build import build

        print("built:", build())


setup(
    name="maggy_amd",
    version="0.1.0",
    description=("MI355X-native distribution-transparent experiment "
                 "engine (maggy-compatible API)"),
    packages=find_packages(include=["maggy_amd", "maggy_amd.*"]),
    python_requires=">=3.10",
    install_requires=["numpy", "torch"],
    cmdclass={"build_hip": BuildHip},
)
