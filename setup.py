import sys

from setuptools import Command, find_packages, setup


class BuildExtInTree(Command):
    """`python setup.py build_ext --inplace` -> in-tree gfx950 build."""

    user_options = [("inplace", "i", "build in-tree (always on)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        from sparktorch_amd.ops.build import build_extension

        print("built:", build_extension(verbose=True))


setup(
    name="sparktorch_amd",
    version="0.1.0",
    description="MI355X-native distributed PyTorch training bridge (sparktorch-compatible API)",
    packages=find_packages(exclude=["tests"]),
    python_requires=">=3.9",
    install_requires=["torch", "numpy", "dill"],
    cmdclass={"build_ext": BuildExtInTree},
)
