"""setup.py — builds the in-tree HIP extension via tools/build_ext.py (hipcc, gfx950).

`python setup.py build_ext --inplace` is the canonical build command; it shells out to
hipcc directly (HIP-native source, no hipify pass).
"""
import sys

from setuptools import Command, find_packages, setup


class BuildExt(Command):
    user_options = [("inplace", "i", "build in-tree (always true here)"), ("force", "f", "force rebuild")]

    def initialize_options(self):
        self.inplace = True
        self.force = False

    def finalize_options(self):
        pass

    def run(self):
        sys.path.insert(0, "tools")
        import build_ext as be

        be.build(force=bool(self.force))


setup(
    name="draco_amd",
    version="0.1.0",
    packages=find_packages(include=["draco_amd", "draco_amd.*"]),
    cmdclass={"build_ext": BuildExt},
    python_requires=">=3.10",
)
