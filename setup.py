"""Build shim (the reference's setup.py analog): compiles the native
modules in-tree via the gpudpf build driver.  `pip install -e .` is NOT
required — `python setup.py build_ext --inplace` (or `make build`) is
enough; the package is used from the repo root."""

import sys

from setuptools import Command, setup


class BuildExtInplace(Command):
    description = "build gpudpf native modules in-tree (g++ + hipcc gfx950)"
    user_options = [("inplace", "i", "build in-tree (always on)")]

    def initialize_options(self):
        self.inplace = True

    def finalize_options(self):
        pass

    def run(self):
        sys.path.insert(0, ".")
        from gpudpf import _build

        _build.build(force=False)


setup(
    name="gpudpf",
    version="0.1.0",
    description="MI355X-native DPF / 2-server PIR engine",
    packages=["gpudpf", "pir", "pir.datasets"],
    cmdclass={"build_ext": BuildExtInplace},
)
