"""setup.py — builds the in-tree native extension via infinistore_amd._build
(hipcc for gfx950). `python setup.py build_ext --inplace` or just importing
the package triggers the build."""

from pathlib import Path

from setuptools import setup
from setuptools.command.build_ext import build_ext as _build_ext


class BuildNative(_build_ext):
    def run(self):
        import sys

        sys.path.insert(0, str(Path(__file__).resolve().parent))
        from infinistore_amd import _build

        _build.build_native(verbose=True)


setup(
    name="infinistore-amd",
    version="0.1.0",
    description="MI355X-native GPU-direct KV-cache store",
    packages=["infinistore_amd"],
    cmdclass={"build_ext": BuildNative},
    ext_modules=[],
    entry_points={
        "console_scripts": ["infinistore-amd=infinistore_amd.server:main"],
    },
)
