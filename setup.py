# Copyright (c) Flashy-AMD authors.
"""Package setup.  The native gfx950 extension is built in-tree by
``python -m flashy_amd.ops.build`` (invoked automatically by
``setup.py build_ext --inplace``)."""
from pathlib import Path

from setuptools import Command, find_packages, setup
from setuptools.command.build_ext import build_ext as _build_ext


class HipBuildExt(_build_ext):
    def run(self):
        from flashy_amd.ops.build import build
        build(verbose=True)


setup(
    name="flashy_amd",
    version="0.1.0a1",
    description="MI355X-native minimal solver framework for deep learning",
    packages=find_packages(include=["flashy_amd", "flashy_amd.*"]),
    package_data={"flashy_amd.ops": ["*.so", "csrc/*"]},
    python_requires=">=3.9",
    install_requires=["torch", "pyyaml", "numpy"],
    cmdclass={"build_ext": HipBuildExt},
)
