"""setuptools shim: `python setup.py build_ext --inplace` (or pip wheel)
drives the hipcc gfx950 build in build_ext.py and places the extensions
in-tree under starway_amd/."""
from __future__ import annotations

import sys
from pathlib import Path

from setuptools import setup
from setuptools.command.build_ext import build_ext as _build_ext

sys.path.insert(0, str(Path(__file__).resolve().parent))


class HipccBuildExt(_build_ext):
    def run(self):  # noqa: D102
        import build_ext as be

        be.build()

    def get_output_mapping(self):  # in-tree build: nothing to copy
        return {}


setup(
    name="starway-amd",
    version="0.1.0",
    packages=["starway_amd", "starway_amd.benchmarks"],
    cmdclass={"build_ext": HipccBuildExt},
    # A dummy ext module so build_ext runs under `pip install`/wheel builds.
    ext_modules=[],
    package_data={"starway_amd": ["*.so", "*.pyi"]},
)
