"""Build hook: `python setup.py build_ext --inplace` compiles the HIP
extension for gfx950 in-tree (csrc/ -> kfac_amd/_kfaccore.so)."""

from __future__ import annotations

from setuptools import Command
from setuptools import setup


class BuildHip(Command):
    """Compile the gfx950 HIP extension with hipcc (no hipify)."""

    user_options: list = []

    def initialize_options(self) -> None:
        pass

    def finalize_options(self) -> None:
        pass

    def run(self) -> None:
        from kfac_amd._build import build

        build(force=True)


setup(cmdclass={'build_ext': BuildHip})
