"""Every module in the package imports cleanly (guards against stray
syntax/import regressions in rarely-exercised modules)."""

from __future__ import annotations

import importlib
import os
import pkgutil
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def test_all_modules_import() -> None:
    import kfac_amd

    failures = []
    for info in pkgutil.walk_packages(
        kfac_amd.__path__, prefix='kfac_amd.',
    ):
        try:
            importlib.import_module(info.name)
        except Exception as e:  # pragma: no cover
            failures.append((info.name, repr(e)))
    assert not failures, failures


def test_examples_import() -> None:
    for mod in (
        'examples.utils',
        'examples.vision.engine',
        'examples.vision.optimizers',
        'examples.vision.datasets',
        'examples.language.dataset',
        'examples.language.engine',
    ):
        importlib.import_module(mod)
