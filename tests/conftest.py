"""Pytest configuration: register the gpu marker; repo-root imports."""

from __future__ import annotations

import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

# The test harness forks workers (testing/distributed.py). A fork while the
# parent's OpenMP/intra-op thread pool holds a lock deadlocks the child, so
# keep the parent single-threaded for the whole session.
torch.set_num_threads(1)


def pytest_configure(config: pytest.Config) -> None:
    config.addinivalue_line(
        'markers',
        'gpu: test requires a ROCm GPU (run on MI355X via gpurun)',
    )


def pytest_collection_modifyitems(
    config: pytest.Config,
    items: list[pytest.Item],
) -> None:
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason='no GPU available')
    for item in items:
        if 'gpu' in item.keywords:
            item.add_marker(skip)
