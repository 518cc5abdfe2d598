"""Integration convergence gate (analog of the reference's MNIST gate:
final quality with K-FAC better than without,
reference tests/integration/mnist_integration_test.py:104-176 — no MNIST
download offline, so a fixed synthetic classification task stands in).
"""

from __future__ import annotations

import sys

import torch

sys.path.insert(0, '.')

from testing.distributed import run_distributed  # noqa: E402


def _train_once(use_kfac: bool) -> float:
    from kfac_amd import KFACPreconditioner
    from testing.models import LeNet

    torch.manual_seed(3)
    # fixed separable-ish synthetic task: images from 10 class templates
    templates = torch.randn(10, 1, 28, 28)
    n = 256
    labels = torch.arange(n) % 10
    x = templates[labels] + 0.3 * torch.randn(n, 1, 28, 28)

    torch.manual_seed(4)
    model = LeNet()
    opt = torch.optim.SGD(model.parameters(), lr=0.02)
    precon = None
    if use_kfac:
        precon = KFACPreconditioner(
            model, factor_update_steps=1, inv_update_steps=5, lr=0.02,
        )
    for _ in range(30):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), labels)
        loss.backward()
        if precon is not None:
            precon.step()
        opt.step()
    with torch.no_grad():
        final = torch.nn.functional.cross_entropy(model(x), labels).item()
        acc = (model(x).argmax(1) == labels).float().mean().item()
    return final, acc


def _gate() -> None:
    loss_kfac, acc_kfac = _train_once(True)
    loss_sgd, acc_sgd = _train_once(False)
    # K-FAC must reach at least as good a fit in the same step budget
    assert loss_kfac < loss_sgd, (loss_kfac, loss_sgd)
    assert acc_kfac >= acc_sgd - 1e-6, (acc_kfac, acc_sgd)


def test_kfac_beats_sgd_gate() -> None:
    run_distributed(1, _gate, timeout=300)
