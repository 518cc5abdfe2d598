"""Model-family shape/parameter gates (CPU).

The model zoo parity surface: torchvision-free ResNet-50/101/152,
CIFAR option-A ResNets (reference examples/vision/cifar_resnet.py),
Transformer LM (reference examples/language/transformer.py), and the
GPT-NeoX-125M bench model.
"""

from __future__ import annotations

import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kfac_amd.models import resnet50  # noqa: E402
from kfac_amd.models import TransformerModel  # noqa: E402
from kfac_amd.models.cifar_resnet import cifar_resnet  # noqa: E402
from kfac_amd.models.gptneox import gptneox_125m  # noqa: E402


def _nparams(m: torch.nn.Module) -> int:
    return sum(p.numel() for p in m.parameters())


def test_resnet50_matches_torchvision_count() -> None:
    m = resnet50()
    assert _nparams(m) == 25_557_032  # torchvision resnet50 exact
    out = m(torch.randn(2, 3, 224, 224))
    assert out.shape == (2, 1000)


@pytest.mark.parametrize(
    'depth,expected',
    [(20, 269_722), (32, 464_154), (56, 853_018)],
)
def test_cifar_resnet_counts(depth: int, expected: int) -> None:
    m = cifar_resnet(depth)
    assert _nparams(m) == expected  # published He et al. CIFAR sizes
    assert m(torch.randn(2, 3, 32, 32)).shape == (2, 10)


def test_cifar_resnet_rejects_bad_depth() -> None:
    with pytest.raises(ValueError):
        cifar_resnet(21)


def test_transformer_lm_forward() -> None:
    m = TransformerModel(100, 32, 4, 64, 2, 0.0)
    x = torch.randint(0, 100, (12, 3))  # (seq, batch)
    out = m(x)
    assert out.shape == (12, 3, 100)


def test_gptneox_125m_size_and_forward() -> None:
    m = gptneox_125m()
    n = _nparams(m)
    # 125M-class: within 25% of the nominal size
    assert 100e6 < n < 200e6, n
    x = torch.randint(0, 50304, (2, 16))
    out = m(x)
    assert out.shape == (2, 16, 50304)
    assert torch.isfinite(out).all()


def test_resnet_deep_variant_counts() -> None:
    from kfac_amd.models import resnet101
    from kfac_amd.models import resnet152

    # torchvision-exact counts
    assert _nparams(resnet101()) == 44_549_160
    assert _nparams(resnet152()) == 60_192_808
