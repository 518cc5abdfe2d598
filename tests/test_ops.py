"""Unit tests for kfac_amd.ops reference implementations.

Each op is validated against a literal transcription of the upstream
semantics (kfac/layers/utils.py get_cov, modules.py get_a_factor /
get_g_factor) computed with plain torch ops.
"""

from __future__ import annotations

import pytest
import torch
import torch.nn.functional as F

from kfac_amd.layers.utils import append_bias_ones
from kfac_amd.layers.utils import get_cov
from kfac_amd.ops import reference as ref


@pytest.fixture(autouse=True)
def _seed_rng():
    # per-test seeding: module-level seeding runs at import (collection)
    # time and earlier tests shift the global RNG stream
    torch.manual_seed(0)


def literal_cov(a: torch.Tensor) -> torch.Tensor:
    cov = a.t() @ (a / a.size(0))
    return (cov + cov.t()) / 2


def test_cov_linear_matches_literal() -> None:
    a = torch.randn(64, 17)
    out = torch.zeros(18, 18)
    ref.cov_linear(a, bias=True, out=out, beta=0.0, coeff=1.0 / 64)
    expected = literal_cov(append_bias_ones(a))
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)


def test_cov_linear_accumulate_and_beta() -> None:
    a1 = torch.randn(32, 8)
    a2 = torch.randn(32, 8)
    out = torch.zeros(8, 8)
    ref.cov_linear(a1, bias=False, out=out, beta=0.0, coeff=1.0 / 32)
    ref.cov_linear(a2, bias=False, out=out, beta=1.0, coeff=1.0 / 32)
    expected = literal_cov(a1) + literal_cov(a2)
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)
    # beta=0.5 halves the existing accumulation
    ref.cov_linear(a1, bias=False, out=out, beta=0.5, coeff=1.0 / 32)
    expected = expected / 2 + literal_cov(a1)
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize(
    'shape,k,s,p,bias',
    [
        ((2, 3, 8, 8), (3, 3), (1, 1), (1, 1), True),
        ((2, 4, 9, 9), (3, 3), (2, 2), (0, 0), False),
        ((1, 2, 7, 5), (5, 3), (1, 2), (2, 1), True),
        ((3, 1, 6, 6), (1, 1), (1, 1), (0, 0), False),
    ],
)
def test_cov_conv_a_matches_reference_semantics(shape, k, s, p, bias) -> None:
    """A-factor equals: patches -> /spatial (after ones append) -> get_cov."""
    x = torch.randn(*shape)
    patches = ref.extract_patches(x, k, s, p)
    spatial = patches.size(1) * patches.size(2)
    a = patches.reshape(-1, patches.size(-1))
    if bias:
        a = append_bias_ones(a)
    a = a / spatial
    expected = get_cov(a)

    n = shape[1] * k[0] * k[1] + int(bias)
    out = torch.zeros(n, n)
    ref.cov_conv_a(
        x, kernel_size=k, stride=s, padding=p, bias=bias, out=out, beta=0.0,
    )
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)


def test_cov_conv_g_matches_reference_semantics() -> None:
    g = torch.randn(4, 6, 5, 5)
    rows = g.permute(0, 2, 3, 1).reshape(-1, 6) / 25
    expected = get_cov(rows)
    out = torch.zeros(6, 6)
    ref.cov_conv_g(g, out=out, beta=0.0)
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)


def test_extract_patches_matches_unfold() -> None:
    x = torch.randn(2, 3, 10, 10)
    patches = ref.extract_patches(x, (3, 3), (1, 1), (1, 1))
    # F.unfold gives (N, C*kh*kw, L) with C-major ordering of the patch dim
    unf = F.unfold(x, (3, 3), padding=(1, 1)).transpose(1, 2)
    torch.testing.assert_close(
        patches.reshape(2, -1, 27), unf, rtol=1e-6, atol=1e-6,
    )


def test_precond_eigen_matches_literal() -> None:
    m, n = 6, 9
    grad = torch.randn(m, n)
    a = torch.randn(n, n)
    a = a @ a.t() / n
    g = torch.randn(m, m)
    g = g @ g.t() / m
    da, qa = torch.linalg.eigh(a)
    dg, qg = torch.linalg.eigh(g)
    damping = 1e-3
    v1 = qg.t() @ grad @ qa
    v2 = v1 / (torch.outer(dg, da) + damping)
    expected = qg @ v2 @ qa.t()

    out = ref.precond_eigen(grad, qa, qg, da=da, dg=dg, damping=damping)
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)

    dgda = 1 / (torch.outer(dg, da) + damping)
    out2 = ref.precond_eigen(grad, qa, qg, dgda=dgda)
    torch.testing.assert_close(out2, expected, rtol=1e-5, atol=1e-6)


def test_precond_inverse_matches_literal() -> None:
    grad = torch.randn(4, 7)
    a_inv = torch.randn(7, 7)
    g_inv = torch.randn(4, 4)
    expected = g_inv @ grad @ a_inv
    out = ref.precond_inverse(grad, a_inv, g_inv)
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-6)


def test_inv_damped() -> None:
    n = 12
    x = torch.randn(n, n)
    x = x @ x.t() / n
    inv = ref.inv_damped(x, 0.1)
    eye = inv @ (x + 0.1 * torch.eye(n))
    torch.testing.assert_close(eye, torch.eye(n), rtol=1e-4, atol=1e-4)


def test_eigh_clamps_negative() -> None:
    x = torch.diag(torch.tensor([-1.0, 2.0, 3.0]))
    d, q = ref.eigh(x)
    assert (d >= 0).all()


def test_triu_roundtrip() -> None:
    n = 9
    x = torch.randn(n, n)
    x = (x + x.t()) / 2
    v = ref.triu_pack(x)
    assert v.numel() == n * (n + 1) // 2
    y = ref.triu_unpack(v, n)
    torch.testing.assert_close(x, y)


def test_triu_pack_rejects_nonsquare() -> None:
    with pytest.raises(ValueError):
        ref.triu_pack(torch.randn(3, 4))


def test_grad_scale_from_accum() -> None:
    from kfac_amd import ops

    accum = torch.tensor(4.0)
    s = ops.grad_scale_from_accum(accum, kl_clip=1.0, lr=1.0)
    torch.testing.assert_close(s, torch.tensor(0.5))
    # below clip -> capped at 1
    accum = torch.tensor(0.0001)
    s = ops.grad_scale_from_accum(accum, kl_clip=1.0, lr=1.0)
    torch.testing.assert_close(s, torch.tensor(1.0))
    # zero accum -> 1 (matches reference vg_sum==0 -> 1.0)
    s = ops.grad_scale_from_accum(torch.tensor(0.0), kl_clip=1.0, lr=1.0)
    torch.testing.assert_close(s, torch.tensor(1.0))


def test_get_triu_fill_triu_module_api() -> None:
    from kfac_amd.distributed import fill_triu, get_triu

    n = 7
    x = torch.randn(n, n)
    x = (x + x.t()) / 2
    v = get_triu(x)
    y = fill_triu((n, n), v)
    torch.testing.assert_close(x, y)


def test_module_helper_get_factor_api() -> None:
    from kfac_amd.layers.modules import LinearModuleHelper
    from kfac_amd.layers.utils import get_cov
    from kfac_amd.ops.reference import append_bias_ones

    lin = torch.nn.Linear(5, 3)
    h = LinearModuleHelper(lin)
    a = torch.randn(12, 5)
    fa = h.get_a_factor(a)
    torch.testing.assert_close(
        fa, get_cov(append_bias_ones(a)), rtol=1e-5, atol=1e-6,
    )
    g = torch.randn(12, 3)
    fg = h.get_g_factor(g)
    torch.testing.assert_close(fg, get_cov(g), rtol=1e-5, atol=1e-6)


def test_require_ext_fails_loudly(monkeypatch) -> None:
    """On a GPU box without the built extension, ops must raise — no
    silent eager fallback (round requirement)."""
    from kfac_amd import ops as _ops

    monkeypatch.setattr(_ops, '_EXT', None)
    monkeypatch.setattr(_ops, '_EXT_TRIED', True)
    monkeypatch.delenv('KFAC_AMD_ALLOW_EAGER', raising=False)
    monkeypatch.delenv('KFAC_AMD_FORCE_EAGER', raising=False)
    with pytest.raises(RuntimeError, match='not built'):
        _ops._require_ext('cov_linear')
    # the two debug escapes return None instead
    monkeypatch.setenv('KFAC_AMD_ALLOW_EAGER', '1')
    assert _ops._require_ext('cov_linear') is None


def test_refine_inverse_converges_and_certifies() -> None:
    """Newton-Schulz warm inverse: quadratic convergence from a drifted
    start, residual certificate, refusal on garbage starts."""
    from kfac_amd import ops as _ops

    torch.manual_seed(2)
    n = 96
    b = torch.randn(n, n)
    f = b @ b.T / n + 0.5 * torch.eye(n)
    m = f + 1e-3 * torch.eye(n)
    exact = torch.linalg.inv(m)

    # drifted start (the EMA-update scenario)
    pert = torch.randn(n, n) * 1e-3
    x0 = exact + 0.5 * (pert + pert.T)
    x, ok = _ops.refine_inverse(m, x0)
    assert ok
    torch.testing.assert_close(x, exact, rtol=1e-4, atol=1e-6)
    torch.testing.assert_close(x, x.T)  # symmetric by construction

    # exact start stays exact
    x, ok = _ops.refine_inverse(m, exact.clone())
    assert ok
    torch.testing.assert_close(x, exact, rtol=1e-5, atol=1e-7)

    # garbage start must be REFUSED, not returned
    bad = torch.randn(n, n)
    _, ok = _ops.refine_inverse(m, bad)
    assert not ok


def test_inverse_layer_warm_matches_cold(monkeypatch) -> None:
    """KFACInverseLayer phases warm-started from the previous inverse
    must match exact recomputation through a multi-phase run."""
    from kfac_amd.distributed import TorchDistributedCommunicator
    from kfac_amd.enums import AllreduceMethod
    from kfac_amd.layers.inverse import KFACInverseLayer
    from kfac_amd.layers.modules import LinearModuleHelper

    results = {}
    for warm in (True, False):
        monkeypatch.setenv('KFAC_AMD_WARM_INV', '1' if warm else '0')
        torch.manual_seed(4)
        module = torch.nn.Linear(24, 12)
        layer = KFACInverseLayer(
            LinearModuleHelper(module),
            tdc=TorchDistributedCommunicator(),
            allreduce_method=AllreduceMethod.ALLREDUCE,
        )
        invs = []
        for phase in range(4):
            x = torch.randn(64, 24)
            g = torch.randn(64, 12)
            layer.save_layer_input([x])
            layer.save_layer_grad_output((g,))
            layer.update_a_factor(0.95)
            layer.update_g_factor(0.95)
            layer.compute_a_inv(damping=1e-3)
            layer.compute_g_inv(damping=1e-3)
            invs.append((layer.a_inv.clone(), layer.g_inv.clone()))
        results[warm] = invs
    for (aw, gw), (ac, gc) in zip(results[True], results[False]):
        # certificate: ||M X - I||_F <= 1e-6 sqrt(n) -> entrywise ~1e-6
        torch.testing.assert_close(aw, ac, rtol=1e-3, atol=2e-6)
        torch.testing.assert_close(gw, gc, rtol=1e-3, atol=2e-6)
