"""GPU numerics tests: HIP kernels vs the fp32 torch reference.

Every CDNA4 kernel in csrc/ is validated here against
kfac_amd.ops.reference computed in fp32 on the same device (SURVEY.md §4:
numerics tests compare the HIP kernel against a plain PyTorch fp32
reference of the same op). Inputs are asymmetric random matrices so
operand/output transposes cannot pass silently.
"""

from __future__ import annotations

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope='module', autouse=True)
def _require_ext():
    import torch

    if not torch.cuda.is_available():
        pytest.skip('no GPU')
    from kfac_amd import ops

    assert ops.extension_available(), (
        'HIP extension must be built on GPU boxes (fail loudly, no eager '
        'fallback)'
    )


@pytest.fixture(autouse=True)
def _seed_rng():
    # deterministic per-test RNG regardless of suite ordering
    torch.manual_seed(0)


def _ext():
    from kfac_amd import _kfaccore

    return _kfaccore


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize('m,k,bias', [(64, 16, True), (300, 65, False), (1024, 129, True), (37, 7, True)])
def test_cov_linear(dtype, m, k, bias) -> None:
    from kfac_amd.ops import reference as ref

    torch.manual_seed(0)
    a = torch.randn(m, k, device='cuda', dtype=dtype)
    n = k + int(bias)
    out = torch.randn(n, n, device='cuda')
    out = (out + out.t()).contiguous()
    expected = out.clone()
    ref.cov_linear(a.float(), bias=bias, out=expected, beta=0.5, coeff=1.0 / m)
    _ext().cov_linear(a.reshape(-1, k), out, bias, 0.5, 1.0 / m)
    torch.cuda.synchronize()
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    torch.testing.assert_close(out, expected, rtol=tol, atol=tol)


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
@pytest.mark.parametrize(
    'shape,kern,stride,pad,bias',
    [
        ((4, 3, 16, 16), (3, 3), (1, 1), (1, 1), True),
        ((2, 8, 14, 14), (3, 3), (2, 2), (1, 1), False),
        ((2, 3, 32, 32), (7, 7), (2, 2), (3, 3), False),
        ((3, 5, 9, 9), (1, 1), (1, 1), (0, 0), True),
    ],
)
def test_cov_conv_a(dtype, shape, kern, stride, pad, bias) -> None:
    from kfac_amd.ops import reference as ref

    torch.manual_seed(1)
    x = torch.randn(*shape, device='cuda', dtype=dtype)
    n = shape[1] * kern[0] * kern[1] + int(bias)
    out = torch.zeros(n, n, device='cuda')
    expected = torch.zeros_like(out)
    ref.cov_conv_a(
        x.float(), kernel_size=kern, stride=stride, padding=pad, bias=bias,
        out=expected, beta=0.0,
    )
    _ext().cov_conv_a(
        x, out, kern[0], kern[1], stride[0], stride[1], pad[0], pad[1],
        bias, 0.0, 1.0,
    )
    torch.cuda.synchronize()
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    torch.testing.assert_close(out, expected, rtol=tol, atol=tol)


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_cov_conv_g(dtype) -> None:
    from kfac_amd.ops import reference as ref

    torch.manual_seed(2)
    g = torch.randn(8, 24, 7, 7, device='cuda', dtype=dtype)
    out = torch.zeros(24, 24, device='cuda')
    expected = torch.zeros_like(out)
    ref.cov_conv_g(g.float(), out=expected, beta=0.0)
    _ext().cov_conv_g(g, out, 0.0, 1.0)
    torch.cuda.synchronize()
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    torch.testing.assert_close(out, expected, rtol=tol, atol=tol)


@pytest.mark.parametrize('m,n', [(6, 26), (64, 65), (100, 200), (1000, 2049), (16, 16)])
def test_precond_eigen_fused(m, n) -> None:
    """Validate against an fp64 reference with realistic (orthogonal) bases.

    Random dense qa/qg amplify fp32 cancellation ~1000x; real K-FAC bases
    are orthogonal eigenvector matrices, so test with those.
    """
    torch.manual_seed(3)
    grad = torch.randn(m, n, device='cuda')
    sa = torch.randn(n, n, device='cuda')
    qa = torch.linalg.eigh(sa + sa.t())[1].contiguous()
    sg = torch.randn(m, m, device='cuda')
    qg = torch.linalg.eigh(sg + sg.t())[1].contiguous()
    dgda = torch.rand(m, n, device='cuda') + 0.5
    # fp64 ground truth
    e64 = (
        qg.double()
        @ ((qg.double().t() @ grad.double() @ qa.double()) * dgda.double())
        @ qa.double().t()
    )
    out = _ext().precond_eigen_fused(grad, qa, qg, dgda)
    torch.cuda.synchronize()
    torch.testing.assert_close(
        out.double(), e64, rtol=1e-3, atol=1e-4 * float(e64.abs().max()),
    )


def test_precond_eigen_dgda_bf16_grad() -> None:
    """bf16-grad round trip through the non-prediv eigen precondition.

    Bases are orthogonal (eigenvectors of symmetric matrices) as in real
    K-FAC state — random dense bases amplify rounding ~1000x (see
    test_precond_eigen_fused) and do not represent the op's input class.
    """
    from kfac_amd.ops import reference as ref

    torch.manual_seed(4)
    m, n = 128, 257
    grad = torch.randn(m, n, device='cuda', dtype=torch.bfloat16)
    sa = torch.randn(n, n, device='cuda')
    qa = torch.linalg.eigh(sa + sa.t())[1].contiguous()
    sg = torch.randn(m, m, device='cuda')
    qg = torch.linalg.eigh(sg + sg.t())[1].contiguous()
    dg = torch.rand(m, device='cuda') + 0.1
    da = torch.rand(n, device='cuda') + 0.1
    expected = ref.precond_eigen(grad, qa, qg, da=da, dg=dg, damping=1e-3)
    out = _ext().precond_eigen(grad, qa, qg, dg, da, 1e-3)
    torch.cuda.synchronize()
    assert out.dtype == torch.bfloat16
    torch.testing.assert_close(
        out.float(), expected.float(), rtol=2e-2, atol=2e-2,
    )


def test_precond_inverse() -> None:
    from kfac_amd.ops import reference as ref

    torch.manual_seed(5)
    m, n = 100, 131
    grad = torch.randn(m, n, device='cuda')
    a_inv = torch.randn(n, n, device='cuda')
    g_inv = torch.randn(m, m, device='cuda')
    expected = ref.precond_inverse(grad, a_inv, g_inv)
    out = _ext().precond_inverse(grad, a_inv, g_inv)
    torch.cuda.synchronize()
    torch.testing.assert_close(out, expected, rtol=2e-4, atol=2e-4)


@pytest.mark.parametrize('dtype', [torch.float32, torch.bfloat16])
def test_kl_clip_accum(dtype) -> None:
    torch.manual_seed(6)
    p = torch.randn(1000, 257, device='cuda', dtype=dtype)
    g = torch.randn(1000, 257, device='cuda', dtype=dtype)
    accum = torch.zeros((), device='cuda')
    _ext().kl_clip_accum(accum, p, g)
    torch.cuda.synchronize()
    expected = (p.float() * g.float()).sum()
    torch.testing.assert_close(accum, expected, rtol=1e-3, atol=1e-2)


def test_triu_roundtrip_gpu() -> None:
    torch.manual_seed(7)
    n = 131
    x = torch.randn(n, n, device='cuda')
    x = (x + x.t()).contiguous()
    v = _ext().triu_pack(x)
    assert v.numel() == n * (n + 1) // 2
    y = _ext().triu_unpack(v, n)
    torch.cuda.synchronize()
    torch.testing.assert_close(x, y)
    # matches the CPU reference packing order
    from kfac_amd.ops import reference as ref

    torch.testing.assert_close(v.cpu(), ref.triu_pack(x.cpu()))


def test_ops_dispatch_routes_to_ext() -> None:
    """kfac_amd.ops on GPU tensors must hit the extension, not eager torch."""
    from kfac_amd import ops
    from kfac_amd.ops import reference as ref

    a = torch.randn(128, 32, device='cuda')
    out = torch.zeros(33, 33, device='cuda')
    ops.cov_linear(a, bias=True, out=out, beta=0.0, coeff=1.0 / 128)
    expected = torch.zeros_like(out)
    ref.cov_linear(a, bias=True, out=expected, beta=0.0, coeff=1.0 / 128)
    torch.cuda.synchronize()
    torch.testing.assert_close(out, expected, rtol=1e-5, atol=1e-5)


def test_precond_eigen_grouped_matches_per_layer() -> None:
    """Grouped 4-launch chain == per-layer fused chain for mixed shapes."""
    torch.manual_seed(11)
    shapes = [(64, 147), (256, 2304), (1000, 2049), (512, 513), (64, 64)]
    grads, qas, qgs, dgdas = [], [], [], []
    for m, n in shapes:
        grads.append(torch.randn(m, n, device='cuda'))
        sa = torch.randn(n, n, device='cuda')
        qas.append(torch.linalg.eigh(sa + sa.t())[1].contiguous())
        sg = torch.randn(m, m, device='cuda')
        qgs.append(torch.linalg.eigh(sg + sg.t())[1].contiguous())
        dgdas.append(torch.rand(m, n, device='cuda') + 0.5)
    outs = _ext().precond_eigen_grouped(grads, qas, qgs, dgdas)
    for i in range(len(shapes)):
        expected = _ext().precond_eigen_fused(grads[i], qas[i], qgs[i], dgdas[i])
        torch.cuda.synchronize()
        torch.testing.assert_close(outs[i], expected, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize('n,batch', [(64, 12), (48, 6), (37, 3)])
def test_syevj_small_vs_torch(n, batch) -> None:
    """Hand-written batched Jacobi vs torch.linalg.eigh ground truth."""
    torch.manual_seed(13)
    a = torch.randn(batch, n, n, device='cuda')
    stack = 0.95 * torch.eye(n, device='cuda').expand(batch, n, n).clone()
    stack = stack + 0.05 * (a @ a.transpose(1, 2)) / n
    w, v = _ext().syevj_small(stack, 20, 1e-5)
    torch.cuda.synchronize()
    # reconstruction: V diag(w) V^T == A
    recon = v @ torch.diag_embed(w) @ v.transpose(1, 2)
    torch.testing.assert_close(recon, stack, rtol=1e-4, atol=1e-5)
    # orthogonality
    eye = v.transpose(1, 2) @ v
    torch.testing.assert_close(
        eye, torch.eye(n, device='cuda').expand(batch, n, n),
        rtol=1e-4, atol=1e-4,
    )
    # eigenvalue multiset matches torch (order-free compare)
    d_ref = torch.linalg.eigh(stack)[0]
    torch.testing.assert_close(
        torch.sort(w, dim=1).values, d_ref, rtol=1e-4, atol=1e-5,
    )


def test_syevj_small_clamps_negative() -> None:
    n = 32
    a = torch.diag(
        torch.linspace(-1.0, 1.0, n, device='cuda'),
    ).unsqueeze(0).contiguous()
    w, v = _ext().syevj_small(a, 20, 1e-5)
    torch.cuda.synchronize()
    assert (w >= 0).all()


def test_syevd_batched_repeated_calls_track_data() -> None:
    """Repeated ops.eigh_batched calls with the same (B, n) shape but
    DIFFERENT data must track the new inputs (persistent workspace
    buffers are cached per shape; this is the gate that exposed the
    round-1 hipGraph replay as unsound before it was deleted)."""
    from kfac_amd import ops

    torch.manual_seed(7)
    for trial in range(3):
        b, n = 3, 200
        a = torch.randn(b, n, n, device='cuda')
        stack = (a @ a.transpose(1, 2)) / n + torch.eye(n, device='cuda')
        d, q = ops.eigh_batched(stack)
        d_ref, _ = torch.linalg.eigh(stack)
        torch.cuda.synchronize()
        torch.testing.assert_close(d, d_ref, rtol=1e-4, atol=1e-4)
        recon = q @ torch.diag_embed(d) @ q.transpose(1, 2)
        torch.testing.assert_close(recon, stack, rtol=1e-4, atol=1e-4)
        eye = torch.eye(n, device='cuda').expand(b, n, n)
        torch.testing.assert_close(
            q @ q.transpose(1, 2), eye, rtol=1e-4, atol=1e-4,
        )


def test_eigh_batched_small_jacobi() -> None:
    from kfac_amd import ops

    torch.manual_seed(8)
    a = torch.randn(5, 48, 48, device='cuda')
    stack = (a @ a.transpose(1, 2)) / 48 + torch.eye(48, device='cuda')
    d, q = ops.eigh_batched(stack)
    torch.cuda.synchronize()
    recon = q @ torch.diag_embed(d) @ q.transpose(1, 2)
    torch.testing.assert_close(recon, stack, rtol=1e-3, atol=1e-3)


def test_eigh_batched_degenerate_identity_screen() -> None:
    """Fully-decayed scalar-identity factors (with denormal
    off-diagonals) make rocSOLVER syevd fail silently (info != 0) and
    even torch.linalg.eigh mangle the basis — the near-diagonal screen
    must return the EXACT (diag, I) answer instead."""
    from kfac_amd import ops

    b, n = 4, 256
    stack = 0.569 * torch.eye(n, device='cuda').expand(b, n, n).contiguous()
    # denormal off-diagonal dust (the observed failure pattern)
    stack = stack + 1e-42 * torch.randn(b, n, n, device='cuda').abs()
    stack = 0.5 * (stack + stack.transpose(1, 2))
    d, q = ops.eigh_batched(stack)
    torch.cuda.synchronize()
    eye = torch.eye(n, device='cuda').expand(b, n, n)
    torch.testing.assert_close(q, eye)
    torch.testing.assert_close(
        d, stack.diagonal(dim1=-2, dim2=-1), rtol=0, atol=0,
    )
