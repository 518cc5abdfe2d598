"""Per-layer pipeline tests (coverage model: reference tests/layers/).

Drives the full KFAC layer state machine
(save -> update -> reduce -> inv -> broadcast -> precondition ->
update_grad) across {eigen, inverse} x {world 1, 4} x {broadcast on/off}
x {bucketed, symmetry-aware, grad-scaler, prediv, fp64} and checks the
math against plain torch ops.
"""

from __future__ import annotations

import sys

import pytest
import torch
import torch.distributed as dist

sys.path.insert(0, '.')

from kfac_amd.distributed import TorchDistributedCommunicator  # noqa: E402
from kfac_amd.enums import AllreduceMethod  # noqa: E402
from kfac_amd.layers.eigen import KFACEigenLayer  # noqa: E402
from kfac_amd.layers.inverse import KFACInverseLayer  # noqa: E402
from kfac_amd.layers.modules import LinearModuleHelper  # noqa: E402
from testing.distributed import run_distributed  # noqa: E402


def _drive_layer(
    layer_type: str = 'eigen',
    bucketed: bool = False,
    symmetry_aware: bool = False,
    grad_scaler: float | None = None,
    prediv: bool = False,
    dtype: torch.dtype = torch.float32,
    broadcast: bool = False,
) -> None:
    """Run the whole pipeline; assert against a torch reference."""
    world = dist.get_world_size() if dist.is_initialized() else 1
    rank = dist.get_rank() if dist.is_initialized() else 0
    torch.manual_seed(7)  # same on all ranks

    in_dim, out_dim, batch = 9, 5, 16
    module = torch.nn.Linear(in_dim, out_dim).to(dtype)
    module.weight.grad = torch.randn(out_dim, in_dim, dtype=dtype)
    module.bias.grad = torch.randn(out_dim, dtype=dtype)
    orig_w = module.weight.grad.clone()
    orig_b = module.bias.grad.clone()

    # rank-specific minibatch slice
    torch.manual_seed(100 + rank)
    x = torch.randn(batch, in_dim, dtype=dtype)
    g_out = torch.randn(batch, out_dim, dtype=dtype)
    if grad_scaler is not None:
        g_out = g_out * grad_scaler

    tdc = TorchDistributedCommunicator()
    kwargs = dict(
        tdc=tdc,
        allreduce_method=(
            AllreduceMethod.ALLREDUCE_BUCKETED
            if bucketed
            else AllreduceMethod.ALLREDUCE
        ),
        symmetry_aware=symmetry_aware,
        grad_scaler=(lambda: grad_scaler) if grad_scaler is not None else None,
        factor_dtype=dtype,
        inv_dtype=torch.float64 if dtype == torch.float64 else torch.float32,
    )
    if layer_type == 'eigen':
        layer = KFACEigenLayer(
            LinearModuleHelper(module), prediv_eigenvalues=prediv, **kwargs,
        )
    else:
        layer = KFACInverseLayer(LinearModuleHelper(module), **kwargs)

    layer.save_layer_input([x])
    layer.save_layer_grad_output((g_out,))
    layer.update_a_factor(alpha=0.95)
    layer.update_g_factor(alpha=0.95)
    layer.reduce_a_factor()
    layer.reduce_g_factor()
    tdc.flush_allreduce_buckets()

    damping = 1e-3
    src = 0
    if rank == src or not broadcast:
        layer.compute_a_inv(damping=damping)
        layer.compute_g_inv(damping=damping)
    if broadcast and world > 1:
        layer.broadcast_a_inv(src=src)
        layer.broadcast_g_inv(src=src)
    if rank == src or not broadcast:
        layer.preconditioned_grad(damping=damping)
    if broadcast and world > 1:
        layer.broadcast_grad(src=src)
    result = layer.grad.clone()
    layer.update_grad(scale=0.5)

    # ---- torch reference (average of per-rank factor contributions) ----
    fdt = torch.float32 if dtype != torch.float64 else torch.float64
    a_sum = torch.zeros(in_dim + 1, in_dim + 1, dtype=fdt)
    g_sum = torch.zeros(out_dim, out_dim, dtype=fdt)
    for r in range(world):
        torch.manual_seed(100 + r)
        xr = torch.randn(batch, in_dim, dtype=dtype)
        gr = torch.randn(batch, out_dim, dtype=dtype)
        # grad scaler applied then unscaled -> net identity on the factor
        xb = torch.cat([xr, xr.new_ones(batch, 1)], dim=1).to(fdt)
        a_sum += (xb.t() @ xb) / batch
        grf = gr.to(fdt)
        g_sum += (grf.t() @ grf) / batch
    a_new = a_sum / world
    g_new = g_sum / world
    a_fac = 0.95 * torch.eye(in_dim + 1, dtype=fdt) + 0.05 * a_new
    g_fac = 0.95 * torch.eye(out_dim, dtype=fdt) + 0.05 * g_new
    grad = torch.cat([orig_w, orig_b.view(-1, 1)], 1).to(fdt)
    if layer_type == 'eigen':
        da, qa = torch.linalg.eigh(a_fac)
        dg, qg = torch.linalg.eigh(g_fac)
        da = da.clamp(min=0)
        dg = dg.clamp(min=0)
        v1 = qg.t() @ grad @ qa
        v2 = v1 / (torch.outer(dg, da) + damping)
        expected = qg @ v2 @ qa.t()
    else:
        a_inv = torch.linalg.inv(
            a_fac + damping * torch.eye(in_dim + 1, dtype=fdt),
        )
        g_inv = torch.linalg.inv(
            g_fac + damping * torch.eye(out_dim, dtype=fdt),
        )
        expected = g_inv @ grad @ a_inv

    tol = 1e-4 if dtype != torch.float64 else 1e-8
    torch.testing.assert_close(
        result.to(fdt), expected, rtol=tol, atol=tol,
    )
    # update_grad wrote scaled grads back
    torch.testing.assert_close(
        module.weight.grad.to(fdt), 0.5 * expected[:, :-1], rtol=tol, atol=tol,
    )
    torch.testing.assert_close(
        module.bias.grad.to(fdt), 0.5 * expected[:, -1], rtol=tol, atol=tol,
    )


def test_eigen_single() -> None:
    run_distributed(1, _drive_layer, 'eigen')


def test_inverse_single() -> None:
    run_distributed(1, _drive_layer, 'inverse')


def test_eigen_prediv() -> None:
    run_distributed(1, _drive_layer, 'eigen', False, False, None, True)


def test_eigen_fp64() -> None:
    run_distributed(
        1, _drive_layer, 'eigen', dtype=torch.float64,
    )


def test_eigen_grad_scaler() -> None:
    run_distributed(1, _drive_layer, 'eigen', grad_scaler=3.0)


@pytest.mark.parametrize('bucketed', [False, True])
@pytest.mark.parametrize('symmetry_aware', [False, True])
def test_eigen_world4(bucketed: bool, symmetry_aware: bool) -> None:
    run_distributed(4, _drive_layer, 'eigen', bucketed, symmetry_aware)


def test_eigen_world4_broadcast() -> None:
    """MEM-OPT style: inverses+grad computed on rank 0, broadcast out."""
    run_distributed(4, _drive_layer, 'eigen', broadcast=True)


def test_inverse_world4_broadcast_symmetric() -> None:
    run_distributed(
        4, _drive_layer, 'inverse', False, True, broadcast=True,
    )


def test_conv1d_end_to_end() -> None:
    """Conv1d support (extension beyond the reference): factors, eigen
    precondition and a converging training loop on CPU."""
    import torch

    from kfac_amd import KFACPreconditioner

    torch.manual_seed(0)
    model = torch.nn.Sequential(
        torch.nn.Conv1d(4, 8, 3, padding=1),
        torch.nn.ReLU(),
        torch.nn.Conv1d(8, 8, 5, stride=2, padding=2),
        torch.nn.Flatten(),
        torch.nn.Linear(8 * 8, 3),
    )
    p = KFACPreconditioner(model, factor_update_steps=1, inv_update_steps=2)
    assert len(p._layers) == 3
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    x = torch.randn(16, 4, 16)
    y = torch.randint(0, 3, (16,))
    losses = []
    for _ in range(15):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        p.step()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]
    # A factor shape: in*k + bias column
    layer = next(
        lay for _, (n, lay) in p._layers.items() if n == '0'
    )
    assert layer.a_factor.shape == (4 * 3 + 1, 4 * 3 + 1)


def test_conv3d_matches_conv2d_when_depth1() -> None:
    """Conv3d with kd=1 on depth-1 input is exactly a Conv2d: the K-FAC
    factors must agree with the Conv2d helper's."""
    import torch

    from kfac_amd.layers.modules import (
        Conv2dModuleHelper,
        Conv3dModuleHelper,
    )

    torch.manual_seed(3)
    c2 = torch.nn.Conv2d(3, 6, (3, 3), stride=(2, 2), padding=(1, 1))
    c3 = torch.nn.Conv3d(3, 6, (1, 3, 3), stride=(1, 2, 2), padding=(0, 1, 1))
    h2 = Conv2dModuleHelper(c2)
    h3 = Conv3dModuleHelper(c3)
    x2 = torch.randn(4, 3, 10, 10)
    x3 = x2.unsqueeze(2)  # depth 1
    n = 3 * 9 + 1
    a2 = torch.zeros(n, n)
    a3 = torch.zeros(n, n)
    h2.accumulate_a_factor(x2, a2, 0.0, 1.0)
    h3.accumulate_a_factor(x3, a3, 0.0, 1.0)
    torch.testing.assert_close(a2, a3, rtol=1e-5, atol=1e-6)
    g2 = torch.randn(4, 6, 5, 5)
    g3 = g2.unsqueeze(2)
    o2 = torch.zeros(6, 6)
    o3 = torch.zeros(6, 6)
    h2.accumulate_g_factor(g2, o2, 0.0, 1.0)
    h3.accumulate_g_factor(g3, o3, 0.0, 1.0)
    torch.testing.assert_close(o2, o3, rtol=1e-5, atol=1e-6)


def test_conv3d_end_to_end() -> None:
    import torch

    from kfac_amd import KFACPreconditioner

    torch.manual_seed(1)
    model = torch.nn.Sequential(
        torch.nn.Conv3d(2, 4, 3, padding=1),
        torch.nn.ReLU(),
        torch.nn.Flatten(),
        torch.nn.Linear(4 * 4 * 6 * 6, 3),
    )
    p = KFACPreconditioner(model, factor_update_steps=1, inv_update_steps=2)
    assert len(p._layers) == 2
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    x = torch.randn(8, 2, 4, 6, 6)
    y = torch.randint(0, 3, (8,))
    losses = []
    for _ in range(15):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        p.step()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]


def test_nonsymmetric_eigen_path() -> None:
    """K6: a module helper reporting non-symmetric factors routes
    through torch.linalg.eig (.real parts); the preconditioned gradient
    is basis-invariant, so it must match the symmetric eigh path
    (reference tests/layers/layers_test.py test_nonsymmetric_eigen)."""

    class NonSymHelper(LinearModuleHelper):
        def has_symmetric_factors(self) -> bool:
            return False

    torch.manual_seed(3)
    in_dim, out_dim, batch = 7, 4, 32
    results = {}
    for helper_cls in (LinearModuleHelper, NonSymHelper):
        torch.manual_seed(3)
        module = torch.nn.Linear(in_dim, out_dim)
        module.weight.grad = torch.randn(out_dim, in_dim)
        module.bias.grad = torch.randn(out_dim)
        layer = KFACEigenLayer(
            helper_cls(module),
            tdc=TorchDistributedCommunicator(),
            allreduce_method=AllreduceMethod.ALLREDUCE,
        )
        assert layer.symmetric_factors == (helper_cls is LinearModuleHelper)
        x = torch.randn(batch, in_dim)
        g = torch.randn(batch, out_dim)
        layer.save_layer_input([x])
        layer.save_layer_grad_output((g,))
        layer.update_a_factor(0.95)
        layer.update_g_factor(0.95)
        layer.compute_a_inv(damping=1e-3)
        layer.compute_g_inv(damping=1e-3)
        assert layer.qa.dtype == torch.float32
        assert (layer.da >= 0).all() and (layer.dg >= 0).all()
        layer.preconditioned_grad(damping=1e-3)
        results[helper_cls.__name__] = layer.grad.clone()
    torch.testing.assert_close(
        results['NonSymHelper'],
        results['LinearModuleHelper'],
        rtol=1e-4,
        atol=1e-5,
    )
