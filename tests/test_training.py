"""End-to-end training convergence gates (reference tests/training_test.py).

Every case runs inside a forked worker (even world_size=1): running
autograd in the pytest parent and later forking deadlocks the child's
autograd engine — the same hazard the reference works around
(tests/training_test.py:68-75).
"""

from __future__ import annotations

import pickle
import sys

import pytest
import tempfile

import torch

sys.path.insert(0, '.')

from kfac_amd import KFACPreconditioner  # noqa: E402
from kfac_amd.enums import DistributedStrategy  # noqa: E402
from testing.distributed import run_distributed  # noqa: E402
from testing.models import LeNet  # noqa: E402
from testing.models import TinyModel  # noqa: E402


def _train(
    model: torch.nn.Module,
    x: torch.Tensor,
    y: torch.Tensor,
    steps: int = 20,
    dist_avg: bool = True,
    **kfac_kwargs,
) -> list[float]:
    optimizer = torch.optim.SGD(model.parameters(), lr=0.01)
    precon = KFACPreconditioner(
        model,
        factor_update_steps=1,
        inv_update_steps=2,
        lr=0.01,
        **kfac_kwargs,
    )
    criterion = torch.nn.CrossEntropyLoss()
    losses = []
    for _ in range(steps):
        optimizer.zero_grad()
        loss = criterion(model(x), y)
        loss.backward()
        if dist_avg and torch.distributed.is_initialized():
            world = torch.distributed.get_world_size()
            for p in model.parameters():
                torch.distributed.all_reduce(p.grad)
                p.grad /= world
        precon.step()
        optimizer.step()
        losses.append(loss.item())
    return losses


def _tiny_case(**kwargs) -> None:
    torch.manual_seed(42)
    model = TinyModel()
    x = torch.randn(32, 10)
    y = torch.randint(0, 3, (32,))
    losses = _train(model, x, y, **kwargs)
    assert losses[0] > losses[-1], losses


def _lenet_case() -> None:
    torch.manual_seed(42)
    model = LeNet()
    x = torch.randn(16, 1, 28, 28)
    y = torch.randint(0, 10, (16,))
    losses = _train(model, x, y, steps=10)
    assert losses[0] > losses[-1], losses


def _dist_case(strategy_name: str, **kfac_kwargs) -> None:
    strategy = DistributedStrategy[strategy_name]
    torch.manual_seed(42)
    model = TinyModel()
    for p in model.parameters():
        torch.distributed.broadcast(p.data, src=0)
    x = torch.randn(32, 10)
    y = torch.randint(0, 3, (32,))
    losses = _train(
        model, x, y, steps=10, grad_worker_fraction=strategy, **kfac_kwargs,
    )
    assert losses[0] > losses[-1], losses


def test_tiny_model_loss_decreases() -> None:
    run_distributed(1, _tiny_case)


def test_lenet_conv_loss_decreases() -> None:
    run_distributed(1, _lenet_case)


def test_tiny_model_inverse_method() -> None:
    run_distributed(1, _tiny_case, compute_method='inverse')


def test_distributed_training_hybrid() -> None:
    run_distributed(2, _dist_case, 'HYBRID_OPT')


def test_distributed_training_hybrid_world4() -> None:
    # world 4 at fraction 0.5: 2 grad workers per layer, gradient
    # broadcasts active — the partitioned grouped-precondition path's
    # protocol (grad workers precondition, then broadcast) at the
    # grid shape BASELINE.json names for 8-GPU runs.
    run_distributed(4, _dist_case, 'HYBRID_OPT')


def test_distributed_training_mem_opt() -> None:
    run_distributed(2, _dist_case, 'MEM_OPT')


def test_distributed_training_comm_opt() -> None:
    run_distributed(4, _dist_case, 'COMM_OPT')


def _single_reference(path: str, **kfac_kwargs) -> None:
    torch.manual_seed(7)
    model = TinyModel()
    x = torch.randn(32, 10)
    y = torch.randint(0, 3, (32,))
    sd0 = {k: v.clone() for k, v in model.state_dict().items()}
    # strategy kwargs are world-size-dependent; only the compute method
    # changes the single-process math
    ref_kwargs = {
        k: v for k, v in kfac_kwargs.items() if k == 'compute_method'
    }
    losses = _train(model, x, y, steps=5, dist_avg=False, **ref_kwargs)
    with open(path, 'wb') as f:
        pickle.dump((sd0, x, y, losses), f)


def _half_batch_distributed(path: str, **kfac_kwargs) -> None:
    with open(path, 'rb') as fh:
        sd0, x, y, losses_single = pickle.load(fh)
    rank = torch.distributed.get_rank()
    model = TinyModel()
    model.load_state_dict(sd0)
    half = x.size(0) // 2
    xs = x[rank * half : (rank + 1) * half]
    ys = y[rank * half : (rank + 1) * half]
    losses = _train(model, xs, ys, steps=5, **kfac_kwargs)
    lt = torch.tensor(losses)
    torch.distributed.all_reduce(lt)
    lt /= 2
    torch.testing.assert_close(
        lt, torch.tensor(losses_single), rtol=1e-3, atol=1e-4,
    )


@pytest.mark.parametrize(
    'kfac_kwargs',
    [
        {},
        {'grad_worker_fraction': DistributedStrategy.HYBRID_OPT},
        {'grad_worker_fraction': DistributedStrategy.MEM_OPT},
        {'compute_method': 'inverse'},
        {'symmetry_aware': True},
        # prediv (eigenvalue outer product) requires colocated factors
        {'colocate_factors': False, 'compute_eigenvalue_outer_product': False},
        {'update_factors_in_hook': False},
    ],
    ids=[
        'comm-opt',
        'hybrid-opt',
        'mem-opt',
        'inverse',
        'symmetry-aware',
        'no-colocate',
        'factors-in-step',
    ],
)
def test_distributed_matches_single_process(kfac_kwargs) -> None:
    """World-2 training on half batches == single-process on the full batch.

    Validates the whole distributed pipeline numerically for every
    worker-placement strategy AND the explicit-inverse method: factor
    allreduce-averaging, inverse/eigen broadcasts, gradient broadcasts
    (HYBRID/MEM-OPT) and DDP-style grad averaging together reproduce
    the single-process K-FAC trajectory.
    """
    with tempfile.NamedTemporaryFile(suffix='.pkl') as f:
        run_distributed(1, _single_reference, f.name, **kfac_kwargs)
        run_distributed(2, _half_batch_distributed, f.name, **kfac_kwargs)


def test_distributed_training_hybrid_world8() -> None:
    # the driver's 8-GPU scaling run shape: world 8, HYBRID 0.5 ->
    # 4 gradient workers per layer in a 4x2 grid, inverse AND gradient
    # broadcasts active. Protocol-only on CPU gloo; RCCL first contact
    # is the driver's run by design.
    run_distributed(8, _dist_case, 'HYBRID_OPT')


def test_distributed_training_mem_opt_world8() -> None:
    # BASELINE.json's grad_worker_fraction=1/8 config shape: every
    # layer has ONE grad worker broadcasting to the other 7 ranks.
    run_distributed(8, _dist_case, 'MEM_OPT')


def test_distributed_training_inverse_method_hybrid() -> None:
    """INVERSE method across ranks: explicit A_inv/G_inv broadcast from
    the inverse workers to the grad-worker groups (layers/inverse.py)."""
    run_distributed(2, _dist_case, 'HYBRID_OPT', compute_method='inverse')


def test_distributed_training_inverse_method_mem_opt_world4() -> None:
    run_distributed(4, _dist_case, 'MEM_OPT', compute_method='inverse')


def test_distributed_training_symmetry_aware_hybrid() -> None:
    """Triu-packed factor allreduce + inverse broadcast wire format."""
    run_distributed(
        2, _dist_case, 'HYBRID_OPT', symmetry_aware=True,
    )


def _single_reference_lenet(path: str) -> None:
    torch.manual_seed(9)
    model = LeNet()
    x = torch.randn(16, 1, 28, 28)
    y = torch.randint(0, 10, (16,))
    sd0 = {k: v.clone() for k, v in model.state_dict().items()}
    losses = _train(model, x, y, steps=4, dist_avg=False)
    with open(path, 'wb') as f:
        pickle.dump((sd0, x, y, losses), f)


def _half_batch_distributed_lenet(path: str) -> None:
    with open(path, 'rb') as fh:
        sd0, x, y, losses_single = pickle.load(fh)
    rank = torch.distributed.get_rank()
    model = LeNet()
    model.load_state_dict(sd0)
    half = x.size(0) // 2
    xs = x[rank * half : (rank + 1) * half]
    ys = y[rank * half : (rank + 1) * half]
    losses = _train(model, xs, ys, steps=4)
    lt = torch.tensor(losses)
    torch.distributed.all_reduce(lt)
    lt /= 2
    torch.testing.assert_close(
        lt, torch.tensor(losses_single), rtol=1e-3, atol=1e-4,
    )


def test_distributed_conv_matches_single_process() -> None:
    """Conv2d factor path (im2col A, spatial-averaged G) under world-2:
    half-batch training must reproduce the single-process trajectory."""
    with tempfile.NamedTemporaryFile(suffix='.pkl') as f:
        run_distributed(1, _single_reference_lenet, f.name)
        run_distributed(2, _half_batch_distributed_lenet, f.name)


def _resume_case(strategy_name: str) -> None:
    """Interrupt at step 6, checkpoint, rebuild, resume 4 more steps —
    must match an uninterrupted 10-step run exactly."""
    strategy = DistributedStrategy[strategy_name]
    rank = torch.distributed.get_rank()
    world = torch.distributed.get_world_size()
    g = torch.Generator().manual_seed(13)
    xs = [torch.randn(16, 10, generator=g) for _ in range(10)]
    ys = [torch.randint(0, 3, (16,), generator=g) for _ in range(10)]

    def build():
        torch.manual_seed(3)
        model = TinyModel()
        precon = KFACPreconditioner(
            model,
            factor_update_steps=1,
            inv_update_steps=2,
            lr=0.05,
            grad_worker_fraction=strategy,
        )
        return model, precon

    def run(model, precon, lo, hi):
        opt = torch.optim.SGD(model.parameters(), lr=0.05)
        out = []
        half = 16 // world
        for x, y in zip(xs[lo:hi], ys[lo:hi]):
            xb = x[rank * half : (rank + 1) * half]
            yb = y[rank * half : (rank + 1) * half]
            opt.zero_grad()
            loss = torch.nn.functional.cross_entropy(model(xb), yb)
            loss.backward()
            for p_ in model.parameters():
                torch.distributed.all_reduce(p_.grad)
                p_.grad /= world
            precon.step()
            opt.step()
            out.append(loss.item())
        return out

    model, precon = build()
    full = run(model, precon, 0, 10)

    model, precon = build()
    part = run(model, precon, 0, 6)
    sd = precon.state_dict()
    msd = {k: v.clone() for k, v in model.state_dict().items()}
    model2, precon2 = build()
    model2.load_state_dict(msd)
    precon2.load_state_dict(sd, compute_inverses=True)
    assert precon2.steps == precon.steps
    part += run(model2, precon2, 6, 10)
    torch.testing.assert_close(
        torch.tensor(part), torch.tensor(full), rtol=1e-4, atol=1e-6,
    )


@pytest.mark.parametrize('strategy', ['COMM_OPT', 'HYBRID_OPT'])
def test_distributed_resume_matches_uninterrupted(strategy: str) -> None:
    run_distributed(2, _resume_case, strategy)


def _accum_single_reference(path: str) -> None:
    torch.manual_seed(17)
    model = TinyModel()
    x = torch.randn(32, 10)
    y = torch.randint(0, 3, (32,))
    sd0 = {k: v.clone() for k, v in model.state_dict().items()}
    losses = _train(model, x, y, steps=5, dist_avg=False)
    with open(path, 'wb') as f:
        pickle.dump((sd0, x, y, losses), f)


def _accum_distributed(path: str) -> None:
    """world-2, accumulation_steps=2: each rank sees two 8-sample
    micro-batches per optimizer step. Losses use sum-reduction divided
    by the GLOBAL batch so every sample's grad-output scale matches the
    single 32-sample pass exactly — G factors are covariances of the
    raw grad-outputs, so per-sample scaling is part of the semantics.
    Factors accumulate per mini-step and must average to the full-batch
    covariance."""
    with open(path, 'rb') as fh:
        sd0, x, y, losses_single = pickle.load(fh)
    rank = torch.distributed.get_rank()
    model = TinyModel()
    model.load_state_dict(sd0)
    precon = KFACPreconditioner(
        model,
        factor_update_steps=1,
        inv_update_steps=2,
        lr=0.01,
        accumulation_steps=2,
    )
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    quarter = 8
    losses = []
    for _ in range(5):
        opt.zero_grad()
        total = 0.0
        for micro in range(2):
            lo = (2 * rank + micro) * quarter
            xb, yb = x[lo : lo + quarter], y[lo : lo + quarter]
            loss = (
                torch.nn.functional.cross_entropy(
                    model(xb), yb, reduction='sum',
                )
                / 32
            )
            loss.backward()
            total += loss.item()
        for p_ in model.parameters():
            torch.distributed.all_reduce(p_.grad)
        precon.step()
        opt.step()
        losses.append(total)
    lt = torch.tensor(losses)
    torch.distributed.all_reduce(lt)
    torch.testing.assert_close(
        lt, torch.tensor(losses_single), rtol=1e-3, atol=1e-4,
    )


def test_gradient_accumulation_distributed_matches_single() -> None:
    """2 ranks x 2 micro-batches == one 32-sample single-process step."""
    with tempfile.NamedTemporaryFile(suffix='.pkl') as f:
        run_distributed(1, _accum_single_reference, f.name)
        run_distributed(2, _accum_distributed, f.name)


def _amp_scaled_case() -> None:
    """grad_scaler semantics: a loss scaled by S (grad-outputs S-scaled
    in the hooks, corrected by the 1/S^2 covariance coefficient) plus an
    unscale_ of .grad before step() must reproduce the unscaled run."""
    scale = 1024.0
    torch.manual_seed(31)
    model_a = TinyModel()
    model_b = TinyModel()
    model_b.load_state_dict(model_a.state_dict())
    x = torch.randn(32, 10)
    y = torch.randint(0, 3, (32,))

    pa = KFACPreconditioner(
        model_a, factor_update_steps=1, inv_update_steps=2, lr=0.01,
    )
    pb = KFACPreconditioner(
        model_b,
        factor_update_steps=1,
        inv_update_steps=2,
        lr=0.01,
        grad_scaler=lambda: scale,
    )
    oa = torch.optim.SGD(model_a.parameters(), lr=0.01)
    ob = torch.optim.SGD(model_b.parameters(), lr=0.01)
    for _ in range(6):
        oa.zero_grad()
        torch.nn.functional.cross_entropy(model_a(x), y).backward()
        pa.step()
        oa.step()

        ob.zero_grad()
        (torch.nn.functional.cross_entropy(model_b(x), y) * scale).backward()
        for q in model_b.parameters():  # scaler.unscale_(optimizer)
            q.grad /= scale
        pb.step()
        ob.step()
    for ka, kb in zip(
        model_a.state_dict().values(), model_b.state_dict().values(),
    ):
        torch.testing.assert_close(ka, kb, rtol=1e-4, atol=1e-6)


def test_amp_grad_scaler_matches_unscaled() -> None:
    run_distributed(1, _amp_scaled_case)


def _reduced_precision_case() -> None:
    """inv_dtype=bf16 second-order state trains stably and is actually
    stored reduced (reference supports reduced-precision
    eigendecomposition storage)."""
    torch.manual_seed(5)
    model = TinyModel()
    x = torch.randn(32, 10)
    y = torch.randint(0, 3, (32,))
    precon = KFACPreconditioner(
        model,
        factor_update_steps=1,
        inv_update_steps=2,
        lr=0.01,
        inv_dtype=torch.bfloat16,
    )
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    losses = []
    for _ in range(12):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(model(x), y)
        loss.backward()
        precon.step()
        opt.step()
        losses.append(loss.item())
    assert losses[0] > losses[-1], losses
    layer = next(iter(precon._layers.values()))[1]
    assert layer.qa.dtype == torch.bfloat16
    assert layer.qg.dtype == torch.bfloat16


def test_reduced_precision_inverse_state() -> None:
    run_distributed(1, _reduced_precision_case)


def _ddp_half_batch(path: str) -> None:
    """Real DistributedDataParallel wrapping (the README/examples usage):
    registration walks the DDP wrapper, DDP itself averages gradients."""
    with open(path, 'rb') as fh:
        sd0, x, y, losses_single = pickle.load(fh)
    rank = torch.distributed.get_rank()
    model = TinyModel()
    model.load_state_dict(sd0)
    ddp = torch.nn.parallel.DistributedDataParallel(model)
    precon = KFACPreconditioner(
        ddp, factor_update_steps=1, inv_update_steps=2, lr=0.01,
    )
    opt = torch.optim.SGD(ddp.parameters(), lr=0.01)
    half = x.size(0) // 2
    xs = x[rank * half : (rank + 1) * half]
    ys = y[rank * half : (rank + 1) * half]
    losses = []
    for _ in range(5):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(ddp(xs), ys)
        loss.backward()  # DDP averages gradients here
        precon.step()
        opt.step()
        losses.append(loss.item())
    lt = torch.tensor(losses)
    torch.distributed.all_reduce(lt)
    lt /= 2
    torch.testing.assert_close(
        lt, torch.tensor(losses_single), rtol=1e-3, atol=1e-4,
    )


def test_ddp_wrapped_matches_single_process() -> None:
    with tempfile.NamedTemporaryFile(suffix='.pkl') as f:
        run_distributed(1, _single_reference, f.name)
        run_distributed(2, _ddp_half_batch, f.name)


def _inverse_warm_cold_case() -> None:
    """The batched warm (Newton-Schulz) inverse phase must reproduce the
    exact-inverse trajectory across multiple phases."""
    import os

    params = {}
    for warm in (True, False):
        os.environ['KFAC_AMD_WARM_INV'] = '1' if warm else '0'
        try:
            torch.manual_seed(11)
            model = TinyModel()
            g = torch.Generator().manual_seed(6)
            precon = KFACPreconditioner(
                model,
                factor_update_steps=1,
                inv_update_steps=2,
                lr=0.01,
                compute_method='inverse',
            )
            opt = torch.optim.SGD(model.parameters(), lr=0.01)
            for _ in range(8):
                x = torch.randn(32, 10, generator=g)
                y = torch.randint(0, 3, (32,), generator=g)
                opt.zero_grad()
                loss = torch.nn.functional.cross_entropy(model(x), y)
                loss.backward()
                precon.step()
                opt.step()
        finally:
            os.environ.pop('KFAC_AMD_WARM_INV', None)
        counts = [
            getattr(layer, '_warm_inv_phases_a', 0)
            for _, layer in precon._layers.values()
        ]
        # the warm path must actually ENGAGE (certified refinements) in
        # the warm run — otherwise this equivalence proves nothing
        if warm:
            assert max(counts) >= 2, counts
        else:
            assert max(counts) == 0, counts
        params[warm] = {k: v.clone() for k, v in model.state_dict().items()}
    for k in params[True]:
        torch.testing.assert_close(
            params[True][k], params[False][k], rtol=1e-4, atol=1e-6,
        )


def test_inverse_warm_phase_matches_exact() -> None:
    run_distributed(1, _inverse_warm_cold_case)
