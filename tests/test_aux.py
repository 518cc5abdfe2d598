"""Unit tests for aux subsystems: scheduler, hyperparams, tracing,
registration, module helpers, preconditioner validation + checkpointing.

Coverage model: reference tests/scheduler_test.py, hyperparams_test.py,
tracing_test.py, layers/register_test.py, layers/modules_test.py,
base_preconditioner_test.py.
"""

from __future__ import annotations

import sys

import pytest
import torch

sys.path.insert(0, '.')

from kfac_amd import KFACPreconditioner  # noqa: E402
from kfac_amd.distributed import TorchDistributedCommunicator  # noqa: E402
from kfac_amd.enums import AssignmentStrategy  # noqa: E402
from kfac_amd.hyperparams import exp_decay_factor_averaging  # noqa: E402
from kfac_amd.layers.eigen import KFACEigenLayer  # noqa: E402
from kfac_amd.layers.modules import Conv2dModuleHelper  # noqa: E402
from kfac_amd.layers.modules import LinearModuleHelper  # noqa: E402
from kfac_amd.layers.register import any_match  # noqa: E402
from kfac_amd.layers.register import get_flattened_modules  # noqa: E402
from kfac_amd.layers.register import register_modules  # noqa: E402
from kfac_amd.scheduler import LambdaParamScheduler  # noqa: E402
from kfac_amd import tracing  # noqa: E402
from testing.models import LeNet  # noqa: E402
from testing.models import TinyModel  # noqa: E402


# ---------------------------------------------------------------- hyperparams


@pytest.fixture(autouse=True)
def _seed_rng():
    # deterministic per-test RNG regardless of suite ordering
    torch.manual_seed(0)


def test_exp_decay_factor_averaging() -> None:
    fn = exp_decay_factor_averaging()
    assert fn(0) == fn(1) == 0.0
    assert fn(2) == 0.5
    assert fn(100) == pytest.approx(0.95)  # capped at min_value
    with pytest.raises(ValueError):
        fn(-1)
    with pytest.raises(ValueError):
        exp_decay_factor_averaging(min_value=0)


# ---------------------------------------------------------------- scheduler

def test_lambda_param_scheduler() -> None:
    model = TinyModel()
    p = KFACPreconditioner(model, damping=0.01, lr=0.1)
    sched = LambdaParamScheduler(
        p,
        damping_lambda=lambda step: 0.5,
        lr_lambda=lambda step: 2.0,
        factor_update_steps_lambda=lambda step: 3.0,
    )
    sched.step()
    assert p.damping == pytest.approx(0.005)
    assert p.lr == pytest.approx(0.2)
    assert p.factor_update_steps == 3


def test_scheduler_step_count_truncation_guard() -> None:
    # A decay schedule that would truncate factor_update_steps to 0 must
    # raise instead of silently producing a modulo-by-zero time bomb
    # (the reference's scheduler truncates silently).
    model = TinyModel()
    p = KFACPreconditioner(model, factor_update_steps=1)
    sched = LambdaParamScheduler(
        p, factor_update_steps_lambda=lambda step: 0.5,
    )
    with pytest.raises(ValueError, match='stay >= 1'):
        sched.step()
    assert p.factor_update_steps == 1  # unchanged on failure


def test_scheduler_rejects_callable_params() -> None:
    model = TinyModel()
    p = KFACPreconditioner(model, damping=lambda s: 0.01)
    with pytest.raises(ValueError):
        LambdaParamScheduler(p, damping_lambda=lambda s: 0.5)


def test_callable_hyperparams_resolve_with_step() -> None:
    model = TinyModel()
    p = KFACPreconditioner(
        model,
        damping=lambda s: 0.1 / (s + 1),
        factor_update_steps=lambda s: 2,
    )
    assert p.damping == pytest.approx(0.1)
    p._steps = 9
    assert p.damping == pytest.approx(0.01)
    assert p.factor_update_steps == 2


# ---------------------------------------------------------------- tracing

def test_trace_decorator() -> None:
    tracing.clear_trace()

    @tracing.trace()
    def f(x: int) -> int:
        return x * 2

    assert f(3) == 6
    assert f(4) == 8
    t = tracing.get_trace()
    (name,) = t.keys()
    assert 'f' in name
    assert t[name] >= 0
    total = tracing.get_trace(average=False)[name]
    assert total >= t[name]
    tracing.log_trace()
    tracing.clear_trace()
    assert tracing.get_trace() == {}


# ---------------------------------------------------------------- register

def test_register_modules_counts() -> None:
    tdc = TorchDistributedCommunicator()
    layers = register_modules(LeNet(), KFACEigenLayer, skip_layers=[], tdc=tdc)
    # 2 conv + 3 linear
    assert len(layers) == 5


def test_register_skip_by_name_and_class() -> None:
    tdc = TorchDistributedCommunicator()
    layers = register_modules(
        LeNet(), KFACEigenLayer, skip_layers=['conv.*'], tdc=tdc,
    )
    assert len(layers) == 3
    layers = register_modules(
        LeNet(), KFACEigenLayer, skip_layers=['Conv2d'], tdc=tdc,
    )
    assert len(layers) == 3
    layers = register_modules(
        LeNet(), KFACEigenLayer, skip_layers=['.*'], tdc=tdc,
    )
    assert len(layers) == 0


def test_register_skips_frozen() -> None:
    model = LeNet()
    model.fc1.weight.requires_grad_(False)
    tdc = TorchDistributedCommunicator()
    layers = register_modules(model, KFACEigenLayer, skip_layers=[], tdc=tdc)
    assert len(layers) == 4


def test_any_match_is_fullmatch() -> None:
    assert any_match('conv1', ['conv1'])
    assert not any_match('xconv1y', ['conv1'])
    assert any_match('xconv1y', ['.*conv1.*'])


def test_get_flattened_modules_leaves_only() -> None:
    names = [n for n, _ in get_flattened_modules(LeNet())]
    assert 'conv1' in names and 'fc3' in names
    assert '' not in names  # root is not a leaf


# ---------------------------------------------------------------- helpers

def test_linear_helper_shapes_and_grads() -> None:
    lin = torch.nn.Linear(7, 3)
    h = LinearModuleHelper(lin)
    assert h.a_factor_shape == (8, 8)
    assert h.g_factor_shape == (3, 3)
    lin.weight.grad = torch.randn(3, 7)
    lin.bias.grad = torch.randn(3)
    g = h.get_grad()
    assert g.shape == (3, 8)
    torch.testing.assert_close(g[:, -1], lin.bias.grad)
    new = torch.randn(3, 8)
    h.set_grad(new)
    torch.testing.assert_close(lin.weight.grad, new[:, :-1])
    torch.testing.assert_close(lin.bias.grad, new[:, -1])


def test_conv_helper_shapes_and_grads() -> None:
    conv = torch.nn.Conv2d(2, 4, 3, padding=1)
    h = Conv2dModuleHelper(conv)
    assert h.a_factor_shape == (2 * 9 + 1, 2 * 9 + 1)
    assert h.g_factor_shape == (4, 4)
    conv.weight.grad = torch.randn(4, 2, 3, 3)
    conv.bias.grad = torch.randn(4)
    g = h.get_grad()
    assert g.shape == (4, 19)
    h.set_grad(g)
    torch.testing.assert_close(
        conv.weight.grad, g[:, :-1].reshape(4, 2, 3, 3),
    )


# -------------------------------------------------------- preconditioner API

def test_preconditioner_validation() -> None:
    model = TinyModel()
    with pytest.raises(ValueError):
        KFACPreconditioner(model, factor_update_steps=0)
    with pytest.raises(ValueError):
        KFACPreconditioner(model, inv_update_steps=-1)
    with pytest.raises(ValueError):
        KFACPreconditioner(model, damping=0)
    with pytest.raises(ValueError):
        KFACPreconditioner(model, factor_decay=1.5)
    with pytest.raises(ValueError):
        KFACPreconditioner(model, kl_clip=0.0)
    with pytest.raises(ValueError):
        KFACPreconditioner(model, allreduce_bucket_cap_mb=-1)
    with pytest.raises(ValueError):
        KFACPreconditioner(
            model,
            compute_eigenvalue_outer_product=True,
            colocate_factors=False,
        )
    with pytest.raises(ValueError):
        KFACPreconditioner(model, grad_worker_fraction=2.0)


def test_preconditioner_string_enums() -> None:
    model = TinyModel()
    p = KFACPreconditioner(
        model,
        assignment_strategy='memory',
        compute_method='inverse',
        compute_eigenvalue_outer_product=False,
    )
    assert p.assignment_strategy == AssignmentStrategy.MEMORY
    assert repr(p)  # smoke: sorted-config repr


def test_state_dict_roundtrip_and_resume() -> None:
    torch.manual_seed(0)
    model = TinyModel()
    x = torch.randn(16, 10)
    y = torch.randint(0, 3, (16,))
    p = KFACPreconditioner(model, factor_update_steps=1, inv_update_steps=1)
    loss = torch.nn.functional.cross_entropy(model(x), y)
    loss.backward()
    p.step()

    sd = p.state_dict()
    assert sd['steps'] == 1
    assert 'damping' in sd and 'layers' in sd
    assert set(next(iter(sd['layers'].values())).keys()) == {'A', 'G'}

    model2 = TinyModel()
    p2 = KFACPreconditioner(model2, factor_update_steps=1, inv_update_steps=1)
    p2.load_state_dict(sd, compute_inverses=True)
    assert p2.steps == 1
    for (_, (n1, l1)), (_, (n2, l2)) in zip(
        p._layers.items(), p2._layers.items(),
    ):
        torch.testing.assert_close(l1.a_factor, l2.a_factor)
        torch.testing.assert_close(l1.g_factor, l2.g_factor)
        # inverses recomputed on load
        assert l2.qa is not None and l2.qg is not None


def test_state_dict_without_factors() -> None:
    model = TinyModel()
    p = KFACPreconditioner(model)
    sd = p.state_dict(include_factors=False)
    assert 'layers' not in sd
    p2 = KFACPreconditioner(TinyModel())
    with pytest.warns(UserWarning):
        p2.load_state_dict(sd, compute_inverses=True)


def test_load_state_dict_layer_count_mismatch() -> None:
    p = KFACPreconditioner(TinyModel())
    sd = p.state_dict()
    sd['layers'] = {'one': {'A': None, 'G': None}}
    p2 = KFACPreconditioner(TinyModel())
    with pytest.raises(ValueError):
        p2.load_state_dict(sd)


def test_memory_usage_accounting() -> None:
    torch.manual_seed(0)
    model = TinyModel()
    p = KFACPreconditioner(model, factor_update_steps=1, inv_update_steps=1)
    x = torch.randn(8, 10)
    torch.nn.functional.cross_entropy(model(x), torch.randint(0, 3, (8,))).backward()
    p.step()
    usage = p.memory_usage()
    assert usage['a_factors'] > 0
    assert usage['g_factors'] > 0
    assert usage['a_inverses'] > 0
    assert usage['total'] == sum(v for k, v in usage.items() if k != 'total')


def test_reset_batch() -> None:
    torch.manual_seed(0)
    model = TinyModel()
    p = KFACPreconditioner(
        model, factor_update_steps=1, update_factors_in_hook=False,
    )
    x = torch.randn(8, 10)
    torch.nn.functional.cross_entropy(model(x), torch.randint(0, 3, (8,))).backward()
    assert any(l._a_count > 0 for _, l in p._layers.values())
    p.reset_batch()
    assert all(l._a_count == 0 for _, l in p._layers.values())


def test_gradient_accumulation_factors() -> None:
    """accumulation_steps=2: factor = EMA over the mean of 2 minibatches."""
    torch.manual_seed(1)
    lin_model = TinyModel()
    p = KFACPreconditioner(
        lin_model, factor_update_steps=1, inv_update_steps=1,
        accumulation_steps=2, lr=0.01,
    )
    x1 = torch.randn(8, 10)
    x2 = torch.randn(8, 10)
    y = torch.randint(0, 3, (8,))
    for x in (x1, x2):
        torch.nn.functional.cross_entropy(lin_model(x), y).backward()
    # after 2 mini-steps the hook-boundary fired: factors EMA'd
    name, layer = next(iter(p._layers.values()))
    a = layer.a_factor
    assert a is not None
    # manual: mean of the two minibatch covariances, identity-init EMA
    xs = [torch.cat([x, x.new_ones(8, 1)], dim=1) for x in (x1, x2)]
    mean_cov = sum(xb.t() @ xb / 8 for xb in xs) / 2
    expected = 0.95 * torch.eye(11) + 0.05 * mean_cov
    torch.testing.assert_close(a, expected, rtol=1e-5, atol=1e-6)
    p.step()


def test_update_factors_in_hook_equivalence() -> None:
    """update_factors_in_hook True and False produce identical grads."""
    results = {}
    for in_hook in (True, False):
        torch.manual_seed(5)
        model = TinyModel()
        p = KFACPreconditioner(
            model, factor_update_steps=1, inv_update_steps=1, lr=0.01,
            update_factors_in_hook=in_hook,
        )
        x = torch.randn(16, 10)
        y = torch.randint(0, 3, (16,))
        torch.nn.functional.cross_entropy(model(x), y).backward()
        p.step()
        results[in_hook] = {
            n: prm.grad.clone() for n, prm in model.named_parameters()
        }
    for n in results[True]:
        torch.testing.assert_close(results[True][n], results[False][n])


def test_empty_registration_step() -> None:
    """A model with every layer skipped still steps cleanly."""
    model = TinyModel()
    p = KFACPreconditioner(model, skip_layers=['.*'])
    assert len(p._layers) == 0
    x = torch.randn(4, 10)
    torch.nn.functional.cross_entropy(model(x), torch.randint(0, 3, (4,))).backward()
    p.step()
    assert p.steps == 1
    assert p.memory_usage()['total'] == 0


def test_kl_clip_none_disables_scaling() -> None:
    torch.manual_seed(2)
    model = TinyModel()
    p = KFACPreconditioner(
        model, factor_update_steps=1, inv_update_steps=1, kl_clip=None,
    )
    x = torch.randn(8, 10)
    torch.nn.functional.cross_entropy(model(x), torch.randint(0, 3, (8,))).backward()
    p.step()  # must not raise; grads preconditioned unscaled
    for _, prm in model.named_parameters():
        assert torch.isfinite(prm.grad).all()


def _trace_worker_synced() -> None:
    import torch.distributed as dist

    from kfac_amd import tracing

    @tracing.trace(sync=True)
    def traced_barrier_op() -> int:
        return dist.get_rank()

    traced_barrier_op()
    traced_barrier_op()
    t = tracing.get_trace(average=True)
    key = next(k for k in t if k.endswith('traced_barrier_op'))
    assert t[key] >= 0.0
    tracing.clear_trace()
    assert tracing.get_trace() == {}


def test_trace_synced_world2() -> None:
    """Reference parity: synced (dist.barrier-bracketed) tracing at world 2
    (tracing_test.py)."""
    from testing.distributed import run_distributed

    run_distributed(2, _trace_worker_synced)


def _harness_failing_worker() -> None:
    import torch.distributed as dist

    if dist.get_rank() == 1:
        raise AssertionError('intentional failure on rank 1')


def test_harness_surfaces_worker_failure() -> None:
    """The fork harness itself is tested (reference
    tests/testing/distributed_wrapper_test.py): a failing rank must fail
    the test, not hang or pass silently."""
    import pytest as _pytest

    from testing.distributed import run_distributed

    with _pytest.raises(AssertionError, match='ranks'):
        run_distributed(2, _harness_failing_worker, timeout=60.0)


def test_state_dict_excludes_callable_hyperparams() -> None:
    """Reference parity (base_preconditioner.py:215-247): callable
    hyperparameters are not serializable and are omitted from state."""
    model = TinyModel()
    p = KFACPreconditioner(
        model,
        factor_update_steps=1,
        inv_update_steps=1,
        damping=lambda step: 0.001 * 0.99**step,
        lr=0.1,
    )
    sd = p.state_dict(include_factors=False)
    assert 'damping' not in sd
    assert sd['lr'] == 0.1
    assert sd['steps'] == 0


def test_shared_module_registered_once() -> None:
    """A module reused at two points in the graph gets ONE K-FAC layer
    (named_modules dedup; reference register.py:20-28)."""
    import torch.nn as nn

    from kfac_amd.layers.register import get_flattened_modules

    lin = nn.Linear(4, 4)
    model = nn.Sequential(lin, nn.ReLU(), lin)
    mods = get_flattened_modules(model)
    assert sum(1 for _, m in mods if m is lin) == 1


def test_grouped_conv_skipped_gracefully() -> None:
    """Grouped/depthwise Conv2d is skipped with a warning instead of
    crashing mid-precondition (the reference has no groups handling and
    fails with an opaque shape error)."""
    import warnings as _warnings

    m = torch.nn.Sequential(
        torch.nn.Conv2d(8, 16, 3, padding=1, groups=4),
        torch.nn.Flatten(),
        torch.nn.Linear(16 * 8 * 8, 4),
    )
    with _warnings.catch_warnings(record=True) as rec:
        _warnings.simplefilter('always')
        p = KFACPreconditioner(m, factor_update_steps=1, inv_update_steps=1)
    assert any('grouped' in str(w.message) for w in rec)
    assert len(p._layers) == 1  # only the Linear
    x = torch.randn(2, 8, 8, 8)
    torch.nn.functional.cross_entropy(
        m(x), torch.randint(0, 4, (2,)),
    ).backward()
    p.step()  # completes; conv trains unpreconditioned


def test_dilated_and_string_padded_conv2d_skipped() -> None:
    """Conv2d with dilation != 1 or 'same'/'valid' padding is skipped with
    a warning: the fused im2col models neither, and computing a wrong A
    factor silently would be worse than not preconditioning."""
    import warnings as _warnings

    for conv in (
        torch.nn.Conv2d(4, 8, 3, dilation=2),
        torch.nn.Conv2d(4, 8, 3, padding='same'),
    ):
        m = torch.nn.Sequential(
            conv, torch.nn.Flatten(), torch.nn.Linear(8 * 4 * 4, 2),
        )
        with _warnings.catch_warnings(record=True) as rec:
            _warnings.simplefilter('always')
            p = KFACPreconditioner(
                m, factor_update_steps=1, inv_update_steps=1,
            )
        assert any('Conv2d supports' in str(w.message) for w in rec)
        assert len(p._layers) == 1  # only the Linear


def test_bucket_key_includes_average_flag() -> None:
    """Tensors reduced with different average flags must not share a
    bucket (the per-bucket scale is recorded at first open)."""
    from kfac_amd.distributed import TorchDistributedCommunicator

    tdc = TorchDistributedCommunicator()
    a = torch.ones(4)
    k1 = (tdc._group_key(None)) + (a.dtype, a.device, True)
    k2 = (tdc._group_key(None)) + (a.dtype, a.device, False)
    assert k1 != k2


def test_state_dict_second_order_resume() -> None:
    """include_second_order embeds eigendecompositions so a resume can
    skip the inverse recomputation (extension over the reference)."""
    torch.manual_seed(4)
    model = TinyModel()
    p = KFACPreconditioner(
        model, factor_update_steps=1, inv_update_steps=1, lr=0.1,
    )
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    x = torch.randn(16, 10)
    y = torch.randint(0, 3, (16,))
    for _ in range(2):
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        p.step()
        opt.step()
    sd = p.state_dict(include_second_order=True)
    layers_sd = sd['layers']
    assert any(
        k.startswith('so_') for v in layers_sd.values() for k in v
    )
    # plain format unchanged when not requested
    sd_plain = p.state_dict()
    assert all(
        set(v) == {'A', 'G'} for v in sd_plain['layers'].values()
    )

    model2 = TinyModel()
    p2 = KFACPreconditioner(
        model2, factor_update_steps=1, inv_update_steps=1, lr=0.1,
    )
    calls = {'n': 0}
    orig = type(list(p2._layers.values())[0][1]).compute_a_inv

    def counting(self, *a, **k):
        calls['n'] += 1
        return orig(self, *a, **k)

    import unittest.mock as mock

    with mock.patch.object(
        type(list(p2._layers.values())[0][1]), 'compute_a_inv', counting,
    ):
        p2.load_state_dict(sd, compute_inverses=True)
    assert calls['n'] == 0, 'recomputation should be skipped'
    for _, layer in p2._layers.values():
        assert layer.has_second_order_state()
    # training continues from the restored state
    opt2 = torch.optim.SGD(model2.parameters(), lr=0.1)
    opt2.zero_grad()
    torch.nn.functional.cross_entropy(model2(x), y).backward()
    p2.step()
    opt2.step()


def test_scheduler_rejects_disabled_param() -> None:
    """Scheduling a None (disabled) hyperparameter fails at construction,
    not with a TypeError deep in step()."""
    model = TinyModel()
    p = KFACPreconditioner(model, kl_clip=None)
    with pytest.raises(ValueError, match='disabled'):
        LambdaParamScheduler(p, kl_clip_lambda=lambda s: 0.99)
