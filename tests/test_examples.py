"""Example CLIs smoke tests (subprocess, synthetic data, CPU)."""

from __future__ import annotations

import os
import pytest
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args: list[str], timeout: int = 240) -> subprocess.CompletedProcess:
    return subprocess.run(
        [sys.executable, *args],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=timeout,
    )


def test_cifar_example_runs_and_checkpoints() -> None:
    with tempfile.TemporaryDirectory() as td:
        args = [
            'examples/torch_cifar10_resnet.py',
            '--epochs', '1',
            '--max-steps-per-epoch', '3',
            '--batch-size', '16',
            '--val-batch-size', '16',
            '--model', 'resnet20',
            '--kfac-inv-update-steps', '2',
            '--checkpoint-dir', td,
            '--checkpoint-freq', '1',
        ]
        r = _run(args)
        assert r.returncode == 0, r.stderr[-2000:]
        assert os.path.exists(os.path.join(td, 'checkpoint_1.pth.tar'))
        # resume path
        r2 = _run(args)
        assert r2.returncode == 0, r2.stderr[-2000:]


def test_language_model_example_runs() -> None:
    r = _run(
        [
            'examples/torch_language_model.py',
            '--epochs', '1',
            '--steps-per-epoch', '4',
            '--batch-size', '4',
            '--vocab', '256',
        ],
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert 'ppl=' in r.stdout


def test_cifar_example_no_kfac() -> None:
    with tempfile.TemporaryDirectory() as td:
        r = _run(
            [
                'examples/torch_cifar10_resnet.py',
                '--epochs', '1',
                '--max-steps-per-epoch', '2',
                '--batch-size', '8',
                '--model', 'resnet20',
                '--kfac-inv-update-steps', '0',
                '--checkpoint-dir', td,
            ],
        )
        assert r.returncode == 0, r.stderr[-2000:]


def test_gpt_neox_mlp_example_runs() -> None:
    """2-way tensor-parallel example CLI under torchrun/gloo."""
    sys.path.insert(0, REPO)
    from testing.distributed import find_free_port

    port = str(find_free_port())
    env = dict(os.environ)
    env['MASTER_ADDR'] = '127.0.0.1'
    env['MASTER_PORT'] = port
    r = subprocess.run(
        [
            sys.executable,
            '-m',
            'torch.distributed.run',
            '--nproc-per-node',
            '2',
            '--master-addr',
            '127.0.0.1',
            '--master-port',
            port,
            'examples/torch_gpt_neox_mlp.py',
            '--steps',
            '12',
            '--backend',
            'gloo',
        ],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=240,
        env=env,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    assert 'done' in r.stdout


def test_imagenet_example_runs() -> None:
    """ImageNet CLI: 2 synthetic steps of resnet50 on CPU."""
    r = _run(
        [
            'examples/torch_imagenet_resnet.py',
            '--synthetic',
            '--epochs', '1',
            '--max-steps-per-epoch', '2',
            '--batch-size', '2',
            '--val-batch-size', '2',
            '--kfac-inv-update-steps', '2',
        ],
        timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]


def test_example_utils_units() -> None:
    """Unit semantics of the shared example utilities (previously only
    covered through the CLI smokes)."""
    import torch

    sys.path.insert(0, REPO)
    from examples.utils import accuracy
    from examples.utils import create_lr_schedule
    from examples.utils import LabelSmoothLoss
    from examples.utils import Metric

    # LabelSmoothLoss at smoothing=0 == plain cross entropy
    torch.manual_seed(0)
    x = torch.randn(16, 10)
    y = torch.randint(0, 10, (16,))
    torch.testing.assert_close(
        LabelSmoothLoss(0.0)(x, y),
        torch.nn.functional.cross_entropy(x, y),
    )
    # smoothing mixes toward uniform: loss increases on confident preds
    conf = torch.full((4, 10), -10.0)
    t = torch.arange(4) % 10
    conf[torch.arange(4), t] = 10.0
    assert LabelSmoothLoss(0.1)(conf, t) > LabelSmoothLoss(0.0)(conf, t)

    # Metric averages over updates (single-process path)
    m = Metric('loss')
    m.update(torch.tensor(2.0))
    m.update(torch.tensor(4.0))
    assert float(m.avg) == 3.0

    # warmup ramps from 1/workers to 1, then one factor of alpha per
    # decay epoch passed (epoch 25 is past BOTH 10 and 20 -> alpha^2;
    # this was inverted before this test existed)
    sched = create_lr_schedule(8, warmup_epochs=4, decay_schedule=[10, 20])
    assert sched(0) == 1.0 / 8
    assert sched(5) == 1.0
    assert sched(10) == pytest.approx(0.1)
    assert sched(15) == pytest.approx(0.1)
    assert sched(25) == pytest.approx(0.01)

    assert float(accuracy(x, x.argmax(dim=1))) == 1.0


def test_save_checkpoint_roundtrip(tmp_path) -> None:
    import torch

    sys.path.insert(0, REPO)
    from examples.utils import save_checkpoint
    from kfac_amd import KFACPreconditioner
    from testing.models import TinyModel

    model = TinyModel()
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    precon = KFACPreconditioner(model)
    path = str(tmp_path / 'ckpt.pth.tar')
    save_checkpoint(model, opt, precon, [], path, epoch=3)
    state = torch.load(path, weights_only=False)
    assert state['epoch'] == 3
    assert set(state['model'].keys()) == set(model.state_dict().keys())
    assert state['preconditioner']['steps'] == 0


def test_vision_engine_accumulation_boundaries() -> None:
    """engine.train steps the optimizer/preconditioner only at
    accumulation boundaries and scales micro-losses by 1/steps."""
    import torch

    sys.path.insert(0, REPO)
    from examples.vision.engine import train
    from kfac_amd import KFACPreconditioner
    from testing.models import TinyModel

    torch.manual_seed(0)
    model = TinyModel()
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    precon = KFACPreconditioner(
        model, factor_update_steps=1, inv_update_steps=1, lr=0.05,
    )
    steps = {'n': 0}
    orig_step = opt.step

    def counting_step(*a, **k):
        steps['n'] += 1
        return orig_step(*a, **k)

    opt.step = counting_step
    data = [
        (torch.randn(8, 10), torch.randint(0, 3, (8,)))
        for _ in range(6)
    ]
    train(
        epoch=0,
        model=model,
        optimizer=opt,
        preconditioner=precon,
        loss_func=torch.nn.CrossEntropyLoss(),
        train_loader=data,
        device=torch.device('cpu'),
        accumulation_steps=2,
        log_interval=1000,
    )
    assert steps['n'] == 3  # 6 micro-batches / 2
    assert precon.steps == 3  # preconditioner stepped only at boundaries


def test_language_dataset_units() -> None:
    import torch

    sys.path.insert(0, REPO)
    from examples.language.dataset import corpus_batch
    from examples.language.dataset import synthetic_batch

    dev = torch.device('cpu')
    x1, y1 = synthetic_batch(100, 4, 8, dev, seed=5)
    x2, y2 = synthetic_batch(100, 4, 8, dev, seed=5)
    torch.testing.assert_close(x1, x2)  # reproducible per seed
    assert x1.shape == (8, 4) and y1.shape == (8 * 4,)
    # target is the next-token shift of the input stream
    torch.testing.assert_close(x1[1:].reshape(-1), y1[: 7 * 4])

    corpus = torch.arange(1000) % 50
    cx, cy = corpus_batch(corpus, batch=2, seq=5, step=0, device=dev)
    assert cx.shape == (5, 2) and cy.shape == (10,)
    torch.testing.assert_close(cy.view(5, 2)[:-1], cx[1:])  # shifted by one
