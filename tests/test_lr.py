"""LR schedule trajectories vs the reference formulas.

step30: /root/reference/benchmark/imagenet/imagenet_pytorch.py:225-229
warmup: /root/reference/benchmark/imagenet/imagenet_horovod.py:258-275
"""

import pytest
import torch

from ddlbench_amd.utils.lr import apply_lr, make_lr_schedule


def test_constant():
    f = make_lr_schedule("constant")
    assert all(f(e, 0.0) == 1.0 for e in (1, 10, 100))


def test_step30_trajectory():
    f = make_lr_schedule("step30")
    # epochs 1..30 -> 1.0; 31..60 -> 0.1; 61..90 -> 0.01
    assert f(1, 0.0) == 1.0
    assert f(30, 0.0) == 1.0
    assert f(31, 0.0) == pytest.approx(0.1)
    assert f(60, 0.0) == pytest.approx(0.1)
    assert f(61, 0.0) == pytest.approx(0.01)


def test_warmup_ramp_and_steps():
    n = 8
    f = make_lr_schedule("warmup", world_size=n, warmup_epochs=5)
    # base LR is lr0*N; epoch 1 start must give lr0 -> factor 1/N
    assert f(1, 0.0) == pytest.approx(1.0 / n)
    # monotone ramp within warmup
    vals = [f(e, fr) for e in (1, 2, 3, 4, 5) for fr in (0.0, 0.5)]
    assert all(b >= a for a, b in zip(vals, vals[1:]))
    # end of warmup reaches the full scaled LR
    assert f(6, 0.0) == pytest.approx(1.0)
    # reference steps at 30/60/80 (0-based)
    assert f(31, 0.0) == pytest.approx(1e-1)
    assert f(61, 0.0) == pytest.approx(1e-2)
    assert f(81, 0.0) == pytest.approx(1e-3)


def test_warmup_world1_is_flat_then_steps():
    f = make_lr_schedule("warmup", world_size=1, warmup_epochs=5)
    assert f(1, 0.0) == pytest.approx(1.0)
    assert f(3, 0.7) == pytest.approx(1.0)


def test_apply_lr_sets_groups():
    m = torch.nn.Linear(2, 2)
    opt = torch.optim.SGD(m.parameters(), lr=0.4)
    base = [g["lr"] for g in opt.param_groups]
    apply_lr(opt, base, 0.1)
    assert opt.param_groups[0]["lr"] == pytest.approx(0.04)
    apply_lr(opt, base, 1.0)  # factors are not cumulative
    assert opt.param_groups[0]["lr"] == pytest.approx(0.4)


def test_trainer_applies_schedule():
    """End-to-end: Trainer sets group lr from the schedule each epoch."""
    from ddlbench_amd.config import BenchConfig
    from ddlbench_amd.engine import Trainer

    cfg = BenchConfig(dataset="mnist", arch="resnet18", device="cpu",
                      lr=0.5, lr_schedule="step30", epochs=1,
                      batch_size=4, log_interval=0, num_workers=0)
    model = torch.nn.Sequential(torch.nn.Flatten(),
                                torch.nn.Linear(28 * 28, 10))
    opt = torch.optim.SGD(model.parameters(), lr=0.5)
    tr = Trainer(cfg, model, opt, torch.device("cpu"))
    x = torch.randn(8, 1, 28, 28)
    y = torch.randint(10, (8,))
    loader = [(x[:4], y[:4]), (x[4:], y[4:])]
    tr.train_epoch(loader, epoch=31)  # step30 -> 0.1x
    assert opt.param_groups[0]["lr"] == pytest.approx(0.05)
    tr.train_epoch(loader, epoch=1)
    assert opt.param_groups[0]["lr"] == pytest.approx(0.5)
