"""Checkpoint / resume round-trip."""

import torch

from ddlbench_amd.ops.sgd import FusedSGD
from ddlbench_amd.utils.checkpoint import (load_stage_checkpoint,
                                           save_stage_checkpoint)


def test_roundtrip(tmp_path):
    torch.manual_seed(0)
    m = torch.nn.Linear(4, 4)
    opt = FusedSGD(m.parameters(), lr=0.1, momentum=0.9, backend="torch")
    for _ in range(2):
        opt.zero_grad()
        m(torch.randn(2, 4)).sum().backward()
        opt.step()
    path = save_stage_checkpoint(str(tmp_path), 0, 3, "resnet18", m, opt,
                                 0.5)
    assert path.endswith("checkpoint.0.pth.tar")

    m2 = torch.nn.Linear(4, 4)
    opt2 = FusedSGD(m2.parameters(), lr=0.1, momentum=0.9, backend="torch")
    state = load_stage_checkpoint(str(tmp_path), 0, m2, opt2)
    assert state["epoch"] == 3 and state["arch"] == "resnet18"
    torch.testing.assert_close(m2.weight, m.weight)
    # momentum buffers restored: one more identical step matches
    x = torch.randn(2, 4)
    for mm, oo in ((m, opt), (m2, opt2)):
        oo.zero_grad()
        mm(x).sum().backward()
        oo.step()
    torch.testing.assert_close(m2.weight, m.weight)


def test_missing_returns_none(tmp_path):
    m = torch.nn.Linear(2, 2)
    assert load_stage_checkpoint(str(tmp_path), 5, m) is None


def test_single_strategy_resume(tmp_path):
    """run_single writes checkpoints and resumes past finished epochs."""
    from ddlbench_amd.config import BenchConfig
    from ddlbench_amd.strategies import run_single
    cfg = BenchConfig(dataset="mnist", arch="resnet18", epochs=1,
                      batch_size=8, synthetic_scale=0.0005, device="cpu",
                      num_workers=0, log_interval=0,
                      checkpoint_dir=str(tmp_path))
    run_single(cfg)
    import os
    assert os.path.exists(tmp_path / "checkpoint.0.pth.tar")
    cfg2 = BenchConfig(dataset="mnist", arch="resnet18", epochs=1,
                       batch_size=8, synthetic_scale=0.0005, device="cpu",
                       num_workers=0, log_interval=0,
                       checkpoint_dir=str(tmp_path), resume=True)
    res = run_single(cfg2)  # epoch already done -> returns without training
    assert res["sec_per_epoch"] == 0.0


def test_bnact_num_batches_tracked_syncs_on_state_dict():
    from ddlbench_amd.ops.modules import BNAct
    m = BNAct(4)
    m.train()
    x = torch.randn(2, 4, 3, 3)
    m(x)
    m(x)
    sd = m.state_dict()
    assert sd["num_batches_tracked"].item() == 2
    m2 = BNAct(4)
    m2.load_state_dict(sd)
    assert m2.num_batches_tracked.item() == 2


def test_mixed_dtype_param_groups_cpu():
    """FusedSGD with bf16 + f32 params in one group (the BNAct-keeps-f32
    situation under a bf16 model)."""
    from ddlbench_amd.ops.sgd import FusedSGD
    torch.manual_seed(0)
    a = torch.nn.Parameter(torch.randn(4, 4, dtype=torch.bfloat16))
    b = torch.nn.Parameter(torch.randn(4))
    opt = FusedSGD([a, b], lr=0.1, momentum=0.9, backend="torch")
    (a.float().sum() + b.sum()).backward()
    pa, pb = a.detach().clone(), b.detach().clone()
    opt.step()
    assert not torch.equal(a, pa) and not torch.equal(b, pb)


def test_versioned_optimizer_checkpoint_roundtrip(tmp_path):
    """Stage checkpoint through the VersionedOptimizer wrapper."""
    from ddlbench_amd.parallel.pipeline.stash import VersionedOptimizer
    torch.manual_seed(0)
    m = torch.nn.Linear(4, 4)
    opt = VersionedOptimizer(FusedSGD(m.parameters(), lr=0.1,
                                      momentum=0.9, backend="torch"))
    for _ in range(2):
        opt.zero_grad(set_to_none=False)
        m(torch.randn(2, 4)).sum().backward()
        opt.step()
    save_stage_checkpoint(str(tmp_path), 1, 5, "x", m, opt)
    m2 = torch.nn.Linear(4, 4)
    opt2 = VersionedOptimizer(FusedSGD(m2.parameters(), lr=0.1,
                                       momentum=0.9, backend="torch"))
    st = load_stage_checkpoint(str(tmp_path), 1, m2, opt2)
    assert st["epoch"] == 5
    torch.testing.assert_close(m2.weight, m.weight)
