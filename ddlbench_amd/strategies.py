"""Strategy runners — one per execution strategy.

single    — single-device baseline (reference *_pytorch.py, §2.3)
ddp       — data-parallel, bucketed RCCL all-reduce (ref *_horovod.py, §2.4)
gpipe     — synchronous micro-batch pipeline (ref *_gpipe.py, §2.5)
pipedream — asynchronous 1F1B pipeline (ref pipedream-fork, §2.10)
"""

from __future__ import annotations

import torch

from ddlbench_amd.config import BenchConfig
from ddlbench_amd.data import make_loaders
from ddlbench_amd.engine import (Trainer, compute_dtype, make_optimizer,
                                 resolve_device)
from ddlbench_amd.models import build_model
from ddlbench_amd.utils import BenchLogger


def _set_threads(cfg: BenchConfig) -> None:
    if cfg.cores_per_gpu > 0:
        torch.set_num_threads(cfg.cores_per_gpu)


def run_single(cfg: BenchConfig) -> dict:
    """Single-device baseline (reference mnist_pytorch.py:163-226)."""
    _set_threads(cfg)
    torch.manual_seed(cfg.seed)
    device = resolve_device(cfg)
    model = build_model(cfg.dataset, cfg.arch).to(device)
    if compute_dtype(cfg) != torch.float32:
        model = model.to(compute_dtype(cfg))
    if cfg.channels_last and device.type == "cuda":
        model = model.to(memory_format=torch.channels_last)
    optimizer = make_optimizer(cfg, model)
    start_epoch = 1
    if cfg.resume and cfg.checkpoint_dir:
        from ddlbench_amd.utils.checkpoint import load_stage_checkpoint
        state = load_stage_checkpoint(cfg.checkpoint_dir, 0, model,
                                      optimizer)
        if state is not None:
            start_epoch = state["epoch"] + 1
    train_loader, test_loader, _ = make_loaders(
        cfg, pin_memory=device.type == "cuda")
    trainer = Trainer(cfg, model, optimizer, device)

    def on_epoch_end(epoch, metrics):
        if cfg.checkpoint_dir:
            from ddlbench_amd.utils.checkpoint import save_stage_checkpoint
            save_stage_checkpoint(cfg.checkpoint_dir, 0, epoch, cfg.arch,
                                  model, optimizer,
                                  metrics.get("valid_accuracy", 0.0))

    return trainer.fit(train_loader, test_loader,
                       start_epoch=start_epoch, on_epoch_end=on_epoch_end)


def run_ddp(cfg: BenchConfig) -> dict:
    """Data-parallel over RCCL/xGMI, one process per GPU
    (reference mnist_horovod.py train flow)."""
    from ddlbench_amd.parallel import (BucketedDataParallel,
                                       allreduce_mean_scalar,
                                       init_distributed)
    _set_threads(cfg)
    env = init_distributed()
    torch.manual_seed(cfg.seed + env.rank)
    device = resolve_device(cfg, env.local_rank)
    model = build_model(cfg.dataset, cfg.arch).to(device)
    if compute_dtype(cfg) != torch.float32:
        model = model.to(compute_dtype(cfg))
    if cfg.channels_last and device.type == "cuda":
        model = model.to(memory_format=torch.channels_last)
    dp = BucketedDataParallel(model)
    # reference scales LR by world size (mnist_horovod.py:226)
    optimizer = make_optimizer(cfg, model, lr_scale=env.world_size)
    start_epoch = 1
    if cfg.resume and cfg.checkpoint_dir:
        from ddlbench_amd.utils.checkpoint import load_stage_checkpoint
        state = load_stage_checkpoint(cfg.checkpoint_dir, 0, model,
                                      optimizer)
        if state is not None:
            start_epoch = state["epoch"] + 1
    train_loader, test_loader, sampler = make_loaders(
        cfg, world_size=env.world_size, rank=env.rank,
        pin_memory=device.type == "cuda")
    trainer = Trainer(cfg, dp, optimizer, device,
                      logger=BenchLogger(env.rank),
                      world_size=env.world_size,
                      allreduce_metrics=lambda v: allreduce_mean_scalar(
                          v, device if device.type == "cuda" else None))

    def on_epoch_end(epoch, metrics):
        # rank 0 only (reference: rank_in_stage==0 saves)
        if cfg.checkpoint_dir and env.rank == 0:
            from ddlbench_amd.utils.checkpoint import save_stage_checkpoint
            save_stage_checkpoint(cfg.checkpoint_dir, 0, epoch, cfg.arch,
                                  model, optimizer,
                                  metrics.get("valid_accuracy", 0.0))

    return trainer.fit(train_loader, test_loader, sampler,
                       start_epoch=start_epoch, on_epoch_end=on_epoch_end)


def run_gpipe(cfg: BenchConfig) -> dict:
    """Synchronous micro-batch pipeline, single process × all visible
    devices (reference mnist_gpipe.py:213-225)."""
    from ddlbench_amd.parallel.pipeline.gpipe import run_gpipe_training
    _set_threads(cfg)
    torch.manual_seed(cfg.seed)
    return run_gpipe_training(cfg)


def run_pipedream(cfg: BenchConfig) -> dict:
    """Asynchronous 1F1B pipeline, one process per GPU
    (reference main_with_runtime.py:406-500)."""
    from ddlbench_amd.parallel.pipeline.runner import run_1f1b_training
    _set_threads(cfg)
    return run_1f1b_training(cfg)


RUNNERS = {
    "single": run_single,
    "ddp": run_ddp,
    "gpipe": run_gpipe,
    "pipedream": run_pipedream,
}


def run(cfg: BenchConfig) -> dict:
    return RUNNERS[cfg.strategy](cfg)
