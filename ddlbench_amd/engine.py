"""Shared training/validation engine.

Implements the reference's measurement protocol:
per-LOGINTER samples/sec + memory prints, epoch timing bracketed by device
synchronization, top-1 validation, and the final summary line — one engine
shared by the single-GPU, DP, and pipeline strategies instead of twelve
copy-pasted scripts.
"""

from __future__ import annotations

import time
from typing import Callable, Optional

import torch

from ddlbench_amd.config import BenchConfig
from ddlbench_amd.ops import functional as NF
from ddlbench_amd.ops.modules import set_default_backend
from ddlbench_amd.utils import AverageMeter, BenchLogger, accuracy, gpu_memory_gb


def resolve_device(cfg: BenchConfig, local_rank: int = 0) -> torch.device:
    if cfg.device == "cpu":
        return torch.device("cpu")
    if cfg.device == "auto":
        if torch.cuda.is_available():
            return torch.device("cuda", local_rank)
        return torch.device("cpu")
    d = torch.device(cfg.device)
    if d.type == "cuda" and d.index is None:
        d = torch.device("cuda", local_rank)
    return d


def compute_dtype(cfg: BenchConfig) -> torch.dtype:
    return torch.bfloat16 if cfg.dtype == "bfloat16" else torch.float32


def sync(device: torch.device) -> None:
    if device.type == "cuda":
        torch.cuda.synchronize(device)


class Trainer:
    """Single-process training loop (the `single` strategy), and the base
    for DP (gradient hooks injected via the step_fn seam)."""

    def __init__(self, cfg: BenchConfig, model: torch.nn.Module,
                 optimizer: torch.optim.Optimizer,
                 device: torch.device,
                 logger: Optional[BenchLogger] = None,
                 loss_fn: Optional[Callable] = None,
                 world_size: int = 1,
                 allreduce_metrics: Optional[Callable] = None):
        set_default_backend(cfg.kernel_backend)
        self.cfg = cfg
        self.model = model
        self.optimizer = optimizer
        self.device = device
        self.log = logger or BenchLogger()
        # fused softmax-CE HIP kernel on GPU, F.cross_entropy on CPU
        self.loss_fn = loss_fn or (
            lambda out, tgt: NF.cross_entropy(out, tgt,
                                              backend=cfg.kernel_backend))
        self.world_size = world_size
        # hook: DP engine averages metrics across ranks (reference:
        # hvd.allreduce for metric averaging, mnist_horovod.py:129-132)
        self.allreduce_metrics = allreduce_metrics or (lambda v: v)
        self.dtype = compute_dtype(cfg)
        # set when the model is wrapped in BucketedDataParallel
        self.dp = model if hasattr(model, "finalize_backward") else None
        # LR schedule over the optimizer's configured base LR (already
        # x world_size for DDP) — reference imagenet_pytorch.py:225-229,
        # imagenet_horovod.py:258-275
        from ddlbench_amd.utils.lr import make_lr_schedule
        self.lr_schedule = make_lr_schedule(
            cfg.lr_schedule, world_size=world_size,
            warmup_epochs=cfg.warmup_epochs)
        self._base_lrs = [g["lr"] for g in optimizer.param_groups]
        self._warmup_active = cfg.lr_schedule == "warmup"

    def _to_device(self, x: torch.Tensor, y: torch.Tensor):
        x = x.to(self.device, dtype=self.dtype, non_blocking=True)
        if self.cfg.channels_last:
            x = x.contiguous(memory_format=torch.channels_last)
        y = y.to(self.device, non_blocking=True)
        return x, y

    def train_epoch(self, loader, epoch: int) -> tuple:
        self.model.train()
        cfg = self.cfg
        n_batches = len(loader)
        seen = 0
        # loss accumulates ON DEVICE; .item() only at log points so the
        # steady-state loop never host-syncs (the reference's per-batch
        # loss read under-reports samples/sec; cf. runner.py deferral)
        loss_sum = None
        if self.device.type == "cuda":
            torch.cuda.reset_peak_memory_stats(self.device)
        from ddlbench_amd.utils.lr import apply_lr
        apply_lr(self.optimizer, self._base_lrs,
                 self.lr_schedule(epoch, 0.0))
        # the warmup ramp is per-batch in the reference
        # (imagenet_horovod.py:262: epoch + (batch+1)/len(loader))
        per_batch_lr = (self._warmup_active
                        and epoch <= self.cfg.warmup_epochs)
        sync(self.device)
        tick = time.perf_counter()
        window_start, window_samples = tick, 0
        for i, (x, y) in enumerate(loader):
            if per_batch_lr:
                apply_lr(self.optimizer, self._base_lrs,
                         self.lr_schedule(epoch, (i + 1) / n_batches))
            x, y = self._to_device(x, y)
            out = self.model(x)
            loss = self.loss_fn(out, y)
            if self.dp is not None:
                self.dp.zero_grad_buckets()
                loss.backward()  # bucket all-reduces overlap backward
                self.dp.finalize_backward()
            else:
                self.optimizer.zero_grad(set_to_none=True)
                loss.backward()
            self.optimizer.step()
            with torch.no_grad():
                contrib = loss.detach() * y.size(0)
                loss_sum = contrib if loss_sum is None \
                    else loss_sum.add_(contrib)
            seen += y.size(0)
            window_samples += y.size(0)
            if cfg.log_interval and (i + 1) % cfg.log_interval == 0:
                sync(self.device)
                now = time.perf_counter()
                sps = window_samples * self.world_size / (now - window_start)
                alloc, reserved, total = gpu_memory_gb(self.device)
                self.log.train_step(epoch, cfg.epochs,
                                    int(100.0 * (i + 1) / n_batches),
                                    sps, alloc, reserved, total)
                if self.dp is not None \
                        and hasattr(self.dp, "pop_reduce_times"):
                    rt = self.dp.pop_reduce_times()
                    if rt:  # DDLB_LOG_REDUCE=1 (extract_reduce_times)
                        self.log.info("reduce_times_ms: " + " ".join(
                            f"{v:.3f}" for v in rt))
                window_start, window_samples = time.perf_counter(), 0
        sync(self.device)
        elapsed = time.perf_counter() - tick
        sps = seen * self.world_size / elapsed
        avg_loss = (float(loss_sum.item()) / seen) if seen else 0.0
        return avg_loss, sps, elapsed

    @torch.no_grad()
    def validate(self, loader) -> tuple:
        self.model.eval()
        losses = AverageMeter()
        accs = AverageMeter()
        for x, y in loader:
            x, y = self._to_device(x, y)
            out = self.model(x)
            loss = self.loss_fn(out, y)
            acc1 = accuracy(out, y)[0]
            losses.update(loss.item(), y.size(0))
            accs.update(acc1, y.size(0))
        return (self.allreduce_metrics(losses.avg),
                self.allreduce_metrics(accs.avg))

    def fit(self, train_loader, test_loader, train_sampler=None,
            start_epoch: int = 1, on_epoch_end=None) -> dict:
        cfg = self.cfg
        epoch_sps, epoch_secs = [], []
        val_acc = val_loss = 0.0
        for epoch in range(start_epoch, cfg.epochs + 1):
            if train_sampler is not None:
                train_sampler.set_epoch(epoch)
            train_loss, sps, secs = self.train_epoch(train_loader, epoch)
            val_loss, val_acc = self.validate(test_loader)
            epoch_sps.append(sps)
            epoch_secs.append(secs)
            self.log.epoch(epoch, cfg.epochs, train_loss, sps,
                           val_loss, val_acc)
            if on_epoch_end is not None:
                on_epoch_end(epoch, {"valid_accuracy": val_acc,
                                     "valid_loss": val_loss})
        avg_sps = sum(epoch_sps) / max(len(epoch_sps), 1)
        avg_secs = sum(epoch_secs) / max(len(epoch_secs), 1)
        self.log.final(val_acc, avg_sps, avg_secs)
        return {"valid_accuracy": val_acc, "samples_per_sec": avg_sps,
                "sec_per_epoch": avg_secs}


def make_optimizer(cfg: BenchConfig, model: torch.nn.Module,
                   lr_scale: float = 1.0) -> torch.optim.Optimizer:
    """SGD+momentum(+weight decay) — fused HIP step when on GPU."""
    from ddlbench_amd.ops.sgd import FusedSGD
    return FusedSGD(model.parameters(), lr=cfg.lr * lr_scale,
                    momentum=cfg.momentum, weight_decay=cfg.weight_decay,
                    backend=cfg.kernel_backend)
