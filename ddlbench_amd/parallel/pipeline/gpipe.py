"""GPipe-style synchronous micro-batch pipeline — single process,
multi-device, HIP streams.

The reference wraps a sequential model in torchgpipe
(/root/reference/benchmark/mnist/mnist_gpipe.py:213-225: balance_by_time,
GPipe(chunks=MICROBATCHES), input on devices[0], loss on devices[-1]).
This is the MI355X-native rebuild:

* stages are contiguous slices of the flattened model, one per device;
* forward walks the fill schedule (stage j handles micro-batch i at clock
  i+j), issuing each stage's kernels and the inter-device copy from one
  host thread — every device's HIP queue stays busy because launches are
  asynchronous and copies use per-device side streams + events;
* backward is driven by autograd over the summed micro-batch losses;
  PyTorch's engine runs one worker thread per device, so micro-batch
  backwards pipeline across stages without extra machinery;
* optional per-micro-batch activation checkpointing ("except_last",
  torchgpipe's default) trades recompute for memory.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint as _ckpt

from ddlbench_amd.parallel.pipeline.balance import balance_by_time


class _StageCopy(torch.autograd.Function):
    """Device-to-device copy on a dedicated stream with proper
    stream-ordering in both directions."""

    @staticmethod
    def forward(ctx, x, dst_device, streams):
        ctx.src_device = x.device
        ctx.streams = streams
        if x.device == dst_device:
            return x
        if x.is_cuda or dst_device.type == "cuda":
            copy_stream = streams.get(dst_device)
            if copy_stream is not None:
                copy_stream.wait_stream(torch.cuda.current_stream(x.device))
                with torch.cuda.stream(copy_stream):
                    out = x.to(dst_device, non_blocking=True)
                torch.cuda.current_stream(dst_device).wait_stream(copy_stream)
                out.record_stream(torch.cuda.current_stream(dst_device))
                return out
        return x.to(dst_device)

    @staticmethod
    def backward(ctx, dy):
        if dy.device == ctx.src_device:
            return dy, None, None
        return dy.to(ctx.src_device, non_blocking=True), None, None


class GPipeModel(nn.Module):
    def __init__(self, seq: nn.Sequential, balance: List[int],
                 devices: Optional[List[torch.device]] = None,
                 chunks: int = 1, checkpoint: str = "except_last"):
        super().__init__()
        assert sum(balance) == len(seq), "balance must cover the model"
        assert checkpoint in ("never", "always", "except_last")
        self.chunks = chunks
        self.checkpoint = checkpoint
        if devices is None:
            n = torch.cuda.device_count()
            devices = ([torch.device("cuda", i) for i in range(n)]
                       if n else [torch.device("cpu")])
        # one device per stage, cycling if fewer devices than stages
        self.devices = [devices[i % len(devices)]
                        for i in range(len(balance))]
        mods = list(seq)
        self.stages = nn.ModuleList()
        off = 0
        for si, size in enumerate(balance):
            stage = nn.Sequential(*mods[off:off + size]).to(self.devices[si])
            self.stages.append(stage)
            off += size
        self._copy_streams = {}
        for d in set(self.devices):
            if d.type == "cuda":
                self._copy_streams[d] = torch.cuda.Stream(d)

    @property
    def in_device(self) -> torch.device:
        return self.devices[0]

    @property
    def out_device(self) -> torch.device:
        return self.devices[-1]

    def _run_stage(self, si: int, x: torch.Tensor,
                   is_last_chunk: bool) -> torch.Tensor:
        stage = self.stages[si]
        use_ckpt = (self.checkpoint == "always" or
                    (self.checkpoint == "except_last" and
                     si < len(self.stages) - 1))
        if use_ckpt and self.training and torch.is_grad_enabled():
            return _ckpt(stage, x, use_reentrant=False)
        return stage(x)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        micro = list(torch.chunk(x, self.chunks, dim=0))
        n_m, n_s = len(micro), len(self.stages)
        # buf[i] = current activation of micro-batch i
        buf = [None] * n_m
        # fill-steady-drain schedule: clock t runs (i, j=t-i)
        for t in range(n_m + n_s - 1):
            lo = max(0, t - n_s + 1)
            hi = min(t, n_m - 1)
            # reverse order: later stages first, so their kernels are
            # queued before this clock's stage-0 work floods the host
            for i in range(hi, lo - 1, -1):
                j = t - i
                inp = micro[i] if j == 0 else buf[i]
                inp = _StageCopy.apply(inp, self.devices[j],
                                       self._copy_streams)
                buf[i] = self._run_stage(j, inp, i == n_m - 1)
        return torch.cat(buf, dim=0)


def build_gpipe(cfg, seq: nn.Sequential, sample: torch.Tensor,
                devices=None) -> GPipeModel:
    n_dev = (len(devices) if devices
             else (torch.cuda.device_count() or 1))
    n_parts = min(n_dev, len(seq))
    balance = balance_by_time(
        n_parts, seq, sample,
        device=devices[0] if devices else
        (torch.device("cuda", 0) if torch.cuda.is_available()
         else torch.device("cpu")))
    return GPipeModel(seq, balance, devices=devices, chunks=cfg.microbatches)


def run_gpipe_training(cfg) -> dict:
    """Reference run_epoch flow (mnist_gpipe.py:87-99): input to
    devices[0], target/loss on devices[-1]."""
    from ddlbench_amd.data import make_loaders, synthetic_batch
    from ddlbench_amd.engine import Trainer, compute_dtype, make_optimizer
    from ddlbench_amd.models import build_sequential
    from ddlbench_amd.utils import BenchLogger

    torch.manual_seed(cfg.seed)
    seq = build_sequential(cfg.dataset, cfg.arch)
    dtype = compute_dtype(cfg)
    if dtype != torch.float32:
        seq = seq.to(dtype)
    sample, _ = synthetic_batch(cfg, batch_size=cfg.batch_size,
                                dtype=dtype)
    model = build_gpipe(cfg, seq, sample)
    optimizer = make_optimizer(cfg, model)

    # reference contract: BATCH_SIZE is the micro-batch size; the loader
    # batch is micro x MICROBATCHES (mnist_gpipe.py:35-41)
    import dataclasses
    loader_cfg = dataclasses.replace(
        cfg, batch_size=cfg.batch_size * cfg.microbatches)
    train_loader, test_loader, _ = make_loaders(
        loader_cfg, pin_memory=model.in_device.type == "cuda")
    trainer = Trainer(cfg, model, optimizer, model.out_device,
                      logger=BenchLogger(0))
    return trainer.fit(train_loader, test_loader)
