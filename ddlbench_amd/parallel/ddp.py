"""BucketedDataParallel — gradient all-reduce DP engine on RCCL over xGMI.

The MI355X-native replacement for the reference's Horovod layer
(hvd.DistributedOptimizer with tensor fusion —
/root/reference/benchmark/mnist/mnist_horovod.py:230-236):

* parameters' ``.grad`` are persistent views into flat per-bucket buffers,
  so gradient accumulation writes straight into the all-reduce payload —
  zero copy-in/copy-out;
* buckets are filled in reverse parameter order (≈ backward completion
  order); when a bucket's last grad lands (post-accumulate-grad hook) its
  all-reduce launches asynchronously on RCCL's stream and overlaps the
  rest of backward;
* bucket size defaults to 32 MiB: the 8-GPU xGMI mesh is 7 point-to-point
  links per GPU (≈153 GB/s each) and ring all-reduce is per-link bound, so
  buckets must be large enough to amortize per-collective latency while
  leaving several buckets per backward for overlap (tunable via
  ``bucket_mb`` / env DDLB_BUCKET_MB for sweeps);
* averaging = SUM all-reduce + local divide (gloo has no AVG).

Works on the gloo backend for CPU-only multi-process tests.
"""

from __future__ import annotations

import os
import time
from typing import List

import torch
import torch.distributed as dist
import torch.nn as nn


class _Bucket:
    __slots__ = ("params", "flat", "views", "pending", "work", "comm",
                 "t0", "ev0", "ev1")

    def __init__(self) -> None:
        self.params: List[torch.nn.Parameter] = []
        self.flat: torch.Tensor = None
        self.views = {}
        self.pending = 0
        self.work = None
        self.comm = None  # fp32 reduction scratch (DDLB_BUCKET_FP32=1)
        self.t0 = 0.0     # reduce-time logging (DDLB_LOG_REDUCE=1)
        self.ev0 = None
        self.ev1 = None


class BucketedDataParallel(nn.Module):
    def __init__(self, module: nn.Module, bucket_mb: float = 0.0,
                 process_group=None, average: bool = True,
                 log_reduce_times: bool = False):
        super().__init__()
        self.module = module
        self.pg = process_group
        self.average = average
        if bucket_mb <= 0:
            bucket_mb = float(os.environ.get("DDLB_BUCKET_MB", "32"))
        self.bucket_bytes = int(bucket_mb * 2**20)
        self._fp32_reduce = os.environ.get("DDLB_BUCKET_FP32", "0") == "1"
        # per-bucket all-reduce span logging (the reference extracts the
        # same curves from a patched DDP's log_reduce_times stdout,
        # profiler utils/all_reduce/extract_reduce_times.py:7-30)
        self._log_reduce = log_reduce_times or \
            os.environ.get("DDLB_LOG_REDUCE", "0") == "1"
        self._reduce_ms: List[float] = []
        self.world_size = (dist.get_world_size(self.pg)
                           if dist.is_initialized() else 1)
        self._buckets: List[_Bucket] = []
        self._param_bucket = {}
        self._hooks = []
        if self.world_size > 1:
            self._broadcast_state()
            self._build_buckets()

    # -- init-time ------------------------------------------------------
    def _broadcast_state(self) -> None:
        """Rank-0 state to all ranks (reference: hvd.broadcast_parameters
        + broadcast_optimizer_state, mnist_horovod.py:230-231)."""
        for t in list(self.module.parameters()) + list(self.module.buffers()):
            if t.is_floating_point() or t.dtype in (torch.int64, torch.int32):
                dist.broadcast(t.data, src=0, group=self.pg)

    def _build_buckets(self) -> None:
        params = [p for p in self.module.parameters() if p.requires_grad]
        # reverse order ≈ autograd completion order
        by_dtype = {}
        for p in reversed(params):
            by_dtype.setdefault(p.dtype, []).append(p)
        for dtype, plist in by_dtype.items():
            cur = _Bucket()
            size = 0
            for p in plist:
                nbytes = p.numel() * p.element_size()
                if cur.params and size + nbytes > self.bucket_bytes:
                    self._finish_bucket(cur, dtype)
                    cur = _Bucket()
                    size = 0
                cur.params.append(p)
                size += nbytes
            if cur.params:
                self._finish_bucket(cur, dtype)
        for p in params:
            h = p.register_post_accumulate_grad_hook(self._grad_ready)
            self._hooks.append(h)

    @staticmethod
    def _layout_view(flat_slice: torch.Tensor, p: torch.Tensor):
        """View a flat bucket slice with the param's memory format.

        A channels_last param needs a channels_last-strided grad view:
        a row-major view_as() would make autograd accumulate NCHW-order
        data while FusedSGD walks the param NHWC-order — and FusedSGD's
        layout remediation would rebind p.grad to a fresh tensor,
        orphaning the bucket (grads then never reach the all-reduce)."""
        if (p.dim() == 4 and not p.is_contiguous()
                and p.is_contiguous(memory_format=torch.channels_last)):
            n, c, h, w = p.shape
            return flat_slice.view(n, h, w, c).permute(0, 3, 1, 2)
        return flat_slice.view_as(p)

    def _finish_bucket(self, bucket: _Bucket, dtype) -> None:
        device = bucket.params[0].device
        total = sum(p.numel() for p in bucket.params)
        bucket.flat = torch.zeros(total, dtype=dtype, device=device)
        off = 0
        for p in bucket.params:
            n = p.numel()
            view = self._layout_view(bucket.flat[off:off + n], p)
            bucket.views[p] = view
            # persistent .grad view: accumulation writes into the payload
            p.grad = view
            off += n
            self._param_bucket[p] = bucket
        bucket.pending = len(bucket.params)
        self._buckets.append(bucket)

    # -- backward-time --------------------------------------------------
    def _grad_ready(self, p: torch.nn.Parameter) -> None:
        b = self._param_bucket[p]
        b.pending -= 1
        if b.pending == 0:
            # Reduction dtype: param dtype (bf16) by default — the same
            # choice torch DDP makes, and the bandwidth-right one for
            # the 7-link xGMI mesh. The reference's horovod reduced
            # fp32; DDLB_BUCKET_FP32=1 reproduces that numerics at 2x
            # the wire bytes (docs/MULTIGPU.md).
            if self._log_reduce:
                if b.flat.is_cuda:
                    b.ev0 = torch.cuda.Event(enable_timing=True)
                    b.ev0.record()
                else:
                    b.t0 = time.perf_counter()
            if self._fp32_reduce and b.flat.dtype != torch.float32:
                b.comm = b.flat.to(torch.float32)
                b.work = dist.all_reduce(b.comm, op=dist.ReduceOp.SUM,
                                         group=self.pg, async_op=True)
            else:
                b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                         group=self.pg, async_op=True)
            b.pending = len(b.params)

    def finalize_backward(self) -> None:
        """Wait for in-flight reductions and average. Call between
        loss.backward() and optimizer.step().

        Contract: exactly one backward per zero_grad_buckets()/
        finalize_backward() pair — a bucket's all-reduce launches when
        its last grad lands, so a second accumulation pass would race
        the in-flight collective (the reference's horovod path has the
        same one-pass default, batches_per_allreduce=1,
        imagenet_horovod.py:36)."""
        if self.world_size <= 1:
            return
        ev_pairs = []
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None
                if self._log_reduce:
                    if b.ev0 is not None:
                        b.ev1 = torch.cuda.Event(enable_timing=True)
                        b.ev1.record()
                        ev_pairs.append((b.ev0, b.ev1))
                        b.ev0 = None
                    else:
                        self._reduce_ms.append(
                            (time.perf_counter() - b.t0) * 1e3)
                if b.comm is not None:
                    if self.average:
                        b.comm.div_(self.world_size)
                    b.flat.copy_(b.comm)
                    b.comm = None
                elif self.average:
                    b.flat.div_(self.world_size)
            # restore grads dropped by a zero_grad(set_to_none=True)
            for p in b.params:
                if p.grad is None:
                    p.grad = b.views[p]
        if ev_pairs:
            torch.cuda.synchronize()
            self._reduce_ms.extend(e0.elapsed_time(e1)
                                   for e0, e1 in ev_pairs)

    def pop_reduce_times(self) -> List[float]:
        """Drain logged per-bucket all-reduce spans (ms), bucket-ready →
        collective-complete as seen by the compute stream — overlap with
        the rest of backward included. Empty unless DDLB_LOG_REDUCE=1 or
        log_reduce_times=True."""
        out, self._reduce_ms = self._reduce_ms, []
        return out

    def zero_grad_buckets(self) -> None:
        if self.world_size <= 1:
            # set_to_none avoids one fill launch per param AND the
            # read-modify-write grad accumulation pass
            self.module.zero_grad(set_to_none=True)
            return
        for b in self._buckets:
            b.flat.zero_()
            for p in b.params:
                if p.grad is None:
                    p.grad = b.views[p]

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)
