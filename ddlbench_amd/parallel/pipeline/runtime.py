"""1F1B asynchronous pipeline runtime — one process per GPU over RCCL.

The MI355X rebuild of the reference's StageRuntime
(/root/reference/pipedream-fork/runtime/runtime.py): warmup-forward /
one-forward-one-backward steady state / backward drain
(main_with_runtime.py:432-494), weight-version consistency, per-stage
data parallelism, and round-robin replica routing — on true RCCL p2p
send/recv over xGMI (comm.py) instead of helper-thread broadcast pairs,
and copy-on-step weight versioning (stash.py) instead of state_dict
deques.

Replication routing: minibatch m is processed by replica (m mod R_s) of
stage s; the producing replica sends to replica (m mod R_{s+1}) of the
next stage (the reference's round-robin messaging schedule,
communication.py:455-521, made static). Minibatch counts are trimmed to
a multiple of lcm(R_0..R_k) by the runner (the reference's gcd/lcm fix,
runtime.py:663-690)."""

from __future__ import annotations

import math
from collections import deque
from dataclasses import dataclass
from typing import Callable, List, Optional, Tuple

import torch

from ddlbench_amd.parallel.pipeline.comm import PipelineTransport


@dataclass
class StagePlan:
    """Static layout of the pipeline (identical on every rank)."""
    replicas: List[int]                  # R_s per stage

    @property
    def num_stages(self) -> int:
        return len(self.replicas)

    @property
    def world_size(self) -> int:
        return sum(self.replicas)

    def stage_of_rank(self, rank: int) -> Tuple[int, int]:
        """(stage_id, rank_in_stage)."""
        off = 0
        for s, r in enumerate(self.replicas):
            if rank < off + r:
                return s, rank - off
            off += r
        raise ValueError(rank)

    def rank_of(self, stage: int, replica: int) -> int:
        return sum(self.replicas[:stage]) + replica

    def stage_ranks(self, stage: int) -> List[int]:
        base = sum(self.replicas[:stage])
        return list(range(base, base + self.replicas[stage]))

    def edges(self) -> List[Tuple[int, int]]:
        """All (src_rank, dst_rank) forward edges that ever carry a
        minibatch under round-robin routing."""
        out = []
        for s in range(self.num_stages - 1):
            ra, rb = self.replicas[s], self.replicas[s + 1]
            period = ra * rb // math.gcd(ra, rb)
            pairs = {(m % ra, m % rb) for m in range(period)}
            for a, b in sorted(pairs):
                out.append((self.rank_of(s, a), self.rank_of(s + 1, b)))
        return out

    def lcm_replicas(self) -> int:
        l = 1
        for r in self.replicas:
            l = l * r // math.gcd(l, r)
        return l

    def num_warmup(self, stage: int) -> int:
        """In-flight depth for 1F1B (reference runtime.py:133-176:
        (ranks after me) / (ranks in my stage))."""
        after = sum(self.replicas[stage + 1:])
        return max(after // self.replicas[stage], 0)


@dataclass
class TensorSpec:
    """Static shape/dtype of one tensor on a pipeline edge. ``grad``
    marks float tensors whose gradient travels back on the bwd channel;
    pass-through tensors (lengths, token ids, masks) set grad=False and
    ride the pipeline forward only (the reference routes e.g.
    target_length the same way, runtime.py:540-543)."""
    shape: Tuple[int, ...]
    dtype: "torch.dtype" = None
    grad: bool = True


def _as_specs(x, default_dtype) -> Optional[List[TensorSpec]]:
    """Accept torch.Size (single-tensor API) or a list of TensorSpec."""
    if x is None:
        return None
    if isinstance(x, (list, tuple)) and x and isinstance(x[0], TensorSpec):
        return [TensorSpec(tuple(t.shape), t.dtype or default_dtype, t.grad)
                for t in x]
    return [TensorSpec(tuple(x), default_dtype, True)]


class RuntimeStats:
    """fwd/bwd counters: compute time, send/recv bytes (the reference's
    runtime_utilities.RuntimeStats)."""

    def __init__(self) -> None:
        self.reset()

    def reset(self) -> None:
        self.fwd_count = 0
        self.bwd_count = 0
        self.send_bytes = 0
        self.recv_bytes = 0

    def as_dict(self) -> dict:
        return {"fwd_count": self.fwd_count, "bwd_count": self.bwd_count,
                "send_bytes": self.send_bytes,
                "recv_bytes": self.recv_bytes}


class StageRuntime:
    def __init__(self, plan: StagePlan, rank: int,
                 module: torch.nn.Module,
                 transport: PipelineTransport,
                 in_shape,   # torch.Size | List[TensorSpec] | None
                 out_shape,  # torch.Size | List[TensorSpec]
                 device: torch.device,
                 dtype: torch.dtype,
                 loss_fn: Optional[Callable] = None,
                 dp_wrapper=None):
        self.plan = plan
        self.rank = rank
        self.stage, self.replica = plan.stage_of_rank(rank)
        self.module = module
        self.transport = transport
        self.in_specs = _as_specs(in_shape, dtype)
        self.out_specs = _as_specs(out_shape, dtype)
        self.device = device
        self.dtype = dtype
        self.loss_fn = loss_fn
        self.dp = dp_wrapper  # BucketedDataParallel over the stage group
        self.is_first = self.stage == 0
        self.is_last = self.stage == plan.num_stages - 1
        # in-flight state: (x, out_or_loss, send_work, mb_index, extras)
        self.inflight: deque = deque()
        self.stats = RuntimeStats()

    # ---- routing helpers ----------------------------------------------
    def _prev_rank(self, mb: int) -> int:
        s = self.stage - 1
        return self.plan.rank_of(s, mb % self.plan.replicas[s])

    def _next_rank(self, mb: int) -> int:
        s = self.stage + 1
        return self.plan.rank_of(s, mb % self.plan.replicas[s])

    def my_minibatches(self, total: int) -> List[int]:
        r = self.plan.replicas[self.stage]
        return [m for m in range(total) if m % r == self.replica]

    # ---- forward / backward --------------------------------------------
    def run_forward(self, mb: int, input_provider=None,
                    target_provider=None, training: bool = True):
        if self.is_first:
            xs = input_provider(mb)
            if torch.is_tensor(xs):
                xs = (xs,)
            xs = tuple(
                t.to(self.device,
                     dtype=self.dtype if t.is_floating_point() else None,
                     non_blocking=True) for t in xs)
        else:
            ch = self.transport.channel(self._prev_rank(mb), self.rank,
                                        "fwd")
            bufs = []
            for spec in self.in_specs:
                buf = torch.empty(spec.shape, device=self.device,
                                  dtype=spec.dtype)
                ch.irecv(buf).wait()
                self.stats.recv_bytes += buf.numel() * buf.element_size()
                bufs.append(buf)
            xs = tuple(bufs)
        if training and not self.is_first:
            for t, spec in zip(xs, self.in_specs):
                if spec.grad:
                    t.requires_grad_(True)
        with torch.enable_grad() if training else torch.no_grad():
            out = self.module(*xs)
        outs = (out,) if torch.is_tensor(out) else tuple(out)
        send_work = None
        loss = None
        extras = {}
        if self.is_last:
            if self.loss_fn is not None and target_provider is not None:
                y = target_provider(mb).to(self.device, non_blocking=True)
                loss = self.loss_fn(outs[0] if len(outs) == 1 else outs, y)
                extras["target"] = y
                extras["output"] = outs[0].detach()
        else:
            ch = self.transport.channel(self.rank, self._next_rank(mb),
                                        "fwd")
            send_work = [ch.isend(t.detach()) for t in outs]
            for t in outs:
                self.stats.send_bytes += t.numel() * t.element_size()
        self.inflight.append((xs, loss if self.is_last else outs,
                              send_work, mb, extras))
        self.stats.fwd_count += 1
        return loss, extras

    def run_backward(self):
        xs, out, send_work, mb, _ = self.inflight.popleft()
        if send_work is not None:
            for w in send_work:
                w.wait()  # activations must be delivered before reuse
        if self.is_last:
            out.backward()  # out is the loss
        else:
            ch = self.transport.channel(self.rank, self._next_rank(mb),
                                        "bwd")
            grads, grad_outs = [], []
            for t, spec in zip(out, self.out_specs):
                if not spec.grad:
                    continue
                gbuf = torch.empty(spec.shape, device=self.device,
                                   dtype=spec.dtype)
                ch.irecv(gbuf).wait()
                grads.append(gbuf)
                grad_outs.append(t)
            torch.autograd.backward(grad_outs, grads)
        if not self.is_first:
            ch = self.transport.channel(self._prev_rank(mb), self.rank,
                                        "bwd")
            for t, spec in zip(xs, self.in_specs):
                if not spec.grad:
                    continue
                g = t.grad if t.grad is not None else torch.zeros_like(t)
                # wait so the grad buffer lifetime is safe (upstream is
                # blocked on it anyway)
                ch.isend(g).wait()
                self.stats.send_bytes += g.numel() * g.element_size()
        self.stats.bwd_count += 1
        return mb

    def pop_eval(self):
        """Retire the oldest eval-mode forward (wait for its sends)."""
        x, out, send_work, mb, extras = self.inflight.popleft()
        if send_work is not None:
            for w in send_work:
                w.wait()
        return mb, extras

    # ---- eval ack clocking (reference run_ack, runtime.py:630-654) ----
    def send_ack(self, mb: int) -> None:
        ch = self.transport.channel(self._prev_rank(mb), self.rank, "bwd")
        ch.isend(torch.zeros(1, dtype=self.dtype,
                             device=self.device)).wait()

    def recv_ack(self, mb: int) -> None:
        ch = self.transport.channel(self.rank, self._next_rank(mb), "bwd")
        buf = torch.empty(1, dtype=self.dtype, device=self.device)
        ch.irecv(buf).wait()
