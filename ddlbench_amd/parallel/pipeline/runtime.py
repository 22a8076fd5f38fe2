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
import torch.distributed as dist

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
                 in_shape: Optional[torch.Size],
                 out_shape: torch.Size,
                 device: torch.device,
                 dtype: torch.dtype,
                 loss_fn: Optional[Callable] = None,
                 dp_wrapper=None):
        self.plan = plan
        self.rank = rank
        self.stage, self.replica = plan.stage_of_rank(rank)
        self.module = module
        self.transport = transport
        self.in_shape = in_shape
        self.out_shape = out_shape
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
            x = input_provider(mb).to(self.device, dtype=self.dtype,
                                      non_blocking=True)
        else:
            buf = torch.empty(self.in_shape, device=self.device,
                              dtype=self.dtype)
            ch = self.transport.channel(self._prev_rank(mb), self.rank,
                                        "fwd")
            ch.irecv(buf).wait()
            self.stats.recv_bytes += buf.numel() * buf.element_size()
            x = buf
        if training and not self.is_first:
            x.requires_grad_(True)
        with torch.enable_grad() if training else torch.no_grad():
            out = self.module(x)
        send_work = None
        loss = None
        extras = {}
        if self.is_last:
            if self.loss_fn is not None and target_provider is not None:
                y = target_provider(mb).to(self.device, non_blocking=True)
                loss = self.loss_fn(out, y)
                extras["target"] = y
                extras["output"] = out.detach()
        else:
            ch = self.transport.channel(self.rank, self._next_rank(mb),
                                        "fwd")
            send_work = ch.isend(out.detach())
            self.stats.send_bytes += out.numel() * out.element_size()
        self.inflight.append((x, loss if self.is_last else out, send_work,
                              mb, extras))
        self.stats.fwd_count += 1
        return loss, extras

    def run_backward(self):
        x, out, send_work, mb, _ = self.inflight.popleft()
        if send_work is not None:
            send_work.wait()  # activation must be delivered before reuse
        if self.is_last:
            out.backward()  # out is the loss
        else:
            gbuf = torch.empty(self.out_shape, device=self.device,
                               dtype=self.dtype)
            ch = self.transport.channel(self.rank, self._next_rank(mb),
                                        "bwd")
            ch.irecv(gbuf).wait()
            out.backward(gbuf)
        if not self.is_first:
            ch = self.transport.channel(self._prev_rank(mb), self.rank,
                                        "bwd")
            # send input grad upstream; wait so the grad buffer lifetime
            # is safe (upstream is blocked on it anyway — the pipeline's
            # critical path is unaffected)
            ch.isend(x.grad).wait()
            self.stats.send_bytes += x.grad.numel() * x.grad.element_size()
        self.stats.bwd_count += 1
        return mb

    def pop_eval(self):
        """Retire the oldest eval-mode forward (wait for its send)."""
        x, out, send_work, mb, extras = self.inflight.popleft()
        if send_work is not None:
            send_work.wait()
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
