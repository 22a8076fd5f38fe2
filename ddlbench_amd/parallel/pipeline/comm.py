"""P2P activation/gradient transport for pipeline stages.

The reference emulates p2p with per-tensor helper threads doing
2-rank broadcasts (intra-node) or CPU-staged send/recv (inter-node), and
sends a shape message before every payload
(/root/reference/pipedream-fork/runtime/communication.py:635-712).
MI355X-native rebuild:

* true RCCL point-to-point send/recv over xGMI — no CPU staging, no
  broadcast-pair hack, no Python helper threads;
* shapes are static (known from one dry run), so no shape messages —
  receives are posted into preallocated buffers;
* one dedicated process group per (pipeline edge, direction): each group
  is its own RCCL communicator with its own internal stream, so forward
  activations, backward gradients, and eval acks never serialize against
  each other and cannot deadlock on call-order (RCCL, like NCCL, ignores
  p2p tags — ordering is per-communicator);
* on gloo (CPU tests) the same code path works with world-default group
  + tags.
"""

from __future__ import annotations

from typing import Dict, List, Tuple

import torch
import torch.distributed as dist


class EdgeChannel:
    """One direction of one pipeline edge (src rank -> dst rank)."""

    def __init__(self, src: int, dst: int, group, tag: int):
        self.src = src
        self.dst = dst
        self.group = group
        self.tag = tag

    def isend(self, t: torch.Tensor):
        return dist.isend(t.contiguous(), self.dst, group=self.group,
                          tag=self.tag)

    def irecv(self, buf: torch.Tensor):
        return dist.irecv(buf, self.src, group=self.group, tag=self.tag)


class PipelineTransport:
    """Builds per-edge-direction channels. All ranks must call the
    constructor with the same edge list (new_group is collective)."""

    def __init__(self, edges: List[Tuple[int, int]], backend: str):
        """edges: list of (src, dst) rank pairs that exchange forward
        activations; the reverse direction (gradients/acks) gets its own
        group."""
        self.channels: Dict[Tuple[int, int, str], EdgeChannel] = {}
        use_groups = backend == "nccl"
        for i, (src, dst) in enumerate(edges):
            for direction, (a, b) in (("fwd", (src, dst)),
                                      ("bwd", (dst, src))):
                if use_groups:
                    g = dist.new_group([src, dst])
                    tag = 0
                else:
                    g = None  # gloo: default group + tags
                    tag = 2 * i + (0 if direction == "fwd" else 1)
                self.channels[(src, dst, direction)] = EdgeChannel(
                    a, b, g, tag)

    def channel(self, src: int, dst: int, direction: str) -> EdgeChannel:
        return self.channels[(src, dst, direction)]


class _LocalWork:
    def wait(self):
        return None


class _LocalChannel:
    def __init__(self, q):
        self.q = q

    def isend(self, t: torch.Tensor):
        self.q.append(t.detach().clone())
        return _LocalWork()

    def irecv(self, buf: torch.Tensor):
        buf.copy_(self.q.popleft())
        return _LocalWork()


class LocalTransport:
    """In-process transport: every stage lives in ONE process. Lets a
    single GPU run a multi-stage 1F1B schedule (schedule, weight
    versioning and copy ordering under real HIP streams) without
    multi-process RCCL — the sends/recvs become deque hand-offs, so the
    caller must interleave the stages' run_forward/run_backward in a
    valid pipeline order (a recv before its send raises IndexError
    rather than deadlocking)."""

    def __init__(self, edges: List[Tuple[int, int]]):
        from collections import deque
        self.channels = {}
        for (src, dst) in edges:
            for direction in ("fwd", "bwd"):
                self.channels[(src, dst, direction)] = _LocalChannel(
                    deque())

    def channel(self, src: int, dst: int, direction: str):
        return self.channels[(src, dst, direction)]


def dry_run_shapes(stages: List[torch.nn.Module], sample: torch.Tensor,
                   device=torch.device("cpu")) -> List[torch.Size]:
    """Output shape of each stage for one micro/minibatch — run once at
    startup so no shape ever travels on the wire (the reference already
    precomputes training_tensor_shapes but sends shapes anyway;
    main_with_runtime.py:168-192)."""
    shapes = []
    with torch.no_grad():
        x = sample.to(device)
        for st in stages:
            # a rank may have moved only ITS stage's layers to the GPU
            # (they are shared with the full chain) — follow each
            # slice's parameters so the probe never crosses devices
            p = next(st.parameters(), None)
            if p is not None and x.device != p.device:
                x = x.to(p.device)
            x = st(x)
            shapes.append(x.shape)
    return shapes
