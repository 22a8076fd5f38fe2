#!/usr/bin/env python3
"""Communication micro-benchmarks: p2p and collective bandwidth sweeps.

The reference's tests/communication (point_to_point.py size sweep
10..8e8 floats, all_to_all.py allreduce bandwidth —
/root/reference/pipedream-fork/runtime/tests/communication/) rebuilt as
one torchrun-launched tool on RCCL over xGMI:

  python -m torch.distributed.run --nproc-per-node 2 --master-addr \
      127.0.0.1 benchmark/comm/comm_bench.py --op p2p
  ... --op allreduce | reduce_scatter | all_gather | broadcast | all_to_all

Prints one line per size: bytes, time/iter, algorithmic GB/s, bus GB/s.
On an 8-GPU xGMI mesh the ring all-reduce bus bandwidth is bounded by a
single p2p link (~153 GB/s spec per link, 7 links/GPU) — this sweep is
what bucket-size tuning reads."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

from ddlbench_amd.parallel import init_distributed  # noqa: E402


def bus_factor(op: str, world: int) -> float:
    """algbw -> busbw factor (standard nccl-tests definitions)."""
    if op == "allreduce":
        return 2.0 * (world - 1) / world
    if op in ("reduce_scatter", "all_gather", "all_to_all"):
        return (world - 1) / world
    return 1.0


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--op", default="allreduce",
                   choices=["p2p", "allreduce", "reduce_scatter",
                            "all_gather", "broadcast", "all_to_all"])
    p.add_argument("--min-bytes", type=int, default=1 << 10)
    p.add_argument("--max-bytes", type=int, default=1 << 28)
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--dtype", default="bfloat16")
    args = p.parse_args()

    env = init_distributed()
    world = env.world_size
    assert world >= 2, "launch with torchrun --nproc-per-node >= 2"
    dev = (torch.device("cuda", env.local_rank)
           if torch.cuda.is_available() else torch.device("cpu"))
    dtype = (torch.bfloat16 if args.dtype == "bfloat16" and dev.type == "cuda"
             else torch.float32)
    esz = torch.tensor([], dtype=dtype).element_size()

    def sync():
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)

    size = args.min_bytes
    results = []
    while size <= args.max_bytes:
        n = max(size // esz, world)
        n -= n % world
        x = torch.rand(n, dtype=torch.float32).to(dev, dtype)

        def run_once():
            if args.op == "p2p":
                peer = env.rank ^ 1
                if peer >= world:
                    return
                if env.rank % 2 == 0:
                    dist.send(x, peer)
                    dist.recv(x, peer)
                else:
                    dist.recv(x, peer)
                    dist.send(x, peer)
            elif args.op == "allreduce":
                dist.all_reduce(x)
            elif args.op == "reduce_scatter":
                out = x[: n // world].clone()
                dist.reduce_scatter_tensor(out, x)
            elif args.op == "all_gather":
                out = torch.empty_like(x)
                dist.all_gather_into_tensor(out, x[: n // world])
            elif args.op == "broadcast":
                dist.broadcast(x, src=0)
            elif args.op == "all_to_all":
                out = torch.empty_like(x)
                dist.all_to_all_single(out, x)

        for _ in range(args.warmup):
            run_once()
        sync()
        dist.barrier()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            run_once()
        sync()
        dist.barrier()
        dt = (time.perf_counter() - t0) / args.iters
        nbytes = n * esz
        if args.op == "p2p":
            nbytes *= 2  # round trip
        algbw = nbytes / dt / 1e9
        busbw = algbw * bus_factor(args.op, world)
        if env.rank == 0:
            line = {"op": args.op, "bytes": nbytes, "us": dt * 1e6,
                    "algbw_GBs": round(algbw, 2),
                    "busbw_GBs": round(busbw, 2), "world": world,
                    "dtype": str(dtype).replace("torch.", "")}
            results.append(line)
            print(json.dumps(line), flush=True)
        size *= 4
    if env.rank == 0 and os.environ.get("COMM_BENCH_OUT"):
        with open(os.environ["COMM_BENCH_OUT"], "w") as f:
            json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
