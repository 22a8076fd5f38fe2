#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 ImageNet-shape data-parallel training.

Driver contract:
  python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run (one rank per
GPU over RCCL). Rank 0 prints ONE JSON line with the whole-job aggregate
images/sec (BASELINE.json metric: "images/sec (whole node) ResNet-50
ImageNet-shape at 1/2/4/8 MI355X").

Timed region: K full training steps (forward, loss, backward with
overlapped bucketed all-reduce, fused SGD update), bracketed by
barrier + torch.cuda.synchronize on both sides; MAX elapsed over ranks.
Synthetic ImageNet-shape data (3x224x224), random-init weights, bf16
compute with fp32 optimizer state.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=256,
                   help="per-GPU batch size")
    p.add_argument("--model", default="resnet50")
    p.add_argument("--dataset", default="imagenet",
                   choices=["imagenet", "cifar10", "mnist", "highres"],
                   help="input shape / class count")
    p.add_argument("--dtype", default="bfloat16",
                   choices=["bfloat16", "float32"])
    p.add_argument("--device", default="auto",
                   help="auto|cuda|cpu (cpu only for plumbing tests)")
    p.add_argument("--kernel-backend", default="auto",
                   choices=["auto", "native", "torch"])
    p.add_argument("--strategy", default="ddp", choices=["ddp", "gpipe"],
                   help="ddp: one rank per GPU over RCCL (the driver "
                        "contract); gpipe: single-process micro-batch "
                        "pipeline across all visible GPUs")
    p.add_argument("--microbatches", type=int, default=8,
                   help="gpipe chunk count")
    p.add_argument("--conv", default="mfma", choices=["miopen", "mfma"],
                   help="conv backend: the in-tree MFMA implicit-GEMM "
                        "kernels (default — the hand-written CDNA4 hot "
                        "path) or the MIOpen library")
    p.add_argument("--hipgraph", default="off", choices=["auto", "off"],
                   help="capture the whole training step as a hipGraph "
                        "when single-GPU (measured ~1% BEHIND eager at "
                        "b256 — the grads-persistent capture pays an "
                        "accumulate pass; kept for smaller per-GPU "
                        "batches where launch gaps dominate)")
    p.add_argument("--memory-format", default="channels_last",
                   choices=["channels_last", "contiguous"],
                   help="channels_last (NHWC) keeps MIOpen on its native "
                        "xdlops conv solvers and our BN kernels on the "
                        "coalesced NHWC path")
    return p.parse_args()


def main():
    args = parse_args()
    from ddlbench_amd.config import BenchConfig
    from ddlbench_amd.engine import compute_dtype, resolve_device
    from ddlbench_amd.models import build_model
    from ddlbench_amd.ops import functional as NF
    from ddlbench_amd.ops.modules import set_default_backend
    from ddlbench_amd.ops.sgd import FusedSGD
    from ddlbench_amd.parallel import BucketedDataParallel, init_distributed

    env = init_distributed()
    world = env.world_size
    cfg = BenchConfig(dataset=args.dataset, arch=args.model, strategy="ddp",
                      batch_size=args.batch, dtype=args.dtype,
                      kernel_backend=args.kernel_backend, num_workers=0)
    set_default_backend(cfg.kernel_backend)
    device = resolve_device(cfg, env.local_rank)
    dtype = compute_dtype(cfg)
    if device.type == "cpu":
        dtype = torch.float32  # plumbing mode only

    torch.backends.cudnn.benchmark = True  # MIOpen find for best solvers
    channels_last = (args.memory_format == "channels_last"
                     and device.type == "cuda")
    torch.manual_seed(1234 + env.rank)
    if args.strategy == "gpipe":
        from ddlbench_amd.models import build_sequential
        from ddlbench_amd.parallel.pipeline.gpipe import build_gpipe
        assert world == 1, "gpipe strategy is single-process multi-device"
        seq = build_sequential(cfg.dataset, cfg.arch)
        if dtype != torch.float32:
            seq = seq.to(dtype)
        sample = torch.randn((max(args.batch // args.microbatches, 1),)
                             + tuple(cfg.shape), dtype=dtype)
        import dataclasses
        gcfg = dataclasses.replace(cfg, microbatches=args.microbatches)
        model = build_gpipe(gcfg, seq, sample.to(device))
        device = model.out_device
    else:
        model = build_model(cfg.dataset, cfg.arch).to(device)
        if dtype != torch.float32:
            model = model.to(dtype)
        if channels_last:
            model = model.to(memory_format=torch.channels_last)
    if args.conv == "mfma" and device.type == "cuda":
        from ddlbench_amd.ops.conv import convert_convs
        n_conv = convert_convs(model, dtype)
        if env.rank == 0:
            print(f"# mfma conv kernels on {n_conv} layers",
                  file=sys.stderr)
    dp = BucketedDataParallel(model)
    opt = FusedSGD(model.parameters(), lr=0.1 * world, momentum=0.9,
                   weight_decay=1e-4, backend=cfg.kernel_backend)

    # synthetic device-resident batches, rotated so no step reuses the
    # previous step's input (no caching of outputs; all compute runs)
    n_pool = 4
    pool = []
    g = torch.Generator().manual_seed(7 + env.rank)
    for _ in range(n_pool):
        x = torch.randn((args.batch,) + tuple(cfg.shape), generator=g)
        y = torch.randint(cfg.num_classes, (args.batch,), generator=g)
        x = x.to(device, dtype=dtype)
        if channels_last:
            x = x.contiguous(memory_format=torch.channels_last)
        pool.append((x, y.to(device)))

    def step(i):
        x, y = pool[i % n_pool]
        out = dp(x)
        loss = NF.cross_entropy(out, y, backend=cfg.kernel_backend)
        dp.zero_grad_buckets()
        loss.backward()
        dp.finalize_backward()
        opt.step()
        return loss

    # hipGraph-captured step (single GPU): replays the whole
    # fwd+loss+bwd+fused-SGD step as one graph — removes every
    # inter-kernel launch gap (round-1 trace: ~11% GPU idle at b256).
    # Grads stay allocated (set_to_none=False) so every captured
    # pointer is stable across replays.
    def gstep(i):
        x, y = pool[i % n_pool]
        out = dp(x)
        loss = NF.cross_entropy(out, y, backend=cfg.kernel_backend)
        model.zero_grad(set_to_none=False)
        loss.backward()
        opt.step()
        return loss

    use_graph = (args.hipgraph != "off" and device.type == "cuda"
                 and world == 1 and args.strategy == "ddp")
    step_fn = step
    if use_graph:
        for i in range(max(args.warmup, 3)):
            gstep(i)
        torch.cuda.synchronize(device)
        try:
            gpool = torch.cuda.graphs.graph_pool_handle()
            graphs = []
            for i in range(n_pool):
                gr = torch.cuda.CUDAGraph()
                with torch.cuda.graph(gr, pool=gpool):
                    gstep(i)
                graphs.append(gr)
            torch.cuda.synchronize(device)

            def step_fn(i, _g=graphs):
                _g[i % n_pool].replay()
            if env.rank == 0:
                print("# hipGraph-captured step (4 graphs)",
                      file=sys.stderr)
        except Exception as e:  # noqa: BLE001 - eager fallback
            if env.rank == 0:
                print(f"# hipGraph capture failed ({e}); eager",
                      file=sys.stderr)
            step_fn = step

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    for i in range(args.warmup):
        step_fn(i)
    barrier_sync()
    # per-step boundaries via device events (no host syncs inside the
    # timed region; queried after the closing synchronize)
    events = None
    if device.type == "cuda":
        events = [torch.cuda.Event(enable_timing=True)
                  for _ in range(args.steps + 1)]
    t0 = time.perf_counter()
    if events:
        events[0].record()
    for i in range(args.steps):
        step_fn(args.warmup + i)
        if events:
            events[i + 1].record()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    step_ms = None
    if events:
        step_ms = [events[i].elapsed_time(events[i + 1])
                   for i in range(args.steps)]

    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device.type == "cuda" else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    if env.rank == 0:
        total_images = args.steps * args.batch * world
        value = total_images / elapsed
        print(json.dumps({
            "metric": "images/sec",
            "value": value,
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            **({"ms_per_step_std": round(float(torch.tensor(
                    step_ms).std(unbiased=False)), 4),
                "ms_per_step_max": round(max(step_ms), 3)}
               if step_ms else {}),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": str(dtype).replace("torch.", ""),
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * world,
                "input": "x".join(str(d) for d in cfg.shape),
                "parallelism": (f"dp{world}" if args.strategy == "ddp"
                                else f"gpipe{torch.cuda.device_count() or 1}"
                                     f"x{args.microbatches}mb"),
            },
        }), flush=True)


if __name__ == "__main__":
    main()
