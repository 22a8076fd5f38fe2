#!/usr/bin/env python3
"""Fused-op microbenchmarks vs eager PyTorch: BN+ReLU(+add), CE, SGD.

Reports ms and achieved HBM GB/s per op over ResNet-50 activation
shapes in both layouts — the evidence for the memory-bound fusion wins
and the input to kernel tuning (MI355X HBM3E ceiling ~6.3 TB/s)."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch  # noqa: E402

SHAPES = [  # (N, C, H, W) resnet50 @224
    (256, 64, 112, 112), (256, 256, 56, 56), (256, 512, 28, 28),
    (256, 1024, 14, 14), (256, 2048, 7, 7),
]


def timeit(fn, iters, warmup):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--dtype", default="bfloat16")
    args = p.parse_args()
    assert torch.cuda.is_available()
    from ddlbench_amd.ops import functional as NF
    dev = torch.device("cuda", 0)
    dt = torch.bfloat16 if args.dtype == "bfloat16" else torch.float32
    esz = torch.tensor([], dtype=dt).element_size()

    for nhwc in (False, True):
        fmt = torch.channels_last if nhwc else torch.contiguous_format
        for N, C, H, W in SHAPES:
            x = torch.randn(N, C, H, W, device=dev, dtype=dt) \
                .contiguous(memory_format=fmt)
            res = torch.randn_like(x)
            g = torch.rand(C, device=dev) + 0.5
            b = torch.randn(C, device=dev)
            rm = torch.zeros(C, device=dev)
            rv = torch.ones(C, device=dev)
            nb = N * C * H * W * esz

            def fused_fwd():
                return NF.bn_act(x, g, b, rm, rv, True, 0.1, 1e-5,
                                 "relu", res, backend="native")

            def eager_fwd():
                y = torch.nn.functional.batch_norm(x, rm, rv, g, b, True,
                                                   0.1, 1e-5)
                return torch.relu(y + res)

            t_f = timeit(fused_fwd, args.iters, args.warmup)
            t_e = timeit(eager_fwd, args.iters, args.warmup)
            # fwd traffic: read x (2x: stats+apply), read res, write y
            row = {"op": "bn_add_relu_fwd",
                   "layout": "nhwc" if nhwc else "nchw",
                   "shape": f"{N}x{C}x{H}x{W}",
                   "fused_ms": round(t_f * 1e3, 3),
                   "eager_ms": round(t_e * 1e3, 3),
                   "fused_GBs": round(4 * nb / t_f / 1e9, 0),
                   "speedup": round(t_e / t_f, 2)}
            print(json.dumps(row), flush=True)

            # backward
            xg = x.clone().requires_grad_(True)
            rg = res.clone().requires_grad_(True)
            gg = g.clone().requires_grad_(True)
            bg = b.clone().requires_grad_(True)
            y = NF.bn_act(xg, gg, bg, rm.clone(), rv.clone(), True, 0.1,
                          1e-5, "relu", rg, backend="native")
            dy = torch.randn_like(y)

            def fused_bwd():
                grads = torch.autograd.grad(y, [xg, rg, gg, bg], dy,
                                            retain_graph=True)
                return grads

            x2 = x.clone().requires_grad_(True)
            r2 = res.clone().requires_grad_(True)
            g2 = g.clone().requires_grad_(True)
            b2 = b.clone().requires_grad_(True)
            y2 = torch.relu(torch.nn.functional.batch_norm(
                x2, rm.clone(), rv.clone(), g2, b2, True, 0.1, 1e-5) + r2)

            def eager_bwd():
                return torch.autograd.grad(y2, [x2, r2, g2, b2], dy,
                                           retain_graph=True)

            t_f = timeit(fused_bwd, args.iters, args.warmup)
            t_e = timeit(eager_bwd, args.iters, args.warmup)
            row = {"op": "bn_add_relu_bwd",
                   "layout": "nhwc" if nhwc else "nchw",
                   "shape": f"{N}x{C}x{H}x{W}",
                   "fused_ms": round(t_f * 1e3, 3),
                   "eager_ms": round(t_e * 1e3, 3),
                   "fused_GBs": round(6 * nb / t_f / 1e9, 0),
                   "speedup": round(t_e / t_f, 2)}
            print(json.dumps(row), flush=True)

    # cross entropy
    for B, K in [(256, 1000), (512, 1000), (512, 32320)]:
        logits = torch.randn(B, K, device=dev, dtype=dt,
                             requires_grad=True)
        tgt = torch.randint(K, (B,), device=dev)
        t_f = timeit(lambda: NF.cross_entropy(logits, tgt,
                                              backend="native"),
                     args.iters, args.warmup)
        t_e = timeit(lambda: torch.nn.functional.cross_entropy(
            logits, tgt), args.iters, args.warmup)
        print(json.dumps({"op": "cross_entropy_fwd",
                          "shape": f"{B}x{K}",
                          "fused_ms": round(t_f * 1e3, 4),
                          "eager_ms": round(t_e * 1e3, 4),
                          "speedup": round(t_e / t_f, 2)}), flush=True)

    # fused SGD on resnet50-sized param set
    from ddlbench_amd.models import build_model
    from ddlbench_amd.ops.sgd import FusedSGD
    m = build_model("imagenet", "resnet50").to(dev).to(dt)
    for p_ in m.parameters():
        p_.grad = torch.randn_like(p_)
    o_f = FusedSGD(m.parameters(), lr=0.1, momentum=0.9,
                   weight_decay=1e-4, backend="native")
    t_f = timeit(o_f.step, args.iters, args.warmup)
    m2 = build_model("imagenet", "resnet50").to(dev).to(dt)
    for p_ in m2.parameters():
        p_.grad = torch.randn_like(p_)
    o_e = torch.optim.SGD(m2.parameters(), lr=0.1, momentum=0.9,
                          weight_decay=1e-4)
    t_e = timeit(o_e.step, args.iters, args.warmup)
    print(json.dumps({"op": "sgd_step_resnet50",
                      "fused_ms": round(t_f * 1e3, 3),
                      "eager_ms": round(t_e * 1e3, 3),
                      "speedup": round(t_e / t_f, 2)}), flush=True)


if __name__ == "__main__":
    main()
