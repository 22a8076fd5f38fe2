#!/usr/bin/env python3
"""Per-shape conv microbenchmark: MFMA implicit-GEMM kernel vs MIOpen.

Times forward and data-grad over the ResNet-50 layer shapes (bf16,
channels_last) and prints one JSON line per shape with ms and effective
TFLOP/s for both backends — the measurement that drives the conv
dispatch policy."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch  # noqa: E402

# (C, H/W, K, R, stride) — ResNet-50 body at 224^2 (stem excluded: C=3)
RESNET50_SHAPES = [
    (64, 56, 64, 1, 1), (64, 56, 64, 3, 1), (64, 56, 256, 1, 1),
    (256, 56, 64, 1, 1), (256, 56, 128, 1, 1), (128, 56, 128, 3, 2),
    (128, 28, 512, 1, 1), (512, 28, 128, 1, 1), (512, 28, 256, 1, 1),
    (256, 28, 256, 3, 2), (256, 14, 1024, 1, 1), (1024, 14, 256, 1, 1),
    (1024, 14, 512, 1, 1), (512, 14, 512, 3, 2), (512, 7, 2048, 1, 1),
    (2048, 7, 512, 1, 1),
]


def bench_op(fn, iters, warmup):
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
    p.add_argument("--batch", type=int, default=256)
    p.add_argument("--iters", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    args = p.parse_args()
    assert torch.cuda.is_available()
    torch.backends.cudnn.benchmark = True
    from ddlbench_amd.ops.conv import conv2d_mfma
    dev = torch.device("cuda", 0)
    cl = torch.channels_last

    for C, HW, K, R, stride in RESNET50_SHAPES:
        pad = (R - 1) // 2
        N = args.batch
        x = torch.randn(N, C, HW, HW, device=dev,
                        dtype=torch.bfloat16).contiguous(memory_format=cl)
        w = torch.randn(K, C, R, R, device=dev,
                        dtype=torch.bfloat16).contiguous(memory_format=cl)
        OH = (HW + 2 * pad - R) // stride + 1
        flops = 2.0 * N * OH * OH * K * C * R * R
        dy = torch.randn(N, K, OH, OH, device=dev,
                         dtype=torch.bfloat16).contiguous(memory_format=cl)
        w_perm = w.permute(1, 2, 3, 0).contiguous()

        from ddlbench_amd.ops import require_extension
        ext = require_extension()
        res = {"shape": f"C{C}_HW{HW}_K{K}_R{R}_s{stride}",
               "batch": N}
        # forward: v2 (deep pipeline, default), v1 (128-tile), MIOpen
        os.environ["DDLB_CONV_V2"] = "1"
        t = bench_op(lambda: ext.conv_igemm_fwd(x, w, stride, pad),
                     args.iters, args.warmup)
        res["mfma2_fwd_ms"] = round(t * 1e3, 3)
        res["mfma2_fwd_tf"] = round(flops / t / 1e12, 1)
        os.environ["DDLB_CONV_V2"] = "0"
        t = bench_op(lambda: ext.conv_igemm_fwd(x, w, stride, pad),
                     args.iters, args.warmup)
        res["mfma_fwd_ms"] = round(t * 1e3, 3)
        res["mfma_fwd_tf"] = round(flops / t / 1e12, 1)
        t = bench_op(lambda: torch.nn.functional.conv2d(
            x, w, None, stride, pad), args.iters, args.warmup)
        res["miopen_fwd_ms"] = round(t * 1e3, 3)
        res["miopen_fwd_tf"] = round(flops / t / 1e12, 1)
        # dgrad
        os.environ["DDLB_CONV_V2"] = "1"
        t = bench_op(lambda: ext.conv_igemm_dgrad(
            dy, w_perm, N, C, HW, HW, stride, pad),
            args.iters, args.warmup)
        res["mfma2_dgrad_ms"] = round(t * 1e3, 3)
        res["mfma2_dgrad_tf"] = round(flops / t / 1e12, 1)
        os.environ["DDLB_CONV_V2"] = "0"
        t = bench_op(lambda: ext.conv_igemm_dgrad(
            dy, w_perm, N, C, HW, HW, stride, pad),
            args.iters, args.warmup)
        res["mfma_dgrad_ms"] = round(t * 1e3, 3)
        res["mfma_dgrad_tf"] = round(flops / t / 1e12, 1)
        os.environ["DDLB_CONV_V2"] = "1"
        t = bench_op(lambda: torch.ops.aten.convolution_backward(
            dy, x, w, None, [stride, stride], [pad, pad], [1, 1], False,
            [0, 0], 1, [True, False, False]), args.iters, args.warmup)
        res["miopen_dgrad_ms"] = round(t * 1e3, 3)
        res["miopen_dgrad_tf"] = round(flops / t / 1e12, 1)
        # wgrad: v2 (tr_b16 pipeline) then v1 (scalar gathers)
        os.environ["DDLB_WGRAD_V2"] = "1"
        t = bench_op(lambda: ext.conv_igemm_wgrad(x, dy, R, R, stride,
                                                  pad),
                     args.iters, args.warmup)
        res["mfma2_wgrad_ms"] = round(t * 1e3, 3)
        res["mfma2_wgrad_tf"] = round(flops / t / 1e12, 1)
        os.environ["DDLB_WGRAD_V2"] = "0"
        t = bench_op(lambda: ext.conv_igemm_wgrad(x, dy, R, R, stride,
                                                  pad),
                     args.iters, args.warmup)
        res["mfma_wgrad_ms"] = round(t * 1e3, 3)
        res["mfma_wgrad_tf"] = round(flops / t / 1e12, 1)
        os.environ["DDLB_WGRAD_V2"] = "1"
        t = bench_op(lambda: torch.ops.aten.convolution_backward(
            dy, x, w, None, [stride, stride], [pad, pad], [1, 1], False,
            [0, 0], 1, [False, True, False]), args.iters, args.warmup)
        res["miopen_wgrad_ms"] = round(t * 1e3, 3)
        res["miopen_wgrad_tf"] = round(flops / t / 1e12, 1)
        print(json.dumps(res), flush=True)


if __name__ == "__main__":
    main()
