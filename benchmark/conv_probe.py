#!/usr/bin/env python3
"""Minimal conv kernel loop for PMC profiling (one kernel dominates the
trace window so per-kernel counter rows are unambiguous).

  rocprofv3 --kernel-trace --pmc SQ_WAVE_CYCLES ... -- \
      python benchmark/conv_probe.py --shape C512_HW28_K256_R1_s1 --op fwd
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import torch  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--shape", default="C512_HW28_K256_R1_s1")
    p.add_argument("--op", default="fwd", choices=["fwd", "dgrad", "wgrad"])
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--batch", type=int, default=256)
    args = p.parse_args()
    parts = dict(kv.split("_", 0) for kv in [])  # noqa
    toks = args.shape.split("_")
    C = int(toks[0][1:]); HW = int(toks[1][2:]); K = int(toks[2][1:])
    R = int(toks[3][1:]); stride = int(toks[4][1:])
    pad = (R - 1) // 2
    N = args.batch
    dev = torch.device("cuda", 0)
    cl = torch.channels_last
    x = torch.randn(N, C, HW, HW, device=dev,
                    dtype=torch.bfloat16).contiguous(memory_format=cl)
    w = torch.randn(K, C, R, R, device=dev,
                    dtype=torch.bfloat16).contiguous(memory_format=cl)
    OH = (HW + 2 * pad - R) // stride + 1
    dy = torch.randn(N, K, OH, OH, device=dev,
                     dtype=torch.bfloat16).contiguous(memory_format=cl)
    from ddlbench_amd.ops import require_extension
    ext = require_extension()
    if args.op == "fwd":
        fn = lambda: ext.conv_igemm_fwd(x, w, stride, pad)  # noqa
    elif args.op == "dgrad":
        w_perm = w.permute(1, 2, 3, 0).contiguous()
        fn = lambda: ext.conv_igemm_dgrad(  # noqa
            dy, w_perm, N, C, HW, HW, stride, pad)
    else:
        fn = lambda: ext.conv_igemm_wgrad(x, dy, R, R, stride, pad)  # noqa
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    import time
    t0 = time.perf_counter()
    for _ in range(args.iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    fl = 2.0 * N * OH * OH * K * C * R * R
    print(f"{args.shape} {args.op}: {dt*1e3:.3f} ms  {fl/dt/1e12:.1f} TF")


if __name__ == "__main__":
    main()
