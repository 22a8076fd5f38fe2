#!/usr/bin/env python3
"""Summarize a rocprofv3 kernel trace into a steady-state table.

Input: the kernel_trace.csv from
  rocprofv3 --kernel-trace --stats --output-format csv -d DIR -- <cmd>
MIOpen's find phase dominates whole-run totals, so this tool windows the
LAST --window-ms of the trace (the timed steps run last) and aggregates
busy time per kernel, grouping ours vs library kernels.

  python run/analyze_trace.py gpurun_out/prof/runc/*_kernel_trace.csv \
      --window-ms 140 [--markdown]
"""

import argparse
import csv
import json
import sys
from collections import defaultdict

GROUPS = [
    ("ddlbench_bn", lambda n: n.startswith("void bn_") or
     n.startswith("bn_")),
    ("ddlbench_conv", lambda n: "conv_igemm" in n or "conv_wgrad" in n or
     "wgrad_combine" in n or "dw3x3" in n),
    ("ddlbench_pool", lambda n: "maxpool_fwd_nhwc" in n or
     "maxpool_bwd_nhwc" in n),
    ("ddlbench_sgd_adam", lambda n: "fused_sgd" in n or "fused_adam" in n),
    ("ddlbench_ce_seq", lambda n: n.startswith("void ce_") or
     "revert_varlen" in n or "varlen_mask" in n),
    ("miopen_conv", lambda n: n.startswith("igemm") or "naive_conv" in n or
     ("ck" in n and "conv" in n.lower())),
    ("rocblas_gemm", lambda n: n.startswith("Cijk") or
     "gemm" in n.lower()),
    ("torch_elementwise", lambda n: "elementwise" in n or "Functor" in n or
     "fillBuffer" in n or "SubTensor" in n),
    ("pool_norm_other_torch", lambda n: "pool" in n or "reduce" in n
     or "at::native" in n),
]


def classify(name: str) -> str:
    for g, pred in GROUPS:
        if pred(name):
            return g
    return "other"


def main():
    p = argparse.ArgumentParser()
    p.add_argument("trace")
    p.add_argument("--window-ms", type=float, default=140.0)
    p.add_argument("--top", type=int, default=25)
    p.add_argument("--markdown", action="store_true")
    a = p.parse_args()

    rows = list(csv.DictReader(open(a.trace)))
    if not rows:
        sys.exit("empty trace")
    rows.sort(key=lambda r: int(r["Start_Timestamp"]))
    t1 = int(rows[-1]["End_Timestamp"])
    cut = t1 - a.window_ms * 1e6

    per_kernel = defaultdict(lambda: [0, 0])
    per_group = defaultdict(lambda: [0, 0])
    for r in rows:
        s, e = int(r["Start_Timestamp"]), int(r["End_Timestamp"])
        if s < cut:
            continue
        name = r["Kernel_Name"]
        per_kernel[name[:90]][0] += e - s
        per_kernel[name[:90]][1] += 1
        g = classify(name)
        per_group[g][0] += e - s
        per_group[g][1] += 1
    tot = sum(v[0] for v in per_kernel.values()) or 1

    if a.markdown:
        print(f"busy {tot / 1e6:.1f} ms in the last {a.window_ms:.0f} ms\n")
        print("| group | ms | % | calls |\n|---|---|---|---|")
        for g, (d, c) in sorted(per_group.items(), key=lambda kv: -kv[1][0]):
            print(f"| {g} | {d / 1e6:.2f} | {d / tot * 100:.1f} | {c} |")
        print("\n| ms | % | calls | kernel |\n|---|---|---|---|")
        for n, (d, c) in sorted(per_kernel.items(),
                                key=lambda kv: -kv[1][0])[:a.top]:
            print(f"| {d / 1e6:.2f} | {d / tot * 100:.1f} | {c} | "
                  f"`{n.replace('|', '/')}` |")
    else:
        print(json.dumps({
            "busy_ms": round(tot / 1e6, 2),
            "window_ms": a.window_ms,
            "groups": {g: {"ms": round(d / 1e6, 2),
                           "pct": round(d / tot * 100, 1), "calls": c}
                       for g, (d, c) in sorted(per_group.items(),
                                               key=lambda kv: -kv[1][0])},
        }, indent=1))


if __name__ == "__main__":
    main()
