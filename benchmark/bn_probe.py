#!/usr/bin/env python3
"""Within-process A/B probe of the BN kernel variants.

Times the raw bindings per phase (stats+apply = train fwd; apply only =
eval fwd; reduce+finalize+dx = bwd) for each variant interleaved in one
process (cdna_hip_programming.md §5.4 rule 24)."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch  # noqa: E402

SHAPES = [(256, 64, 112, 112), (256, 256, 56, 56), (256, 512, 28, 28),
          (256, 1024, 14, 14), (256, 2048, 7, 7)]


def t_ms(fn, iters=15, warmup=4):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    from ddlbench_amd.ops import require_extension
    ext = require_extension()
    dev = torch.device("cuda", 0)
    cl = torch.channels_last
    for N, C, H, W in SHAPES:
        x = torch.randn(N, C, H, W, device=dev,
                        dtype=torch.bfloat16).contiguous(memory_format=cl)
        res = torch.randn_like(x)
        dy = torch.randn_like(x)
        g = torch.rand(C, device=dev) + 0.5
        b = torch.randn(C, device=dev)
        rm = torch.zeros(C, device=dev)
        rv = torch.ones(C, device=dev)
        outs = ext.bn_act_fwd(x, res, g, b, rm, rv, True, 0.1,
                              1e-5, 1, True)
        y, mean, invstd = outs[0], outs[1], outs[2]
        mask = outs[3] if len(outs) > 3 else None
        nb = N * C * H * W * 2 / 1e9
        row = {"shape": f"{N}x{C}x{H}x{W}"}
        for variant in (1, 2):
            ext.set_bn_variant(variant)
            tf = t_ms(lambda: ext.bn_act_fwd(x, res, g, b, rm, rv, True,
                                             0.1, 1e-5, 1, True))
            te = t_ms(lambda: ext.bn_act_fwd(x, res, g, b, rm, rv, False,
                                             0.1, 1e-5, 1, True))
            tb = t_ms(lambda: ext.bn_act_bwd(dy, y, x, mean, invstd, g, 1,
                                             True, True, True, None))
            tbm = (t_ms(lambda: ext.bn_act_bwd(dy, None, x, mean, invstd,
                                               g, 1, True, True, True,
                                               mask))
                   if mask is not None else float("nan"))
            row[f"v{variant}_fwd_ms"] = round(tf, 3)
            row[f"v{variant}_fwd_TBs"] = round(4 * nb / tf, 2)
            row[f"v{variant}_apply_ms"] = round(te, 3)
            row[f"v{variant}_apply_TBs"] = round(3 * nb / te, 2)
            row[f"v{variant}_stats_TBs"] = round(nb / max(tf - te, 1e-5), 2)
            row[f"v{variant}_bwd_ms"] = round(tb, 3)
            row[f"v{variant}_bwd_TBs"] = round(8 * nb / tb, 2)
            row[f"v{variant}_bwdmask_ms"] = round(tbm, 3)
        ext.set_bn_variant(0)
        print(json.dumps(row), flush=True)


if __name__ == "__main__":
    main()
