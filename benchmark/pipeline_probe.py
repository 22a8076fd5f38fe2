#!/usr/bin/env python3
"""Single-GPU 2-stage pipeline timing probe (VERDICT round-1 item 4):
GPipe fill-drain and 1F1B with versioned weights, both stages on
cuda:0, per-phase wall times."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import torch  # noqa: E402


def sync():
    torch.cuda.synchronize()


def main():
    dev = torch.device("cuda", 0)
    out = {}

    # ---- GPipe: 2 partitions on one device ----------------------------
    from ddlbench_amd.config import BenchConfig
    from ddlbench_amd.models import build_sequential
    from ddlbench_amd.parallel.pipeline.gpipe import build_gpipe
    cfg = BenchConfig(dataset="cifar10", arch="resnet18",
                      microbatches=8, batch_size=128)
    torch.manual_seed(0)
    seq = build_sequential("cifar10", "resnet18").to(torch.bfloat16)
    sample = torch.randn(16, 3, 32, 32, dtype=torch.bfloat16)
    model = build_gpipe(cfg, seq, sample.to(dev),
                        devices=[dev, dev])
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    x = torch.randn(128, 3, 32, 32, device=dev, dtype=torch.bfloat16)
    y = torch.randint(10, (128,), device=dev)
    for _ in range(3):
        opt.zero_grad()
        torch.nn.functional.cross_entropy(
            model(x).float(), y).backward()
        opt.step()
    sync()
    t0 = time.perf_counter()
    iters = 20
    for _ in range(iters):
        opt.zero_grad()
        torch.nn.functional.cross_entropy(
            model(x).float(), y).backward()
        opt.step()
    sync()
    out["gpipe_2stage_1gpu_ms"] = round(
        (time.perf_counter() - t0) / iters * 1e3, 3)
    out["gpipe_balance"] = [len(s) for s in model.stages]

    # ---- 1F1B: 2 StageRuntimes, LocalTransport, per-phase times -------
    import tests.test_pipeline_gpu as tp
    full, mods, (rt0, rt1), B = tp._two_stage_runtimes(dev)
    from ddlbench_amd.ops.sgd import FusedSGD
    from ddlbench_amd.parallel.pipeline.stash import VersionedOptimizer
    opts = [VersionedOptimizer(FusedSGD(m.parameters(), lr=0.01,
                                        momentum=0.9, backend="native"),
                               versioned=(i == 0))
            for i, m in enumerate(mods)]
    xb = torch.randn(B, 6, device=dev)
    yb = torch.randint(3, (B,), device=dev)
    phases = {"s0_fwd": 0.0, "s1_fwd": 0.0, "s1_bwd": 0.0,
              "s0_bwd": 0.0, "steps": 0.0}
    n = 40
    rt0.run_forward(0, lambda i: xb, lambda i: yb, training=True)
    sync()
    for m in range(n):
        t = time.perf_counter()
        if m + 1 < n:
            rt0.run_forward(m + 1, lambda i: xb, lambda i: yb,
                            training=True)
        sync()
        t1 = time.perf_counter()
        phases["s0_fwd"] += t1 - t
        rt1.run_forward(m, lambda i: xb, lambda i: yb, training=True)
        sync()
        t2 = time.perf_counter()
        phases["s1_fwd"] += t2 - t1
        opts[1].zero_grad(set_to_none=False)
        rt1.run_backward()
        sync()
        t3 = time.perf_counter()
        phases["s1_bwd"] += t3 - t2
        opts[0].zero_grad(set_to_none=False)
        rt0.run_backward()
        sync()
        t4 = time.perf_counter()
        phases["s0_bwd"] += t4 - t3
        opts[1].step()
        opts[0].step()
        sync()
        phases["steps"] += time.perf_counter() - t4
    out["1f1b_2stage_1gpu_per_mb_us"] = {
        k: round(v / n * 1e6, 1) for k, v in phases.items()}
    print(json.dumps(out))


if __name__ == "__main__":
    main()
