#!/usr/bin/env python3
"""Print layer summaries of every model x dataset combination
(the reference's benchmark/network_summary.py:26-33)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch  # noqa: E402

from ddlbench_amd.config import DATASET_SHAPES  # noqa: E402
from ddlbench_amd.models import ARCHS, build_model  # noqa: E402
from ddlbench_amd.utils.summary import format_summary, summarize  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("-b", "--benchmark", default="all",
                   choices=["all"] + sorted(DATASET_SHAPES))
    p.add_argument("-a", "--arch", default="all",
                   choices=["all"] + list(ARCHS))
    args = p.parse_args()
    datasets = (sorted(DATASET_SHAPES) if args.benchmark == "all"
                else [args.benchmark])
    archs = list(ARCHS) if args.arch == "all" else [args.arch]
    for ds in datasets:
        c, h, w, _, _, _ = DATASET_SHAPES[ds]
        for arch in archs:
            model = build_model(ds, arch)
            rows = summarize(model, torch.randn(1, c, h, w))
            print(f"\n=== {ds} / {arch} ===")
            print(format_summary(rows))


if __name__ == "__main__":
    main()
