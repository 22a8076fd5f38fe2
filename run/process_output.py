#!/usr/bin/env python3
"""Parse benchmark stdout logs into a JSON summary.

The counterpart of the reference's
pipedream-fork/runtime/scripts/process_output.py:40-60 — recovers
per-interval samples/sec, per-epoch metrics, and the final summary from
the fixed-format result log.

    python run/process_output.py out/<stamp>/run.log
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

from ddlbench_amd.utils import parse_result_line  # noqa: E402


def extract_reduce_times(path: str):
    """Per-bucket all-reduce spans (ms) from `reduce_times_ms:` lines
    (DDLB_LOG_REDUCE=1) — the counterpart of the reference's
    profiler utils/all_reduce/extract_reduce_times.py:7-30."""
    times = []
    with open(path) as f:
        for line in f:
            if "reduce_times_ms:" in line:
                vals = line.split("reduce_times_ms:", 1)[1].split()
                times.extend(float(v) for v in vals)
    return times


def parse_log(path: str) -> dict:
    out = {"train": [], "epochs": [], "final": None}
    with open(path) as f:
        for line in f:
            d = parse_result_line(line)
            if d is None:
                continue
            kind = d.pop("kind")
            if kind == "train":
                out["train"].append(d)
            elif kind == "epoch":
                out["epochs"].append(d)
            else:
                out["final"] = d
    reduce_ms = extract_reduce_times(path)
    if reduce_ms:
        out["reduce_times_ms"] = reduce_ms
    return out


def main():
    p = argparse.ArgumentParser()
    p.add_argument("logs", nargs="+")
    args = p.parse_args()
    for path in args.logs:
        summary = parse_log(path)
        print(json.dumps({"log": path, **summary}, indent=1))


if __name__ == "__main__":
    main()
