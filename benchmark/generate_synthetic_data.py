#!/usr/bin/env python3
"""Materialize synthetic datasets to disk (optional).

The reference pre-generates random JPEG ImageFolder trees once per node
(/root/reference/benchmark/generate_synthetic_data.py). Our datasets are
tensor-native and generated on the fly (ddlbench_amd/data/synthetic.py),
so this tool exists only for CLI parity and for workflows that want an
on-disk dataset: it writes sharded .pt tensor files with the reference's
shapes/sizes.

    python generate_synthetic_data.py mnist|cifar10|imagenet|highres \
        [--out DIR] [--scale F] [--shard-size N]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch  # noqa: E402

from ddlbench_amd.config import DATASET_SHAPES  # noqa: E402
from ddlbench_amd.data import SyntheticImageDataset  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("dataset", choices=sorted(DATASET_SHAPES))
    p.add_argument("--out", default=os.environ.get("DATADIR", "data"))
    p.add_argument("--scale", type=float, default=0.01,
                   help="fraction of the full dataset size")
    p.add_argument("--shard-size", type=int, default=2048)
    p.add_argument("--seed", type=int, default=42)
    args = p.parse_args()

    for split, train in (("train", True), ("val", False)):
        ds = SyntheticImageDataset(args.dataset, train=train,
                                   seed=args.seed, scale=args.scale)
        out_dir = os.path.join(args.out, args.dataset, split)
        os.makedirs(out_dir, exist_ok=True)
        for shard_start in range(0, len(ds), args.shard_size):
            n = min(args.shard_size, len(ds) - shard_start)
            xs = torch.stack([ds[shard_start + i][0] for i in range(n)])
            ys = torch.tensor([ds[shard_start + i][1] for i in range(n)])
            path = os.path.join(out_dir,
                                f"shard_{shard_start:08d}.pt")
            torch.save({"x": xs, "y": ys}, path)
        print(f"{args.dataset}/{split}: {len(ds)} samples -> {out_dir}")


if __name__ == "__main__":
    main()
