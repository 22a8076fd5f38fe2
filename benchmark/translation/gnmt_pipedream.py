#!/usr/bin/env python3
"""GNMT through the 1F1B pipeline (torchrun, one rank per GPU).

The reference's translation/main_with_runtime.py counterpart
(SURVEY.md §2.12): profiled stage balance, tuple-I/O 1F1B runtime,
fixed padded sequence lengths for static edge shapes."""
import argparse, os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
from ddlbench_amd.gnmt_runner import run_gnmt_pipeline

if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--vocab", type=int, default=32320)
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--layers", type=int, default=4)
    p.add_argument("--minibatches", type=int, default=64)
    p.add_argument("--src-len", type=int, default=48)
    p.add_argument("--tgt-len", type=int, default=48)
    p.add_argument("--lr", type=float, default=2.5e-4)
    p.add_argument("--dtype", default="float32",
                   choices=["float32", "bfloat16"])
    p.add_argument("--device", default="auto")
    p.add_argument("--data-dir", default=os.environ.get("DATADIR", ""),
                   help="parallel corpus root (train.src/train.tgt); "
                        "synthetic token stream when empty")
    a = p.parse_args()
    run_gnmt_pipeline(
        epochs=int(os.environ.get("EPOCHS", 3)),
        batch_size=int(os.environ.get("BATCH_SIZE", 64)),
        n_minibatches=a.minibatches, vocab=a.vocab, hidden=a.hidden,
        layers=a.layers, lr=a.lr, dtype=a.dtype, device=a.device,
        src_len_max=a.src_len, tgt_len=a.tgt_len,
        log_interval=int(os.environ.get("LOGINTER", 0)),
        data_dir=a.data_dir)
