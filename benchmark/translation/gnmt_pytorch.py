#!/usr/bin/env python3
"""GNMT translation benchmark, single device.

MI355X-native counterpart of the reference's pipedream-only GNMT
workload (pipedream-fork/runtime/translation/); env contract
EPOCHS/BATCH_SIZE/LOGINTER honoured."""
import argparse, os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
from ddlbench_amd.gnmt_runner import run_gnmt

def main(ddp=False):
    p = argparse.ArgumentParser()
    p.add_argument("--vocab", type=int, default=32320)
    p.add_argument("--hidden", type=int, default=1024)
    p.add_argument("--layers", type=int, default=4)
    p.add_argument("--dataset-size", type=int, default=2000)
    p.add_argument("--max-len", type=int, default=50)
    p.add_argument("--lr", type=float, default=2.5e-4)
    p.add_argument("--dtype", default="float32", choices=["float32", "bfloat16"])
    p.add_argument("--device", default="auto")
    p.add_argument("--kernel-backend", default="auto")
    p.add_argument("--data-dir", default=os.environ.get("DATADIR", ""),
                   help="parallel corpus root (train.src/train.tgt); "
                        "synthetic token stream when empty")
    p.add_argument("--optimizer", default="sgd",
                   choices=["sgd", "adam"],
                   help="adam = the reference GNMT optimizer "
                        "(fused multi-tensor step either way)")
    a = p.parse_args()
    run_gnmt(epochs=int(os.environ.get("EPOCHS", 3)),
             batch_size=int(os.environ.get("BATCH_SIZE", 64)),
             log_interval=int(os.environ.get("LOGINTER", 25)),
             dataset_size=a.dataset_size, vocab=a.vocab, hidden=a.hidden,
             layers=a.layers, lr=a.lr, dtype=a.dtype, device=a.device,
             max_len=a.max_len, ddp=ddp, kernel_backend=a.kernel_backend,
             data_dir=a.data_dir, optimizer=a.optimizer)

if __name__ == "__main__":
    main(ddp=False)
