#!/usr/bin/env python3
"""cifar10 x pytorch benchmark entrypoint.

MI355X-native counterpart of the reference's benchmark/cifar10/cifar10_pytorch.py
(env contract: DATADIR/EPOCHS/BATCH_SIZE/LOGINTER/CORES_GPU[/MICROBATCHES];
flags: -a/--arch, -s/--synthetic_data, --lr, --momentum). Strategy: single.
"""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
from ddlbench_amd.cli import main

if __name__ == "__main__":
    main("cifar10", "single")
