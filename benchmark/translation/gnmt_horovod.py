#!/usr/bin/env python3
"""GNMT translation benchmark, data-parallel (RCCL over xGMI);
launch under torchrun, one rank per GPU."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))
from gnmt_pytorch import main

if __name__ == "__main__":
    main(ddp=True)
