"""Per-stage checkpoint / resume.

Reference contract (/root/reference/pipedream-fork/runtime/
main_with_runtime.py:393-403, 580-584): rank_in_stage 0 of every stage
writes checkpoint.<stage>.pth.tar after each epoch with {epoch, arch,
state_dict, optimizer state, best metric}; resume loads the
stage-matched file. Same file naming here; single-device strategies use
stage 0."""

from __future__ import annotations

import os
from typing import Optional

import torch


def checkpoint_path(ckpt_dir: str, stage: int) -> str:
    return os.path.join(ckpt_dir, f"checkpoint.{stage}.pth.tar")


def save_stage_checkpoint(ckpt_dir: str, stage: int, epoch: int, arch: str,
                          module: torch.nn.Module,
                          optimizer=None, best_metric: float = 0.0) -> str:
    os.makedirs(ckpt_dir, exist_ok=True)
    state = {
        "epoch": epoch,
        "arch": arch,
        "stage": stage,
        "state_dict": module.state_dict(),
        # reference key name (main_with_runtime.py:393-403) so
        # checkpoints stay cross-readable with reference tooling
        "best_prec1": best_metric,
    }
    if optimizer is not None:
        inner = getattr(optimizer, "inner", optimizer)
        state["optimizer"] = inner.state_dict()
    path = checkpoint_path(ckpt_dir, stage)
    tmp = path + ".tmp"
    torch.save(state, tmp)
    os.replace(tmp, path)  # atomic: never a torn checkpoint
    return path


def load_stage_checkpoint(ckpt_dir: str, stage: int,
                          module: torch.nn.Module,
                          optimizer=None,
                          map_location="cpu") -> Optional[dict]:
    path = checkpoint_path(ckpt_dir, stage)
    if not os.path.exists(path):
        return None
    state = torch.load(path, map_location=map_location, weights_only=False)
    if "best_prec1" not in state:  # round-1 checkpoints used best_metric
        state["best_prec1"] = state.get("best_metric", 0.0)
    module.load_state_dict(state["state_dict"])
    if optimizer is not None and "optimizer" in state:
        inner = getattr(optimizer, "inner", optimizer)
        inner.load_state_dict(state["optimizer"])
    return state
