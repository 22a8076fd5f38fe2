"""Result-log contract.

The reference's stdout format is parsed downstream, so we reproduce it
byte-compatibly (/root/reference/benchmark/mnist/mnist_pytorch.py:79-83,
94-97, 225-226):

  train | %d/%d epoch (%d%%) | %.3f samples/sec (estimated) | mem (GB): %.3f (%.3f) / %.3f
  %d/%d epoch | train loss:%.3f %.3f samples/sec | valid loss:%.3f accuracy:%.3f
  valid accuracy: %.4f | %.3f samples/sec, %.3f sec/epoch (average)
"""

from __future__ import annotations

import re
import sys
from typing import Optional


class BenchLogger:
    """Stdout logger reproducing the reference benchmark log format."""

    def __init__(self, rank: int = 0, stream=None) -> None:
        self.rank = rank
        self.stream = stream or sys.stdout

    def _emit(self, line: str) -> None:
        if self.rank == 0:
            print(line, file=self.stream, flush=True)

    def train_step(self, epoch: int, epochs: int, pct: int,
                   samples_per_sec: float,
                   mem_alloc_gb: float, mem_reserved_gb: float,
                   mem_total_gb: float) -> None:
        self._emit(
            "train | %d/%d epoch (%d%%) | %.3f samples/sec (estimated) | "
            "mem (GB): %.3f (%.3f) / %.3f"
            % (epoch, epochs, pct, samples_per_sec,
               mem_alloc_gb, mem_reserved_gb, mem_total_gb))

    def epoch(self, epoch: int, epochs: int, train_loss: float,
              samples_per_sec: float, valid_loss: float,
              valid_accuracy: float) -> None:
        self._emit(
            "%d/%d epoch | train loss:%.3f %.3f samples/sec | "
            "valid loss:%.3f accuracy:%.3f"
            % (epoch, epochs, train_loss, samples_per_sec,
               valid_loss, valid_accuracy))

    def final(self, valid_accuracy: float, samples_per_sec: float,
              sec_per_epoch: float) -> None:
        self._emit(
            "valid accuracy: %.4f | %.3f samples/sec, %.3f sec/epoch (average)"
            % (valid_accuracy, samples_per_sec, sec_per_epoch))

    def info(self, msg: str) -> None:
        self._emit(msg)


_FINAL_RE = re.compile(
    r"valid accuracy: ([\d.]+) \| ([\d.]+) samples/sec, ([\d.]+) sec/epoch")
_EPOCH_RE = re.compile(
    r"(\d+)/(\d+) epoch \| train loss:([-\d.]+) ([\d.]+) samples/sec \| "
    r"valid loss:([-\d.]+) accuracy:([\d.]+)")
_TRAIN_RE = re.compile(
    r"train \| (\d+)/(\d+) epoch \((\d+)%\) \| ([\d.]+) samples/sec")


def parse_result_line(line: str) -> Optional[dict]:
    """Parse one reference-format log line into a dict (or None).

    The counterpart of the reference's process_output.py — lets tests and
    the run harness recover metrics from stdout logs."""
    m = _FINAL_RE.search(line)
    if m:
        return {"kind": "final", "valid_accuracy": float(m.group(1)),
                "samples_per_sec": float(m.group(2)),
                "sec_per_epoch": float(m.group(3))}
    m = _EPOCH_RE.search(line)
    if m:
        return {"kind": "epoch", "epoch": int(m.group(1)),
                "epochs": int(m.group(2)), "train_loss": float(m.group(3)),
                "samples_per_sec": float(m.group(4)),
                "valid_loss": float(m.group(5)),
                "valid_accuracy": float(m.group(6))}
    m = _TRAIN_RE.search(line)
    if m:
        return {"kind": "train", "epoch": int(m.group(1)),
                "epochs": int(m.group(2)), "pct": int(m.group(3)),
                "samples_per_sec": float(m.group(4))}
    return None
