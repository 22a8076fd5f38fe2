"""LocalTransport + 2-stage schedules on CPU (the CPU-suite mirror of
tests/test_pipeline_gpu.py — same harness, device=cpu)."""

import torch

import tests.test_pipeline_gpu as tg


def _cpu(monkeypatch):
    monkeypatch.setattr(tg, "_dev", lambda: torch.device("cpu"))
    # strip the gpu mark effect: call the functions directly


def test_local_two_stage_exact(monkeypatch):
    _cpu(monkeypatch)
    tg.test_1f1b_two_stages_one_gpu_exact()


def test_local_two_stage_pipelined(monkeypatch):
    _cpu(monkeypatch)
    tg.test_1f1b_pipelined_weight_versioning_one_gpu()


def test_local_gpipe_two_partitions(monkeypatch):
    _cpu(monkeypatch)
    tg.test_gpipe_two_stages_one_gpu()
