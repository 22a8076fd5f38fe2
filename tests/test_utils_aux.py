"""Aux subsystems: summary, act/grad logger, sparsity, process_output,
FusedAdam CPU semantics."""

import subprocess
import sys

import torch

from ddlbench_amd.models import build_model
from ddlbench_amd.ops.adam import FusedAdam, FusedAdamW
from ddlbench_amd.utils.actlogger import (ActivationGradientLogger,
                                          measure_sparsity)
from ddlbench_amd.utils.summary import format_summary, summarize


def test_summary_counts_all_params():
    m = build_model("mnist", "resnet18")
    rows = summarize(m, torch.randn(1, 1, 28, 28))
    total = sum(r["params"] for r in rows)
    assert total == sum(p.numel() for p in m.parameters())
    text = format_summary(rows)
    assert "Total params" in text


def test_actlogger_roundtrip(tmp_path):
    m = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.ReLU(),
                            torch.nn.Linear(8, 2))
    lg = ActivationGradientLogger(m, str(tmp_path), every_n_epochs=1)
    lg.start()
    m(torch.randn(3, 4)).sum().backward()
    lg.stop()
    path = lg.dump(1)
    lg.close()
    import pickle
    with open(path, "rb") as f:
        d = pickle.load(f)
    assert d["activations"] and d["gradients"]


def test_sparsity():
    t = torch.zeros(100)
    t[:25] = 1.0
    s = measure_sparsity(t)
    assert s["sparsity"] == 0.75
    assert s["compression_ratio"] == 2.0


def test_process_output_cli(tmp_path):
    log = tmp_path / "run.log"
    log.write_text(
        "train | 1/3 epoch (50%) | 100.000 samples/sec (estimated) | "
        "mem (GB): 1.000 (2.000) / 288.000\n"
        "1/3 epoch | train loss:2.300 100.000 samples/sec | "
        "valid loss:2.200 accuracy:0.100\n"
        "valid accuracy: 0.1000 | 100.000 samples/sec, 5.000 sec/epoch "
        "(average)\n")
    out = subprocess.run(
        [sys.executable, "run/process_output.py", str(log)],
        capture_output=True, text=True, check=True)
    assert '"samples_per_sec": 100.0' in out.stdout
    assert '"final"' in out.stdout


def test_fused_adam_matches_torch():
    torch.manual_seed(0)
    a = torch.nn.Linear(6, 6)
    b = torch.nn.Linear(6, 6)
    b.load_state_dict(a.state_dict())
    oa = FusedAdam(a.parameters(), lr=1e-2, weight_decay=1e-2,
                   backend="torch")
    ob = torch.optim.Adam(b.parameters(), lr=1e-2, weight_decay=1e-2)
    for _ in range(5):
        x = torch.randn(4, 6)
        for m, o in ((a, oa), (b, ob)):
            o.zero_grad()
            m(x).pow(2).sum().backward()
            o.step()
    for pa, pb in zip(a.parameters(), b.parameters()):
        torch.testing.assert_close(pa, pb, rtol=1e-5, atol=1e-6)


def test_fused_adamw_matches_torch():
    torch.manual_seed(0)
    a = torch.nn.Linear(6, 6)
    b = torch.nn.Linear(6, 6)
    b.load_state_dict(a.state_dict())
    oa = FusedAdamW(a.parameters(), lr=1e-2, weight_decay=1e-2,
                    backend="torch")
    ob = torch.optim.AdamW(b.parameters(), lr=1e-2, weight_decay=1e-2)
    for _ in range(5):
        x = torch.randn(4, 6)
        for m, o in ((a, oa), (b, ob)):
            o.zero_grad()
            m(x).pow(2).sum().backward()
            o.step()
    for pa, pb in zip(a.parameters(), b.parameters()):
        torch.testing.assert_close(pa, pb, rtol=1e-5, atol=1e-6)


def test_conv_mfma_module_cpu_fallback():
    """Conv2dMFMA uses the library conv on CPU (same math)."""
    import torch.nn as nn
    from ddlbench_amd.ops.conv import Conv2dMFMA, convert_convs, mfma_eligible
    torch.manual_seed(0)
    conv = nn.Conv2d(16, 32, 3, stride=1, padding=1, bias=False)
    assert mfma_eligible(conv)
    wrapped = Conv2dMFMA(conv)
    x = torch.randn(2, 16, 8, 8)
    torch.testing.assert_close(wrapped(x), conv(x))
    # converter counts + skips the C=3 stem
    model = nn.Sequential(nn.Conv2d(3, 16, 3, bias=False),
                          nn.Conv2d(16, 16, 3, bias=False))
    assert convert_convs(model) == 1
    assert isinstance(model[0], nn.Conv2d)
    assert isinstance(model[1], Conv2dMFMA)


def test_extract_reduce_times(tmp_path):
    sys.path.insert(0, "run")
    from process_output import extract_reduce_times
    log = tmp_path / "run.log"
    log.write_text(
        "train | 1/3 epoch (50%) | 100.000 samples/sec (estimated) | "
        "mem (GB): 1.000 (2.000) / 288.000\n"
        "reduce_times_ms: 1.250 0.750\n"
        "reduce_times_ms: 2.000\n")
    assert extract_reduce_times(str(log)) == [1.25, 0.75, 2.0]
