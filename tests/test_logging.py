"""Log contract: byte-format parity with the reference + parser round-trip."""

import io

from ddlbench_amd.utils import BenchLogger, parse_result_line


def _capture(fn):
    buf = io.StringIO()
    log = BenchLogger(rank=0, stream=buf)
    fn(log)
    return buf.getvalue()


def test_train_line_format():
    out = _capture(lambda l: l.train_step(2, 3, 57, 1234.5678, 1.5, 2.25, 288.0))
    assert out == ("train | 2/3 epoch (57%) | 1234.568 samples/sec (estimated)"
                   " | mem (GB): 1.500 (2.250) / 288.000\n")
    d = parse_result_line(out)
    assert d == {"kind": "train", "epoch": 2, "epochs": 3, "pct": 57,
                 "samples_per_sec": 1234.568}


def test_epoch_line_format():
    out = _capture(lambda l: l.epoch(1, 3, 2.3456, 999.9, 2.1, 0.4512))
    assert out == ("1/3 epoch | train loss:2.346 999.900 samples/sec | "
                   "valid loss:2.100 accuracy:0.451\n")
    d = parse_result_line(out)
    assert d["kind"] == "epoch" and d["valid_accuracy"] == 0.451


def test_final_line_format():
    out = _capture(lambda l: l.final(0.9312, 4567.8, 12.345))
    assert out == ("valid accuracy: 0.9312 | 4567.800 samples/sec, "
                   "12.345 sec/epoch (average)\n")
    d = parse_result_line(out)
    assert d == {"kind": "final", "valid_accuracy": 0.9312,
                 "samples_per_sec": 4567.8, "sec_per_epoch": 12.345}


def test_nonzero_rank_is_silent():
    buf = io.StringIO()
    BenchLogger(rank=1, stream=buf).final(1.0, 1.0, 1.0)
    assert buf.getvalue() == ""


def test_parse_garbage_returns_none():
    assert parse_result_line("hello world") is None
