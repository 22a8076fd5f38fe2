"""YAML driver launcher."""
import subprocess
import sys


def test_driver_dry_run(tmp_path):
    cfg = tmp_path / "c.yml"
    cfg.write_text("benchmark: imagenet\nframework: horovod\ngpus: 4\n")
    out = subprocess.run(
        [sys.executable, "run/driver.py", str(cfg), "--dry-run"],
        capture_output=True, text=True, check=True)
    assert "--nproc-per-node=4" in out.stdout
    assert "imagenet_horovod.py" in out.stdout


def test_driver_rejects_bad_config(tmp_path):
    cfg = tmp_path / "c.yml"
    cfg.write_text("benchmark: nope\nframework: pytorch\n")
    out = subprocess.run(
        [sys.executable, "run/driver.py", str(cfg), "--dry-run"],
        capture_output=True, text=True)
    assert out.returncode != 0


def test_driver_runs_example():
    out = subprocess.run(
        [sys.executable, "run/driver.py", "run/configs/example.yml"],
        capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
