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


def test_driver_translation_dry_run(tmp_path):
    """GNMT via yml config — the reference's entry for this workload
    (runtime/translation/driver_configs/*.yml)."""
    cfg = tmp_path / "c.yml"
    cfg.write_text("benchmark: translation\nframework: pipedream\n"
                   "gpus: 2\n")
    out = subprocess.run(
        [sys.executable, "run/driver.py", str(cfg), "--dry-run"],
        capture_output=True, text=True, check=True)
    assert "gnmt_pipedream.py" in out.stdout
    assert "--nproc-per-node=2" in out.stdout
    # gpipe is not a GNMT mode
    cfg.write_text("benchmark: translation\nframework: gpipe\n")
    out = subprocess.run(
        [sys.executable, "run/driver.py", str(cfg), "--dry-run"],
        capture_output=True, text=True)
    assert out.returncode != 0


def test_driver_runs_gnmt_pipeline_config():
    out = subprocess.run(
        [sys.executable, "run/driver.py",
         "run/configs/gnmt_pipeline.yml"],
        capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
