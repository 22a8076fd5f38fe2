"""run.sh front-end smoke (CPU)."""
import subprocess


def test_run_sh_pytorch_cpu(tmp_path):
    out = subprocess.run(
        ["bash", "run/run.sh", "-b", "mnist", "-f", "pytorch",
         "-m", "resnet18", "-e", "1", "-B", "32", "-p", "0"],
        capture_output=True, text=True, timeout=900,
        env={"PATH": "/usr/bin:/bin:/usr/local/bin",
             "DDLB_DEVICE": "cpu", "HOME": str(tmp_path)})
    # the entrypoint runs on whatever device exists; here CPU
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "valid accuracy:" in out.stdout


def test_run_sh_rejects_bad_framework():
    out = subprocess.run(
        ["bash", "run/run.sh", "-f", "nope"],
        capture_output=True, text=True, timeout=60)
    assert out.returncode != 0


def test_run_sh_horovod_two_ranks_cpu(tmp_path):
    """The torchrun launch path of the harness (2 gloo ranks on CPU)."""
    import os
    env = dict(os.environ)
    env.update(HOME=str(tmp_path), MASTER_PORT="29541")
    out = subprocess.run(
        ["bash", "run/run.sh", "-b", "mnist", "-f", "horovod", "-g", "2",
         "-m", "resnet18", "-e", "1", "-B", "16", "-p", "0"],
        capture_output=True, text=True, timeout=900, env=env)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-1000:]
    assert "valid accuracy:" in out.stdout


def test_run_sh_real_data_requires_datadir(tmp_path):
    """`-s` (reference real-data semantics) must fail fast without
    DATADIR."""
    out = subprocess.run(
        ["bash", "run/run.sh", "-b", "cifar10", "-f", "pytorch", "-s"],
        capture_output=True, text=True, timeout=60,
        env={"PATH": "/usr/bin:/bin:/usr/local/bin",
             "HOME": str(tmp_path)})
    assert out.returncode != 0
    assert "DATADIR" in (out.stdout + out.stderr)


def test_run_sh_real_data_trains_from_tree(tmp_path):
    """`run.sh -b cifar10 -s` trains from an on-disk class tree
    (VERDICT round-1 item 5 'Done' criterion)."""
    import os

    import numpy as np
    rng = np.random.default_rng(0)
    root = tmp_path / "data"
    for split, n in (("train", 12), ("val", 6)):
        for cls in ("a", "b"):
            d = root / split / cls
            d.mkdir(parents=True)
            for i in range(n):
                np.save(d / f"{i}.npy",
                        rng.integers(0, 255, (32, 32, 3),
                                     dtype=np.uint8))
    env = dict(os.environ)
    env.update(HOME=str(tmp_path), DATADIR=str(root))
    out = subprocess.run(
        ["bash", "run/run.sh", "-b", "cifar10", "-f", "pytorch",
         "-m", "resnet18", "-e", "1", "-B", "4", "-p", "0", "-s"],
        capture_output=True, text=True, timeout=900, env=env,
        cwd=".")
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "valid accuracy:" in out.stdout


def test_run_sh_gpipe_cpu(tmp_path):
    """gpipe front-end path (single process; 1 partition on CPU)."""
    import os
    env = dict(os.environ)
    env.update(HOME=str(tmp_path))
    out = subprocess.run(
        ["bash", "run/run.sh", "-b", "mnist", "-f", "gpipe",
         "-m", "resnet18", "-e", "1", "-B", "8", "-M", "2", "-p", "0"],
        capture_output=True, text=True, timeout=900, env=env)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "valid accuracy:" in out.stdout


def test_run_sh_pipedream_two_ranks_cpu(tmp_path):
    """pipedream front-end: profile -> partition (analysis printed) ->
    1F1B runtime, 2 gloo ranks on CPU."""
    import os
    env = dict(os.environ)
    env.update(HOME=str(tmp_path), MASTER_PORT="29551")
    out = subprocess.run(
        ["bash", "run/run.sh", "-b", "mnist", "-f", "pipedream",
         "-g", "2", "-m", "resnet18", "-e", "1", "-B", "8", "-M", "4",
         "-p", "0"],
        capture_output=True, text=True, timeout=900, env=env)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "valid accuracy:" in out.stdout
    assert "2 GPUs" in out.stdout  # the partition analysis line
