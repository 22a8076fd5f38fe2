"""bench.py driver contract (CPU plumbing mode)."""
import json
import subprocess
import sys

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _run(*extra):
    out = subprocess.run(
        [sys.executable, "bench.py", "--device", "cpu", "--steps", "2",
         "--warmup", "1", "--batch", "4", *extra],
        capture_output=True, text=True, timeout=900, check=True)
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line)


def test_ddp_contract_fields():
    d = _run()
    assert REQUIRED.issubset(d)
    assert d["metric"] == "images/sec"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["config"]["model"] == "resnet50"
    assert d["config"]["global_batch"] == 4
    assert d["value"] > 0 and d["ms_per_step"] > 0


def test_gpipe_strategy_contract():
    d = _run("--strategy", "gpipe", "--microbatches", "2")
    assert d["config"]["parallelism"].startswith("gpipe")
    assert d["value"] > 0


def test_dataset_flag():
    d = _run("--dataset", "cifar10", "--model", "mobilenetv2")
    assert d["config"]["input"] == "3x32x32"


def test_torchrun_8rank_driver_command(tmp_path):
    """The exact driver launch shape: torch.distributed.run --nnodes=1
    --nproc-per-node 8 ... bench.py --gpus 8 (CPU/gloo plumbing) must
    produce ONE valid whole-job JSON line from rank 0
    (VERDICT round-1 item 3: cold-start readiness for the first
    8-GPU run)."""
    import os
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
         "--master-port", "29617", "bench.py", "--gpus", "8",
         "--device", "cpu", "--steps", "2", "--warmup", "1",
         "--batch", "2", "--model", "resnet18", "--dataset", "mnist"],
        capture_output=True, text=True, timeout=900, check=True, env=env)
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"exactly one JSON line, got: {out.stdout}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 8
    assert d["config"]["parallelism"] == "dp8"
    assert d["value"] > 0
