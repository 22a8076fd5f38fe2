#!/usr/bin/env python3
"""YAML-config experiment driver.

Parity with the reference's runtime/driver.py: read a
config describing the job, validate it, build per-worker command lines
with the rank math, launch one process per GPU, and record run metadata
(git hash, command) in the output directory. Single-node MI355X flavor:
workers are local processes over RCCL (no ssh/docker layer).

Config example (run/configs/example.yml):

    benchmark: imagenet        # mnist|cifar10|imagenet|highres
    framework: horovod         # pytorch|horovod|gpipe|pipedream
    model: resnet50
    gpus: 8
    epochs: 3
    batch_size: 256
    dtype: bfloat16
    log_interval: 25
    synthetic_scale: 0.01
    output_dir: out/driver
"""

import argparse
import datetime
import os
import subprocess
import sys

import yaml

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ("benchmark", "framework")
DEFAULTS = dict(model="resnet50", gpus=1, epochs=3, batch_size=0,
                dtype="float32", log_interval=25, synthetic_scale=0.01,
                output_dir="out/driver", master_port=29531,
                extra_args=[])
VALID_BM = ("mnist", "cifar10", "imagenet", "highres", "translation")
VALID_FW = ("pytorch", "horovod", "gpipe", "pipedream")


def load_config(path: str) -> dict:
    with open(path) as f:
        cfg = yaml.safe_load(f)
    for k in REQUIRED:
        if k not in cfg:
            raise ValueError(f"config missing required key {k!r}")
    if cfg["benchmark"] not in VALID_BM:
        raise ValueError(f"benchmark must be one of {VALID_BM}")
    if cfg["framework"] not in VALID_FW:
        raise ValueError(f"framework must be one of {VALID_FW}")
    if cfg["benchmark"] == "translation" and cfg["framework"] == "gpipe":
        # GNMT runs single / DP / 1F1B (the reference ran it only
        # through the pipedream driver)
        raise ValueError("translation supports pytorch|horovod|pipedream")
    out = dict(DEFAULTS)
    out.update(cfg)
    return out


def build_command(cfg: dict):
    ds = cfg["benchmark"]
    if ds == "translation":
        # GNMT: yml-config launch is the reference's entry for this
        # workload (runtime/translation/driver_configs/*.yml)
        script = os.path.join(ROOT, "benchmark", "translation",
                              f"gnmt_{cfg['framework']}.py")
        args = ["--dtype", cfg["dtype"]]
        args += list(map(str, cfg["extra_args"]))
        if cfg["framework"] == "pipedream" and cfg["gpus"] > 1 or \
                cfg["framework"] == "horovod" and cfg["gpus"] > 1:
            return [sys.executable, "-m", "torch.distributed.run",
                    "--nnodes=1", f"--nproc-per-node={cfg['gpus']}",
                    "--master-addr", "127.0.0.1",
                    "--master-port", str(cfg["master_port"]),
                    script] + args
        return [sys.executable, script] + args
    script_ds = "imagenet" if ds == "highres" else ds
    script = os.path.join(ROOT, "benchmark", script_ds,
                          f"{script_ds}_{cfg['framework']}.py")
    args = ["-a", cfg["model"], "--dtype", cfg["dtype"]]
    args += (["-s", "-1"] if ds == "highres"
             else ["-s", str(cfg["synthetic_scale"])])
    args += list(map(str, cfg["extra_args"]))
    if cfg["framework"] in ("horovod", "pipedream") and cfg["gpus"] > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run",
               "--nnodes=1", f"--nproc-per-node={cfg['gpus']}",
               "--master-addr", "127.0.0.1",
               "--master-port", str(cfg["master_port"]), script] + args
    else:
        cmd = [sys.executable, script] + args
    return cmd


def main():
    p = argparse.ArgumentParser()
    p.add_argument("config")
    p.add_argument("--dry-run", action="store_true")
    a = p.parse_args()
    cfg = load_config(a.config)
    cmd = build_command(cfg)

    stamp = datetime.datetime.now().strftime("%Y-%m-%dT%H.%M.%S")
    out_dir = os.path.join(ROOT, cfg["output_dir"], stamp)
    os.makedirs(out_dir, exist_ok=True)
    try:
        git = subprocess.run(["git", "rev-parse", "HEAD"], cwd=ROOT,
                             capture_output=True, text=True).stdout.strip()
    except OSError:
        git = "n/a"
    with open(os.path.join(out_dir, "info.txt"), "w") as f:
        f.write(f"config: {a.config}\n{yaml.safe_dump(cfg)}")
        f.write(f"git: {git}\ncommand: {' '.join(cmd)}\n")

    if a.dry_run:
        print(" ".join(cmd))
        return 0
    env = dict(os.environ,
               EPOCHS=str(cfg["epochs"]),
               LOGINTER=str(cfg["log_interval"]))
    if cfg["batch_size"]:
        env["BATCH_SIZE"] = str(cfg["batch_size"])
    log_path = os.path.join(out_dir, "run.log")
    with open(log_path, "w") as log:
        proc = subprocess.Popen(cmd, cwd=ROOT, env=env,
                                stdout=log, stderr=subprocess.STDOUT)
        rc = proc.wait()
    print(f"exit {rc}; log: {log_path}")
    return rc


if __name__ == "__main__":
    sys.exit(main())
