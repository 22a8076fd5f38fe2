"""Environment-variable + CLI configuration contract.

Replicates the reference interface (env vars consumed by every benchmark
entrypoint — /root/reference run/run/run_template.sh:70-73 and
benchmark/mnist/mnist_pytorch.py:31-35) while adding explicit dataclass
plumbing so the engines do not read the environment themselves.

Env contract (all optional, with the reference's defaults):
  DATADIR       — dataset root (only used with real data; synthetic is default)
  EPOCHS        — epochs per run (reference default 3)
  BATCH_SIZE    — per-GPU batch size (micro-batch size for the gpipe path)
  LOGINTER      — log every N batches (reference default 25)
  CORES_GPU     — CPU threads per GPU process
  MICROBATCHES  — number of micro-batches (pipeline paths)
"""

from __future__ import annotations

import argparse
import dataclasses
import os

# Dataset shape table — mirrors the reference's synthetic generator
# (/root/reference/benchmark/generate_synthetic_data.py:76-107).
DATASET_SHAPES = {
    #            C   H    W   classes  train_n  test_n
    "mnist":    (1,  28,  28,  10,     60_000,  10_000),
    "cifar10":  (3,  32,  32,  10,     50_000,  10_000),
    "imagenet": (3, 224, 224, 1000, 1_281_167,  50_000),
    "highres":  (3, 512, 512, 1000,    50_000,  10_000),
}

# Reference per-dataset default batch sizes (run_template.sh:186-266).
DEFAULT_BATCH = {"mnist": 128, "cifar10": 64, "imagenet": 32, "highres": 32}
DEFAULT_MICROBATCHES = {"mnist": 24, "cifar10": 32, "imagenet": 12, "highres": 12}


def _env_int(name: str, default: int) -> int:
    v = os.environ.get(name, "")
    try:
        return int(v)
    except ValueError:
        return default


@dataclasses.dataclass
class BenchConfig:
    dataset: str = "mnist"
    arch: str = "resnet18"
    strategy: str = "single"          # single | ddp | gpipe | pipedream
    epochs: int = 3
    batch_size: int = 0               # 0 → dataset default
    log_interval: int = 25
    cores_per_gpu: int = 0            # 0 → leave torch defaults
    microbatches: int = 0             # 0 → dataset default
    synthetic: bool = True
    synthetic_scale: float = 1.0      # fraction of the full dataset size to use
    data_dir: str = ""
    lr: float = 0.01
    lr_schedule: str = "constant"     # constant | step30 | warmup (utils/lr.py)
    warmup_epochs: int = 5            # warmup schedule ramp length
    momentum: float = 0.9
    weight_decay: float = 0.0
    dtype: str = "float32"            # compute dtype: float32 | bfloat16
    channels_last: bool = False
    device: str = "auto"              # auto | cuda | cpu
    seed: int = 42
    num_workers: int = 2
    kernel_backend: str = "auto"      # auto | native | torch (ops dispatch)
    checkpoint_dir: str = ""          # per-stage checkpoints when set
    resume: bool = False              # load stage checkpoints at start
    no_input_pipelining: bool = False  # pipedream: pure model parallel
    straight_pipeline: bool = False    # pipedream: no stage replication

    def __post_init__(self) -> None:
        if self.dataset not in DATASET_SHAPES:
            raise ValueError(f"unknown dataset {self.dataset!r}")
        if self.batch_size <= 0:
            self.batch_size = DEFAULT_BATCH[self.dataset]
        if self.microbatches <= 0:
            self.microbatches = DEFAULT_MICROBATCHES[self.dataset]

    @property
    def shape(self):
        c, h, w, ncls, ntrain, ntest = DATASET_SHAPES[self.dataset]
        return c, h, w

    @property
    def num_classes(self) -> int:
        return DATASET_SHAPES[self.dataset][3]

    @property
    def train_size(self) -> int:
        return max(1, int(DATASET_SHAPES[self.dataset][4] * self.synthetic_scale))

    @property
    def test_size(self) -> int:
        return max(1, int(DATASET_SHAPES[self.dataset][5] * self.synthetic_scale))

    @classmethod
    def from_env(cls, dataset: str, strategy: str, **overrides) -> "BenchConfig":
        """Build a config honouring the reference env-var contract."""
        kw = dict(
            dataset=dataset,
            strategy=strategy,
            epochs=_env_int("EPOCHS", 3),
            batch_size=_env_int("BATCH_SIZE", 0),
            log_interval=_env_int("LOGINTER", 25),
            cores_per_gpu=_env_int("CORES_GPU", 0),
            microbatches=_env_int("MICROBATCHES", 0),
            data_dir=os.environ.get("DATADIR", ""),
        )
        kw.update(overrides)
        return cls(**kw)


def make_parser(default_arch: str = "resnet18") -> argparse.ArgumentParser:
    """argparse surface shared by every benchmark entrypoint.

    Mirrors the reference's flags (-a/--arch, -s/--synthetic_data, --lr,
    --momentum — mnist_pytorch.py:147-161) plus dtype/backend knobs that
    are MI355X-specific.
    """
    p = argparse.ArgumentParser()
    p.add_argument("-a", "--arch", default=default_arch)
    p.add_argument("-s", "--synthetic-data", "--synthetic_data", dest="synthetic_scale",
                   type=float, default=0.01,
                   help="fraction of the full dataset size to synthesize "
                        "(reference uses full-size synthetic trees; keep small "
                        "for smoke runs). -1 selects the highres variant where "
                        "supported.")
    p.add_argument("--real-data", action="store_true",
                   help="train from the class-per-directory tree under "
                        "$DATADIR (reference `run.sh -s` semantics) "
                        "instead of the synthetic stream")
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--lr-schedule", default="constant",
                   choices=["constant", "step30", "warmup"],
                   help="step30 = imagenet /30-epoch decay; warmup = "
                        "horovod 5-epoch ramp to lr*N then 30/60/80 steps")
    p.add_argument("--warmup-epochs", type=int, default=5)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--weight-decay", type=float, default=0.0)
    p.add_argument("--dtype", default="float32", choices=["float32", "bfloat16"])
    p.add_argument("--channels-last", action="store_true")
    p.add_argument("--device", default="auto")
    p.add_argument("--kernel-backend", default="auto",
                   choices=["auto", "native", "torch"])
    p.add_argument("--seed", type=int, default=42)
    return p


def config_from_args(dataset: str, strategy: str, args: argparse.Namespace,
                     **overrides) -> BenchConfig:
    highres = getattr(args, "synthetic_scale", 1.0) == -1
    kw = dict(
        arch=args.arch,
        synthetic=not getattr(args, "real_data", False),
        synthetic_scale=(1.0 if highres else max(args.synthetic_scale, 0.0) or 1.0),
        lr=args.lr,
        lr_schedule=getattr(args, "lr_schedule", "constant"),
        warmup_epochs=getattr(args, "warmup_epochs", 5),
        momentum=args.momentum,
        weight_decay=args.weight_decay,
        dtype=args.dtype,
        channels_last=args.channels_last,
        device=args.device,
        kernel_backend=args.kernel_backend,
        seed=args.seed,
    )
    kw.update(overrides)
    return BenchConfig.from_env("highres" if highres else dataset, strategy, **kw)
