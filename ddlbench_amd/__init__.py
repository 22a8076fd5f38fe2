"""ddlbench_amd — MI355X-native distributed deep-learning training benchmark.

A from-scratch framework with the capabilities of sara-nl/DDLBench
(see SURVEY.md in the repo root for the blueprint), re-designed for
AMD Instinct MI355X (gfx950, CDNA4):

* single-GPU PyTorch-ROCm baseline (``strategy="single"``)
* data-parallel training whose gradient all-reduce is RCCL over xGMI,
  with bucket fusion sized for the 8-GPU point-to-point mesh
  (``ddlbench_amd.parallel.ddp``)
* synchronous micro-batch pipeline (GPipe-style, single process,
  multi-device, HIP streams — ``ddlbench_amd.parallel.pipeline.gpipe``)
* asynchronous 1F1B pipeline (PipeDream-style profiler → partitioner →
  runtime with weight stashing and RCCL p2p activation transport —
  ``ddlbench_amd.parallel.pipeline``)

The compute hot path is hand-written CDNA4 HIP kernels in
``ddlbench_amd.ops`` (fused BN+ReLU, depthwise conv, NHWC MFMA conv GEMM,
fused SGD step, cross-entropy) — no CUDA shims, no hipify.
"""

__version__ = "0.1.0"

from ddlbench_amd.config import BenchConfig  # noqa: F401
