#!/usr/bin/env bash
# Multi-GPU pre-flight (the docs/MULTIGPU.md checklist, automated).
# CPU steps run anywhere; GPU steps run when a HIP device is visible.
# Intended before the first cold multi-GPU launch on a new node.
set -euo pipefail
cd "$(dirname "$0")/.."

echo "== [CPU] multi-process semantics on gloo (DDP, 1F1B, hybrid)"
python -m pytest tests/test_ddp_cpu.py tests/test_pipedream_cpu.py -q

echo "== [CPU] driver command rehearsal (8-rank torchrun, JSON contract)"
python -m pytest tests/test_bench_contract.py -q

if python -c 'import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)'; then
  echo "== [GPU] full GPU suite (native kernels, pipeline schedules)"
  python -m pytest tests/ -q -m gpu
  echo "== [GPU] single-GPU bench (one JSON line expected)"
  python bench.py --steps 5 --warmup 2
  N=$(python -c 'import torch; print(torch.cuda.device_count())')
  if [ "$N" -ge 2 ]; then
    echo "== [GPU] 2-rank RCCL rendezvous + bucket overlap"
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
      --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" \
      bench.py --gpus 2 --steps 10 --warmup 3
  else
    echo "== single GPU visible: multi-rank RCCL step skipped"
  fi
else
  echo "== no HIP device: GPU steps skipped (CPU checks passed)"
fi
echo "preflight OK"
