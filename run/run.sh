#!/usr/bin/env bash
# DDLBench-compatible front-end (reference: /root/reference/run/run/run.sh).
#
# Flags (same letters as the reference):
#   -b benchmark   mnist | cifar10 | imagenet | highres      (default mnist)
#   -f framework   pytorch | horovod | gpipe | pipedream     (default pytorch)
#                  (horovod = our RCCL data-parallel path; the names keep
#                   the reference CLI contract)
#   -g gpus        GPUs (= ranks) on this node                (default 1)
#   -n nodes       node count — single-node launcher; >1 errors out
#   -p interval    log interval in batches                    (default 25)
#   -m model       resnet18/34/50/101/152 | vgg11/13/16/19 | mobilenetv2
#   -s             REAL data from $DATADIR (reference semantics:
#                  run/run/run.sh:39-41); synthetic stream otherwise
#   -e epochs      epochs                                      (default 3)
#   -B batch       per-GPU batch size (dataset default if unset)
#   -M micro       micro-batch count for pipeline paths
#   -d dtype       float32 | bfloat16                          (default float32)
#
# Creates out/<timestamp>/ with info.txt and the run log, like the
# reference's out/<date>/ (run.sh:78-96). Launch is torchrun-based —
# no SLURM, one process per GPU over RCCL.
set -euo pipefail

BENCH=mnist; FRAMEWORK=pytorch; GPUS=1; NODES=1; LOGINTER=25
MODEL=""; SCALE="0.01"; EPOCHS=3; BATCH=""; MICRO=""; DTYPE=float32
REALDATA=0

while getopts "b:f:g:n:p:m:se:B:M:d:h" opt; do
  case $opt in
    b) BENCH=$OPTARG;;
    f) FRAMEWORK=$OPTARG;;
    g) GPUS=$OPTARG;;
    n) NODES=$OPTARG;;
    p) LOGINTER=$OPTARG;;
    m) MODEL=$OPTARG;;
    s) REALDATA=1;;
    e) EPOCHS=$OPTARG;;
    B) BATCH=$OPTARG;;
    M) MICRO=$OPTARG;;
    d) DTYPE=$OPTARG;;
    h) grep '^#' "$0" | head -25; exit 0;;
    *) exit 1;;
  esac
done

if [ "$NODES" != "1" ]; then
  echo "multi-node launch is not wired in this harness (single node, up to 8 GPUs)" >&2
  exit 1
fi

case $BENCH in
  mnist|cifar10|imagenet|highres) ;;
  *) echo "unknown benchmark $BENCH" >&2; exit 1;;
esac
case $FRAMEWORK in
  pytorch|horovod|gpipe|pipedream) ;;
  *) echo "unknown framework $FRAMEWORK" >&2; exit 1;;
esac

ROOT="$(cd "$(dirname "$0")/.." && pwd)"
DS=$BENCH
EXTRA=()
if [ "$BENCH" = highres ]; then DS=imagenet; EXTRA+=(-s -1); else EXTRA+=(-s "$SCALE"); fi
if [ "$REALDATA" = "1" ]; then
  : "${DATADIR:?-s (real data) needs DATADIR}"
  EXTRA+=(--real-data)
fi
if [ -z "$MODEL" ]; then
  case $DS in imagenet) MODEL=resnet50;; *) MODEL=resnet18;; esac
fi

STAMP=$(date +%Y-%m-%dT%H.%M.%S)
OUT="$ROOT/out/$STAMP"
mkdir -p "$OUT"
{
  echo "benchmark: $BENCH"
  echo "framework: $FRAMEWORK"
  echo "model: $MODEL"
  echo "gpus: $GPUS  nodes: $NODES"
  echo "epochs: $EPOCHS  log interval: $LOGINTER"
  echo "dtype: $DTYPE"
  echo "git: $(git -C "$ROOT" rev-parse --short HEAD 2>/dev/null || echo n/a)"
  echo "date: $STAMP"
} > "$OUT/info.txt"

export EPOCHS LOGINTER
[ -n "$BATCH" ] && export BATCH_SIZE=$BATCH
[ -n "$MICRO" ] && export MICROBATCHES=$MICRO

SCRIPT="$ROOT/benchmark/$DS/${DS}_${FRAMEWORK}.py"
ARGS=(-a "$MODEL" --dtype "$DTYPE" "${EXTRA[@]}")

echo "== $BENCH / $FRAMEWORK / $MODEL on $GPUS GPU(s) -> $OUT"
case $FRAMEWORK in
  pytorch|gpipe)
    python "$SCRIPT" "${ARGS[@]}" 2>&1 | tee "$OUT/run.log";;
  horovod|pipedream)
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$GPUS" \
      --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" \
      "$SCRIPT" "${ARGS[@]}" 2>&1 | tee "$OUT/run.log";;
esac
