#!/bin/bash
# One-lease pre-tuning for the benchmark configs (run on an MI355X box):
#
#   bash tools/pretune.sh
#
# Produces, under gpurun_out/pretune/:
#   tunableop_gpt2.csv      full hipBLASLt GEMM tuning for GPT-2 small
#                           (fwd + bwd shapes) -> commit as
#                           profiles/tunableop_gpt2_gfx950.csv
#   miopen_udb/             MIOpen user find-db + kernel cache for the
#                           ResNet-50 b512 and b8192 conv shapes -> ship
#                           so fresh boxes skip the multi-minute
#                           benchmark-find/compile phase
#
# Budget: ~15-20 min total on one box. Each phase is independently
# resumable; re-running skips completed outputs.
set -ex
cd "$(dirname "$0")/.."
OUT=gpurun_out/pretune
mkdir -p "$OUT"

# ---- 1. TunableOp: GPT-2 GEMMs (fwd+bwd tuned during warmup) ----
if [ ! -f "$OUT/tunableop_gpt2.csv" ]; then
  PYTORCH_TUNABLEOP_ENABLED=1 \
  PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME="$OUT/tunableop_gpt2.csv" \
  timeout 600 python bench.py --model gpt2 --tunableop --steps 3 --warmup 3 || true
fi

# ---- 2. MIOpen find-db + kernel cache for the ResNet configs ----
export MIOPEN_USER_DB_PATH="$PWD/$OUT/miopen_udb"
export MIOPEN_CUSTOM_CACHE_DIR="$PWD/$OUT/miopen_udb/kcache"
mkdir -p "$MIOPEN_USER_DB_PATH" "$MIOPEN_CUSTOM_CACHE_DIR"
timeout 500 python bench.py --model resnet50 --batch-size 512 --steps 3 --warmup 2 || true
timeout 900 python bench.py --model resnet50 --batch-size 8192 --steps 2 --warmup 1 \
  --ckpt-layers layer1,layer2 --no-graph || true

ls -la "$OUT"
echo "pretune done: commit $OUT/tunableop_gpt2.csv to profiles/ and ship $OUT/miopen_udb"
