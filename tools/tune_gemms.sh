#!/bin/bash
# Offline hipBLASLt/rocBLAS GEMM tuning for the flagship bench shapes.
# Run ON A GPU BOX; results land in tunableop/tunableop_results*.csv and
# are committed so runtime loads them read-only (torch_on_k8s_amd/tunable.py).
#
#   /usr/local/graft/bin/gpurun --timeout 1800 -- 'bash tools/tune_gemms.sh'
set -ex
cd "$(dirname "$0")/.."
export PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=${TUNE_MS:-120}
# mbs8 (bench headline) and mbs4 (grad-accum/elastic configs)
timeout 900 python bench.py --gpus 1 --steps 2 --warmup 1
timeout 600 python bench.py --gpus 1 --steps 1 --warmup 1 --micro-batch 4
mkdir -p gpurun_out/tunableop
cp tunableop/tunableop_results*.csv gpurun_out/tunableop/
