#!/bin/bash
# Extend the TunableOp table with the Llama-3-8B backward GEMM shapes, then
# verify e2e. Run via gpurun.
set -o pipefail
L=gpurun_out/tune_gemms.log
mkdir -p gpurun_out/tune
# seed with the existing (forward-shape) table so its entries are kept
cp kubeflow_amd/ops/tuned/tunableop0.csv gpurun_out/tune/tunableop0.csv
export PYTORCH_TUNABLEOP_ENABLED=1
export PYTORCH_TUNABLEOP_FILENAME=gpurun_out/tune/tunableop.csv
export PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=100
export PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS=30
{
echo "=== tuning pass (1 step, all fwd+bwd shapes) ==="
PYTORCH_TUNABLEOP_TUNING=1 timeout 480 python3 bench.py --gpus 1 --steps 1 --warmup 1 2>&1 | tail -4
echo "=== table ==="
wc -l gpurun_out/tune/tunableop0.csv kubeflow_amd/ops/tuned/tunableop0.csv
echo "=== verify with extended table ==="
PYTORCH_TUNABLEOP_TUNING=0 timeout 260 python3 bench.py --gpus 1 --steps 6 --warmup 2 2>&1 | tail -3
echo "=== baseline recheck (committed table) ==="
unset PYTORCH_TUNABLEOP_ENABLED PYTORCH_TUNABLEOP_TUNING PYTORCH_TUNABLEOP_FILENAME
timeout 260 python3 bench.py --gpus 1 --steps 6 --warmup 2 2>&1 | tail -3
echo "=== done ==="
} > "$L" 2>&1
tail -c 3000 "$L"
