#!/bin/bash
# Round-1 final GPU validation sweep. Run via gpurun; log under gpurun_out/.
set -o pipefail
L=gpurun_out/final_sweep.log
mkdir -p gpurun_out
{
echo "=== pytest -m gpu ==="
timeout 360 python3 -m pytest tests/ -q -m gpu -p no:cacheprovider 2>&1 | tail -25
echo "=== smoke ==="
timeout 90 python3 -c 'import __graft_entry__; __graft_entry__.smoke(); print("SMOKE OK")' 2>&1 | tail -5
echo "=== attn_bench ==="
timeout 150 python3 tests/attn_bench.py 2>&1 | tail -30
echo "=== bench.py 1-GPU headline ==="
timeout 200 python3 bench.py --gpus 1 --steps 8 --warmup 3 2>&1 | tail -5
echo "=== serve_bench ==="
timeout 150 python3 tests/serve_bench.py llama3-8b 16 64 128 2>&1 | tail -10
echo "=== done ==="
} > "$L" 2>&1
tail -c 4000 "$L"
