#!/bin/bash
# Same-box A/B: torch TunableOp (rocBLAS/hipBLASLt GEMM solution sweep,
# enabled via the API -- see bench.py GA_TUNABLEOP) vs the default
# hipBLASLt heuristic, on the headline config.  --steps must be a
# multiple of accum(4) so auto window-fusion stays at the headline 4.
set -u
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
CSV=gpurun_out/tunableop.csv
B="python bench.py --gpus 1 --steps 160 --warmup 32"

echo "=== baseline A ==="
timeout 300 $B 2>gpurun_out/tun_a.err | tail -1

echo "=== tuning pass ==="
GA_TUNABLEOP=tune GA_TUNABLEOP_FILE=$CSV PYTORCH_TUNABLEOP_VERBOSE=1 \
  timeout 900 python bench.py --gpus 1 --steps 32 --warmup 32 \
  >gpurun_out/tun_tuning.log 2>&1
echo "tuning rc=$?; csv: $(wc -l < $CSV 2>/dev/null || echo MISSING) lines"

echo "=== tuned B (replay) ==="
GA_TUNABLEOP=replay GA_TUNABLEOP_FILE=$CSV \
  timeout 300 $B 2>gpurun_out/tun_b.err | tail -1

echo "=== baseline A2 ==="
timeout 300 $B 2>gpurun_out/tun_a2.err | tail -1

echo "=== tuned B2 ==="
GA_TUNABLEOP=replay GA_TUNABLEOP_FILE=$CSV \
  timeout 300 $B 2>gpurun_out/tun_b2.err | tail -1

echo "=== solutions chosen ==="
grep -i "gemm" $CSV 2>/dev/null | head -25
