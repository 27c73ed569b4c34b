#!/bin/bash
# Same-box A/B: PYTORCH_TUNABLEOP (rocBLAS/hipBLASLt solution sweep) vs
# the default hipBLASLt heuristic, on the headline bench config.
# Tuning happens during the EAGER capture-warmup iterations (every GEMM
# shape runs eagerly before hipGraph capture), so the captured graph
# records the tuned solution choices.
set -u
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
CSV=gpurun_out/tunableop_results.csv
B="python bench.py --gpus 1 --steps 150 --warmup 30"

echo "=== baseline A ==="
timeout 300 $B 2>gpurun_out/tun_a.err | tail -1

echo "=== tuning pass (writes $CSV) ==="
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
PYTORCH_TUNABLEOP_FILENAME=$CSV PYTORCH_TUNABLEOP_VERBOSE=1 \
  timeout 600 python bench.py --gpus 1 --steps 30 --warmup 30 \
  >gpurun_out/tun_tuning.log 2>&1
echo "tuning rc=$? ; csv lines: $(wc -l < ${CSV}0 2>/dev/null || wc -l < $CSV 2>/dev/null || echo none)"
ls -la gpurun_out/tunableop* 2>/dev/null

# torch appends an instance suffix; normalize
TCSV=$(ls gpurun_out/tunableop_results* 2>/dev/null | head -1)
echo "=== tuned B (read-only: $TCSV) ==="
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=0 \
PYTORCH_TUNABLEOP_FILENAME=$CSV \
  timeout 300 $B 2>gpurun_out/tun_b.err | tail -1

echo "=== baseline A2 ==="
timeout 300 $B 2>gpurun_out/tun_a2.err | tail -1

echo "=== tuned B2 ==="
PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=0 \
PYTORCH_TUNABLEOP_FILENAME=$CSV \
  timeout 300 $B 2>gpurun_out/tun_b2.err | tail -1

echo "=== tuned kernels chosen ==="
grep -m 20 "GemmTunableOp" "$TCSV" 2>/dev/null || head -30 "$TCSV" 2>/dev/null
