#!/bin/bash
# Confirm/refute the ~+1% from GA_TUNABLEOP=replay-with-no-results
# (TunableOp enabled, tuning off, lookup always misses -> Default
# solution).  4x interleaved reps + per-kernel stats for both modes.
set -u
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
B="python bench.py --gpus 1 --steps 160 --warmup 32"
for i in 1 2 3 4; do
  echo "--- rep $i base ---"
  timeout 300 $B 2>/dev/null | tail -1 | python -c 'import json,sys; print("base", json.load(sys.stdin)["value"])'
  echo "--- rep $i tun ---"
  GA_TUNABLEOP=replay GA_TUNABLEOP_FILE=/nonexistent.csv \
  timeout 300 $B 2>/dev/null | tail -1 | python -c 'import json,sys; print("tun ", json.load(sys.stdin)["value"])'
done
export TMPDIR=/tmp
cd /tmp
timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/pa -o pa -- \
  python /root/repo/bench.py --gpus 1 --steps 60 --warmup 20 >/dev/null 2>&1
timeout 420 env GA_TUNABLEOP=replay GA_TUNABLEOP_FILE=/nonexistent.csv \
  rocprofv3 --kernel-trace --stats -d /tmp/pb -o pb -- \
  python /root/repo/bench.py --gpus 1 --steps 60 --warmup 20 >/dev/null 2>&1
for m in pa pb; do
  echo "=== $m GEMM stats ==="
  f=$(ls /tmp/$m/*kernel_stats.csv 2>/dev/null | head -1)
  [ -z "$f" ] && { ls /tmp/$m; continue; }
  grep -i "Cijk\|gemm" "$f" | head -8
  cp "$f" "/root/repo/gpurun_out/tunconfirm_${m}_stats.csv"
done
