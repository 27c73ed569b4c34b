#!/bin/bash
# Round-end validation bundle: full GPU suite, driver-style bench, smoke,
# model sweep, steady-tail trace summary (writes under gpurun_out/).
set -u
cd "$(dirname "$0")/.."
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out

echo "=== pytest -m gpu ==="
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -3

echo "=== smoke ==="
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke(); print('SMOKE OK')" 2>&1 | tail -1

echo "=== driver-style bench (defaults) ==="
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 2>/dev/null | tail -1

echo "=== headline 300 steps ==="
timeout 300 python bench.py --gpus 1 --steps 300 --warmup 40 2>/dev/null | tail -1

echo "=== masked / dropout ==="
timeout 300 python bench.py --gpus 1 --steps 160 --warmup 32 --masked on 2>/dev/null | tail -1 | python -c "import json,sys; print('masked', json.load(sys.stdin)['value'])"
timeout 300 python bench.py --gpus 1 --steps 160 --warmup 32 --dropout 0.1 2>/dev/null | tail -1 | python -c "import json,sys; print('dropout', json.load(sys.stdin)['value'])"

echo "=== bert-base / bert-large seq512 ==="
timeout 300 python bench.py --gpus 1 --model bert-base --seq-len 512 --accum 16 --steps 48 --warmup 16 2>/dev/null | tail -1 | python -c "import json,sys; print('base512', json.load(sys.stdin)['value'])"
timeout 300 python bench.py --gpus 1 --model bert-large --seq-len 512 --accum 32 --steps 64 --warmup 16 2>/dev/null | tail -1 | python -c "import json,sys; print('large512', json.load(sys.stdin)['value'])"

echo "=== steady-tail trace ==="
(cd /tmp && export TMPDIR=/tmp && timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/tr2 -o tr2 -- python /root/repo/bench.py --gpus 1 --steps 60 --warmup 20 >/tmp/tr2.log 2>&1)
db=$(ls /tmp/tr2/*.db 2>/dev/null | head -1)
[ -n "$db" ] && python tools/trace_summary.py "$db" --tail 0.12 --windows 15 > gpurun_out/r02c_final_trace.md 2>&1 && tail -24 gpurun_out/r02c_final_trace.md
