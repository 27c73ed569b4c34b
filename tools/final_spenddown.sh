#!/bin/bash
# Last GPU call of the round: suite + smoke + headline reps + one more
# same-box FFN A/B datapoint.
set -u
cd "$(dirname "$0")/.."
export HSA_ENABLE_IPC_MODE_LEGACY=0
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -1
timeout 200 python -c "import __graft_entry__; __graft_entry__.smoke(); print('SMOKE OK')" 2>&1 | tail -1
B="python bench.py --gpus 1 --steps 200 --warmup 40"
for i in 1 2 3; do
  timeout 300 $B 2>/dev/null | tail -1 | python -c "import json,sys; print('headline', json.load(sys.stdin)['value'])"
done
GA_CUSTOM_FFN=0 timeout 300 $B 2>/dev/null | tail -1 | python -c "import json,sys; print('ffn-off ', json.load(sys.stdin)['value'])"
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 2>/dev/null | tail -1 | python -c "import json,sys; print('driver-style', json.load(sys.stdin)['value'])"
