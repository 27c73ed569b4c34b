#!/usr/bin/env bash
# Stream-ordering / race smoke: run the GPU suite
# with every kernel launch and copy serialized. A pass here means no result
# depends on asynchronous completion order (the ROCm stand-in for a
# compute-sanitizer race check); a numerics diff vs the normal run is an
# ordering bug in the engine's stream/event plumbing (wgrad side stream,
# colreduce batching, graph capture).
#
# Usage (on a GPU box):  bash tools/ordering_smoke.sh
set -euo pipefail
cd "$(dirname "$0")/.."
export AMD_SERIALIZE_KERNEL=3   # wait after each kernel launch
export AMD_SERIALIZE_COPY=3     # wait after each copy
export HIP_LAUNCH_BLOCKING=1
python -m pytest tests -m gpu -q -x
echo "ordering smoke: OK (suite green under full serialization)"
