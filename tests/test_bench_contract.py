"""Driver contract rehearsal for bench.py on CPU.

The round-end driver runs `python bench.py --gpus N ...` (N>1 via
torch.distributed.run, one rank per GPU over RCCL) and parses ONE JSON
line from rank 0.  On a CPU-only box the same launch falls back to gloo,
so the full multi-rank path -- self-launch, torchrun rendezvous at
127.0.0.1, DP wiring, aggregation across ranks, single JSON line --
is rehearsable here exactly as the driver will drive it on the 8-GPU node.
"""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(ROOT, "bench.py")

REQUIRED = {
    "metric": "samples_per_sec_per_node",
    "unit": "samples/s",
    "higher_is_better": True,
    "scaling": "weak",
    "data": "synthetic",
}


def run_bench(extra):
    out = subprocess.run(
        [sys.executable, BENCH, "--steps", "4", "--warmup", "1",
         "--micro-batch", "2", "--seq-len", "64"] + extra,
        capture_output=True, text=True, timeout=540, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected ONE json line, got {lines!r}"
    return json.loads(lines[0])


def check_schema(j, n_gpus):
    for k, v in REQUIRED.items():
        assert j[k] == v, (k, j[k])
    assert j["n_gpus"] == n_gpus
    assert j["steps"] == 4 and j["warmup"] == 1
    assert j["value"] > 0 and j["ms_per_step"] > 0
    cfg = j["config"]
    assert cfg["model"] == "bert-small" and cfg["seq_len"] == 64
    assert cfg["parallelism"] == f"dp{n_gpus}"
    # whole-job aggregate: global_batch = micro_batch * accum * world
    assert cfg["global_batch"] == 2 * cfg["accum"] * n_gpus


def test_bench_single_rank_cpu():
    j = run_bench(["--gpus", "1"])
    check_schema(j, 1)


@pytest.mark.timeout(540)
def test_bench_self_launch_dp2_cpu():
    # exactly what `python bench.py --gpus 2` does on the driver's node:
    # re-exec through torch.distributed.run, 2 ranks, rank 0 prints
    j = run_bench(["--gpus", "2"])
    check_schema(j, 2)


@pytest.mark.timeout(540)
def test_bench_driver_style_torchrun_entry():
    # the driver's own N>1 invocation shape: torchrun wraps bench.py, so
    # WORLD_SIZE is already set and the self-launch branch must NOT re-exec
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29631", BENCH, "--gpus", "2", "--steps", "4",
         "--warmup", "1", "--micro-batch", "2", "--seq-len", "64"],
        capture_output=True, text=True, timeout=540, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1
    check_schema(json.loads(lines[0]), 2)
