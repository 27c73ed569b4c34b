"""In-tree build of the gfx950 HIP extension.

Shells out to ``setup.py build_ext --inplace`` at the repo root so the
compiled ``_ga_hip`` .so lands next to this package (NOT in a JIT cache) and
travels with repo snapshots to GPU boxes.

Usage: ``python -m gradient_accumulation_tf_estimator_amd.ops.build``
"""

from __future__ import annotations

import os
import pathlib
import subprocess
import sys

HERE = pathlib.Path(__file__).resolve().parent
REPO = HERE.parent.parent


def build(verbose: bool = False) -> None:
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    env.setdefault("MAX_JOBS", "8")
    cmd = [sys.executable, str(REPO / "setup.py"), "build_ext", "--inplace"]
    res = subprocess.run(cmd, cwd=str(REPO), env=env,
                         capture_output=not verbose, text=True)
    if res.returncode != 0:
        raise RuntimeError(
            f"HIP extension build failed (rc={res.returncode}):\n"
            f"{(res.stdout or '')[-4000:]}\n{(res.stderr or '')[-4000:]}"
        )


if __name__ == "__main__":
    build(verbose=True)
    print("built gradient_accumulation_tf_estimator_amd.ops._ga_hip")
