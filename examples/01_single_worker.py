"""Ablation 01: single worker, batch 200, no accumulation.
Reference: distributedExample/01_single_worker_with_estimator.py (B=200, K=1)."""
from mnist_common import run

if __name__ == "__main__":
    run("01_single_worker", batch_size=200, accum=1)
