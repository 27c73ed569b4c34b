"""Ablation 02: single worker, batch 100 x accumulation K=2 (effective 200).
Reference: distributedExample/02_single_worker_with_estimator_gaccum.py."""
from mnist_common import run

if __name__ == "__main__":
    run("02_single_worker_gaccum", batch_size=100, accum=2)
