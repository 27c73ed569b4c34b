"""Ablation 03: 2 data-parallel workers x batch 100, no accumulation.
Reference: distributedExample/03_multi_worker_with_estimator.py.
Launch: torchrun --standalone --nproc-per-node 2 examples/03_multi_worker.py"""
from mnist_common import run
from gradient_accumulation_tf_estimator_amd.data.input_fn import InputContext
from gradient_accumulation_tf_estimator_amd.parallel.launch import cleanup, init_distributed

if __name__ == "__main__":
    ctx = init_distributed()
    run("03_multi_worker", batch_size=100, accum=1,
        input_context=InputContext(ctx.world_size, ctx.rank))
    cleanup()
