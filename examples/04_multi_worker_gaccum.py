"""Ablation 04: 2 workers x batch 50 x K=2 (effective 200) -- gradient
accumulation UNDER data parallelism.
Reference: distributedExample/04_multi_worker_with_estimator_gaccum.py.
Launch: torchrun --standalone --nproc-per-node 2 examples/04_multi_worker_gaccum.py"""
from mnist_common import run
from gradient_accumulation_tf_estimator_amd.data.input_fn import InputContext
from gradient_accumulation_tf_estimator_amd.parallel.launch import cleanup, init_distributed

if __name__ == "__main__":
    ctx = init_distributed()
    run("04_multi_worker_gaccum", batch_size=50, accum=2,
        input_context=InputContext(ctx.world_size, ctx.rank))
    cleanup()
