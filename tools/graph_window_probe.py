"""Measure per-replay overhead of the micro-step hipGraph: is capturing a
whole K-window (multiple micro-steps per graph) worth it?

Captures 1-, 2- and 4-micro-step accumulate graphs of the bench model and
compares per-step replay time. (The multi-step graphs reuse the same static
inputs -- timing only.)

MEASURED (MI355X): 1-step 712.8 us, 2-step 711.1, 4-step 709.9 per
micro-step -- replay overhead is ~1.5 us/step, so window-sized graphs are
not worth the input-plumbing complexity. Kept as the measurement record."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.models.bert import CONFIGS, BertForSequenceClassification
from gradient_accumulation_tf_estimator_amd.ops import fused as fops

fops.set_grouped_wgrad(True)
torch.manual_seed(0)
cfg = CONFIGS["bert-small"]()
model = BertForSequenceClassification(cfg).to("cuda", torch.bfloat16)
op = create_optimizer(model, 2e-5, 10**6, 100,
                      gradient_accumulation_multiplier=4, clip_norm=1.0,
                      backend="hip")
eng = op.engine
ids = torch.randint(0, cfg.vocab_size, (8, 128), device="cuda")
lab = torch.randint(0, 2, (8,), device="cuda")

def one_step():
    loss = model.loss(ids, lab)
    loss.backward()
    eng.accumulate()
    eng._join_wgrad_stream()

s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(s):
    for _ in range(3):
        one_step()
torch.cuda.current_stream().wait_stream(s)
torch.cuda.synchronize()

graphs = {}
for n in (1, 2, 4):
    g = torch.cuda.CUDAGraph()
    pool = graphs[1].pool() if 1 in graphs else None
    kw = {"pool": pool} if pool else {}
    with torch.cuda.graph(g, **kw):
        for _ in range(n):
            one_step()
    graphs[n] = g
torch.cuda.synchronize()

for n, g in graphs.items():
    g.replay(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    REP = 400 // n
    for _ in range(REP):
        g.replay()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / (REP * n) * 1e6
    print(f"{n}-step graph: {us:8.2f} us per micro-step")
