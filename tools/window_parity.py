"""Window-graph loop vs eager micro-steps: bit-level state parity check."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
from gradient_accumulation_tf_estimator_amd import create_optimizer
from gradient_accumulation_tf_estimator_amd.engine.graphs import GraphedTrainLoop
from gradient_accumulation_tf_estimator_amd.models.bert import CONFIGS, BertForSequenceClassification
from gradient_accumulation_tf_estimator_amd.ops import fused as fops

fops.set_grouped_wgrad(True)
K, B, S, STEPS = 4, 8, 128, 8
cfg = CONFIGS["bert-small"]()

def build():
    torch.manual_seed(3)
    m = BertForSequenceClassification(cfg).to("cuda", torch.bfloat16)
    op = create_optimizer(m, 2e-5, 10**6, 100,
                          gradient_accumulation_multiplier=K, clip_norm=1.0,
                          backend="hip")
    return m, op

gen = torch.Generator().manual_seed(11)
ids = torch.randint(0, cfg.vocab_size, (STEPS, B, S), generator=gen).cuda()
lab = torch.randint(0, 2, (STEPS, B), generator=gen).cuda()

# eager arm
mA, opA = build()
lossesA = []
for i in range(STEPS):
    l = mA.loss(ids[i], lab[i])
    opA.step(l)
    lossesA.append(float(l.detach().float()))

# window arm
mB, opB = build()
slots = [ids[0].clone() for _ in range(K)]
labslots = [lab[0].clone() for _ in range(K)]
loop = GraphedTrainLoop(opB.engine, lambda k: mB.loss(slots[k], labslots[k]),
                        window=True)
lossesB = []
for i in range(STEPS):
    pos = i % K
    if pos == 0:
        for k in range(K - 1):
            slots[k].copy_(ids[i + k]); labslots[k].copy_(lab[i + k])
    elif pos == K - 1:
        slots[K - 1].copy_(ids[i]); labslots[K - 1].copy_(lab[i])
    l = loop.step()
    lossesB.append(float(l.detach().float()))

torch.cuda.synchronize()
import numpy as np
la, lb = np.array(lossesA), np.array(lossesB)
dm = (opA.engine.state.master - opB.engine.state.master).abs().max().item()
print("losses A:", np.round(la, 5))
print("losses B:", np.round(lb, 5))
print(f"max |master diff| = {dm:.3e}  max|loss diff| = {np.abs(la-lb).max():.3e}")
assert np.abs(la - lb).max() < 1e-4 and dm < 1e-6, "WINDOW PARITY FAIL"
print("window parity OK")
