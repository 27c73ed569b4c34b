"""Learning-rate schedule: linear (polynomial power=1) decay with linear warmup.

Reproduces /root/reference/optimization.py:29-54 exactly:

* decay:  ``lr = init_lr * (1 - step / num_train_steps)`` clamped at 0
  (``tf.train.polynomial_decay`` with power=1, end_learning_rate=0; the step
  is clipped to ``num_train_steps``).
* warmup: while ``step < num_warmup_steps``,
  ``lr = init_lr * step / num_warmup_steps`` (so lr == 0 at step 0), blended
  via the 0/1 ``is_warmup`` mask.

``step`` counts *micro*-steps: the reference increments ``global_step`` once
per session.run on both the accumulate and apply branches
(optimization.py:99-103, another-example.py:142,154), so the schedule moves
every micro-step, not every optimizer update. SURVEY.md section 2.2 item 5.
"""

from __future__ import annotations


def learning_rate(
    step: int,
    init_lr: float,
    num_train_steps: int,
    num_warmup_steps: int = 0,
) -> float:
    s = min(step, num_train_steps)
    lr = init_lr * (1.0 - s / float(num_train_steps)) if num_train_steps > 0 else init_lr
    if num_warmup_steps and step < num_warmup_steps:
        lr = init_lr * (float(step) / float(num_warmup_steps))
    return lr
