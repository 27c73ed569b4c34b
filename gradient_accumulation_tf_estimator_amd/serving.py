"""hipGraph-captured batched inference (the serving-side counterpart of
engine/graphs.py).

The reference's deployment story ends at ``estimator.predict``
(another-example.py:385-388); on MI355X an eager predict step at serving
batch sizes is launch-bound the same way the training micro-step was, so
the same remedy applies: capture the forward once per (batch, seq) shape
and replay per request with a single device-side input copy.

    pred = GraphedPredictor(model, example_batch)   # capture
    logits = pred(ids_batch)                        # replay

Shapes must match the capture (pad the tail batch); falls back to plain
eager forward on CPU / for non-bf16 models.
"""

from __future__ import annotations

from typing import Callable, Optional

import torch


class GraphedPredictor:
    """Captures ``fwd_fn(static_inputs) -> output`` into one hipGraph.

    ``example`` fixes the input shape/dtype/device; every call copies the
    request into the static buffer and replays. ``fwd_fn`` defaults to
    ``model.__call__`` (for :class:`BertForSequenceClassification` that is
    the logits path).
    """

    def __init__(self, model: torch.nn.Module, example: torch.Tensor,
                 fwd_fn: Optional[Callable] = None, warmup_iters: int = 3,
                 example_mask: Optional[torch.Tensor] = None):
        """``example_mask`` ([B,S] 1/0) captures the key-padding-mask path
        (real serving batches are padded); requests then pass ``mask=``."""
        self.model = model
        base = fwd_fn or model
        if example_mask is not None:
            self.fwd = lambda x, m: base(x, attention_mask=m)
        else:
            self.fwd = lambda x, m: base(x)
        self.graphed = example.is_cuda
        model.eval()
        self._static_in = example.clone()
        self._static_mask = None if example_mask is None else example_mask.clone()
        if not self.graphed:
            self._graph = None
            return
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(warmup_iters):
                self.fwd(self._static_in, self._static_mask)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph), torch.no_grad():
            self._static_out = self.fwd(self._static_in, self._static_mask)
        torch.cuda.synchronize()

    @torch.no_grad()
    def __call__(self, inputs: torch.Tensor,
                 mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        if (mask is None) != (self._static_mask is None):
            raise ValueError("predictor was captured "
                             + ("WITH" if self._static_mask is not None else "WITHOUT")
                             + " a mask; pass requests the same way")
        if self._graph is None:
            return self.fwd(inputs, mask)
        if inputs.shape != self._static_in.shape:
            raise ValueError(
                f"GraphedPredictor captured shape {tuple(self._static_in.shape)}"
                f", got {tuple(inputs.shape)} -- pad the batch or capture a "
                "second predictor for this shape")
        self._static_in.copy_(inputs)
        if mask is not None:
            self._static_mask.copy_(mask)
        self._graph.replay()
        return self._static_out
