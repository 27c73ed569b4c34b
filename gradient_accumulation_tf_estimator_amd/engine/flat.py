"""Flat parameter/gradient/optimizer-state layout for the accumulation engine.

The reference keeps one ``accum_grads`` fp32 buffer per trainable variable
(``/root/reference/optimization.py:78``) plus Adam ``adam_m``/``adam_v`` slots
(``optimization.py:137-148``). On MI355X we instead lay every per-parameter
buffer out as a *slice of one flat device tensor* so that the whole
accumulate / global-norm / AdamW-apply engine runs as a handful of
grid-stride HIP kernels over contiguous HBM instead of one launch per
variable (the reference's graph lowers to per-variable ``AssignAdd``/Adam
op chains -- see SURVEY.md section 2.3).

Layout decisions (MI355X-first):

* Parameters are ordered ``[weight-decay params..., no-decay params...]``.
  The fused AdamW kernel then needs only a single element-index boundary
  (``decay_boundary``) to decide whether to add ``wd * p`` -- no per-element
  mask bytes, no per-tensor dispatch table. The decay split reproduces the
  reference's regex exclusion (``optimization.py:179-187``).
* Every slice is padded to a multiple of ``ALIGN`` elements so each
  parameter slice starts 256-byte aligned -- float4/short8 vectorized loads
  in the HIP kernels never straddle a parameter boundary mid-vector.
  Padding elements are zero in every buffer and stay zero under the update
  (m=v=g=0 -> u=0; p=0 -> wd*p=0), so they are harmless in the global norm
  and in the fused apply.
"""

from __future__ import annotations

import re
from dataclasses import dataclass
from typing import Dict, List, Sequence, Tuple

import torch

# 64 elements = 256 B for fp32, 128 B for bf16: keeps every param slice
# aligned for the widest vector loads the kernels use (16 B/lane).
ALIGN = 64

# Reference exclusion list, optimization.py:65
DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY = ("LayerNorm", "layer_norm", "bias")


def _round_up(n: int, a: int = ALIGN) -> int:
    return (n + a - 1) // a * a


def use_weight_decay(name: str, exclude: Sequence[str]) -> bool:
    """Reference ``_do_use_weight_decay`` semantics (optimization.py:179-187):
    decay applies unless any exclusion regex matches the parameter name."""
    for r in exclude:
        if re.search(r, name) is not None:
            return False
    return True


@dataclass
class ParamSlice:
    name: str
    shape: torch.Size
    numel: int
    offset: int  # element offset into the flat buffers
    padded: int  # padded numel (multiple of ALIGN)
    decay: bool


@dataclass
class FlatLayout:
    slices: List[ParamSlice]
    total: int  # total padded elements
    decay_boundary: int  # elements < boundary get weight decay
    grad_lo: int = 0  # contiguous region whose grads flow through .grad
    grad_hi: int = 0

    @property
    def names(self) -> List[str]:
        return [s.name for s in self.slices]


def build_layout(
    named_params: Sequence[Tuple[str, torch.Tensor]],
    exclude_from_weight_decay: Sequence[str] = DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY,
    direct_names: Sequence[str] = (),
    pad_total_to: int = ALIGN,
) -> FlatLayout:
    """Order params and assign aligned flat offsets.

    Grouping: [decay-direct | decay-grad | nodecay-grad | nodecay-direct].
    ``direct_names`` are params whose grads bypass ``.grad`` (fused modules
    write them straight into the accum buffer, ops/fused.py), so the K1
    accumulate kernel only has to touch the contiguous middle
    ``[grad_lo, grad_hi)`` region; the decay boundary stays a single index.
    """
    direct = set(direct_names)
    decay_group = [(n, p) for n, p in named_params if use_weight_decay(n, exclude_from_weight_decay)]
    nodecay_group = [(n, p) for n, p in named_params if not use_weight_decay(n, exclude_from_weight_decay)]
    ordered = (
        [(n, p, True) for n, p in decay_group if n in direct]
        + [(n, p, True) for n, p in decay_group if n not in direct]
        + [(n, p, False) for n, p in nodecay_group if n not in direct]
        + [(n, p, False) for n, p in nodecay_group if n in direct]
    )
    slices: List[ParamSlice] = []
    off = 0
    decay_boundary = 0
    grad_lo = grad_hi = None
    for name, p, decay in ordered:
        n = p.numel()
        padded = _round_up(n)
        slices.append(ParamSlice(name, p.shape, n, off, padded, decay))
        if name not in direct:
            if grad_lo is None:
                grad_lo = off
            grad_hi = off + padded
        off += padded
        if decay:
            decay_boundary = off
    if grad_lo is None:
        grad_lo = grad_hi = 0
    # tail padding (zeros in every buffer, invariant under the update) so
    # the total divides into aligned equal shards for reduce-scatter DP
    off = _round_up(off, max(pad_total_to, ALIGN))
    lay = FlatLayout(slices=slices, total=off, decay_boundary=decay_boundary)
    lay.grad_lo, lay.grad_hi = grad_lo, grad_hi
    return lay


class FlatState:
    """Owns the flat buffers and the param/grad views into them.

    Buffers (all length ``layout.total``):
      * ``master``  fp32 -- master weights (the AdamW update target).
      * ``m``, ``v`` fp32 -- Adam slots (named ``adam_m``/``adam_v`` in the
        reference, optimization.py:137-148).
      * ``accum``   fp32 -- the accumulation buffer (``accum_grads``,
        optimization.py:78).
      * ``grads``   param-dtype -- flat gradient buffer; every ``p.grad`` is
        re-pointed to a view of it so autograd writes gradients contiguously.
      * ``model``   param-dtype -- flat model weights; every ``p.data`` is
        re-pointed to a view. When params are fp32 this IS ``master`` (no
        duplicate storage); when bf16, the fused apply kernel writes the
        down-cast copy.
    """

    def __init__(
        self,
        named_params: Sequence[Tuple[str, torch.Tensor]],
        exclude_from_weight_decay: Sequence[str] = DEFAULT_EXCLUDE_FROM_WEIGHT_DECAY,
        direct_names: Sequence[str] = (),
        pad_total_to: int = ALIGN,
    ):
        named_params = [(n, p) for n, p in named_params if p.requires_grad]
        if not named_params:
            raise ValueError("no trainable parameters")
        devices = {p.device for _, p in named_params}
        dtypes = {p.dtype for _, p in named_params}
        if len(devices) != 1:
            raise ValueError(f"all params must share a device, got {devices}")
        if len(dtypes) != 1:
            raise ValueError(f"all params must share a dtype, got {dtypes}")
        self.device = next(iter(devices))
        self.dtype = next(iter(dtypes))
        if self.dtype not in (torch.float32, torch.bfloat16):
            raise ValueError(f"unsupported param dtype {self.dtype}")

        self.layout = build_layout(named_params, exclude_from_weight_decay,
                                   direct_names, pad_total_to)
        N = self.layout.total
        dev = self.device

        self.model = torch.zeros(N, dtype=self.dtype, device=dev)
        if self.dtype == torch.float32:
            self.master = self.model  # shared storage: update writes params directly
        else:
            self.master = torch.zeros(N, dtype=torch.float32, device=dev)
        self.m = torch.zeros(N, dtype=torch.float32, device=dev)
        self.v = torch.zeros(N, dtype=torch.float32, device=dev)
        self.accum = torch.zeros(N, dtype=torch.float32, device=dev)
        self.grads = torch.zeros(N, dtype=self.dtype, device=dev)

        # Re-point params and grads into the flat buffers.
        self._params: List[torch.Tensor] = []
        by_name = dict(named_params)
        for s in self.layout.slices:
            p = by_name[s.name]
            view = self.model[s.offset : s.offset + s.numel].view(s.shape)
            with torch.no_grad():
                view.copy_(p.data)
            p.data = view
            p.grad = self.grads[s.offset : s.offset + s.numel].view(s.shape)
            if self.dtype != torch.float32:
                self.master[s.offset : s.offset + s.numel].view(s.shape).copy_(
                    p.data.to(torch.float32)
                )
            self._params.append(p)

        self._slice_by_param = {
            id(p): s for s, p in zip(self.layout.slices, (by_name[s.name] for s in self.layout.slices))
        }

    def accum_view(self, param: torch.Tensor) -> torch.Tensor:
        """Flat fp32 accumulation-buffer slice for a parameter -- the target
        for ops whose backward accumulates gradients directly (ops/fused.py)."""
        s = self._slice_by_param.get(id(param))
        if s is None:
            raise KeyError("parameter is not managed by this FlatState")
        return self.accum[s.offset : s.offset + s.numel]

    @property
    def decay_boundary(self) -> int:
        return self.layout.decay_boundary

    def state_dict(self) -> Dict:
        d = {
            "master": self.master,
            "m": self.m,
            "v": self.v,
            "accum": self.accum,
            "names": self.layout.names,
            "dtype": str(self.dtype),
        }
        if self.master is not self.model:
            d["model"] = self.model
        return d

    @staticmethod
    def _copy_flat(dst: torch.Tensor, src: torch.Tensor) -> None:
        """Copy tolerating different TAIL padding (a checkpoint written at a
        different DP shard-alignment pads to a different total; the pad is
        zeros by construction)."""
        n = min(dst.numel(), src.numel())
        dst[:n].copy_(src[:n])
        if src.numel() > n and float(src[n:].abs().sum()) != 0.0:
            raise ValueError("checkpoint longer than layout with non-zero tail")
        if dst.numel() > n:
            dst[n:].zero_()

    def load_state_dict(self, d: Dict) -> None:
        if d["names"] != self.layout.names:
            raise ValueError("checkpoint parameter layout does not match model")
        self._copy_flat(self.master, d["master"])
        self._copy_flat(self.m, d["m"])
        self._copy_flat(self.v, d["v"])
        self._copy_flat(self.accum, d["accum"])
        if self.master is not self.model:
            if "model" in d:
                self._copy_flat(self.model, d["model"])
            else:
                self.model.copy_(self.master.to(self.dtype))
