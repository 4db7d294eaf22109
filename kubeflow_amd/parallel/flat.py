"""Flat parameter/gradient space — MI355X-first memory layout.

288 GB of HBM3E per GPU favors a few huge allocations over thousands of
small ones: every parameter of the model is re-homed as a view into ONE
contiguous bf16 buffer, gradients accumulate into a matching flat bf16
buffer, and optimizer state (fp32 master, m, v) lives in matching flat fp32
buffers. Consequences:

  * fused AdamW is ONE kernel launch over the whole model per step;
  * DDP buckets are contiguous ranges of the flat grad buffer — all-reduce
    needs no flatten/unflatten copies at all;
  * parameters are ordered in REVERSE registration order, which is
    (approximately) backward-completion order, so sequential buckets become
    ready sequentially during backward.

Replaces what the reference delegates to per-pod DDP wrappers (SURVEY.md
§2.14: DP/DDP is the one required v1 strategy).
"""
from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn

ALIGN = 64  # element alignment of each param slice (vector-width friendly)


def _aligned(n: int) -> int:
    return (n + ALIGN - 1) // ALIGN * ALIGN


class FlatParamSpace:
    def __init__(self, model: nn.Module, dtype: torch.dtype | None = None):
        params: List[Tuple[str, nn.Parameter]] = [
            (n, p) for n, p in model.named_parameters() if p.requires_grad]
        params.reverse()  # backward-completion order
        if dtype is None:
            dtype = params[0][1].dtype
        self.names = [n for n, _ in params]
        device = params[0][1].device
        total = sum(_aligned(p.numel()) for _, p in params)
        self.numel = total
        self.device = device
        self.data = torch.zeros(total, dtype=dtype, device=device)
        self.grad = torch.zeros(total, dtype=dtype, device=device)
        self.slices: List[Tuple[int, int]] = []
        off = 0
        for _, p in params:
            n = p.numel()
            self.data[off:off + n].copy_(p.data.reshape(-1).to(dtype))
            p.data = self.data[off:off + n].view(p.shape)
            p.grad = self.grad[off:off + n].view(p.shape)
            self.slices.append((off, n))
            off += _aligned(n)
        self.params = [p for _, p in params]

    def zero_grad(self):
        self.grad.zero_()

    def param_offset_end(self, i: int) -> int:
        """End offset (aligned) of param i in the flat space."""
        off, n = self.slices[i]
        return off + _aligned(n)

    def build_wd_mask(self, decay_dim_ge: int = 2) -> torch.Tensor:
        """fp32 {0,1} mask: weight decay only for params with dim >= 2
        (matrices); norms/biases are not decayed — matches common AdamW
        practice for transformer training."""
        mask = torch.zeros(self.numel, dtype=torch.float32, device=self.device)
        for (off, n), p in zip(self.slices, self.params):
            if p.dim() >= decay_dim_ge:
                mask[off:off + n].fill_(1.0)
        return mask

    def state_dict_tensors(self):
        """(name, view) pairs in registration order for checkpointing."""
        return list(zip(self.names, (self.data[o:o + n] for o, n in self.slices)))
