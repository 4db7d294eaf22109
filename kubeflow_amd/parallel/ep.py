"""Expert parallelism — experts sharded across GPUs, tokens exchanged by
collectives.

Completes SURVEY §2.14's strategy enum (after tp/pp/sp). Layout: each rank
owns n_experts/N experts and its own data-parallel token shard. Per MoE
block:

    xg = all_gather(x_local)          # every rank sees the global token set
    partial = sum over LOCAL experts of gated expert outputs on xg
    y_local = reduce_scatter_sum(partial)   # back to this rank's tokens

This "gather-compute-scatter" dataflow is exact (no capacity factor, no
dropped tokens) and maps onto two xGMI collectives per block; the classic
variable-size token all-to-all is a bandwidth optimization of the same
math, left for when expert counts grow (documented seam).

Gradient semantics (EpTrainer wiring in runtime/trainer.py):
  * expert parameters are rank-local; the reduce-scatter's backward
    all-gather already accumulates every rank's loss contribution into
    them, so they need only the 1/N data-parallel scaling;
  * everything else (attention, norms, gate, embeddings) is replicated and
    needs the usual DDP average across ranks.

Both collectives are their own adjoints' transposes:
  all_gather_cat:     forward concat-gather, backward reduce-scatter
  reduce_scatter_sum: forward reduce-scatter, backward all-gather
gloo (CPU CI) lacks reduce_scatter — fall back to all_reduce + slice.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any

import torch
import torch.distributed as dist


@dataclass
class EpContext:
    group: Any
    rank: int
    world: int

    @classmethod
    def from_group(cls, group=None) -> "EpContext":
        if not dist.is_initialized():
            raise RuntimeError("EpContext requires torch.distributed init")
        return cls(group=group, rank=dist.get_rank(group),
                   world=dist.get_world_size(group))

    # ------------------------------------------------------- trainer hooks
    def _split_slices(self, flat):
        if not hasattr(self, "_exp"):
            exp, rep = [], []
            for (off, n), p in zip(flat.slices, flat.params):
                (exp if getattr(p, "_ep_local", False) else rep).append(
                    (off, n))
            self._exp, self._rep = exp, rep
        return self._exp, self._rep

    def sync_grads(self, flat):
        """Data-parallel grad semantics: replicated params average across
        ranks; expert grads already hold every rank's loss contribution
        (the reduce-scatter backward all-gathers them), so they only get
        the 1/N scaling."""
        exp, rep = self._split_slices(flat)
        for off, n in rep:
            dist.all_reduce(flat.grad[off:off + n], group=self.group)
        for off, n in rep:
            flat.grad[off:off + n].div_(self.world)
        for off, n in exp:
            flat.grad[off:off + n].div_(self.world)

    def global_grad_norm(self, flat) -> torch.Tensor:
        exp, rep = self._split_slices(flat)

        def _sq(slices):
            t = torch.zeros((), dtype=torch.float32, device=flat.grad.device)
            for off, n in slices:
                t += torch.linalg.vector_norm(
                    flat.grad[off:off + n], dtype=torch.float32) ** 2
            return t

        s = _sq(exp)
        dist.all_reduce(s, group=self.group)
        return torch.sqrt(s + _sq(rep))

    def sync_replicated(self, model: torch.nn.Module):
        src = (dist.get_global_rank(self.group, 0)
               if self.group is not None else 0)
        for p in model.parameters():
            if not getattr(p, "_ep_local", False):
                dist.broadcast(p.data, src=src, group=self.group)


def _all_gather_cat(x: torch.Tensor, group) -> torch.Tensor:
    world = dist.get_world_size(group)
    chunks = [torch.empty_like(x) for _ in range(world)]
    dist.all_gather(chunks, x.contiguous(), group=group)
    return torch.cat(chunks, dim=0)


def _reduce_scatter_sum(xg: torch.Tensor, group) -> torch.Tensor:
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if dist.get_backend(group) == "nccl":
        out = torch.empty(xg.shape[0] // world, *xg.shape[1:],
                          dtype=xg.dtype, device=xg.device)
        dist.reduce_scatter_tensor(out, xg.contiguous(), group=group)
        return out
    xg = xg.contiguous()
    dist.all_reduce(xg, group=group)
    n = xg.shape[0] // world
    return xg[rank * n:(rank + 1) * n].clone()


class _AllGatherCat(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.ep_group = group
        return _all_gather_cat(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _reduce_scatter_sum(grad, ctx.ep_group), None


class _ReduceScatterSum(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xg, group):
        ctx.ep_group = group
        return _reduce_scatter_sum(xg, group)

    @staticmethod
    def backward(ctx, grad):
        return _all_gather_cat(grad, ctx.ep_group), None


def all_gather_cat(x: torch.Tensor, ctx: EpContext) -> torch.Tensor:
    return _AllGatherCat.apply(x, ctx.group) if ctx.world > 1 else x


def reduce_scatter_sum(xg: torch.Tensor, ctx: EpContext) -> torch.Tensor:
    return _ReduceScatterSum.apply(xg, ctx.group) if ctx.world > 1 else xg


def shard_expert_state_dict(full_sd: dict, n_experts: int, rank: int,
                            world: int) -> dict:
    """Slice a full MoE state dict's expert banks ([E, ...] leading dim)
    to this rank's contiguous expert block; everything else passes
    through (tests, checkpoint import)."""
    local = n_experts // world
    lo = rank * local
    out = {}
    for k, v in full_sd.items():
        if ".experts_" in k:
            out[k] = v[lo:lo + local].clone()
        else:
            out[k] = v.clone()
    return out
