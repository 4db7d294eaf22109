"""Tensor parallelism over xGMI — Megatron-style head/ffn sharding.

SURVEY.md §2.14 scopes DDP as the only strategy any BASELINE config
requires; TP is the first widening of that inventory. Design for MI355X:
xGMI is point-to-point (7 links x ~153 GB/s per GPU), so TP's two
all-reduces per block (after attention-out and after mlp-down, each
B*S*hidden bf16) ride the same ring path as DDP buckets — TP degree should
stay within one xGMI clique, which on this node is all 8 GPUs.

Sharding (LlamaBlock):
  * wqkv: column-parallel by heads — each rank owns n_heads/N q heads and
    n_kv_heads/N kv heads (GQA group size is preserved, so the flash
    kernels run unchanged on local heads).
  * wo: row-parallel over the local heads' columns; partial output is
    all-reduced, then the residual is added (the addmm residual fusion of
    the non-TP path would add the residual N times).
  * w13 (gate++up): column-parallel over ffn, keeping the gate|up halves
    aligned so ops.swiglu sees a [.., 2*ffn_local] tensor.
  * w2: row-parallel over ffn.
  * embeddings / lm_head / norms: replicated. With identical inputs on
    every TP rank, their gradients are bitwise-identical in exact
    arithmetic (the copy_to backward all-reduce restores the full dx
    before it reaches them); `sync_replicated` exists for init and for
    periodic re-sync against nondeterministic-atomics drift on GPU.

The conjugate autograd pair (Megatron f/g):
  copy_to:     forward identity, backward all-reduce  (enters a
               column-parallel region)
  reduce_from: forward all-reduce, backward identity  (leaves a
               row-parallel region)
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, List, Optional, Tuple

import torch
import torch.distributed as dist


@dataclass
class TpContext:
    group: Any
    rank: int
    world: int
    _shard_slices: Optional[List[Tuple[int, int]]] = field(
        default=None, repr=False)
    _repl_slices: Optional[List[Tuple[int, int]]] = field(
        default=None, repr=False)

    @classmethod
    def from_group(cls, group=None) -> "TpContext":
        if not dist.is_initialized():
            raise RuntimeError("TpContext requires torch.distributed init")
        return cls(group=group, rank=dist.get_rank(group),
                   world=dist.get_world_size(group))

    # ------------------------------------------------------------ grad norm
    def global_grad_norm(self, flat) -> torch.Tensor:
        """Global grad norm over a FlatParamSpace whose sharded params carry
        `_tp_sharded`: sum sharded squares across ranks, count replicated
        params once (they hold identical grads on every rank)."""
        if self._shard_slices is None:
            sh, rp = [], []
            for (off, n), p in zip(flat.slices, flat.params):
                (sh if getattr(p, "_tp_sharded", False) else rp).append(
                    (off, n))
            self._shard_slices, self._repl_slices = sh, rp

        def _sq(slices):
            t = torch.zeros((), dtype=torch.float32, device=flat.grad.device)
            for off, n in slices:
                t += torch.linalg.vector_norm(
                    flat.grad[off:off + n], dtype=torch.float32) ** 2
            return t

        s = _sq(self._shard_slices)
        dist.all_reduce(s, group=self.group)
        return torch.sqrt(s + _sq(self._repl_slices))

    # ----------------------------------------------------------- init sync
    def sync_replicated(self, model: torch.nn.Module):
        """Broadcast non-sharded parameters from TP rank 0 so replicated
        state starts identical (per-rank seeds keep shard inits distinct)."""
        src = (dist.get_global_rank(self.group, 0)
               if self.group is not None else 0)
        for p in model.parameters():
            if not getattr(p, "_tp_sharded", False):
                dist.broadcast(p.data, src=src, group=self.group)


class _CopyToTp(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.tp_group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        grad = grad.contiguous()
        dist.all_reduce(grad, group=ctx.tp_group)
        return grad, None


class _ReduceFromTp(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


def copy_to(x: torch.Tensor, ctx: TpContext) -> torch.Tensor:
    return _CopyToTp.apply(x, ctx.group) if ctx.world > 1 else x


def reduce_from(x: torch.Tensor, ctx: TpContext) -> torch.Tensor:
    return _ReduceFromTp.apply(x, ctx.group) if ctx.world > 1 else x


# -------------------------------------------------------------- sharding
def shard_rows(w: torch.Tensor, rank: int, world: int) -> torch.Tensor:
    n = w.shape[0] // world
    return w[rank * n:(rank + 1) * n]


def shard_cols(w: torch.Tensor, rank: int, world: int) -> torch.Tensor:
    n = w.shape[1] // world
    return w[:, rank * n:(rank + 1) * n]


def shard_qkv_rows(w: torch.Tensor, cfg, rank: int, world: int):
    """Slice a fused [ (Hq+2*Hkv)*D, hidden ] qkv weight to this rank's
    heads, keeping the q|k|v section layout."""
    d = cfg.head_dim
    q, k, v = torch.split(
        w, [cfg.n_heads * d, cfg.n_kv_heads * d, cfg.n_kv_heads * d], dim=0)
    return torch.cat([shard_rows(q, rank, world),
                      shard_rows(k, rank, world),
                      shard_rows(v, rank, world)], dim=0)


def shard_gate_up_rows(w: torch.Tensor, cfg, rank: int, world: int):
    """Slice a fused [ 2*ffn, hidden ] gate++up weight keeping halves
    aligned for ops.swiglu."""
    g, u = torch.split(w, [cfg.ffn_dim, cfg.ffn_dim], dim=0)
    return torch.cat([shard_rows(g, rank, world),
                      shard_rows(u, rank, world)], dim=0)


def shard_llama_state_dict(full_sd: dict, cfg, rank: int, world: int) -> dict:
    """Map a full LlamaModel state dict to the TP-local one (tests,
    checkpoint import). Replicated tensors are passed through."""
    out = {}
    for k, w in full_sd.items():
        if k.endswith("wqkv.weight"):
            out[k] = shard_qkv_rows(w, cfg, rank, world).clone()
        elif k.endswith("wo.weight"):
            d = cfg.head_dim
            q_cols = cfg.n_heads * d // world
            out[k] = w[:, rank * q_cols:(rank + 1) * q_cols].clone()
        elif k.endswith("w13.weight"):
            out[k] = shard_gate_up_rows(w, cfg, rank, world).clone()
        elif k.endswith("w2.weight"):
            out[k] = shard_cols(w, rank, world).clone()
        else:
            out[k] = w.clone()
    return out
