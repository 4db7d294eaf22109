"""Ulysses sequence parallelism — all-to-all head-scatter around attention.

Third widening of SURVEY.md §2.14's inventory (after tp.py / pp.py).
Long-context fit for MI355X: activations scale with S, so sharding the
sequence across N GPUs divides per-GPU activation memory by N while every
parameter stays replicated. Attention is the one position-mixing op, so it
is bracketed by two all-to-alls over xGMI:

    x [B, S/N, h] --qkv proj--> [B, S/N, (Hq+2Hkv)d]
      --a2a (scatter heads, gather seq)--> [B, S, (Hq+2Hkv)d / N]
      --rope + flash attention (full sequence, local heads)-->
      [B, S, Hq*d/N]
      --a2a (gather heads, scatter seq)--> [B, S/N, Hq*d]
      --wo + the rest of the block, position-wise on the shard-->

Because parameters are replicated and each rank's loss is the mean over
its own positions, gradient sync is exactly DDP's average — the standard
Trainer + BucketedDDP path needs no change; only the model's attention
call does. An all-to-all is its own adjoint, so backward is the reverse
exchange (the autograd.Functions below).

Constraints: S % N == 0, n_heads % N == 0, n_kv_heads % N == 0.
gloo (CPU CI) lacks all_to_all — the exchange falls back to isend/irecv
pairs; RCCL uses the native all-to-all.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any, List

import torch
import torch.distributed as dist


@dataclass
class SpContext:
    group: Any
    rank: int
    world: int

    @classmethod
    def from_group(cls, group=None) -> "SpContext":
        if not dist.is_initialized():
            raise RuntimeError("SpContext requires torch.distributed init")
        return cls(group=group, rank=dist.get_rank(group),
                   world=dist.get_world_size(group))


def _exchange(chunks: List[torch.Tensor], group) -> List[torch.Tensor]:
    """all_to_all chunk i -> rank i; returns the received chunks (all equal
    shapes). Falls back to isend/irecv for backends without all_to_all."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    # empty() not empty_like(): chunks are slices (non-contiguous) and the
    # receive buffers must be contiguous for the collective
    out = [torch.empty(c.shape, dtype=c.dtype, device=c.device)
           for c in chunks]
    backend = dist.get_backend(group)
    if backend == "nccl":
        dist.all_to_all(out, [c.contiguous() for c in chunks], group=group)
        return out
    # gloo fallback: self-copy + paired isend/irecv
    out[rank].copy_(chunks[rank])
    reqs = []
    for peer in range(world):
        if peer == rank:
            continue
        g = (dist.get_global_rank(group, peer)
             if group is not None else peer)
        reqs.append(dist.isend(chunks[peer].contiguous(), dst=g, group=group))
        reqs.append(dist.irecv(out[peer], src=g, group=group))
    for r in reqs:
        r.wait()
    return out


def _split_qkv_heads(qkv: torch.Tensor, hq: int, hkv: int, d: int,
                     world: int) -> List[torch.Tensor]:
    """[.., (hq+2hkv)d] -> per-destination [.., (hq+2hkv)d/world] chunks,
    keeping each destination's q|k|v section layout."""
    q, k, v = torch.split(qkv, [hq * d, hkv * d, hkv * d], dim=-1)
    ql = q.chunk(world, dim=-1)
    kl = k.chunk(world, dim=-1)
    vl = v.chunk(world, dim=-1)
    return [torch.cat([ql[j], kl[j], vl[j]], dim=-1) for j in range(world)]


def _merge_qkv_heads(chunks: List[torch.Tensor], hq: int, hkv: int, d: int,
                     world: int) -> torch.Tensor:
    """Inverse of _split_qkv_heads (same seq length per chunk)."""
    hql, hkvl = hq // world, hkv // world
    qs, ks, vs = [], [], []
    for c in chunks:
        q, k, v = torch.split(c, [hql * d, hkvl * d, hkvl * d], dim=-1)
        qs.append(q)
        ks.append(k)
        vs.append(v)
    return torch.cat(qs + ks + vs, dim=-1)


class _ScatterHeadsGatherSeq(torch.autograd.Function):
    """[B, S/N, (Hq+2Hkv)d] -> [B, S, (Hq+2Hkv)d/N]"""

    @staticmethod
    def forward(ctx, qkv, sp, hq, hkv, d):
        ctx.sp, ctx.dims = sp, (hq, hkv, d)
        send = _split_qkv_heads(qkv, hq, hkv, d, sp.world)
        recv = _exchange(send, sp.group)
        return torch.cat(recv, dim=1)  # seq chunks in rank order

    @staticmethod
    def backward(ctx, grad):
        sp = ctx.sp
        hq, hkv, d = ctx.dims
        send = list(grad.chunk(sp.world, dim=1))
        recv = _exchange(send, sp.group)
        return (_merge_qkv_heads(recv, hq, hkv, d, sp.world),
                None, None, None, None)


class _GatherHeadsScatterSeq(torch.autograd.Function):
    """[B, S, Hq*d/N] -> [B, S/N, Hq*d]"""

    @staticmethod
    def forward(ctx, o, sp):
        ctx.sp = sp
        send = list(o.chunk(sp.world, dim=1))
        recv = _exchange(send, sp.group)
        return torch.cat(recv, dim=-1)  # head chunks in rank order

    @staticmethod
    def backward(ctx, grad):
        sp = ctx.sp
        send = list(grad.chunk(sp.world, dim=-1))
        recv = _exchange(send, sp.group)
        return torch.cat(recv, dim=1), None


def scatter_heads_gather_seq(qkv, sp: SpContext, hq: int, hkv: int, d: int):
    return _ScatterHeadsGatherSeq.apply(qkv, sp, hq, hkv, d)


def gather_heads_scatter_seq(o, sp: SpContext):
    return _GatherHeadsScatterSeq.apply(o, sp)
