"""Ring attention — context parallelism over point-to-point sends.

Completes the strategy matrix (parallel/strategy.py): where Ulysses
redistributes HEADS around attention (all-to-all, degree capped by
n_kv_heads), ring/CP keeps every head local and rotates K/V blocks around
the ring, merging partial attention with the online-softmax rules — so
context length scales with world size independent of head count.

Layout: every rank holds a CONTIGUOUS sequence chunk (rank r owns global
rows [r*S_loc, (r+1)*S_loc)). Causal contributions: a visiting K/V block
from rank `src` contributes fully when src < r, as the masked diagonal
when src == r, and not at all when src > r — contiguous chunks make the
per-rank causal work uneven (rank 0: one block, rank W-1: W blocks); the
zigzag interleaving that balances it is future work and does not change
the comm pattern.

The forward rotates K/V once around the ring (W-1 exchanges) keeping
running (m, l, O) per query row; the backward makes a second full
rotation carrying (K, V, dK-acc, dV-acc) so each block returns to its
owner with every rank's gradient contribution accumulated — dQ stays
local. Math runs in fp32 (reference-grade: this is the correctness-first
v1 of the strategy; a fused CDNA4 ring kernel plugs in at the
`_block_attn` seam). Works on gloo (CPU CI) and RCCL identically:
exchanges are isend + blocking recv like the PP stage sends.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class RingContext:
    group: object
    rank: int
    world: int

    @classmethod
    def from_group(cls, group=None) -> "RingContext":
        if not dist.is_initialized():
            raise RuntimeError("RingContext requires torch.distributed init")
        return cls(group=group, rank=dist.get_rank(group),
                   world=dist.get_world_size(group))

    def peer(self, r: int) -> int:
        return dist.get_global_rank(self.group, r) if self.group is not None \
            else r


def _rotate(ctx: RingContext, tensors):
    """Send `tensors` to rank+1, receive the same shapes from rank-1."""
    nxt = ctx.peer((ctx.rank + 1) % ctx.world)
    prv = ctx.peer((ctx.rank - 1) % ctx.world)
    out = []
    works = []
    sends = [t.contiguous() for t in tensors]
    for t in sends:
        works.append(dist.isend(t, dst=nxt, group=ctx.group))
    for t in sends:
        r = torch.empty_like(t)
        dist.recv(r, src=prv, group=ctx.group)
        out.append(r)
    for w in works:
        w.wait()
    return out


def _block_scores(q32, k32, scale, diag_mask):
    s = torch.einsum("bqhd,bkhd->bhqk", q32, k32) * scale
    if diag_mask:
        Sl = s.shape[-1]
        m = torch.ones(Sl, Sl, dtype=torch.bool, device=s.device).tril()
        s = s.masked_fill(~m, float("-inf"))
    return s


def _expand_kv(t, g):
    return t.repeat_interleave(g, dim=2) if g > 1 else t


class _RingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, ring: RingContext, causal: bool, scale: float):
        B, Sl, Hq, D = q.shape
        Hkv = k.shape[2]
        g = Hq // Hkv
        q32 = q.float()
        m = torch.full((B, Hq, Sl), float("-inf"), device=q.device)
        l = torch.zeros(B, Hq, Sl, device=q.device)
        o = torch.zeros(B, Hq, Sl, D, device=q.device)
        blk_k, blk_v = k, v
        for j in range(ring.world):
            src = (ring.rank - j) % ring.world
            if not causal or src <= ring.rank:
                k32 = _expand_kv(blk_k.float(), g)
                v32 = _expand_kv(blk_v.float(), g)
                s = _block_scores(q32, k32, scale,
                                  causal and src == ring.rank)
                bm = s.amax(-1)
                m_new = torch.maximum(m, bm)
                # rank r's own (diagonal) block processes first (j=0), so
                # every row's m is finite from the first iteration on;
                # the guards below only cover the j=0 transition itself
                p = torch.exp(s - m_new.unsqueeze(-1))
                p = torch.nan_to_num(p, nan=0.0)
                alpha = torch.where(torch.isinf(m), torch.zeros_like(m),
                                    torch.exp(m - m_new))
                l = l * alpha + p.sum(-1)
                o = o * alpha.unsqueeze(-1) + torch.einsum(
                    "bhqk,bkhd->bhqd", p, v32)
                m = m_new
            if j + 1 < ring.world:
                blk_k, blk_v = _rotate(ring, (blk_k, blk_v))
        lse = m + torch.log(l.clamp_min(1e-30))
        out = (o / l.clamp_min(1e-30).unsqueeze(-1))
        out = out.permute(0, 2, 1, 3).to(q.dtype)  # [B,Sl,Hq,D]
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.ring = ring
        ctx.meta = (causal, scale, g)
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        ring: RingContext = ctx.ring
        causal, scale, g = ctx.meta
        B, Sl, Hq, D = q.shape
        q32 = q.float()
        do32 = dout.float()
        delta = (do32 * out.float()).sum(-1).transpose(1, 2)  # [B,Hq,Sl]
        dq = torch.zeros_like(q32)
        blk_k, blk_v = k, v
        dk_acc = torch.zeros(k.shape, dtype=torch.float32, device=k.device)
        dv_acc = torch.zeros_like(dk_acc)
        for j in range(ring.world):
            src = (ring.rank - j) % ring.world
            if not causal or src <= ring.rank:
                k32 = _expand_kv(blk_k.float(), g)
                v32 = _expand_kv(blk_v.float(), g)
                s = _block_scores(q32, k32, scale,
                                  causal and src == ring.rank)
                p = torch.exp(s - lse.unsqueeze(-1))
                if causal and src == ring.rank:
                    p = torch.nan_to_num(p, nan=0.0)
                # dV (expanded heads) then fold the GQA group back
                dv_e = torch.einsum("bhqk,bqhd->bkhd", p, do32)
                dp = torch.einsum("bqhd,bkhd->bhqk", do32, v32)
                ds = p * (dp - delta.unsqueeze(-1)) * scale
                dq += torch.einsum("bhqk,bkhd->bqhd", ds, k32)
                dk_e = torch.einsum("bhqk,bqhd->bkhd", ds, q32)
                if g > 1:
                    Hkv = k.shape[2]
                    dv_e = dv_e.view(B, Sl, Hkv, g, D).sum(3)
                    dk_e = dk_e.view(B, Sl, Hkv, g, D).sum(3)
                dk_acc += dk_e
                dv_acc += dv_e
            # rotate the block AND its accumulators; after `world` steps
            # each block is back at its owner with all contributions
            blk_k, blk_v, dk_acc, dv_acc = _rotate(
                ring, (blk_k, blk_v, dk_acc, dv_acc))
        return (dq.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype),
                None, None, None)


def ring_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   ring: RingContext, causal: bool = True,
                   scale: float | None = None) -> torch.Tensor:
    """q [B,S_loc,Hq,D], k/v [B,S_loc,Hkv,D] (this rank's chunk) ->
    o [B,S_loc,Hq,D] as if attention ran over the full W*S_loc sequence."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    return _RingAttention.apply(q, k, v, ring, causal, float(scale))
