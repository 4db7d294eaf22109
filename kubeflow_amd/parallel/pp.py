"""Pipeline parallelism — GPipe fill-drain over point-to-point sends.

Second widening of SURVEY.md §2.14's strategy inventory (after tp.py).
MI355X fit: stage boundaries cross exactly one xGMI hop when ranks are
placed contiguously (the gang scheduler's xGMI-contiguous allocation), and
each boundary moves only [B/M, S, hidden] bf16 activations per microbatch —
tiny next to DDP's 16 GB grad traffic, so PP is the low-bandwidth way to
span models past one GPU's 288 GB.

v1 schedule: fill-drain (all microbatch forwards, then all backwards in
reverse). Mathematically identical to single-process gradient accumulation
over the same microbatches — that equivalence is the correctness oracle in
tests/test_pp_gloo.py. 1F1B (less activation memory) is a drop-in schedule
swap later; the stage/boundary machinery here does not change.

Stage r of N owns layers [r*L/N, (r+1)*L/N); rank 0 adds the embedding,
rank N-1 adds final norm + lm_head + loss. All parameters are stage-local
(nothing replicated), so the global grad norm is one all-reduce of squared
stage norms.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any, List, Optional

import torch
import torch.distributed as dist


@dataclass
class PpContext:
    group: Any
    rank: int
    world: int

    @classmethod
    def from_group(cls, group=None) -> "PpContext":
        if not dist.is_initialized():
            raise RuntimeError("PpContext requires torch.distributed init")
        return cls(group=group, rank=dist.get_rank(group),
                   world=dist.get_world_size(group))

    @property
    def is_first(self) -> bool:
        return self.rank == 0

    @property
    def is_last(self) -> bool:
        return self.rank == self.world - 1

    def _peer(self, r: int) -> int:
        return (dist.get_global_rank(self.group, r)
                if self.group is not None else r)

    def send(self, t: torch.Tensor, to_rank: int):
        dist.send(t.contiguous(), dst=self._peer(to_rank), group=self.group)

    def isend(self, t: torch.Tensor, to_rank: int):
        """Non-blocking send; returns (work, tensor) — the tensor must stay
        referenced until work.wait()."""
        t = t.contiguous()
        return (dist.isend(t, dst=self._peer(to_rank), group=self.group), t)

    def recv(self, shape, dtype, device, from_rank: int) -> torch.Tensor:
        t = torch.empty(shape, dtype=dtype, device=device)
        dist.recv(t, src=self._peer(from_rank), group=self.group)
        return t

    def global_grad_norm(self, flat) -> torch.Tensor:
        s = torch.linalg.vector_norm(flat.grad, dtype=torch.float32) ** 2
        dist.all_reduce(s, group=self.group)
        return torch.sqrt(s)

    def broadcast_scalar(self, value: Optional[torch.Tensor],
                         src_rank: int, device) -> torch.Tensor:
        t = (value.detach().to(torch.float32).reshape(1)
             if value is not None
             else torch.zeros(1, dtype=torch.float32, device=device))
        dist.broadcast(t, src=self._peer(src_rank), group=self.group)
        return t[0]


def layer_range(n_layers: int, stage: int, n_stages: int):
    """Contiguous, balanced layer slice for a stage (first stages take the
    remainder — they also host the embedding, so trailing-heavy would be
    worse)."""
    base, rem = divmod(n_layers, n_stages)
    start = stage * base + min(stage, rem)
    return start, start + base + (1 if stage < rem else 0)


class PipelineRunner:
    """Runs one optimizer-step's worth of microbatches through a stage.

    The stage module contract (LlamaStage implements it):
      * first stage:  forward(tokens)            -> activations [b,S,h]
      * middle stage: forward(x)                 -> activations
      * last stage:   forward(x, targets=...)    -> scalar loss
    """

    def __init__(self, stage: torch.nn.Module, ctx: PpContext,
                 microbatches: int, act_dtype: torch.dtype,
                 hidden_size: int, schedule: str = "1f1b"):
        if microbatches < 1:
            raise ValueError("microbatches must be >= 1")
        if schedule not in ("1f1b", "gpipe"):
            raise ValueError(f"unknown pp schedule {schedule!r}")
        self.stage = stage
        self.ctx = ctx
        self.m = microbatches
        self.act_dtype = act_dtype
        self.h = hidden_size
        self.schedule = schedule

    def step(self, tokens: torch.Tensor, targets: torch.Tensor):
        """Forward+backward all microbatches (grads accumulate into the
        stage's params, pre-divided by M like Trainer's grad_accum).
        Returns the mean microbatch loss, broadcast to every rank.

        Schedules (identical math, different memory):
          * gpipe: all forwards, then all backwards — M activations live.
          * 1f1b: warmup forwards then fwd/bwd interleave — at most
            (stages - rank) activations live, so M can grow freely.
        """
        ctx = self.ctx
        B, S = tokens.shape
        if B % self.m:
            raise ValueError(f"batch {B} not divisible by microbatches {self.m}")
        b = B // self.m
        tok_micro = tokens.split(b)
        tgt_micro = targets.split(b)
        device = next(self.stage.parameters()).device

        from collections import deque
        outstanding = deque()  # (x_in, y) FIFO — backward oldest first
        pending = []           # in-flight isend (work, tensor) pairs
        losses = []

        def fwd(i):
            if ctx.is_first:
                x_in = None
                y = self.stage(tok_micro[i].to(device))
            else:
                x_in = ctx.recv((b, S, self.h), self.act_dtype, device,
                                ctx.rank - 1).requires_grad_()
                y = (self.stage(x_in, targets=tgt_micro[i].to(device))
                     if ctx.is_last else self.stage(x_in))
            if ctx.is_last:
                losses.append(y)  # y is the loss
            else:
                pending.append(ctx.isend(y.detach(), ctx.rank + 1))
            outstanding.append((x_in, y))

        def bwd():
            x_in, y = outstanding.popleft()
            if ctx.is_last:
                (y / self.m).backward()
            else:
                dy = ctx.recv((b, S, self.h), self.act_dtype, device,
                              ctx.rank + 1)
                y.backward(dy)
            if not ctx.is_first:
                pending.append(ctx.isend(x_in.grad, ctx.rank - 1))

        if self.schedule == "gpipe":
            for i in range(self.m):
                fwd(i)
            for _ in range(self.m):
                bwd()
        else:  # 1f1b
            warm = min(self.m, ctx.world - 1 - ctx.rank)
            for i in range(warm):
                fwd(i)
            for i in range(warm, self.m):
                fwd(i)
                bwd()
            while outstanding:
                bwd()
        for work, _t in pending:
            work.wait()
        mean_loss = (torch.stack([l.detach() for l in losses]).mean()
                     if ctx.is_last else None)
        return ctx.broadcast_scalar(mean_loss, ctx.world - 1, device)
