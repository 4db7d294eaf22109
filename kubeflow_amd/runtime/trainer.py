"""Training worker runtime — the loop that runs inside a PyTorchJob rank.

This is the half of Kubeflow that lives in the sibling training-operator's
worker pods; here it is a first-class component:
FlatParamSpace + BucketedDDP (RCCL/xGMI) + fused-AdamW (single HIP kernel
over the whole flat model) + cosine LR schedule + checkpoint save/load with
a PyTorchJob-compatible directory layout (checkpoint.py).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch

from kubeflow_amd import ops
from kubeflow_amd.parallel import FlatParamSpace, BucketedDDP


@dataclass
class TrainConfig:
    lr: float = 3e-4
    beta1: float = 0.9
    beta2: float = 0.95
    eps: float = 1e-8
    weight_decay: float = 0.1
    warmup_steps: int = 10
    lr_decay_steps: int = 10000
    min_lr_ratio: float = 0.1
    grad_accum: int = 1
    bucket_mb: Optional[float] = None
    use_wd_mask: bool = True
    grad_clip: float = 1.0  # 0 disables


class _NullDDP:
    """Comm no-op stand-in when the data-parallel degree is 1 (pure TP)."""

    def prepare_step(self):
        pass

    def finalize(self):
        pass


class Trainer:
    """Owns the flat parameter space, DDP overlap and the fused optimizer."""

    def __init__(self, model: torch.nn.Module, cfg: TrainConfig = TrainConfig(),
                 tp_ctx=None, pp_ctx=None, zero: bool = False,
                 dp_group=None, ep_ctx=None):
        import torch.distributed as dist
        self.model = model
        self.cfg = cfg
        self.flat = FlatParamSpace(model)
        self.tp = tp_ctx
        self.pp = pp_ctx
        self.ep = ep_ctx
        self.zero = None
        if ep_ctx is not None:
            # EP: expert params are rank-local (a DDP all-reduce would
            # corrupt them); EpContext.sync_grads runs after backward
            self.ddp = _NullDDP()
            self.p32 = self.flat.data.float()
            self.m = torch.zeros_like(self.p32)
            self.v = torch.zeros_like(self.p32)
            self.wd_mask = (self.flat.build_wd_mask() if cfg.use_wd_mask
                            else None)
            self.step_num = 0
            return
        if (zero and dist.is_initialized() and dist.get_world_size() > 1
                and tp_ctx is None and pp_ctx is None):
            # ZeRO-1: optimizer state sharded 1/N; RS + shard-AdamW + AG
            # replace the DDP all-reduce (parallel/zero.py)
            from kubeflow_amd.parallel.zero import ZeroShard
            self.zero = ZeroShard(self.flat)
            self.ddp = _NullDDP()
            dist.broadcast(self.flat.data, src=0)
            self.p32 = self.zero.data_shard().float()
            self.m = self.zero.new_state()
            self.v = self.zero.new_state()
            if cfg.use_wd_mask:
                full = self.flat.build_wd_mask()
                self.wd_mask = self.zero.shard_of(full).clone()
                del full
            else:
                self.wd_mask = None
            self.step_num = 0
            return
        if tp_ctx is not None and dp_group is not None \
                and dist.get_world_size(dp_group) > 1:
            # TP x DP mesh: DP peers hold the SAME shard, so bucketed
            # all-reduce over the dp group is exactly DDP on each shard
            self.ddp = BucketedDDP(self.flat, bucket_mb=cfg.bucket_mb,
                                   process_group=dp_group)
        elif tp_ctx is not None or pp_ctx is not None:
            # pure TP/PP: ranks hold distinct shards/stages — no grad
            # all-reduce; comm happens inside the model (TP block
            # all-reduces) or the schedule (PP sends).
            self.ddp = _NullDDP()
        else:
            self.ddp = BucketedDDP(self.flat, bucket_mb=cfg.bucket_mb)
        self.p32 = self.flat.data.float()
        self.m = torch.zeros_like(self.p32)
        self.v = torch.zeros_like(self.p32)
        self.wd_mask = self.flat.build_wd_mask() if cfg.use_wd_mask else None
        self.step_num = 0

    # ------------------------------------------------------------ schedule
    def lr_at(self, step: int) -> float:
        cfg = self.cfg
        if step < cfg.warmup_steps:
            return cfg.lr * (step + 1) / max(1, cfg.warmup_steps)
        t = min(1.0, (step - cfg.warmup_steps) /
                max(1, cfg.lr_decay_steps - cfg.warmup_steps))
        floor = cfg.lr * cfg.min_lr_ratio
        return floor + 0.5 * (cfg.lr - floor) * (1 + math.cos(math.pi * t))

    # ---------------------------------------------------------------- step
    def step(self, *batch) -> torch.Tensor:
        """One optimizer step over `grad_accum` micro-batches.

        `batch` is either (inputs, targets) tensors or a callable returning
        them per micro-step. Returns the (device) loss of the last micro.
        """
        cfg = self.cfg
        self.flat.zero_grad()
        loss = None
        for micro in range(cfg.grad_accum):
            if len(batch) == 1 and callable(batch[0]):
                inputs, targets = batch[0](micro)
            else:
                inputs, targets = batch
            final = micro == cfg.grad_accum - 1
            if final:
                self.ddp.prepare_step()
            loss = self.model(inputs, targets)
            if cfg.grad_accum > 1:
                (loss / cfg.grad_accum).backward()
            else:
                loss.backward()
        self.ddp.finalize()
        if self.ep is not None:
            self.ep.sync_grads(self.flat)
        if self.zero is not None:
            self._zero_update()
        else:
            self._clip_and_update()
        return loss.detach()

    # ---------------------------------------------------------- optimizer
    def _clip_and_update(self):
        """Shared grad-clip + fused-AdamW tail of a step (grads already in
        the flat buffer, comm done)."""
        cfg = self.cfg
        if cfg.grad_clip > 0:
            comm = next((c for c in (self.tp, self.pp, self.ep)
                         if c is not None), None)
            if comm is not None:
                gnorm = comm.global_grad_norm(self.flat)
            else:
                gnorm = torch.linalg.vector_norm(self.flat.grad,
                                                 dtype=torch.float32)
            scale = (cfg.grad_clip / (gnorm + 1e-6)).clamp(max=1.0)
            self.flat.grad.mul_(scale.to(self.flat.grad.dtype))
            self.last_grad_norm = gnorm
        self.step_num += 1
        lr = self.lr_at(self.step_num - 1)
        ops.fused_adamw(self.flat.data, self.p32, self.flat.grad, self.m,
                        self.v, self.wd_mask, lr, cfg.beta1, cfg.beta2,
                        cfg.eps, cfg.weight_decay, self.step_num)

    def _zero_update(self):
        """ZeRO-1 step tail: averaged grad shard -> clip -> shard AdamW ->
        param all-gather."""
        cfg = self.cfg
        gs = self.zero.reduce_scatter_grads()
        if cfg.grad_clip > 0:
            gnorm = self.zero.global_grad_norm(gs)
            scale = (cfg.grad_clip / (gnorm + 1e-6)).clamp(max=1.0)
            gs.mul_(scale.to(gs.dtype))
            self.last_grad_norm = gnorm
        self.step_num += 1
        lr = self.lr_at(self.step_num - 1)
        ops.fused_adamw(self.zero.data_shard(), self.p32, gs, self.m,
                        self.v, self.wd_mask, lr, cfg.beta1, cfg.beta2,
                        cfg.eps, cfg.weight_decay, self.step_num)
        self.zero.all_gather_params()

    # ---------------------------------------------------------- checkpoint
    def state_dict(self) -> dict:
        return {
            "step": self.step_num,
            "flat_data": self.flat.data,
            "p32": self.p32,
            "m": self.m,
            "v": self.v,
            "param_names": self.flat.names,
            "rng": torch.get_rng_state(),
            "cuda_rng": (torch.cuda.get_rng_state()
                         if torch.cuda.is_available() else None),
        }

    def load_state_dict(self, sd: dict):
        self.step_num = sd["step"]
        self.flat.data.copy_(sd["flat_data"])
        self.p32.copy_(sd["p32"])
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        torch.set_rng_state(sd["rng"].cpu() if hasattr(sd["rng"], "cpu") else sd["rng"])
        if sd.get("cuda_rng") is not None and torch.cuda.is_available():
            torch.cuda.set_rng_state(sd["cuda_rng"])


class PpTrainer(Trainer):
    """Trainer over one pipeline stage: forward/backward run through the
    1F1B/GPipe schedule (parallel/pp.py); clip + fused AdamW are stage-local
    with a pipeline-global grad norm. `microbatches` plays grad_accum's
    role. With a `dp_group` (PP x DP mesh: same-stage peers across data
    replicas) stage grads are averaged over it after the schedule."""

    def __init__(self, stage: torch.nn.Module, cfg: TrainConfig,
                 pp_ctx, microbatches: int, schedule: str = "1f1b",
                 dp_group=None):
        super().__init__(stage, cfg, pp_ctx=pp_ctx)
        import torch.distributed as dist
        from kubeflow_amd.parallel.pp import PipelineRunner
        self.dp_group = dp_group
        self._dp_world = (dist.get_world_size(dp_group)
                          if dp_group is not None else 1)
        if dp_group is not None and self._dp_world > 1:
            # same-stage peers start identical
            dist.broadcast(self.flat.data,
                           src=dist.get_global_rank(dp_group, 0),
                           group=dp_group)
        self.runner = PipelineRunner(
            stage, pp_ctx, microbatches,
            act_dtype=self.flat.data.dtype,
            hidden_size=stage.cfg.hidden_size, schedule=schedule)

    def step(self, tokens, targets) -> torch.Tensor:
        self.flat.zero_grad()
        loss = self.runner.step(tokens, targets)
        if self.dp_group is not None and self._dp_world > 1:
            import torch.distributed as dist
            dist.all_reduce(self.flat.grad, group=self.dp_group)
            self.flat.grad.div_(self._dp_world)
        self._clip_and_update()
        return loss
