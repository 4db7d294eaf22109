"""Parallelism strategy descriptor — the launcher's extension seam.

SURVEY.md §2.14: DDP is the only strategy any BASELINE config requires.
Implemented: DDP (bucketed all-reduce, ddp.py), TP (Megatron-style
head/ffn sharding, tp.py; composes with DP as a TP x DP mesh when
degree < world_size), PP (1F1B/GPipe stages, pp.py), Ulysses SP (sp.py)
and EP (expert parallel for MoE configs, ep.py) — llama family. Ring-SP/CP
remains a declared-but-reserved enum so PyTorchJob specs stay
forward-compatible, rejected with a clear error until a config demands it
(it needs a ring-attention kernel). The seam is the per-rank environment
the gang launcher already provides (RANK/WORLD_SIZE/LOCAL_RANK + this
descriptor serialized into the worker spec as `parallelism`).

Spec form (PyTorchJob template):
    parallelism: {strategy: ddp}                      # default
    parallelism: {strategy: tp, degree: 8}            # pure TP
    parallelism: {strategy: tp, degree: 2}            # tp2 x dp(world/2)
    parallelism: {strategy: ep, degree: 8}            # MoE expert parallel
"""
from __future__ import annotations

from dataclasses import dataclass
from enum import Enum


class Strategy(str, Enum):
    DDP = "ddp"      # implemented: bucketed all-reduce over RCCL/xGMI
    TP = "tp"        # implemented: Megatron-style head/ffn sharding (tp.py);
                     # degree == world -> pure TP, degree < world -> TP x DP
    PP = "pp"        # implemented: 1F1B/GPipe stages (pp.py);
                     # degree == world -> pure PP, degree < world -> PP x DP
    SP = "sp"        # reserved: sequence/context parallel (ring attention)
    EP = "ep"        # implemented: expert parallel for MoE configs (ep.py);
                     # gather-compute-scatter, pure form in v1
    ULYSSES = "ulysses"  # implemented: attention head-scatter SP (sp.py);
                         # degree == world -> pure, degree < world -> SP x DP


IMPLEMENTED = {Strategy.DDP, Strategy.TP, Strategy.PP, Strategy.ULYSSES,
               Strategy.EP}


@dataclass
class ParallelismSpec:
    strategy: Strategy = Strategy.DDP
    degree: int = 1

    @classmethod
    def from_spec(cls, template: dict) -> "ParallelismSpec":
        p = template.get("parallelism") or {}
        if isinstance(p, str):
            p = {"strategy": p}
        try:
            strategy = Strategy(p.get("strategy", "ddp"))
        except ValueError:
            raise ValueError(
                f"unknown parallelism strategy {p.get('strategy')!r}; "
                f"known: {[s.value for s in Strategy]}")
        spec = cls(strategy=strategy, degree=int(p.get("degree", 1)))
        spec.validate()
        return spec

    def validate(self):
        if self.strategy not in IMPLEMENTED:
            raise NotImplementedError(
                f"parallelism strategy {self.strategy.value!r} is reserved "
                "but not implemented (it needs a ring-attention kernel) — "
                f"implemented: {sorted(s.value for s in IMPLEMENTED)}")
