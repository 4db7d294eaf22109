"""Parallelism strategy descriptor — the launcher's extension seam.

SURVEY.md §2.14: DDP is the only strategy any BASELINE config requires.
Implemented: DDP (bucketed all-reduce, ddp.py), TP (Megatron-style
head/ffn sharding, tp.py; composes with DP as a TP x DP mesh when
degree < world_size), PP (1F1B/GPipe stages, pp.py), Ulysses SP (sp.py)
EP (expert parallel for MoE configs, ep.py) and ring-SP/CP (rotating-KV
context parallel, ring.py — "sp", aliases "ring"/"cp") — llama family.
The seam is the per-rank environment
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
    SP = "sp"        # implemented: ring attention context parallel
                     # (ring.py); "ring"/"cp" accepted as aliases
    EP = "ep"        # implemented: expert parallel for MoE configs (ep.py);
                     # gather-compute-scatter, pure form in v1
    ULYSSES = "ulysses"  # implemented: attention head-scatter SP (sp.py);
                         # degree == world -> pure, degree < world -> SP x DP


IMPLEMENTED = {Strategy.DDP, Strategy.TP, Strategy.PP, Strategy.ULYSSES,
               Strategy.EP, Strategy.SP}

_ALIASES = {"ring": "sp", "cp": "sp"}


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
            raw = p.get("strategy", "ddp")
            strategy = Strategy(_ALIASES.get(raw, raw))
        except ValueError:
            raise ValueError(
                f"unknown parallelism strategy {p.get('strategy')!r}; "
                f"known: {[s.value for s in Strategy]}")
        spec = cls(strategy=strategy, degree=int(p.get("degree", 1)))
        spec.validate()
        return spec

    def validate(self):
        if self.strategy not in IMPLEMENTED:  # future additions fail loudly
            raise NotImplementedError(
                f"parallelism strategy {self.strategy.value!r} is reserved "
                f"— implemented: {sorted(s.value for s in IMPLEMENTED)}")
