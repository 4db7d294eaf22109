"""Checkpoint save/load with a PyTorchJob-compatible directory layout.

Layout (BASELINE.json requires layout compat with PyTorchJob conventions):
    <ckpt_dir>/step-<K>/
        meta.json            {step, world_size, model, timestamp}
        model.pt             bf16 flat parameter space (rank 0)
        optim-rank0.pt       fp32 master + moments + RNG (per rank; DDP keeps
                             them replicated so rank 0's copy is canonical)
    <ckpt_dir>/latest        text file: "step-<K>"

The reference platform itself delegates checkpointing to workloads + PVCs
(SURVEY.md §5 checkpoint/resume); here it is a worker-runtime feature.
"""
from __future__ import annotations

import json
import os
import time

import torch


def save(trainer, ckpt_dir: str, model_name: str, rank: int, world: int):
    step = trainer.step_num
    d = os.path.join(ckpt_dir, f"step-{step}")
    os.makedirs(d, exist_ok=True)
    # TP/PP flat spaces hold DISTINCT shards/stages per rank: every rank
    # must persist its own; replicated (DDP/ZeRO-1) keeps rank 0 canonical
    sharded = (getattr(trainer, "tp", None) is not None
               or getattr(trainer, "pp", None) is not None)
    if sharded:
        torch.save({"flat_data": trainer.flat.data,
                    "param_names": trainer.flat.names},
                   os.path.join(d, f"model-rank{rank}.pt"))
    elif rank == 0:
        torch.save({"flat_data": trainer.flat.data,
                    "param_names": trainer.flat.names}, os.path.join(d, "model.pt"))
    torch.save({"step": step,
                "p32": trainer.p32, "m": trainer.m, "v": trainer.v,
                "rng": torch.get_rng_state(),
                "cuda_rng": (torch.cuda.get_rng_state()
                             if torch.cuda.is_available() else None)},
               os.path.join(d, f"optim-rank{rank}.pt"))
    if rank == 0:
        with open(os.path.join(d, "meta.json"), "w") as f:
            json.dump({"step": step, "world_size": world, "model": model_name,
                       "sharded": sharded, "timestamp": time.time()}, f)
        tmp = os.path.join(ckpt_dir, ".latest.tmp")
        with open(tmp, "w") as f:
            f.write(f"step-{step}")
        os.replace(tmp, os.path.join(ckpt_dir, "latest"))
    return d


def latest_dir(ckpt_dir: str):
    marker = os.path.join(ckpt_dir, "latest")
    if not os.path.exists(marker):
        return None
    with open(marker) as f:
        name = f.read().strip()
    d = os.path.join(ckpt_dir, name)
    return d if os.path.isdir(d) else None


def load(trainer, ckpt_dir: str, rank: int) -> int:
    """Restore trainer state from the latest checkpoint; returns step (0 if
    no checkpoint)."""
    d = latest_dir(ckpt_dir)
    if d is None:
        return 0
    shard_path = os.path.join(d, f"model-rank{rank}.pt")
    if os.path.exists(shard_path):  # TP/PP: this rank's own shard/stage
        model_path = shard_path
    elif os.path.exists(os.path.join(d, "model.pt")):
        model_path = os.path.join(d, "model.pt")  # replicated (DDP/ZeRO)
    else:
        raise FileNotFoundError(
            f"{d}: no model-rank{rank}.pt — sharded (TP/PP) checkpoints "
            "require resuming at the same world size")
    model = torch.load(model_path, map_location="cpu", weights_only=False)
    trainer.flat.data.copy_(model["flat_data"].to(trainer.flat.device))
    opt_path = os.path.join(d, f"optim-rank{rank}.pt")
    if not os.path.exists(opt_path):  # elastic restart with different world
        opt_path = os.path.join(d, "optim-rank0.pt")
    opt = torch.load(opt_path, map_location="cpu", weights_only=False)
    trainer.p32.copy_(opt["p32"].to(trainer.p32.device))
    trainer.m.copy_(opt["m"].to(trainer.m.device))
    trainer.v.copy_(opt["v"].to(trainer.v.device))
    trainer.step_num = opt["step"]
    rng = opt["rng"]
    torch.set_rng_state(rng if isinstance(rng, torch.Tensor) else rng)
    if opt.get("cuda_rng") is not None and torch.cuda.is_available():
        torch.cuda.set_rng_state(opt["cuda_rng"])
    return trainer.step_num
