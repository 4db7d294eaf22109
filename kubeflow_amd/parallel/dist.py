"""Process-group bootstrap: one process per MI355X over RCCL/xGMI.

On ROCm, torch.distributed backend "nccl" IS RCCL; CPU CI uses gloo.
The gang launcher (kubeflow_amd.scheduler) sets RANK/WORLD_SIZE/LOCAL_RANK/
MASTER_ADDR/MASTER_PORT exactly like the reference's training-operator pods
expect ("extension seam in the launcher: per-rank env").
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def init_distributed(backend: str | None = None,
                     timeout_s: int = 600) -> tuple[int, int, torch.device]:
    """Initialize (if WORLD_SIZE>1) and pick this rank's device.

    Returns (rank, world_size, device).
    """
    world = env_world_size()
    rank = env_rank()
    local = env_local_rank()
    if torch.cuda.is_available():
        device = torch.device("cuda", local)
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29510")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s))
    return rank, world, device


def build_mesh(tp_degree: int):
    """Split WORLD into a TP x DP mesh. TP groups are CONTIGUOUS rank
    blocks — the gang scheduler allocates xGMI-adjacent GPUs to adjacent
    ranks, so the latency-critical per-block TP all-reduces stay on
    neighboring links while DP's overlappable grad traffic strides across.

    Returns (tp_group, dp_group, tp_rank, dp_rank). Every rank must call
    this (new_group is collective).
    """
    world, rank = dist.get_world_size(), dist.get_rank()
    if world % tp_degree:
        raise ValueError(f"world {world} not divisible by tp degree "
                         f"{tp_degree}")
    dp_degree = world // tp_degree
    tp_groups = [dist.new_group(list(range(i * tp_degree,
                                           (i + 1) * tp_degree)))
                 for i in range(dp_degree)]
    dp_groups = [dist.new_group(list(range(j, world, tp_degree)))
                 for j in range(tp_degree)]
    tp_rank, dp_rank = rank % tp_degree, rank // tp_degree
    return tp_groups[dp_rank], dp_groups[tp_rank], tp_rank, dp_rank


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def barrier():
    if dist.is_initialized():
        if torch.cuda.is_available() and dist.get_backend() == "nccl":
            dist.barrier(device_ids=[torch.cuda.current_device()])
        else:
            dist.barrier()
