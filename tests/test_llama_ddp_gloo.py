"""Llama-tiny DDP across 2 gloo ranks — the flagship model through the
full distributed path on CPU (same code the 8×MI355X RCCL run executes)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from kubeflow_amd.models import build_model
from kubeflow_amd.runtime import Trainer, TrainConfig


def _worker(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(77 + rank)
        model = build_model("llama-tiny", dtype=torch.float32)
        tr = Trainer(model, TrainConfig(lr=1e-3, warmup_steps=1))
        torch.manual_seed(123 + rank)  # different data per rank
        toks = torch.randint(0, model.cfg.vocab_size, (1, 64))
        tgts = torch.randint(0, model.cfg.vocab_size, (1, 64))
        losses = [float(tr.step(toks, tgts)) for _ in range(3)]
        flat = tr.flat.data.clone()
        gathered = [torch.empty_like(flat) for _ in range(world)]
        dist.all_gather(gathered, flat)
        results[rank] = (losses,
                         bool(torch.allclose(gathered[0], gathered[1],
                                             atol=1e-5)))
    finally:
        dist.destroy_process_group()


def test_llama_tiny_ddp_two_ranks():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_worker, args=(r, world, 29544, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0
        for r in range(world):
            losses, same = results[r]
            assert all(l == l for l in losses)
            assert same, "ranks diverged after DDP steps"


def _elastic_worker(rank, world, port, ckdir, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from kubeflow_amd.runtime import checkpoint as ckpt
        torch.manual_seed(10 + rank)
        model = build_model("llama-tiny", dtype=torch.float32)
        tr = Trainer(model, TrainConfig(lr=1e-3, warmup_steps=1))
        step = ckpt.load(tr, ckdir, rank)
        toks = torch.randint(0, model.cfg.vocab_size, (1, 64))
        tr.step(toks, toks)
        results[rank] = step
    finally:
        dist.destroy_process_group()


def test_elastic_resume_world1_to_world2(tmp_path):
    """A world-1 checkpoint resumes at world 2 (rank1 falls back to the
    rank-0 optimizer shard — DDP state is replicated, SURVEY §5 recovery)."""
    from kubeflow_amd.runtime import checkpoint as ckpt
    torch.manual_seed(10)
    model = build_model("llama-tiny", dtype=torch.float32)
    tr = Trainer(model, TrainConfig(lr=1e-3, warmup_steps=1))
    toks = torch.randint(0, model.cfg.vocab_size, (1, 64))
    for _ in range(2):
        tr.step(toks, toks)
    ckdir = str(tmp_path / "ck")
    ckpt.save(tr, ckdir, "llama-tiny", rank=0, world=1)

    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_elastic_worker,
                             args=(r, 2, 29547, ckdir, results))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0
        assert results[0] == 2 and results[1] == 2
