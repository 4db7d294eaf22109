"""Multi-process DDP correctness on CPU (gloo, world_size=2).

Mirrors SURVEY.md §4's "rebuild mapping" tier-2 strategy: prove the
distributed path by construction on CPU so the RCCL path differs only in
backend string and device.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from kubeflow_amd.models import MnistMLP
from kubeflow_amd.runtime import Trainer, TrainConfig


def _worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(100 + rank)  # different init per rank; bcast fixes
        m = MnistMLP(in_dim=16, hidden=32, n_classes=4)
        tr = Trainer(m, TrainConfig(lr=1e-2, warmup_steps=1,
                                    bucket_mb=0.0001))  # force many buckets
        # params must be identical after DDP broadcast
        flat0 = tr.flat.data.clone()
        gathered = [torch.empty_like(flat0) for _ in range(world)]
        dist.all_gather(gathered, flat0)
        assert torch.equal(gathered[0], gathered[1]), "bcast failed"

        torch.manual_seed(7 + rank)  # different data per rank
        x = torch.randn(8, 16)
        y = torch.randint(0, 4, (8,))
        for _ in range(3):
            loss = tr.step(x, y)
        # after steps, params must STILL be identical across ranks
        flat1 = tr.flat.data.clone()
        dist.all_gather(gathered, flat1)
        same = torch.allclose(gathered[0], gathered[1], atol=1e-6)
        results[rank] = bool(same)
    finally:
        dist.destroy_process_group()


def test_ddp_two_ranks_gloo():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29531
        procs = [ctx.Process(target=_worker, args=(r, world, port, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=120)
        for p in procs:
            assert p.exitcode == 0
        assert results[0] and results[1]


def _grad_worker(rank, world, port, results):
    """Gradients after DDP step == mean of per-rank gradients."""
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(0)
        m = MnistMLP(in_dim=8, hidden=8, n_classes=2)
        tr = Trainer(m, TrainConfig(lr=0.0, warmup_steps=1, weight_decay=0.0))
        torch.manual_seed(50 + rank)
        x = torch.randn(4, 8)
        y = torch.randint(0, 2, (4,))

        # single-rank reference grads for both shards, computed locally
        torch.manual_seed(0)
        m_ref = MnistMLP(in_dim=8, hidden=8, n_classes=2)
        m_ref.load_state_dict(m.state_dict())
        grads = []
        for r in range(world):
            torch.manual_seed(50 + r)
            xr = torch.randn(4, 8)
            yr = torch.randint(0, 2, (4,))
            m_ref.zero_grad()
            m_ref(xr, yr).backward()
            grads.append(torch.cat([p.grad.flatten()
                                    for p in m_ref.parameters()]))
        expect = torch.stack(grads).mean(0)

        tr.ddp.prepare_step()
        loss = m(x, y)
        loss.backward()
        tr.ddp.finalize()
        got = torch.cat([p.grad.flatten() for p in tr.flat.params])
        # note: flat params are reverse-ordered; compare via sorted norms
        results[rank] = (float(got.norm()), float(expect.norm()))
    finally:
        dist.destroy_process_group()


def test_ddp_grad_averaging():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_grad_worker, args=(r, world, 29532, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=120)
        for p in procs:
            assert p.exitcode == 0
        for r in range(world):
            got, expect = results[r]
            assert got == pytest.approx(expect, rel=1e-5)
