"""Tensor-parallel Llama across 2 gloo ranks vs the full single-process
model — numerics parity for forward logits, training-loss trajectory and
the sharded checkpoint mapping. CPU fp32, exact-arithmetic comparisons
(loose allclose only where fp reduction order differs)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from kubeflow_amd.models import build_model
from kubeflow_amd.models.llama import LlamaModel, llama_tiny
from kubeflow_amd.parallel import tp as tpmod
from kubeflow_amd.runtime import Trainer, TrainConfig

SEED = 4242


def _cfg():
    c = llama_tiny()
    c.n_kv_heads = 2  # llama_tiny default kv=1 is not TP2-divisible
    return c


def _full_model():
    torch.manual_seed(SEED)
    return LlamaModel(_cfg(), dtype=torch.float32)


def _tp_worker(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        ctx = tpmod.TpContext.from_group(None)
        full = _full_model()  # same full weights on every rank (same seed)
        cfg = full.cfg
        tpm = LlamaModel(cfg, dtype=torch.float32, tp=ctx)
        tpm.load_state_dict(
            tpmod.shard_llama_state_dict(full.state_dict(), cfg, rank, world))

        torch.manual_seed(SEED + 1)  # identical batch on every TP rank
        toks = torch.randint(0, cfg.vocab_size, (2, 64))
        tgts = torch.randint(0, cfg.vocab_size, (2, 64))

        # ---- forward parity: TP logits == full-model logits
        with torch.no_grad():
            ref = full(toks)
            got = tpm(toks)
        fwd_ok = torch.allclose(ref, got, atol=1e-4, rtol=1e-4)

        # ---- training parity: 3 steps, loss trajectories match
        tr_full = Trainer(_full_model(), TrainConfig(lr=1e-3, warmup_steps=1))
        # rebuild the TP model so both trainers start from the same state
        tpm2 = LlamaModel(cfg, dtype=torch.float32, tp=ctx)
        tpm2.load_state_dict(tpmod.shard_llama_state_dict(
            _full_model().state_dict(), cfg, rank, world))
        tr_tp = Trainer(tpm2, TrainConfig(lr=1e-3, warmup_steps=1),
                        tp_ctx=ctx)
        losses_full, losses_tp = [], []
        for _ in range(3):
            losses_full.append(float(tr_full.step(toks, tgts)))
            losses_tp.append(float(tr_tp.step(toks, tgts)))
        results[rank] = (fwd_ok, losses_full, losses_tp)
    finally:
        dist.destroy_process_group()


def test_tp_two_ranks_matches_full_model():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        procs = [ctx.Process(target=_tp_worker,
                             args=(r, world, 29561, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0
        for r in range(world):
            fwd_ok, lf, lt = results[r]
            assert fwd_ok, "TP forward logits diverged from full model"
            for a, b in zip(lf, lt):
                assert abs(a - b) < 5e-3, (lf, lt)
        # both TP ranks observed the same losses (replicated state in sync)
        assert results[0][2] == pytest.approx(results[1][2], abs=1e-6)


def test_shard_state_dict_shapes():
    cfg = llama_tiny()
    torch.manual_seed(0)
    full = LlamaModel(cfg, dtype=torch.float32)
    sd = full.state_dict()
    n = 2  # n_heads=2, n_kv_heads=1... kv=1 not divisible; use head check
    # llama_tiny has n_kv_heads=1 -> TP2 must be rejected by the block
    from kubeflow_amd.parallel.tp import TpContext
    fake = TpContext(group=None, rank=0, world=2)
    with pytest.raises(ValueError):
        LlamaModel(cfg, dtype=torch.float32, tp=fake)


def test_tp_requires_divisible_heads():
    # a config where TP2 divides everything
    cfg = llama_tiny()
    cfg.n_kv_heads = 2
    torch.manual_seed(0)
    full = LlamaModel(cfg, dtype=torch.float32)
    sd = full.state_dict()
    for rank in range(2):
        local = tpmod.shard_llama_state_dict(sd, cfg, rank, 2)
        w = local["layers.0.wqkv.weight"]
        d = cfg.head_dim
        assert w.shape[0] == (cfg.n_heads // 2 + 2 * cfg.n_kv_heads // 2) * d
        assert local["layers.0.w13.weight"].shape[0] == cfg.ffn_dim
        assert local["layers.0.w2.weight"].shape[1] == cfg.ffn_dim // 2
        assert torch.equal(local["embed.weight"], sd["embed.weight"])
    # the two ranks' qkv shards tile the full weight
    a = tpmod.shard_llama_state_dict(sd, cfg, 0, 2)["layers.0.wqkv.weight"]
    b = tpmod.shard_llama_state_dict(sd, cfg, 1, 2)["layers.0.wqkv.weight"]
    d = cfg.head_dim
    hq, hkv = cfg.n_heads // 2, cfg.n_kv_heads // 2
    q = torch.cat([a[:hq * d], b[:hq * d]])
    assert torch.equal(q, sd["layers.0.wqkv.weight"][:cfg.n_heads * d])


def test_tp_pytorchjob_e2e(tmp_path):
    """Worker-level TP: a 2-replica PyTorchJob with parallelism tp/2 runs
    the real worker path (sync_replicated, TP trainer, gloo) to Succeeded."""
    import time
    from kubeflow_amd.api import new_object
    from kubeflow_amd.api.objects import has_condition
    from kubeflow_amd.platform import Platform

    with Platform(root_dir=str(tmp_path)) as plat:
        plat.store.create(new_object("PyTorchJob", "tp-job", "default", spec={
            "pytorchReplicaSpecs": {"Worker": {
                "replicas": 2, "restartPolicy": "Never",
                "template": {"model": "llama-tiny-mha", "steps": 4,
                             "micro_batch": 2, "seq_len": 64,
                             "gpus_per_replica": 0, "status_every": 2,
                             "save_final": False,
                             "parallelism": {"strategy": "tp",
                                             "degree": 2}}}}}))
        deadline = time.time() + 240
        while time.time() < deadline:
            obj = plat.store.get("PyTorchJob", "tp-job", "default")
            assert not has_condition(obj, "Failed"), obj["status"]
            if has_condition(obj, "Succeeded"):
                break
            time.sleep(0.5)
        assert has_condition(obj, "Succeeded"), obj["status"]


def _tp_ckpt_worker(rank, world, port, ckdir, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from kubeflow_amd.runtime import checkpoint as ckpt
        ctx = tpmod.TpContext.from_group(None)
        cfg = _cfg()
        torch.manual_seed(SEED + rank)  # per-rank shard init
        m1 = LlamaModel(cfg, dtype=torch.float32, tp=ctx)
        ctx.sync_replicated(m1)
        tr1 = Trainer(m1, TrainConfig(lr=1e-3, warmup_steps=1), tp_ctx=ctx)
        torch.manual_seed(SEED + 1)
        toks = torch.randint(0, cfg.vocab_size, (2, 64))
        tgts = torch.randint(0, cfg.vocab_size, (2, 64))
        tr1.step(toks, tgts)
        dist.barrier()
        ckpt.save(tr1, ckdir, "llama-tiny-mha", rank, world)
        dist.barrier()

        torch.manual_seed(SEED + 99 + rank)  # fresh different state
        m2 = LlamaModel(cfg, dtype=torch.float32, tp=ctx)
        tr2 = Trainer(m2, TrainConfig(lr=1e-3, warmup_steps=1), tp_ctx=ctx)
        step = ckpt.load(tr2, ckdir, rank)
        shard_ok = bool(torch.equal(tr1.flat.data, tr2.flat.data))
        # both trainers continue identically after resume
        l1 = float(tr1.step(toks, tgts))
        l2 = float(tr2.step(toks, tgts))
        results[rank] = (step, shard_ok, l1, l2)
    finally:
        dist.destroy_process_group()


def test_tp_checkpoint_roundtrip(tmp_path):
    """TP shards checkpoint per-rank and restore bit-exact (rank 0's shard
    must NOT overwrite rank 1's — the replicated-model layout would)."""
    world = 2
    mpctx = mp.get_context("spawn")
    with mpctx.Manager() as mgr:
        results = mgr.dict()
        procs = [mpctx.Process(target=_tp_ckpt_worker,
                               args=(r, world, 29661, str(tmp_path), results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0
        for r in range(world):
            step, shard_ok, l1, l2 = results[r]
            assert step == 1
            assert shard_ok, f"rank {r}: restored shard differs"
            assert l1 == pytest.approx(l2, abs=1e-6)
