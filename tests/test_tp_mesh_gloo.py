"""TP x DP 2D mesh across 4 gloo ranks (tp2 x dp2) vs a single-process
oracle: DP-averaging two batches over sharded models must equal full-model
gradient accumulation over the same two batches."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from kubeflow_amd.models.llama import LlamaModel, llama_tiny
from kubeflow_amd.parallel import tp as tpmod
from kubeflow_amd.parallel.dist import build_mesh
from kubeflow_amd.runtime import Trainer, TrainConfig

SEED = 8282


def _cfg():
    c = llama_tiny()
    c.n_kv_heads = 2
    return c


def _full_model():
    torch.manual_seed(SEED)
    return LlamaModel(_cfg(), dtype=torch.float32)


def _batch(dp_rank, vocab):
    torch.manual_seed(SEED + 10 + dp_rank)
    return (torch.randint(0, vocab, (1, 64)),
            torch.randint(0, vocab, (1, 64)))


def _mesh_worker(rank, world, port, results):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        tp_group, dp_group, tp_rank, dp_rank = build_mesh(2)
        assert dist.get_world_size(tp_group) == 2
        assert dist.get_world_size(dp_group) == 2
        full = _full_model()
        cfg = full.cfg
        tpm = LlamaModel(cfg, dtype=torch.float32,
                         tp=tpmod.TpContext.from_group(tp_group))
        tpm.load_state_dict(tpmod.shard_llama_state_dict(
            full.state_dict(), cfg, tp_rank, 2))
        ctx = tpm.tp
        tr = Trainer(tpm, TrainConfig(lr=1e-3, warmup_steps=1),
                     tp_ctx=ctx, dp_group=dp_group)

        toks, tgts = _batch(dp_rank, cfg.vocab_size)
        losses = [float(tr.step(toks, tgts)) for _ in range(2)]

        # oracle: full model accumulating BOTH dp batches
        tr_full = Trainer(_full_model(),
                          TrainConfig(lr=1e-3, warmup_steps=1, grad_accum=2))
        micros = [_batch(0, cfg.vocab_size), _batch(1, cfg.vocab_size)]
        for _ in range(2):
            tr_full.step(lambda i: micros[i])
        want = tpmod.shard_llama_state_dict(tr_full.model.state_dict(),
                                            cfg, tp_rank, 2)
        got = tpm.state_dict()
        param_ok = all(torch.allclose(got[k], want[k],
                                      atol=1e-3, rtol=1e-3) for k in want)
        results[rank] = (losses, param_ok, tp_rank, dp_rank)
    finally:
        dist.destroy_process_group()


def test_tp2_dp2_mesh_matches_grad_accum_oracle():
    world = 4
    mpctx = mp.get_context("spawn")
    with mpctx.Manager() as mgr:
        results = mgr.dict()
        procs = [mpctx.Process(target=_mesh_worker,
                               args=(r, world, 29631, results))
                 for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=300)
        for p in procs:
            assert p.exitcode == 0
        for r in range(world):
            losses, param_ok, tp_rank, dp_rank = results[r]
            assert (tp_rank, dp_rank) == (r % 2, r // 2)
            assert param_ok, f"rank {r}: mesh params diverged from oracle"
        # dp peers (same batch partition) saw identical losses
        assert results[0][0] == pytest.approx(results[1][0], abs=1e-5)
        assert results[2][0] == pytest.approx(results[3][0], abs=1e-5)


def test_tp_mesh_pytorchjob_e2e(tmp_path):
    """4-replica PyTorchJob at tp degree 2 -> tp2 x dp2 mesh end-to-end."""
    import time
    from kubeflow_amd.api import new_object
    from kubeflow_amd.api.objects import has_condition
    from kubeflow_amd.platform import Platform

    with Platform(root_dir=str(tmp_path)) as plat:
        plat.store.create(new_object("PyTorchJob", "mesh-job", "default",
                                     spec={"pytorchReplicaSpecs": {"Worker": {
            "replicas": 4, "restartPolicy": "Never",
            "template": {"model": "llama-tiny-mha", "steps": 3,
                         "micro_batch": 1, "seq_len": 64,
                         "gpus_per_replica": 0, "status_every": 1,
                         "save_final": False,
                         "parallelism": {"strategy": "tp",
                                         "degree": 2}}}}}))
        deadline = time.time() + 240
        while time.time() < deadline:
            obj = plat.store.get("PyTorchJob", "mesh-job", "default")
            assert not has_condition(obj, "Failed"), obj["status"]
            if has_condition(obj, "Succeeded"):
                break
            time.sleep(0.5)
        assert has_condition(obj, "Succeeded"), obj["status"]


@pytest.mark.parametrize("strategy,model", [("pp", "llama-tiny"),
                                            ("ulysses", "llama-tiny-mha")])
def test_mesh_pytorchjob_e2e_other_strategies(tmp_path, strategy, model):
    """4-replica PyTorchJob at degree 2 -> (pp|sp)2 x dp2 mesh through the
    real worker path."""
    import time
    from kubeflow_amd.api import new_object
    from kubeflow_amd.api.objects import has_condition
    from kubeflow_amd.platform import Platform

    with Platform(root_dir=str(tmp_path)) as plat:
        plat.store.create(new_object("PyTorchJob", f"{strategy}-mesh",
                                     "default",
                                     spec={"pytorchReplicaSpecs": {"Worker": {
            "replicas": 4, "restartPolicy": "Never",
            "template": {"model": model, "steps": 3,
                         "micro_batch": 2, "seq_len": 64,
                         "gpus_per_replica": 0, "status_every": 1,
                         "save_final": False,
                         "parallelism": {"strategy": strategy,
                                         "degree": 2}}}}}))
        deadline = time.time() + 240
        while time.time() < deadline:
            obj = plat.store.get("PyTorchJob", f"{strategy}-mesh", "default")
            assert not has_condition(obj, "Failed"), obj["status"]
            if has_condition(obj, "Succeeded"):
                break
            time.sleep(0.5)
        assert has_condition(obj, "Succeeded"), obj["status"]
