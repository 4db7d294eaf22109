"""Training worker process entry — what runs inside a PyTorchJob rank.

Launched by the gang launcher as
    python -m kubeflow_amd.runtime.worker --spec <workdir>/spec.json
with RANK/WORLD_SIZE/LOCAL_RANK/MASTER_* set. Heartbeats go to
<workdir>/rank-<r>/status.json (atomic rename); the PyTorchJob controller
derives CR status.conditions from these the way the reference derives
notebook status from pod state + events (apps/common/status.py:10-99).
"""
from __future__ import annotations

import argparse
import json
import os
import signal
import sys
import time

from kubeflow_amd.ops import tunable as _kf_tunable
_kf_tunable.enable()
import torch

from kubeflow_amd.models import build_model
from kubeflow_amd.parallel import dist as kdist
from kubeflow_amd.runtime import Trainer, TrainConfig
from kubeflow_amd.runtime import checkpoint as ckpt


def write_status(rank_dir: str, state: str, step: int = 0, loss=None,
                 metrics=None, error: str = ""):
    payload = {"state": state, "step": step,
               "loss": None if loss is None else float(loss),
               "metrics": metrics or {}, "error": error, "ts": time.time()}
    tmp = os.path.join(rank_dir, ".status.tmp")
    with open(tmp, "w") as f:
        json.dump(payload, f)
    os.replace(tmp, os.path.join(rank_dir, "status.json"))
    if metrics:  # history series for the tensorboard viewer
        with open(os.path.join(rank_dir, "metrics.jsonl"), "a") as f:
            f.write(json.dumps({"step": step, "metrics": metrics}) + "\n")



def run_pytest(spec: dict, rank_dir: str) -> int:
    """CI test task (the kind the ci/ workflow builders emit): runs a
    pytest selection in a subprocess and reports pass/fail through the
    normal rank status seam."""
    import subprocess
    import sys as _sys
    args = list(spec.get("pytest_args") or [])
    repo = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    write_status(rank_dir, "running", 0)
    proc = subprocess.run(
        [_sys.executable, "-m", "pytest", "-q", "-p", "no:cacheprovider",
         "-m", "not gpu", *args],
        cwd=repo, capture_output=True, text=True,
        timeout=float(spec.get("timeout", 1200)))
    tail = (proc.stdout or "")[-2000:]
    with open(os.path.join(rank_dir, "pytest.log"), "w") as f:
        f.write(proc.stdout or "")
        f.write(proc.stderr or "")
    if proc.returncode == 0:
        write_status(rank_dir, "succeeded", 1,
                     metrics={"pytest": tail.splitlines()[-1]
                              if tail.splitlines() else ""})
        return 0
    write_status(rank_dir, "failed", 1,
                 error=f"pytest rc={proc.returncode}: "
                       f"{tail.splitlines()[-1] if tail.splitlines() else ''}")
    return 1


def run_evaluator(spec: dict, workdir: str, rank_dir: str, stop: dict) -> int:
    """TFJob Evaluator role: a sidecar process (NOT part of the training
    gang) that watches the checkpoint directory, loads each new step's
    weights, and reports eval loss over held-out synthetic batches — the
    TF estimator evaluator loop mapped onto the checkpoint seam. Exits 0
    when the trainers' final step appears (or on SIGTERM at job end)."""
    import torch  # local: evaluator runs standalone
    from kubeflow_amd.models import build_model
    from kubeflow_amd.runtime import checkpoint as _ckpt

    device = (torch.device("cuda", 0) if torch.cuda.is_available()
              else torch.device("cpu"))
    dtype = (torch.bfloat16 if device.type == "cuda"
             and spec.get("dtype", "bf16") == "bf16" else torch.float32)
    model = build_model(spec.get("model", "mnist-mlp"), device=device,
                        dtype=dtype)
    cfg = getattr(model, "cfg", None)
    ckpt_dir = spec.get("checkpoint_dir") or os.path.join(workdir,
                                                          "checkpoints")
    final_step = int(spec.get("steps", 100))
    eval_batches = int(spec.get("eval_batches", 4))
    seen = -1
    write_status(rank_dir, "running", 0)
    deadline = time.time() + float(spec.get("eval_timeout", 600))
    while not stop["flag"] and time.time() < deadline:
        d = _ckpt.latest_dir(ckpt_dir)
        step = seen
        if d is not None:
            try:
                step = int(os.path.basename(d).split("-")[1])
            except (IndexError, ValueError):
                step = seen
        if step > seen:
            _ckpt.load_model_weights(model, d)
            model.eval()
            losses = []
            with torch.no_grad():
                for b in range(eval_batches):
                    x, y = synthetic_batch(spec, cfg, device,
                                           rank=1000 + b, step=step)
                    out = model(x)
                    if out.dim() == 3:  # lm
                        loss = torch.nn.functional.cross_entropy(
                            out.float().reshape(-1, out.shape[-1]),
                            y.reshape(-1))
                    else:
                        loss = torch.nn.functional.cross_entropy(
                            out.float(), y)
                    losses.append(float(loss))
            seen = step
            write_status(rank_dir, "running", step,
                         metrics={"eval_loss": sum(losses) / len(losses),
                                  "eval_step": step})
            if step >= final_step:
                break
        else:
            time.sleep(0.2)
    write_status(rank_dir, "succeeded", max(seen, 0),
                 metrics={"eval_step": seen} if seen >= 0 else None)
    return 0


def synthetic_batch(spec: dict, cfg, device, rank: int, step: int):
    """Deterministic-per-(rank,step) synthetic data of the model's shape."""
    seed = (int(spec.get("seed", 0)) * 1000003 + rank * 9176 + step) % (2**31)
    g = torch.Generator(device="cpu").manual_seed(seed)
    kind = spec.get("task") or (
        "mlp" if cfg is None else
        ("classify" if hasattr(cfg, "n_classes") else "lm"))
    if kind == "lm":
        B = int(spec.get("micro_batch", 2))
        S = int(spec.get("seq_len", 4096))
        V = cfg.vocab_size
        toks = torch.randint(0, V, (B, S + 1), generator=g)
        return (toks[:, :-1].to(device), toks[:, 1:].contiguous().to(device))
    if kind == "classify":
        B = int(spec.get("micro_batch", 32))
        S = int(spec.get("seq_len", 128))
        V = cfg.vocab_size
        x = torch.randint(0, V, (B, S), generator=g)
        y = torch.randint(0, cfg.n_classes, (B,), generator=g)
        return x.to(device), y.to(device)
    # mnist-style dense input
    B = int(spec.get("micro_batch", 64))
    x = torch.randn(B, 784, generator=g)
    y = torch.randint(0, 10, (B,), generator=g)
    return x.to(device), y.to(device)


_ASYNC_SAVER = None


def _save_all_ranks(trainer, ckpt_dir, spec, rank, world, final=False):
    """Checkpoint with the `latest` marker committed only after EVERY
    rank's files are durable (checkpoint.py:save commit semantics).

    world==1 defaults to OVERLAPPED saves (checkpoint.AsyncSave): training
    only stalls for the device->host snapshot while the disk-bound write
    runs in the background (KF_ASYNC_CKPT=0 opts out). Multi-rank jobs
    keep the synchronous all-ranks-durable-then-commit protocol."""
    global _ASYNC_SAVER
    from kubeflow_amd.runtime import checkpoint as _ckpt
    use_async = (world == 1
                 and os.environ.get("KF_ASYNC_CKPT", "1") == "1")
    if use_async:
        if _ASYNC_SAVER is None:
            _ASYNC_SAVER = _ckpt.AsyncSave()
        _ASYNC_SAVER.save(trainer, ckpt_dir, spec["model"], rank, world,
                          commit=True)
        if final:
            _ASYNC_SAVER.wait()
        return
    if _ASYNC_SAVER is not None:
        _ASYNC_SAVER.wait()
    kdist.barrier()
    _ckpt.save(trainer, ckpt_dir, spec["model"], rank, world, commit=False)
    kdist.barrier()
    if rank == 0:
        _ckpt.commit_latest(ckpt_dir, trainer.step_num)


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--spec", required=True)
    args = ap.parse_args(argv)
    with open(args.spec) as f:
        spec = json.load(f)

    rank = kdist.env_rank()
    workdir = os.environ.get("KF_JOB_WORKDIR",
                             os.path.dirname(os.path.abspath(args.spec)))
    rank_dir = os.path.join(workdir, f"rank-{rank}")
    os.makedirs(rank_dir, exist_ok=True)
    write_status(rank_dir, "initializing")

    stop = {"flag": False}
    signal.signal(signal.SIGTERM, lambda *_: stop.update(flag=True))

    if spec.get("role") == "Evaluator":
        return run_evaluator(spec, workdir, rank_dir, stop)
    if spec.get("task") == "pytest":
        return run_pytest(spec, rank_dir)

    try:
        from kubeflow_amd.parallel.strategy import ParallelismSpec, Strategy
        pspec = ParallelismSpec.from_spec(spec)  # loud on reserved ones
        rank, world, device = kdist.init_distributed()
        torch.manual_seed(int(spec.get("seed", 0)) + rank)
        dtype = (torch.bfloat16 if device.type == "cuda"
                 and spec.get("dtype", "bf16") == "bf16" else torch.float32)
        tp_ctx = None
        pp_ctx = None
        sp_ctx = None
        ep_ctx = None
        cp_ctx = None
        dp_group = None
        data_rank = rank
        if pspec.strategy == Strategy.TP and world > 1:
            from kubeflow_amd.parallel.tp import TpContext
            deg = pspec.degree if pspec.degree > 1 else world
            if world % deg:
                raise ValueError(f"world_size {world} not divisible by "
                                 f"tp degree {deg}")
            if deg == world:
                tp_ctx = TpContext.from_group(None)  # pure TP
                data_rank = 0
            else:  # TP x DP mesh: contiguous TP blocks, strided DP
                tp_group, dp_group, _tp_rank, dp_rank = kdist.build_mesh(deg)
                tp_ctx = TpContext.from_group(tp_group)
                data_rank = dp_rank
        elif pspec.strategy == Strategy.EP and world > 1:
            from kubeflow_amd.parallel.ep import EpContext
            if pspec.degree not in (1, world):
                raise ValueError(
                    f"pure EP requires degree == world_size ({world}); "
                    f"got {pspec.degree}")
            ep_ctx = EpContext.from_group(None)
        elif pspec.strategy == Strategy.SP and world > 1:
            from kubeflow_amd.parallel.ring import RingContext
            deg = pspec.degree if pspec.degree > 1 else world
            if world % deg:
                raise ValueError(f"world_size {world} not divisible by "
                                 f"ring degree {deg}")
            if deg == world:
                cp_ctx = RingContext.from_group(None)
                data_rank = 0
            else:  # ring x DP mesh: contiguous ring groups, strided DP
                cp_group, _dp_g, _cr, dp_rank = kdist.build_mesh(deg)
                cp_ctx = RingContext.from_group(cp_group)
                data_rank = dp_rank
        elif pspec.strategy == Strategy.ULYSSES and world > 1:
            from kubeflow_amd.parallel.sp import SpContext
            deg = pspec.degree if pspec.degree > 1 else world
            if world % deg:
                raise ValueError(f"world_size {world} not divisible by "
                                 f"ulysses degree {deg}")
            if deg == world:
                sp_ctx = SpContext.from_group(None)  # pure ulysses
                data_rank = 0
            else:  # SP x DP mesh: contiguous seq groups, strided DP.
                # Params stay replicated across ALL ranks and every rank
                # holds a distinct token subset, so the default WORLD-wide
                # DDP grad average is exactly right — only the sequence
                # group and batch index change.
                sp_group, _dp_g, _sr, dp_rank = kdist.build_mesh(deg)
                sp_ctx = SpContext.from_group(sp_group)
                data_rank = dp_rank
        elif pspec.strategy == Strategy.PP and world > 1:
            from kubeflow_amd.parallel.pp import PpContext
            deg = pspec.degree if pspec.degree > 1 else world
            if world % deg:
                raise ValueError(f"world_size {world} not divisible by "
                                 f"pp degree {deg}")
            if not spec["model"].startswith("llama"):
                raise ValueError("pipeline parallelism is implemented for "
                                 f"the llama family only, not {spec['model']!r}")
            if deg == world:
                pp_ctx = PpContext.from_group(None)  # pure PP
                data_rank = 0
            else:  # PP x DP mesh: contiguous stage chains, strided DP
                pp_group, dp_group, _st, dp_rank = kdist.build_mesh(deg)
                pp_ctx = PpContext.from_group(pp_group)
                data_rank = dp_rank
        if pp_ctx is not None:
            from kubeflow_amd.models import model_config
            from kubeflow_amd.models.llama import LlamaStage
            model = LlamaStage(model_config(spec["model"]), pp_ctx.rank,
                               pp_ctx.world, device=device, dtype=dtype)
        else:
            model = build_model(spec["model"], device=device, dtype=dtype,
                                tp=tp_ctx, sp=sp_ctx, ep=ep_ctx, cp=cp_ctx)
        if tp_ctx is not None:
            tp_ctx.sync_replicated(model)
        if ep_ctx is not None:
            ep_ctx.sync_replicated(model)
        cfg = getattr(model, "cfg", None)
        tcfg = TrainConfig(
            lr=float(spec.get("lr", 3e-4)),
            weight_decay=float(spec.get("weight_decay", 0.1)),
            warmup_steps=int(spec.get("warmup_steps", 10)),
            lr_decay_steps=int(spec.get("steps", 100)),
            grad_accum=int(spec.get("grad_accum", 1)),
        )
        if pp_ctx is not None:
            from kubeflow_amd.runtime import PpTrainer
            mb = int(spec.get("micro_batch", 2))
            # default: one microbatch per stage keeps the pipe full and
            # always divides; callers can override via pp_microbatches
            micros = int(spec.get("pp_microbatches", 0)) or (
                pp_ctx.world if mb % pp_ctx.world == 0 else 1)
            trainer = PpTrainer(model, tcfg, pp_ctx, micros,
                                schedule=spec.get("pp_schedule", "1f1b"),
                                dp_group=dp_group)
        else:
            trainer = Trainer(model, tcfg, tp_ctx=tp_ctx, dp_group=dp_group,
                              ep_ctx=ep_ctx,
                              zero=bool(spec.get("zero", False)))

        ckpt_dir = spec.get("checkpoint_dir") or os.path.join(workdir, "checkpoints")
        save_every = int(spec.get("save_every", 0))
        start_step = 0
        if spec.get("resume", True) and ckpt.latest_dir(ckpt_dir):
            start_step = ckpt.load(trainer, ckpt_dir, rank)

        steps = int(spec.get("steps", 100))
        status_every = int(spec.get("status_every", 5))
        write_status(rank_dir, "running", start_step)
        loss = None
        ema = None
        win_t0, win_steps, win_items = time.time(), 0, 0
        for step in range(start_step, steps):
            if stop["flag"]:
                write_status(rank_dir, "failed", step, loss,
                             error="terminated")
                return 143
            # model-parallel peers form one data replica: data_rank is 0
            # (pure) or the dp index (TP/PP x DP meshes)
            x, y = synthetic_batch(spec, cfg, device, data_rank, step)
            shard = sp_ctx or cp_ctx
            if shard is not None:  # ulysses/ring: this rank's seq chunk
                if x.shape[1] % shard.world:
                    raise ValueError(f"seq_len {x.shape[1]} not divisible "
                                     f"by sp/ring degree {shard.world}")
                s = x.shape[1] // shard.world
                x = x[:, shard.rank * s:(shard.rank + 1) * s].contiguous()
                y = y[:, shard.rank * s:(shard.rank + 1) * s].contiguous()
            loss = trainer.step(x, y)
            win_steps += 1
            win_items += x.numel()  # tokens (lm) or features processed
            if (sp_ctx or cp_ctx) is not None:  # global-mean loss
                import torch.distributed as tdist
                lt = loss.to(torch.float32).clone()
                tdist.all_reduce(lt)
                loss = lt / world
            if (step + 1) % status_every == 0 or step + 1 == steps:
                lval = float(loss)
                ema = lval if ema is None else 0.9 * ema + 0.1 * lval
                dt = max(1e-9, time.time() - win_t0)
                gn = getattr(trainer, "last_grad_norm", None)
                write_status(rank_dir, "running", step + 1, lval,
                             metrics={"loss": lval, "loss_ema": ema,
                                      "lr": trainer.lr_at(step),
                                      "grad_norm": (None if gn is None
                                                    else round(float(gn), 4)),
                                      "step_ms": round(dt / win_steps * 1e3,
                                                       2),
                                      "items_per_s": round(win_items / dt,
                                                           1)})
                win_t0, win_steps, win_items = time.time(), 0, 0
            if save_every and (step + 1) % save_every == 0:
                _save_all_ranks(trainer, ckpt_dir, spec, rank, world)
        if spec.get("save_final", True):
            _save_all_ranks(trainer, ckpt_dir, spec, rank, world, final=True)
        elif _ASYNC_SAVER is not None:
            _ASYNC_SAVER.wait()  # drain in-flight periodic saves
        write_status(rank_dir, "succeeded", steps, loss,
                     metrics={"loss": None if loss is None else float(loss),
                              "loss_ema": ema})
        return 0
    except Exception as e:  # surface the error to the controller
        import traceback
        traceback.print_exc()
        write_status(rank_dir, "failed", error=f"{type(e).__name__}: {e}")
        return 1


if __name__ == "__main__":
    sys.exit(main())
