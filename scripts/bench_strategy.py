"""Strategy benchmark — measure TP / Ulysses / ZeRO / PP variants of the
flagship model under torchrun (round-2 xGMI measurements; bench.py stays
the driver's DDP contract).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 scripts/bench_strategy.py \
        --strategy tp --steps 10 --warmup 3 [--micro-batch B] [--seq S]

Prints one JSON line from rank 0. `value` is whole-job tokens/s: for tp/
ulysses/pp the job processes ONE global batch per step (B*S tokens); for
ddp/zero each rank processes its own (B*S*N).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeflow_amd.ops import tunable as _t  # noqa: E402
_t.enable()
import torch  # noqa: E402

from kubeflow_amd.models import build_model, model_config  # noqa: E402
from kubeflow_amd.parallel import dist as kdist  # noqa: E402
from kubeflow_amd.runtime import PpTrainer, Trainer, TrainConfig  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--strategy", default="tp",
                    choices=["ddp", "zero", "tp", "ulysses", "pp", "ep"])
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--micro-batch", type=int, default=0)  # 0 = per-strategy
    ap.add_argument("--seq", type=int, default=4096)
    ap.add_argument("--pp-microbatches", type=int, default=0)
    ap.add_argument("--degree", type=int, default=0,
                    help="tp only: degree < world builds a TP x DP mesh")
    args = ap.parse_args()

    rank, world, device = kdist.init_distributed()
    torch.manual_seed(1234 + rank)
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    S = args.seq
    B = args.micro_batch or {"ddp": 6, "zero": 6, "tp": 8, "ulysses": 2,
                             "pp": 8, "ep": 4}[args.strategy]

    if args.strategy == "tp" and world > 1:
        from kubeflow_amd.parallel.tp import TpContext
        deg = args.degree or world
        if deg == world:
            ctx, dp_group, dp_degree = TpContext.from_group(None), None, 1
        else:
            tp_group, dp_group, _tr, _dr = kdist.build_mesh(deg)
            ctx, dp_degree = TpContext.from_group(tp_group), world // deg
        model = build_model(args.model, device=device, dtype=dtype, tp=ctx)
        ctx.sync_replicated(model)
        trainer = Trainer(model, TrainConfig(warmup_steps=2), tp_ctx=ctx,
                          dp_group=dp_group)
        per_rank_tokens, job_tokens = B * S, B * S * dp_degree
    elif args.strategy == "ulysses" and world > 1:
        from kubeflow_amd.parallel.sp import SpContext
        ctx = SpContext.from_group(None)
        model = build_model(args.model, device=device, dtype=dtype, sp=ctx)
        trainer = Trainer(model, TrainConfig(warmup_steps=2))
        S_local = S // world
        per_rank_tokens, job_tokens = B * S_local, B * S
    elif args.strategy == "ep" and world > 1:
        from kubeflow_amd.parallel.ep import EpContext
        ctx = EpContext.from_group(None)
        model = build_model(args.model, device=device, dtype=dtype, ep=ctx)
        ctx.sync_replicated(model)
        trainer = Trainer(model, TrainConfig(warmup_steps=2), ep_ctx=ctx)
        per_rank_tokens, job_tokens = B * S, B * S * world
    elif args.strategy == "pp" and world > 1:
        from kubeflow_amd.models.llama import LlamaStage
        from kubeflow_amd.parallel.pp import PpContext
        ctx = PpContext.from_group(None)
        model = LlamaStage(model_config(args.model), rank, world,
                           device=device, dtype=dtype)
        micros = args.pp_microbatches or world
        trainer = PpTrainer(model, TrainConfig(warmup_steps=2), ctx, micros)
        per_rank_tokens, job_tokens = B * S, B * S
    else:  # ddp / zero
        model = build_model(args.model, device=device, dtype=dtype)
        trainer = Trainer(model, TrainConfig(warmup_steps=2),
                          zero=args.strategy == "zero")
        per_rank_tokens, job_tokens = B * S, B * S * world

    cfg = model.cfg
    seq_in = (S // world if args.strategy == "ulysses" and world > 1 else S)
    toks = torch.randint(0, cfg.vocab_size, (B, seq_in), device=device)
    tgts = torch.randint(0, cfg.vocab_size, (B, seq_in), device=device)

    for _ in range(args.warmup):
        trainer.step(toks, tgts)
    kdist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        loss = trainer.step(toks, tgts)
    kdist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    el = time.time() - t0

    et = torch.tensor([el], dtype=torch.float64)
    if world > 1:
        import torch.distributed as dist
        if dist.get_backend() == "nccl":
            et = et.to(device)
        dist.all_reduce(et, op=dist.ReduceOp.MAX)
    el = float(et.item())

    if rank == 0:
        print(json.dumps({
            "metric": "strategy_train_tokens_per_s",
            "strategy": args.strategy, "n_gpus": world,
            "value": round(job_tokens * args.steps / el, 2),
            "ms_per_step": round(el / args.steps * 1e3, 2),
            "global_batch": job_tokens // S, "seq_len": S,
            "model": args.model,
            "loss": round(float(loss), 4), "dtype": str(dtype).split(".")[-1],
        }), flush=True)


if __name__ == "__main__":
    main()
