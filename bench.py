#!/usr/bin/env python3
"""bench.py — flagship benchmark: PyTorchJob Llama-3-8B DDP bf16 training.

Measures BASELINE.json's headline metric (tokens/sec at 1/2/4/8 workers) on
synthetic data with random-init weights. Run directly for N=1 or under
`python -m torch.distributed.run --nnodes=1 --nproc-per-node N` for N>1
(reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from the env).

Prints ONE JSON line from rank 0 with the whole-job aggregate.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from kubeflow_amd.ops import tunable as _kf_tunable  # noqa: E402
_kf_tunable.enable()  # must precede the first GEMM

import torch  # noqa: E402

from kubeflow_amd.models import build_model  # noqa: E402
from kubeflow_amd.parallel import dist as kdist  # noqa: E402
from kubeflow_amd.runtime import Trainer, TrainConfig  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default=os.environ.get("KF_BENCH_MODEL", "llama3-8b"))
    ap.add_argument("--seq-len", type=int,
                    default=int(os.environ.get("KF_BENCH_SEQ", "4096")))
    ap.add_argument("--micro-batch", type=int,
                    default=int(os.environ.get("KF_BENCH_MB", "7")))
    # mb sweep on MI355X: mb4 16.4k, mb6 16.8k (r01_mb_sweep.log); round 2
    # in-place CE backward freed the second ~7 GB dlogits buffer, so mb7
    # fits (peak 277.6 GiB) and measures +1.2% over mb6 (20,789 vs 20,542
    # same box); mb8 still OOMs on activations.
    args = ap.parse_args()

    if int(os.environ.get("WORLD_SIZE", "1")) > 1:
        os.environ.setdefault("KF_COMM_STATS", "1")  # comm/overlap stats
    rank, world, device = kdist.init_distributed()
    if world != args.gpus and rank == 0:
        print(f"# note: WORLD_SIZE={world} != --gpus {args.gpus}; using {world}",
              file=sys.stderr)
    n = world
    print(f"# rank={rank}/{world} device={device} "
          f"backend={'nccl' if device.type == 'cuda' and world > 1 else ('gloo' if world > 1 else 'none')} "
          f"bucket_mb={os.environ.get('KF_DDP_BUCKET_MB', '64')} "
          f"zero={os.environ.get('KF_ZERO') == '1'}", file=sys.stderr,
          flush=True)

    torch.manual_seed(1234 + rank)
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    model = build_model(args.model, device=device, dtype=dtype)
    cfg = model.cfg
    # KF_ZERO=1 opts into the ZeRO-1 sharded optimizer (parallel/zero.py);
    # default stays bucketed-overlap DDP pending a measured comparison
    trainer = Trainer(model, TrainConfig(warmup_steps=2, lr=3e-4),
                      zero=os.environ.get("KF_ZERO") == "1")

    B, S = args.micro_batch, args.seq_len
    tokens = torch.randint(0, cfg.vocab_size, (B, S), device=device)
    if hasattr(cfg, "n_classes"):  # bert-family classify head
        targets = torch.randint(0, cfg.n_classes, (B,), device=device)
    else:
        targets = torch.randint(0, cfg.vocab_size, (B, S), device=device)

    t_start = time.time()
    for _ in range(args.warmup):
        loss = trainer.step(tokens, targets)
    kdist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.time()
    comm_ms = []
    for _ in range(args.steps):
        loss = trainer.step(tokens, targets)
        c = getattr(trainer.ddp, "last_comm_ms", None)
        if c is not None:
            comm_ms.append(c)
    kdist.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t1 = time.time()

    elapsed = t1 - t0
    # MAX over ranks (all ranks hit the same barriers; reduce to be exact)
    et = torch.tensor([elapsed], dtype=torch.float64)
    if n > 1:
        import torch.distributed as dist
        if dist.get_backend() == "nccl":
            et = et.to(device)
        dist.all_reduce(et, op=dist.ReduceOp.MAX)
    elapsed = float(et.item())

    ms_per_step = elapsed / args.steps * 1000.0
    tokens_per_step = B * S * n
    value = tokens_per_step * args.steps / elapsed

    if rank == 0:
        out = {
            "metric": "pytorchjob_train_tokens_per_s",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,  # reference publishes no numbers
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": B * n,
                "seq_len": S,
                "parallelism": f"dp{n}",
                "loss": round(float(loss.item()), 4),
            },
        }
        if comm_ms:
            # GPU time the grad all-reduces occupied on the comm stream
            # (they overlap backward; comparing against ms_per_step shows
            # how much headroom the overlap has at this N / bucket size)
            out["comm_ms_per_step"] = round(sum(comm_ms) / len(comm_ms), 2)
            out["bucket_mb"] = float(os.environ.get("KF_DDP_BUCKET_MB", "64"))
        print(json.dumps(out), flush=True)
    if os.environ.get("KF_BENCH_MEM") and device.type == "cuda":
        peak = torch.cuda.max_memory_allocated(device) / (1 << 30)
        print(f"# rank {rank} peak allocated {peak:.1f} GiB", file=sys.stderr,
              flush=True)


if __name__ == "__main__":
    main()
