"""Sync vs overlapped checkpoint timing (weak item 8 evidence)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from kubeflow_amd.models import build_model
from kubeflow_amd.runtime import Trainer, TrainConfig
from kubeflow_amd.runtime import checkpoint as ckpt


def main():
    model_name = sys.argv[1] if len(sys.argv) > 1 else "llama3-1b"
    root = sys.argv[2] if len(sys.argv) > 2 else "/tmp/ckpt-bench"
    dev = torch.device("cuda", 0)
    m = build_model(model_name, device=dev, dtype=torch.bfloat16)
    tr = Trainer(m, TrainConfig(warmup_steps=1))
    toks = torch.randint(0, m.cfg.vocab_size, (2, 2048), device=dev)
    for _ in range(2):
        tr.step(toks, toks)
    torch.cuda.synchronize()
    state_gb = (tr.flat.data.numel() * 2 + tr.p32.numel() * 12) / 1e9

    t0 = time.time()
    ckpt.save(tr, os.path.join(root, "sync"), model_name, 0, 1)
    t_sync = time.time() - t0

    saver = ckpt.AsyncSave()
    t0 = time.time()
    saver.save(tr, os.path.join(root, "async"), model_name, 0, 1)
    t_stall = time.time() - t0          # training-visible stall
    steps = 0
    t1 = time.time()
    while saver._thread is not None and saver._thread.is_alive():
        tr.step(toks, toks)             # training continues under the write
        steps += 1
    torch.cuda.synchronize()
    saver.wait()
    t_bg = time.time() - t0
    print(f"{model_name}: state {state_gb:.1f} GB | sync save {t_sync:.2f} s"
          f" | async stall {t_stall:.2f} s ({100*t_stall/t_sync:.0f}% of"
          f" sync) | write drained after {t_bg:.2f} s with {steps} train"
          " steps overlapped", flush=True)


if __name__ == "__main__":
    main()
