"""Checkpoint IO microbench: libkfio threaded raw writes vs torch.save.
Usage: python scripts/ckpt_io_bench.py [GiB]"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from kubeflow_amd.utils import fastio

gib = float(sys.argv[1]) if len(sys.argv) > 1 else 4.0
n = int(gib * (1 << 30) // 4)
t = torch.randn(n)
print(f"tensor: {gib:.1f} GiB fp32, libkfio native: {fastio.native_available()}")

p1, p2 = "/tmp/ckpt_ts.pt", "/tmp/ckpt_kf.bin"
t0 = time.perf_counter(); torch.save({"flat": t}, p1); dt_ts = time.perf_counter() - t0
t0 = time.perf_counter(); fastio.write_tensor(p2, t); dt_kf = time.perf_counter() - t0
print(f"write: torch.save {gib/dt_ts:.2f} GiB/s ({dt_ts:.2f}s) | "
      f"libkfio {gib/dt_kf:.2f} GiB/s ({dt_kf:.2f}s) | {dt_ts/dt_kf:.1f}x")

o = torch.empty_like(t)
t0 = time.perf_counter(); torch.load(p1, weights_only=False); dt_tl = time.perf_counter() - t0
t0 = time.perf_counter(); fastio.read_into(p2, o); dt_kl = time.perf_counter() - t0
assert torch.equal(t, o)
print(f"read:  torch.load {gib/dt_tl:.2f} GiB/s ({dt_tl:.2f}s) | "
      f"libkfio {gib/dt_kl:.2f} GiB/s ({dt_kl:.2f}s) | {dt_tl/dt_kl:.1f}x")
os.unlink(p1); os.unlink(p2)
