"""Ablation timing of the v4 attention forward kernel (guide m164 method).

Modes: 0=full, 1=no-softmax, 2=no-PV, 3=no-softmax+no-PV, 4=no-staging.
Marginal costs: SM = t(0)-t(1); PV = t(0)-t(2); staging = t(0)-t(4).
"""
import ctypes
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from kubeflow_amd.ops import _backend


def main():
    B, Hq, Hkv, S, D = 4, 32, 8, 4096, 128
    iters = 20
    lib = _backend.require()
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    o = torch.empty_like(q)
    lse = torch.empty(B, Hq, S, dtype=torch.float32, device=dev)
    flops = B * Hq * 2 * (S * S / 2) * D * 2
    stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)

    def run(mode):
        rc = lib.kf_attn_fwd4_abl(
            ctypes.c_int(mode), ctypes.c_void_p(o.data_ptr()),
            ctypes.cast(lse.data_ptr(), ctypes.POINTER(ctypes.c_float)),
            ctypes.c_void_p(q.data_ptr()), ctypes.c_void_p(k.data_ptr()),
            ctypes.c_void_p(v.data_ptr()),
            ctypes.c_int64(B), ctypes.c_int64(S), ctypes.c_int64(Hq),
            ctypes.c_int64(Hkv), ctypes.c_int64(D),
            ctypes.c_int64(0), ctypes.c_int64(0),
            ctypes.c_float(D ** -0.5), ctypes.c_int(1), stream)
        assert rc == 0, rc

    times = {}
    for mode in (0, 1, 2, 3, 4):
        for _ in range(3):
            run(mode)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(iters):
            run(mode)
        torch.cuda.synchronize()
        t = (time.time() - t0) / iters
        times[mode] = t
        print(f"mode {mode}: {t*1000:.3f} ms  ({flops/t/1e12:.1f} TF-equiv)")
    t0 = times[0]
    print(f"marginal: softmax {1e3*(t0-times[1]):.3f} ms | "
          f"PV {1e3*(t0-times[2]):.3f} ms | "
          f"SM+PV {1e3*(t0-times[3]):.3f} ms | "
          f"staging {1e3*(t0-times[4]):.3f} ms")


if __name__ == "__main__":
    main()
