"""Per-shape GEMM efficiency for the llama3-8b training step (mb6 x 4096).

Maps each fwd/dgrad/wgrad GEMM shape to measured TF and % of the 2.5 PF
bf16 dense peak, to locate where hipBLASLt loses the ~49% the step-level
profile shows (profiles/r02_attn_rework.md)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeflow_amd.ops import tunable as _t
_t.enable()

import torch


def bench(M, N, K, ta=False, tb=False, iters=20):
    dev = torch.device("cuda", 0)
    a = torch.randn(K if ta else M, M if ta else K, device=dev,
                    dtype=torch.bfloat16)
    b = torch.randn(N if tb else K, K if tb else N, device=dev,
                    dtype=torch.bfloat16)
    A = a.t() if ta else a
    B = b.t() if tb else b
    for _ in range(5):
        c = A @ B
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        c = A @ B
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    tf = 2 * M * N * K / dt / 1e12
    return tf, dt * 1e3


def main():
    Mtok = 6 * 4096
    H, FFN, V, QKV = 4096, 14336, 128256, 6144
    shapes = [
        # (label, M, N, K, ta, tb)
        ("fwd qkv      ", Mtok, QKV, H, False, True),
        ("fwd attn-out ", Mtok, H, H, False, True),
        ("fwd gate+up  ", Mtok, 2 * FFN, H, False, True),
        ("fwd down     ", Mtok, H, FFN, False, True),
        ("fwd lm_head  ", Mtok, V, H, False, True),
        ("dgrad qkv    ", Mtok, H, QKV, False, False),
        ("dgrad gate+up", Mtok, H, 2 * FFN, False, False),
        ("dgrad down   ", Mtok, FFN, H, False, False),
        ("dgrad lm_head", Mtok, H, V, False, False),
        ("wgrad qkv    ", QKV, H, Mtok, True, False),
        ("wgrad attn-o ", H, H, Mtok, True, False),
        ("wgrad gate+up", 2 * FFN, H, Mtok, True, False),
        ("wgrad down   ", H, FFN, Mtok, True, False),
        ("wgrad lm_head", V, H, Mtok, True, False),
    ]
    total_fl = 0.0
    total_t = 0.0
    for label, M, N, K, ta, tb in shapes:
        tf, ms = bench(M, N, K, ta, tb)
        fl = 2 * M * N * K
        total_fl += fl
        total_t += ms
        print(f"{label} M={M:6d} N={N:6d} K={K:6d} "
              f"{'T' if ta else 'N'}{'T' if tb else 'N'}: "
              f"{tf:7.0f} TF ({100*tf/2500:4.1f}% peak)  {ms:6.2f} ms",
              flush=True)
    print(f"-- aggregate: {total_fl/ (total_t/1e3) / 1e12:.0f} TF over "
          f"{total_t:.1f} ms of pure GEMM")


if __name__ == "__main__":
    main()
