"""Skinny decode-GEMM loop microbench vs hipBLASLt (per shape, M<=16).

CAVEAT (profiles/r02_skinny_gemm.md): the loop keeps W LLC-resident
(256 MB LLC) for both kernels, so big-shape hipBLASLt numbers here are
warmer than in the real decode graph — in-situ kernel times from
rocprofv3 over scripts/decode_profile.py are the deciding measurement.

Usage: python scripts/skinny_bench.py [M]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeflow_amd.ops import tunable as _t
_t.enable()

import torch  # noqa: E402

from kubeflow_amd import ops  # noqa: E402


def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 16
    dev = torch.device("cuda", 0)
    shapes = [("qkv", 6144, 4096), ("wo", 4096, 4096),
              ("w13", 28672, 4096), ("w2", 4096, 14336),
              ("lm_head", 128256, 4096)]
    for name, N, K in shapes:
        x = torch.randn(M, 1, K, device=dev, dtype=torch.bfloat16) * 0.5
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        sk = bench(lambda: ops.skinny_linear(x, w))
        bl = bench(lambda: torch.nn.functional.linear(x, w))
        gb = N * K * 2 / 1e9
        print(f"M{M} {name:8s} N{N} K{K}: skinny {sk:7.1f} us "
              f"({gb / sk * 1e3:5.2f} TB/s)  blaslt {bl:7.1f} us "
              f"({gb / bl * 1e3:5.2f} TB/s)")


if __name__ == "__main__":
    main()
