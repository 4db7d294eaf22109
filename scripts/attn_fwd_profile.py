"""Minimal fwd-only attention loop for rocprofv3 kernel profiling."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from kubeflow_amd import ops


def main():
    B, Hq, Hkv, S, D = 4, 32, 8, 4096, 128
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 10
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
    for _ in range(iters):
        ops.flash_attention(q, k, v, causal=True)
    torch.cuda.synchronize()
    print("done")


if __name__ == "__main__":
    main()
