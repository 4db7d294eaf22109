"""Standalone attention kernel benchmark (GPU): TFLOP/s for fwd and bwd.

Usage: python3 tests/attn_bench.py [B Hq Hkv S iters]
Shape defaults to the Llama-3-8B bench config: B=4, Hq=32, Hkv=8, S=4096.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from kubeflow_amd import ops


def main():
    args = sys.argv[1:]
    B = int(args[0]) if len(args) > 0 else 4
    Hq = int(args[1]) if len(args) > 1 else 32
    Hkv = int(args[2]) if len(args) > 2 else 8
    S = int(args[3]) if len(args) > 3 else 4096
    iters = int(args[4]) if len(args) > 4 else 20
    D = 128
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)

    # causal FLOPs: fwd = 2 matmuls * S^2/2 * D * 2 per (b,h)
    fwd_flops = B * Hq * 2 * (S * S / 2) * D * 2
    bwd_flops = fwd_flops * 2.5  # 5 matmuls vs 2

    # correctness spot-check vs fp32 reference at a smaller shape
    from kubeflow_amd.ops import reference as R
    qs = q[:1, :256, :4].detach()
    ks = k[:1, :256, :2].detach()
    vs = v[:1, :256, :2].detach()
    o_s = ops.flash_attention(qs, ks, vs, causal=True)
    ref = R.sdpa(qs.float().cpu().transpose(1, 2),
                 ks.float().cpu().transpose(1, 2),
                 vs.float().cpu().transpose(1, 2), causal=True).transpose(1, 2)
    err = ((o_s.cpu().float() - ref.float()).norm() / ref.norm()).item()
    print(f"refcheck S=256 relerr={err:.4f}", flush=True)
    assert err < 2e-2, "ATTENTION KERNEL WRONG"

    # fwd timing
    for _ in range(3):
        o = ops.flash_attention(q.detach(), k.detach(), v.detach())
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        o = ops.flash_attention(q.detach(), k.detach(), v.detach())
    torch.cuda.synchronize()
    fwd_t = (time.time() - t0) / iters
    print(f"fwd: {fwd_t*1000:.2f} ms = {fwd_flops/fwd_t/1e12:.1f} TF", flush=True)

    # fwd+bwd timing
    do = torch.randn_like(q)
    for _ in range(2):
        o = ops.flash_attention(q, k, v)
        o.backward(do)
        q.grad = k.grad = v.grad = None
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        o = ops.flash_attention(q, k, v)
        o.backward(do)
        q.grad = k.grad = v.grad = None
    torch.cuda.synchronize()
    tot_t = (time.time() - t0) / iters
    bwd_t = tot_t - fwd_t
    print(f"bwd: {bwd_t*1000:.2f} ms = {bwd_flops/bwd_t/1e12:.1f} TF "
          f"(fwd+bwd {tot_t*1000:.2f} ms)", flush=True)


if __name__ == "__main__":
    main()
