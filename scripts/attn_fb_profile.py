"""fwd+bwd attention loop for rocprof (per-kernel split)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from kubeflow_amd import ops

B, Hq, Hkv, S, D = 4, 32, 8, 4096, 128
dev = torch.device("cuda", 0)
q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16,
                requires_grad=True)
k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16,
                requires_grad=True)
v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16,
                requires_grad=True)
do = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16)
for _ in range(10):
    o = ops.flash_attention(q, k, v)
    o.backward(do)
    q.grad = k.grad = v.grad = None
torch.cuda.synchronize()
print("done")
