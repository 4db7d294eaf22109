import sys, torch
sys.path.insert(0, "/root/repo")
from kubeflow_amd import ops
torch.manual_seed(7)
dev = torch.device("cuda", 0)
B, S, Hq, Hkv, D = 2, 128, 6, 6, 128
kv_len = 37
q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16) * 0.5
k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16) * 0.5
v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)
got = ops.masked_attention(q, k, v, kv_len)
ref = ops.reference.sdpa(q.float().transpose(1,2), k[:, :kv_len].float().transpose(1,2),
                         v[:, :kv_len].float().transpose(1,2), causal=False).transpose(1,2)
d = (got.float() - ref).permute(1, 0, 2, 3).reshape(S, -1)
rf = ref.permute(1, 0, 2, 3).reshape(S, -1)
err_row = d.norm(dim=1) / rf.norm(dim=1).clamp(min=1e-6)
print("total err", ((got.float()-ref).norm()/ref.norm()).item())
print("per-row err head:", err_row[:8].tolist())
print("per-row err tail:", err_row[-8:].tolist())
bad = (err_row > 0.05).nonzero().flatten().tolist()
print("bad rows:", bad[:20], "count", len(bad))
