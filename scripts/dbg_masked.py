import sys, torch
sys.path.insert(0, "/root/repo")
from kubeflow_amd import ops
torch.manual_seed(7)
dev = torch.device("cuda", 0)
B, S, Hq, Hkv, D = 1, 128, 6, 6, 128
kv_len = 37
q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16) * 0.5
k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16) * 0.5
v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16)

def ref_over(n):
    return ops.reference.sdpa(q.float().transpose(1,2), k[:, :n].float().transpose(1,2),
                              v[:, :n].float().transpose(1,2), causal=False).transpose(1,2)

got = ops.masked_attention(q, k, v, kv_len)
for n in (37, 64, 128):
    r = ref_over(n)
    print(f"vs ref[{n}]: err {((got.float()-r).norm()/r.norm()).item():.4f}")
# causal-with-offset hypothesis: row i attends [0, min(i+37, 63)]
qf = q.float().transpose(1,2); kf = k[:, :64].float().transpose(1,2); vf = v[:, :64].float().transpose(1,2)
s = torch.einsum("bhqd,bhkd->bhqk", qf, kf) * (128 ** -0.5)
lim = torch.arange(S, device=dev).unsqueeze(1) + kv_len
mask = torch.arange(64, device=dev).unsqueeze(0) > lim.clamp(max=63)
s = s.masked_fill(mask.unsqueeze(0).unsqueeze(0), float("-inf"))
r2 = torch.softmax(s, dim=-1) @ vf
r2 = r2.transpose(1, 2)
print("vs causal+offset over 64:", ((got.float()-r2).norm()/r2.norm()).item())
# zeroed-tail check
k2, v2 = k.clone(), v.clone()
k2[:, kv_len:] = 0; v2[:, kv_len:] = 0
got2 = ops.masked_attention(q, k2, v2, kv_len)
print("zero-tail differs:", (got.float()-got2.float()).abs().max().item())
