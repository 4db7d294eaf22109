"""Pure-PyTorch reference implementations of the hot ops.

These are the numerics oracle for the HIP kernels (tests compare the CDNA4
kernels against these run in fp32) and the CPU fallback used when no GPU is
present (unit tests in CI run on CPU). On a GPU box the HIP kernels are the
only path — `kubeflow_amd.ops` refuses to fall back silently there.

Reference behavior anchor: the worker hot loop of a Kubeflow PyTorchJob /
InferenceService (BASELINE.json north_star; the reference repo itself contains
no kernels — SURVEY.md §2.13).
"""
from __future__ import annotations

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """y = x * rsqrt(mean(x^2, dim=-1) + eps) * weight, computed in fp32."""
    dt = x.dtype
    x32 = x.float()
    var = x32.pow(2).mean(dim=-1, keepdim=True)
    y = x32 * torch.rsqrt(var + eps)
    return (y * weight.float()).to(dt)


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    dt = x.dtype
    x32 = x.float()
    mu = x32.mean(dim=-1, keepdim=True)
    var = x32.var(dim=-1, unbiased=False, keepdim=True)
    y = (x32 - mu) * torch.rsqrt(var + eps)
    return (y * weight.float() + bias.float()).to(dt)


def rope_cos_sin(seq_len: int, head_dim: int, theta: float = 500000.0,
                 device=None, dtype=torch.float32):
    """Precomputed rotary tables (host-side per CDNA guide: no device trig).

    Returns cos, sin of shape [seq_len, head_dim//2] in fp32.
    """
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, device=device,
                                             dtype=torch.float32) / head_dim))
    t = torch.arange(seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [S, D/2]
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def rope_apply(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               pos_offset: int = 0) -> torch.Tensor:
    """Apply rotary embedding. x: [B, S, H, D]; cos/sin: [>=S, D/2] fp32.

    Pairing is (x[..., :D/2], x[..., D/2:]) — the "rotate_half" convention
    used by Llama-family models.
    """
    B, S, H, D = x.shape
    c = cos[pos_offset:pos_offset + S].view(1, S, 1, D // 2).float()
    s = sin[pos_offset:pos_offset + S].view(1, S, 1, D // 2).float()
    x32 = x.float()
    x1, x2 = x32[..., : D // 2], x32[..., D // 2:]
    out = torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)
    return out.to(x.dtype)


def sdpa(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
         causal: bool = True, scale: float | None = None) -> torch.Tensor:
    """Reference attention (materializes S×S — oracle only, fp32 math).

    q: [B, Hq, S, D]; k, v: [B, Hkv, S, D] (GQA: Hq % Hkv == 0).
    """
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    if scale is None:
        scale = D ** -0.5
    g = Hq // Hkv
    k = k.repeat_interleave(g, dim=1)
    v = v.repeat_interleave(g, dim=1)
    scores = torch.einsum("bhqd,bhkd->bhqk", q.float(), k.float()) * scale
    if causal:
        Sk = k.shape[2]
        mask = torch.ones(S, Sk, dtype=torch.bool, device=q.device).tril(Sk - S)
        scores = scores.masked_fill(~mask, float("-inf"))
    p = torch.softmax(scores, dim=-1)
    out = torch.einsum("bhqk,bhkd->bhqd", p, v.float())
    return out.to(q.dtype)


def softmax_cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                          ignore_index: int = -100):
    """Mean CE over non-ignored targets, fp32 math. logits [T, V], targets [T]."""
    return torch.nn.functional.cross_entropy(
        logits.float(), targets, ignore_index=ignore_index)


def adamw_step(p32: torch.Tensor, g: torch.Tensor, m: torch.Tensor,
               v: torch.Tensor, lr: float, beta1: float, beta2: float,
               eps: float, weight_decay: float, step: int):
    """Decoupled AdamW on fp32 master weights (in-place); returns nothing.

    Matches torch.optim.AdamW semantics with bias correction.
    """
    g32 = g.float()
    m.mul_(beta1).add_(g32, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g32, g32, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (v / bc2).sqrt_().add_(eps)
    p32.mul_(1 - lr * weight_decay)
    p32.addcdiv_(m / bc1, denom, value=-lr)
