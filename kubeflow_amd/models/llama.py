"""Llama-3-family decoder for the PyTorchJob flagship workload.

MI355X-first design decisions:
  * bshd tensor layout end-to-end (no transposes around attention — the HIP
    flash kernel consumes [B,S,H,D] directly).
  * fused QKV and fused gate+up projections: plain library GEMMs go through
    torch.nn.functional.linear (hipBLASLt); everything between GEMMs is a
    hand-written CDNA4 kernel (kubeflow_amd.ops): RMSNorm, RoPE, flash
    attention, fused softmax-CE over the 128256 vocab.
  * bf16 parameters (fp32 master copies live in the optimizer, not here).

Reference-behavior anchor: the model that BASELINE.json's PyTorchJob metric
is quoted on (Llama-3-8B: hidden 4096, 32 layers, 32 q / 8 kv heads,
ffn 14336, vocab 128256).
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from kubeflow_amd import ops
from kubeflow_amd.parallel import sp as spmod
from kubeflow_amd.parallel import tp as tpmod


@dataclass
class LlamaConfig:
    name: str = "llama3-8b"
    vocab_size: int = 128256
    hidden_size: int = 4096
    n_layers: int = 32
    n_heads: int = 32
    n_kv_heads: int = 8
    head_dim: int = 128
    ffn_dim: int = 14336
    rope_theta: float = 500000.0
    max_seq_len: int = 8192
    norm_eps: float = 1e-5
    init_std: float = 0.02
    n_experts: int = 0   # 0 = dense MLP; >0 = mixture-of-experts blocks
    top_k: int = 2       # experts per token (MoE only)

    @property
    def n_params(self) -> int:
        h, v = self.hidden_size, self.vocab_size
        qkv = h * (self.n_heads + 2 * self.n_kv_heads) * self.head_dim
        o = self.n_heads * self.head_dim * h
        mlp = 3 * h * self.ffn_dim * max(1, self.n_experts)
        gate = h * self.n_experts
        return 2 * v * h + self.n_layers * (qkv + o + mlp + gate + 2 * h) + h


def llama3_8b() -> LlamaConfig:
    return LlamaConfig()


def llama3_1b() -> LlamaConfig:
    """Scaled-down config for quick GPU smoke/bench runs."""
    return LlamaConfig(name="llama3-1b", hidden_size=2048, n_layers=16,
                       n_heads=16, n_kv_heads=8, ffn_dim=8192)


def llama3_70b() -> LlamaConfig:
    """Llama-3-70B (70.6B params): needs the model-parallel strategies.
    Replicated training state is 131 GiB bf16 weights + 131 GiB grads +
    788 GiB fp32 master/moments — far past 288 GB/GPU for DDP and still
    ~360 GiB/GPU under ZeRO-1. TP8 or PP8 shard ALL of it 1/8 per GPU
    (~131 GiB total + activations), which fits with room for long
    sequences."""
    return LlamaConfig(name="llama3-70b", hidden_size=8192, n_layers=80,
                       n_heads=64, n_kv_heads=8, ffn_dim=28672)


def llama_tiny(vocab: int = 512) -> LlamaConfig:
    """Tiny config for tests. head_dim stays 128 (HIP kernel contract)."""
    return LlamaConfig(name="llama-tiny", vocab_size=vocab, hidden_size=256,
                       n_layers=2, n_heads=2, n_kv_heads=1, ffn_dim=512,
                       max_seq_len=512)


def llama_moe_tiny(vocab: int = 512) -> LlamaConfig:
    """Tiny mixture-of-experts config (4 experts, top-2) for the EP tests."""
    return LlamaConfig(name="llama-moe-tiny", vocab_size=vocab,
                       hidden_size=256, n_layers=2, n_heads=2, n_kv_heads=1,
                       ffn_dim=512, max_seq_len=512, n_experts=4, top_k=2)


def llama_moe_3b(vocab: int = 128256) -> LlamaConfig:
    """Mid-size MoE (8 experts, top-2, ~3.4B params, ~0.9B active/token) —
    a single-GPU-trainable Mixtral-style config for measuring the MoE
    train path at real scale (the tiny config only covers numerics)."""
    return LlamaConfig(name="llama-moe-3b", vocab_size=vocab,
                       hidden_size=2048, n_layers=16, n_heads=16,
                       n_kv_heads=4, ffn_dim=2048, max_seq_len=8192,
                       n_experts=8, top_k=2)


def llama_tiny_mha(vocab: int = 512) -> LlamaConfig:
    """Tiny MHA config (kv == q heads) whose head/ffn counts divide by 2 —
    used by the tensor-parallel tests and examples."""
    return LlamaConfig(name="llama-tiny-mha", vocab_size=vocab,
                       hidden_size=256, n_layers=2, n_heads=2, n_kv_heads=2,
                       ffn_dim=512, max_seq_len=512)


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.eps = eps

    def forward(self, x):
        return ops.rms_norm(x, self.weight, self.eps)


class MoEMLP(nn.Module):
    """Top-k gated mixture of SwiGLU experts (replaces the dense MLP when
    cfg.n_experts > 0). Expert banks are [E_local, ...] parameter tensors;
    under expert parallelism (parallel/ep.py) each rank owns a contiguous
    E/N block and tokens travel by all-gather / reduce-scatter."""

    def __init__(self, cfg: LlamaConfig, ep=None):
        super().__init__()
        h, f, E = cfg.hidden_size, cfg.ffn_dim, cfg.n_experts
        self.cfg = cfg
        self.ep = ep if (ep is not None and ep.world > 1) else None
        n = self.ep.world if self.ep else 1
        if E % n:
            raise ValueError(f"EP degree {n} must divide n_experts ({E})")
        self.local_e = E // n
        self.e_lo = (self.ep.rank if self.ep else 0) * self.local_e
        self.gate = nn.Linear(h, E, bias=False)  # replicated router
        self.experts_w13 = nn.Parameter(torch.empty(self.local_e, 2 * f, h))
        self.experts_w2 = nn.Parameter(torch.empty(self.local_e, h, f))
        if self.ep:
            self.experts_w13._ep_local = True
            self.experts_w2._ep_local = True

    def forward(self, x):
        from kubeflow_amd.parallel import ep as epmod
        cfg = self.cfg
        B, S, h = x.shape
        xf = x.reshape(B * S, h)
        if self.ep:
            xf = epmod.all_gather_cat(xf, self.ep)
        # router runs (replicated weights) on whatever token set is local
        logits = torch.nn.functional.linear(xf, self.gate.weight)
        topv, topi = logits.topk(cfg.top_k, dim=-1)
        weights = torch.softmax(topv.float(), dim=-1).to(x.dtype)
        out = torch.zeros_like(xf)
        for el in range(self.local_e):
            eg = self.e_lo + el
            sel, slot = (topi == eg).nonzero(as_tuple=True)
            if sel.numel() == 0:
                continue
            rows = xf[sel]
            y = ops.swiglu(
                torch.nn.functional.linear(rows, self.experts_w13[el]))
            y = torch.nn.functional.linear(y, self.experts_w2[el])
            out.index_add_(0, sel, y * weights[sel, slot].unsqueeze(1))
        if self.ep:
            out = epmod.reduce_scatter_sum(out, self.ep)
        return out.view(B, S, h)

    def decode_dense(self, x):
        """Capture-safe decode dispatch: every expert runs on every token
        and the top-k gate weights (scattered to a dense [N, E] mask)
        select the sum. Data-dependent routing (nonzero/index_add) would
        replay the CAPTURED batch's routing under hipGraph; at decode
        batches (N <= 32) running all experts is weight-BW-bound like the
        dense MLP, so this is what makes MoE decode graphable. EP stays
        on the routed path (serving is TP=1/EP=1)."""
        if self.ep:
            return self.forward(x)
        cfg = self.cfg
        N, S, h = x.shape
        xf = x.reshape(N * S, h)
        logits = torch.nn.functional.linear(xf, self.gate.weight)
        topv, topi = logits.topk(cfg.top_k, dim=-1)
        w = torch.softmax(topv.float(), dim=-1).to(x.dtype)
        gatew = torch.zeros(N * S, cfg.n_experts, dtype=x.dtype,
                            device=x.device)
        gatew.scatter_(1, topi, w)
        out = torch.zeros_like(xf)
        for el in range(self.local_e):
            y = ops.swiglu(ops.skinny_linear(xf, self.experts_w13[el]))
            y = ops.skinny_linear(y, self.experts_w2[el])
            out = out + gatew[:, self.e_lo + el].unsqueeze(1) * y
        return out.view(N, S, h)


class LlamaBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp=None, sp=None, ep=None,
                 cp=None):
        super().__init__()
        h, d = cfg.hidden_size, cfg.head_dim
        self.cfg = cfg
        self.tp = tp if (tp is not None and tp.world > 1) else None
        self.sp = sp if (sp is not None and sp.world > 1) else None
        self.cp = cp if (cp is not None and cp.world > 1) else None
        if sum(x is not None for x in (self.tp, self.sp, self.cp)) > 1:
            raise ValueError("tp / ulysses-sp / ring-cp are mutually "
                             "exclusive per block")
        if ep is not None and cfg.n_experts <= 0:
            raise ValueError("expert parallelism requires an MoE config "
                             "(n_experts > 0)")
        if ep is not None and (self.tp is not None or self.sp is not None):
            raise ValueError("ep composes with DP only in v1")
        if self.sp is not None and (cfg.n_heads % self.sp.world or
                                    cfg.n_kv_heads % self.sp.world):
            raise ValueError(
                f"ulysses degree {self.sp.world} must divide heads "
                f"({cfg.n_heads}/{cfg.n_kv_heads})")
        n = self.tp.world if self.tp else 1
        if cfg.n_heads % n or cfg.n_kv_heads % n or cfg.ffn_dim % n:
            raise ValueError(f"TP degree {n} must divide heads "
                             f"({cfg.n_heads}/{cfg.n_kv_heads}) and ffn "
                             f"({cfg.ffn_dim})")
        # local (per-TP-rank) shard sizes; n=1 reproduces the full model
        self.hq, self.hkv = cfg.n_heads // n, cfg.n_kv_heads // n
        self.ffn = cfg.ffn_dim // n
        self.attn_norm = RMSNorm(h, cfg.norm_eps)
        self.wqkv = nn.Linear(h, (self.hq + 2 * self.hkv) * d, bias=False)
        self.wo = nn.Linear(self.hq * d, h, bias=False)
        self.mlp_norm = RMSNorm(h, cfg.norm_eps)
        if cfg.n_experts > 0:
            self.moe = MoEMLP(cfg, ep=ep)
            self.w13 = self.w2 = None
        else:
            self.moe = None
            self.w13 = nn.Linear(h, 2 * self.ffn, bias=False)  # gate ++ up
            self.w2 = nn.Linear(self.ffn, h, bias=False)
        if self.tp:
            for lin in (self.wqkv, self.wo, self.w13, self.w2):
                lin.weight._tp_sharded = True

    def forward(self, x, cos, sin, pos_offset: int = 0, kv_cache=None):
        cfg = self.cfg
        B, S, h = x.shape
        nx = self.attn_norm(x)
        if self.tp:
            nx = tpmod.copy_to(nx, self.tp)
        qkv = F.linear(nx, self.wqkv.weight)
        if kv_cache is not None:
            q, k, v = qkv.split([self.hq * cfg.head_dim,
                                 self.hkv * cfg.head_dim,
                                 self.hkv * cfg.head_dim], dim=-1)
            q = q.view(B, S, self.hq, cfg.head_dim)
            k = k.view(B, S, self.hkv, cfg.head_dim)
            v = v.view(B, S, self.hkv, cfg.head_dim)
            q, k = ops.rope(q, k, cos, sin, pos_offset)
            k, v = kv_cache.update(k, v, pos_offset)
            o = ops.flash_attention(q, k, v, causal=(S > 1))
            o = o.reshape(B, S, self.hq * cfg.head_dim)
        elif self.sp is not None:
            # ulysses: seq-sharded in, full-seq/local-heads attention,
            # seq-sharded out (parallel/sp.py); weights stay replicated so
            # the residual addmm fusion below applies unchanged
            n = self.sp.world
            qkv = spmod.scatter_heads_gather_seq(qkv, self.sp, self.hq,
                                                 self.hkv, cfg.head_dim)
            o = ops.fused_qkv_attention(qkv, cos, sin, self.hq // n,
                                        self.hkv // n, cfg.head_dim)
            o = spmod.gather_heads_scatter_seq(o, self.sp)
        elif self.cp is not None:
            # ring/context parallel: this rank's contiguous chunk attends
            # the full ring-rotated sequence (parallel/ring.py); rope uses
            # the chunk's GLOBAL positions
            from kubeflow_amd.parallel.ring import ring_attention
            q, kk, vv = qkv.split([self.hq * cfg.head_dim,
                                   self.hkv * cfg.head_dim,
                                   self.hkv * cfg.head_dim], dim=-1)
            q = q.view(B, S, self.hq, cfg.head_dim)
            kk = kk.view(B, S, self.hkv, cfg.head_dim)
            vv = vv.view(B, S, self.hkv, cfg.head_dim)
            q, kk = ops.rope(q, kk, cos, sin, self.cp.rank * S)
            o = ring_attention(q, kk, vv, self.cp, causal=True)
            o = o.reshape(B, S, self.hq * cfg.head_dim)
        else:
            o = ops.fused_qkv_attention(qkv, cos, sin, self.hq,
                                        self.hkv, cfg.head_dim)
        if self.tp:
            # row-parallel wo: reduce the partial BEFORE the residual add
            # (the addmm fusion below would add the residual tp.world times)
            attn = tpmod.reduce_from(
                torch.mm(o.view(B * S, -1), self.wo.weight.t()), self.tp)
            x = (x.view(-1, h) + attn).view(B, S, h)
            ny = tpmod.copy_to(self.mlp_norm(x), self.tp)
            y = ops.swiglu(F.linear(ny, self.w13.weight))
            mlp = tpmod.reduce_from(
                torch.mm(y.view(B * S, self.ffn), self.w2.weight.t()),
                self.tp)
            return (x.view(-1, h) + mlp).view(B, S, h)
        # residuals fused into the GEMM epilogue (addmm: C = input + A @ B)
        x = torch.addmm(x.view(-1, h), o.view(B * S, -1),
                        self.wo.weight.t()).view(B, S, h)
        if self.moe is not None:
            return x + self.moe(self.mlp_norm(x))
        y = ops.swiglu(F.linear(self.mlp_norm(x), self.w13.weight))
        return torch.addmm(x.view(-1, h), y.view(B * S, cfg.ffn_dim),
                           self.w2.weight.t()).view(B, S, h)


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig, device=None, dtype=torch.bfloat16,
                 tp=None, sp=None, ep=None, cp=None):
        super().__init__()
        self.cfg = cfg
        self.tp = tp
        self.sp = sp
        self.ep = ep
        factory = dict(device=device, dtype=dtype)
        with torch.device(device if device is not None else "cpu"):
            self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
            self.layers = nn.ModuleList(
                [LlamaBlock(cfg, tp=tp, sp=sp, ep=ep, cp=cp)
                 for _ in range(cfg.n_layers)])
            self.final_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
            self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size,
                                     bias=False)
        self.to(**{k: v for k, v in factory.items() if v is not None})
        cos, sin = ops.rope_cos_sin(cfg.max_seq_len, cfg.head_dim,
                                    cfg.rope_theta, device=device)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        self.init_weights()

    @torch.no_grad()
    def init_weights(self):
        std = self.cfg.init_std
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                p.normal_(0.0, std)
                # scaled init for residual-out projections (GPT-2 style)
                if name.endswith(("wo.weight", "w2.weight", "experts_w2")):
                    p.mul_(1.0 / math.sqrt(2 * self.cfg.n_layers))
            else:
                p.fill_(1.0)

    def forward(self, tokens: torch.Tensor, targets: torch.Tensor = None,
                pos_offset: int = 0, kv_caches=None):
        """tokens [B,S] int64 -> loss (if targets) else logits [B,S,V]."""
        x = self.embed(tokens)
        cos, sin = self.rope_cos, self.rope_sin
        for i, layer in enumerate(self.layers):
            cache = kv_caches[i] if kv_caches is not None else None
            x = layer(x, cos, sin, pos_offset, cache)
        x = self.final_norm(x)
        logits = F.linear(x, self.lm_head.weight)
        if targets is None:
            return logits
        T = logits.shape[0] * logits.shape[1]
        return ops.cross_entropy(logits.view(T, -1), targets.view(T))

    def flops_per_token(self, seq_len: int) -> float:
        """Approximate training FLOPs/token (fwd+bwd) for MFU reporting."""
        n = self.cfg.n_params
        # attention: 2*2*S*D per head-pair term, causal halves it
        attn = (3 * 2 * 2 * self.cfg.n_layers * self.cfg.n_heads *
                self.cfg.head_dim * seq_len / 2)
        return 6 * n + attn


class LlamaStage(nn.Module):
    """One pipeline stage of LlamaModel (parallel/pp.py contract): stage 0
    owns the embedding, the last stage owns final_norm + lm_head + loss,
    every stage owns a contiguous slice of the blocks."""

    def __init__(self, cfg: LlamaConfig, stage: int, n_stages: int,
                 device=None, dtype=torch.bfloat16):
        super().__init__()
        from kubeflow_amd.parallel.pp import layer_range
        self.cfg = cfg
        self.stage_idx, self.n_stages = stage, n_stages
        self.layer_lo, self.layer_hi = layer_range(cfg.n_layers, stage,
                                                   n_stages)
        self.is_first = stage == 0
        self.is_last = stage == n_stages - 1
        factory = dict(device=device, dtype=dtype)
        with torch.device(device if device is not None else "cpu"):
            if self.is_first:
                self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
            self.layers = nn.ModuleList(
                [LlamaBlock(cfg)
                 for _ in range(self.layer_lo, self.layer_hi)])
            if self.is_last:
                self.final_norm = RMSNorm(cfg.hidden_size, cfg.norm_eps)
                self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size,
                                         bias=False)
        self.to(**{k: v for k, v in factory.items() if v is not None})
        cos, sin = ops.rope_cos_sin(cfg.max_seq_len, cfg.head_dim,
                                    cfg.rope_theta, device=device)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        LlamaModel.init_weights(self)  # same init rule, stage-local params

    def forward(self, x, targets: torch.Tensor = None):
        if self.is_first:
            x = self.embed(x)  # x is the token ids here
        for layer in self.layers:
            x = layer(x, self.rope_cos, self.rope_sin)
        if not self.is_last:
            return x
        x = self.final_norm(x)
        logits = F.linear(x, self.lm_head.weight)
        if targets is None:
            return logits
        T = logits.shape[0] * logits.shape[1]
        return ops.cross_entropy(logits.view(T, -1), targets.view(T))


def stage_state_dict(full_sd: dict, cfg: LlamaConfig, stage: int,
                     n_stages: int) -> dict:
    """Slice a full LlamaModel state dict to one LlamaStage (tests,
    checkpoint import)."""
    from kubeflow_amd.parallel.pp import layer_range
    lo, hi = layer_range(cfg.n_layers, stage, n_stages)
    out = {}
    for k, v in full_sd.items():
        if k.startswith("layers."):
            idx = int(k.split(".")[1])
            if lo <= idx < hi:
                out[k.replace(f"layers.{idx}.", f"layers.{idx - lo}.", 1)] = \
                    v.clone()
        elif k.startswith("embed."):
            if stage == 0:
                out[k] = v.clone()
        elif k.startswith(("final_norm.", "lm_head.")):
            if stage == n_stages - 1:
                out[k] = v.clone()
        else:
            out[k] = v.clone()
    return out
