"""BERT-base encoder for Katib fine-tune trials (BASELINE config 4).

Classification fine-tune shape: hidden 768, 12 layers, 12 heads — head_dim 64,
so attention uses the reference path on CPU and torch SDPA-free math; for the
GPU trial workload we use a 128-head-dim variant ("bert-base-hd128") so the
CDNA4 flash kernel (D=128 contract) carries the hot loop. LayerNorm runs
through the hand-written HIP layernorm kernel.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from kubeflow_amd import ops


@dataclass
class BertConfig:
    name: str = "bert-base"
    vocab_size: int = 30522
    hidden_size: int = 768
    n_layers: int = 12
    n_heads: int = 12
    head_dim: int = 64
    ffn_dim: int = 3072
    max_seq_len: int = 512
    n_classes: int = 2
    norm_eps: float = 1e-12
    init_std: float = 0.02


def bert_base() -> BertConfig:
    return BertConfig()


def bert_base_hd128() -> BertConfig:
    """BERT-base-sized encoder with 128-dim heads (6 heads) — the GPU trial
    config: same parameter count/shape class, hot loop on the HIP kernels."""
    return BertConfig(name="bert-base-hd128", n_heads=6, head_dim=128)


def bert_tiny() -> BertConfig:
    return BertConfig(name="bert-tiny", vocab_size=512, hidden_size=128,
                      n_layers=2, n_heads=2, head_dim=64, ffn_dim=256,
                      max_seq_len=128)


class KfLayerNorm(nn.Module):
    """LayerNorm over the HIP kernel (CPU falls back to reference)."""

    def __init__(self, dim: int, eps: float = 1e-12):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x):
        return ops.layer_norm(x, self.weight, self.bias, self.eps)


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        h, hd = cfg.hidden_size, cfg.n_heads * cfg.head_dim
        self.cfg = cfg
        self.wqkv = nn.Linear(h, 3 * hd)
        self.wo = nn.Linear(hd, h)
        self.ln1 = KfLayerNorm(h, eps=cfg.norm_eps)
        self.fc1 = nn.Linear(h, cfg.ffn_dim)
        self.fc2 = nn.Linear(cfg.ffn_dim, h)
        self.ln2 = KfLayerNorm(h, eps=cfg.norm_eps)

    def forward(self, x, kv_len=None):
        cfg = self.cfg
        B, S, _ = x.shape
        q, k, v = self.wqkv(x).split(cfg.n_heads * cfg.head_dim, dim=-1)
        q = q.view(B, S, cfg.n_heads, cfg.head_dim)
        k = k.view(B, S, cfg.n_heads, cfg.head_dim)
        v = v.view(B, S, cfg.n_heads, cfg.head_dim)
        if kv_len is not None and kv_len < S:
            # padded serving batch: attend only the real kv_len tokens
            o = ops.masked_attention(q, k, v, kv_len)
        else:
            o = ops.flash_attention(q, k, v, causal=False)
        o = self.wo(o.reshape(B, S, -1))
        x = self.ln1(x + o)
        return self.ln2(x + self.fc2(F.gelu(self.fc1(x))))


class BertClassifier(nn.Module):
    def __init__(self, cfg: BertConfig, device=None, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        with torch.device(device if device is not None else "cpu"):
            self.tok_embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
            self.pos_embed = nn.Embedding(cfg.max_seq_len, cfg.hidden_size)
            self.embed_ln = KfLayerNorm(cfg.hidden_size, eps=cfg.norm_eps)
            self.layers = nn.ModuleList(
                [BertLayer(cfg) for _ in range(cfg.n_layers)])
            self.classifier = nn.Linear(cfg.hidden_size, cfg.n_classes)
        if device is not None or dtype is not None:
            self.to(device=device, dtype=dtype)
        self.init_weights()

    @torch.no_grad()
    def init_weights(self):
        for p in self.parameters():
            if p.dim() >= 2:
                p.normal_(0.0, self.cfg.init_std)
            else:
                p.zero_()
        for mod in self.modules():
            if isinstance(mod, (nn.LayerNorm, KfLayerNorm)):
                mod.weight.fill_(1.0)
                mod.bias.zero_()

    def forward(self, tokens, targets=None, kv_len=None):
        B, S = tokens.shape
        pos = torch.arange(S, device=tokens.device)
        x = self.embed_ln(self.tok_embed(tokens) + self.pos_embed(pos))
        for layer in self.layers:
            x = layer(x, kv_len=kv_len)
        logits = self.classifier(x[:, 0])  # [CLS]
        if targets is None:
            return logits
        return F.cross_entropy(logits.float(), targets)
