"""Model registry for PyTorchJob / InferenceService / Katib specs.

Jobs name models the way Kubeflow CRs name images: a string key in the spec.
"""
from __future__ import annotations

from typing import Callable, Dict

from .llama import (LlamaConfig, LlamaModel, llama3_8b, llama3_1b,
                    llama3_70b, llama_moe_3b, llama_moe_tiny, llama_tiny,
                    llama_tiny_mha)
from .bert import (BertConfig, BertClassifier, bert_base, bert_base_hd128,
                   bert_tiny)
from .mlp import MnistMLP

__all__ = [
    "LlamaConfig", "LlamaModel", "llama3_8b", "llama3_1b", "llama_tiny",
    "BertConfig", "BertClassifier", "bert_base", "bert_base_hd128",
    "bert_tiny", "MnistMLP", "build_model", "MODEL_REGISTRY",
]


def _llama(cfg_fn):
    def build(device=None, dtype=None, **kw):
        import torch
        return LlamaModel(cfg_fn(), device=device,
                          dtype=dtype or torch.bfloat16, tp=kw.get("tp"),
                          sp=kw.get("sp"), ep=kw.get("ep"),
                          cp=kw.get("cp"))
    return build


def _bert(cfg_fn):
    def build(device=None, dtype=None, **kw):
        import torch
        return BertClassifier(cfg_fn(), device=device,
                              dtype=dtype or torch.bfloat16)
    return build


def _mlp(device=None, dtype=None, **kw):
    m = MnistMLP()
    if device is not None:
        m = m.to(device)
    return m


MODEL_CONFIGS: Dict[str, Callable] = {
    "llama3-8b": llama3_8b,
    "llama3-1b": llama3_1b,
    "llama3-70b": llama3_70b,
    "llama-tiny": llama_tiny,
    "llama-tiny-mha": llama_tiny_mha,
    "llama-moe-tiny": llama_moe_tiny,
    "llama-moe-3b": llama_moe_3b,
    "bert-base": bert_base,
    "bert-base-hd128": bert_base_hd128,
    "bert-tiny": bert_tiny,
}


def model_config(name: str):
    """The config object a registered model is built from (no weights)."""
    if name not in MODEL_CONFIGS:
        raise KeyError(f"no config for model {name!r}; "
                       f"known: {sorted(MODEL_CONFIGS)}")
    return MODEL_CONFIGS[name]()


MODEL_REGISTRY: Dict[str, Callable] = {
    "llama3-8b": _llama(llama3_8b),
    "llama3-1b": _llama(llama3_1b),
    "llama3-70b": _llama(llama3_70b),
    "llama-tiny": _llama(llama_tiny),
    "llama-tiny-mha": _llama(llama_tiny_mha),
    "llama-moe-tiny": _llama(llama_moe_tiny),
    "llama-moe-3b": _llama(llama_moe_3b),
    "bert-base": _bert(bert_base),
    "bert-base-hd128": _bert(bert_base_hd128),
    "bert-tiny": _bert(bert_tiny),
    "mnist-mlp": _mlp,
}


def build_model(name: str, device=None, dtype=None, **kw):
    if name not in MODEL_REGISTRY:
        raise KeyError(f"unknown model {name!r}; known: {sorted(MODEL_REGISTRY)}")
    if (kw.get("tp") is not None or kw.get("sp") is not None
            or kw.get("ep") is not None) and not name.startswith("llama"):
        raise ValueError(f"tensor/sequence/expert parallelism is implemented "
                         f"for the llama family only, not {name!r}")
    return MODEL_REGISTRY[name](device=device, dtype=dtype, **kw)
