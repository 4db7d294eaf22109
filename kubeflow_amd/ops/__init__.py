"""kubeflow_amd.ops — the hot-op library (CDNA4 HIP kernels + CPU reference).

Dispatch policy:
  * CUDA (= ROCm/HIP) tensors -> hand-written gfx950 kernels in libkfops.so,
    loaded via ctypes (`_backend`). Missing library on a GPU box raises —
    no silent eager fallback (KF_NATIVE_KERNELS=0 is the explicit escape
    hatch for bisection).
  * CPU tensors -> the pure-torch reference implementations (differentiable),
    so the full stack runs in CPU-only CI.

All ops are exposed as autograd-capable functions:
  rms_norm, rope, flash_attention, cross_entropy, fused_adamw (no autograd).
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional

import torch

from . import reference
from . import _backend

__all__ = [
    "rms_norm", "layer_norm", "rope", "flash_attention", "attention_decode",
    "flash_attention_rect", "skinny_linear", "kv_store",
    "decode_rope_store", "skinny_linear_q8", "quantize_fp8_rows",
    "dequantize_fp8_rows",
    "cross_entropy", "fused_adamw", "rope_cos_sin", "native_available",
    "masked_attention",
    "swiglu", "fused_qkv_attention",
]

rope_cos_sin = reference.rope_cos_sin


def native_available() -> bool:
    return _backend.try_load() is not None


def _stream() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _p(t: Optional[torch.Tensor]) -> ctypes.c_void_p:
    return ctypes.c_void_p(0 if t is None else t.data_ptr())


def _fp(t: Optional[torch.Tensor]):
    return ctypes.cast(0 if t is None else t.data_ptr(),
                       ctypes.POINTER(ctypes.c_float))


def _ip(t: torch.Tensor):
    return ctypes.cast(t.data_ptr(), ctypes.POINTER(ctypes.c_int64))


def _ip_or_null(t):
    if t is None:
        return ctypes.cast(0, ctypes.POINTER(ctypes.c_int64))
    return ctypes.cast(t.data_ptr(), ctypes.POINTER(ctypes.c_int64))


def _i32p(t):
    if t is None:
        return ctypes.cast(0, ctypes.POINTER(ctypes.c_int32))
    return ctypes.cast(t.data_ptr(), ctypes.POINTER(ctypes.c_int32))


def _use_native(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if _backend.native_enabled():
        _backend.require()
        return True
    return False


# --------------------------------------------------------------- RMSNorm --

class _RMSNormHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        lib = _backend.require()
        shape = x.shape
        x2 = x.contiguous().view(-1, shape[-1])
        rows, cols = x2.shape
        y = torch.empty_like(x2)
        rstd = torch.empty(rows, dtype=torch.float32, device=x.device)
        _backend.check(
            lib.kf_rmsnorm_fwd(_p(y), _fp(rstd), _p(x2), _p(weight), rows,
                               cols, float(eps), _stream()), "rmsnorm_fwd")
        ctx.save_for_backward(x2, weight, rstd)
        ctx.shape = shape
        return y.view(shape)

    @staticmethod
    def backward(ctx, dy):
        lib = _backend.require()
        x2, weight, rstd = ctx.saved_tensors
        rows, cols = x2.shape
        dy2 = dy.contiguous().view(rows, cols)
        dx = torch.empty_like(x2)
        dw = torch.empty_like(weight)
        dw_part = torch.zeros(cols, dtype=torch.float32, device=x2.device)
        _backend.check(
            lib.kf_rmsnorm_bwd(_p(dx), _p(dw), _fp(dw_part), _p(dy2), _p(x2),
                               _p(weight), _fp(rstd), rows, cols, _stream()),
            "rmsnorm_bwd")
        return dx.view(ctx.shape), dw, None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5):
    if _use_native(x):
        return _RMSNormHip.apply(x, weight, eps)
    return reference.rms_norm(x, weight, eps)


# ------------------------------------------------------------------ RoPE --

class _RopeHip(torch.autograd.Function):
    """Joint rotary embedding on (q, k); in-place rotation on fresh clones."""

    @staticmethod
    def forward(ctx, q, k, cos, sin, pos_offset, positions):
        lib = _backend.require()
        B, S, Hq, D = q.shape
        Hkv = k.shape[2]
        # clone(contiguous_format) compacts strided inputs in ONE copy
        # (the QKV-split views are always strided; .contiguous().clone()
        # would copy twice)
        q = q.clone(memory_format=torch.contiguous_format)
        k = k.clone(memory_format=torch.contiguous_format)
        _backend.check(
            lib.kf_rope(_p(q), _p(k), _fp(cos), _fp(sin),
                        _ip_or_null(positions), B, S, Hq, Hkv, D,
                        Hq * D, Hkv * D, pos_offset, 0, _stream()),
            "rope_fwd")
        ctx.save_for_backward(cos, sin, *(
            [positions] if positions is not None else []))
        ctx.dims = (B, S, Hq, Hkv, D, pos_offset)
        return q, k

    @staticmethod
    def backward(ctx, dq, dk):
        lib = _backend.require()
        cos, sin = ctx.saved_tensors[0], ctx.saved_tensors[1]
        positions = ctx.saved_tensors[2] if len(ctx.saved_tensors) > 2 else None
        B, S, Hq, Hkv, D, pos_offset = ctx.dims
        dq = dq.clone(memory_format=torch.contiguous_format)
        dk = dk.clone(memory_format=torch.contiguous_format)
        _backend.check(
            lib.kf_rope(_p(dq), _p(dk), _fp(cos), _fp(sin),
                        _ip_or_null(positions), B, S, Hq, Hkv, D,
                        Hq * D, Hkv * D, pos_offset, 1, _stream()),
            "rope_bwd")
        return dq, dk, None, None, None, None


def rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
         sin: torch.Tensor, pos_offset: int = 0,
         positions: Optional[torch.Tensor] = None):
    """q [B,S,Hq,D], k [B,S,Hkv,D]; cos/sin fp32 [>=S+off, D/2].
    positions: optional int64 [B] per-row base position (serving decode)."""
    if _use_native(q):
        return _RopeHip.apply(q, k, cos, sin, pos_offset, positions)
    if positions is not None:
        qs = [reference.rope_apply(q[i:i+1], cos, sin, int(positions[i]))
              for i in range(q.shape[0])]
        ks = [reference.rope_apply(k[i:i+1], cos, sin, int(positions[i]))
              for i in range(k.shape[0])]
        return torch.cat(qs), torch.cat(ks)
    return (reference.rope_apply(q, cos, sin, pos_offset),
            reference.rope_apply(k, cos, sin, pos_offset))


def kv_store(cache_k: torch.Tensor, cache_v: torch.Tensor,
             k: torch.Tensor, v: torch.Tensor, slots: torch.Tensor,
             positions: torch.Tensor) -> None:
    """Scatter this decode step's k/v token rows into cache row
    [slot[i], position[i]] for each sequence i — one launch instead of two
    advanced-indexing kernels per layer (kv_store.hip). cache_k/v
    [SLOTS,SMAX,Hkv,D]; k/v [N,1,Hkv,D] (or [N,Hkv,D]); slots int32 [N];
    positions int64 [N]. Capture-safe; no autograd (serving only)."""
    n = k.shape[0]
    row = cache_k.shape[2] * cache_k.shape[3]
    if _use_native(k) and k.dtype == torch.bfloat16 and row % 8 == 0:
        lib = _backend.require()
        kk, vv = k.reshape(n, row), v.reshape(n, row)
        if not kk.is_contiguous():
            kk = kk.contiguous()
        if not vv.is_contiguous():
            vv = vv.contiguous()
        _backend.check(
            lib.kf_kv_store(_p(cache_k), _p(cache_v), _p(kk), _p(vv),
                            _i32p(slots), _p(positions), cache_k.shape[1],
                            row, n, _stream()), "kv_store")
        return
    cache_k[slots.long(), positions] = k.reshape(n, *cache_k.shape[2:])
    cache_v[slots.long(), positions] = v.reshape(n, *cache_v.shape[2:])


def _rms_args(x, rms):
    """rms = (RMSNorm module, fp32 gamma tensor) or None -> kernel args.
    The fused path needs K %% 512 == 0 (LDS kernels only)."""
    if rms is None:
        return None, 0.0
    module, g32 = rms
    return g32, float(module.eps)


def skinny_linear(x: torch.Tensor, weight: torch.Tensor,
                  residual: Optional[torch.Tensor] = None,
                  rms=None, fuse_swiglu: bool = False) -> torch.Tensor:
    """Decode-batch linear y = x @ W^T (+ residual) for small leading dims
    (M <= 32): the LDS-staged weight-streaming kernel (skinny_gemm.hip)
    replaces hipBLASLt's ~30-50%-of-BW GEMV path in the serving decode
    step; `residual` fuses the following elementwise add into the
    epilogue. Falls back to F.linear off-GPU or for larger M /
    unsupported shapes. Inference-only (no autograd)."""
    shape = x.shape
    M = 1
    for d in shape[:-1]:
        M *= int(d)
    K = shape[-1] // 2 if fuse_swiglu else shape[-1]
    N = weight.shape[0]
    if (not _use_native(x) or M > 32 or K % 32 or N % 16
            or x.dtype != torch.bfloat16 or (M > 16 and K % 512)
            or ((rms is not None or fuse_swiglu) and K % 512)):
        if rms is not None:
            x = rms[0](x)
        if fuse_swiglu:
            x = swiglu(x)
        y = torch.nn.functional.linear(x, weight)
        return y + residual if residual is not None else y
    lib = _backend.require()
    x2 = x.reshape(M, shape[-1])
    if not x2.is_contiguous():
        x2 = x2.contiguous()
    r2 = None
    if residual is not None:
        r2 = residual.reshape(M, N)
        if not r2.is_contiguous():
            r2 = r2.contiguous()
    g32, eps = _rms_args(x, rms)
    y = torch.empty(M, N, dtype=x.dtype, device=x.device)
    _backend.check(
        lib.kf_skinny_gemm(_p(y), _p(x2), _p(weight), _p(r2), _fp(g32),
                           eps, int(fuse_swiglu), M, N, K,
                           shape[-1], 0, 0, _stream()),
        "skinny_gemm")
    return y.view(*shape[:-1], N)


def quantize_fp8_rows(w: torch.Tensor):
    """Per-output-row OCP e4m3 weight quantization for the W8A16 decode
    path: scale = absmax/448 per row, w8 = round(w/scale) in e4m3fn.
    Returns (w8 uint8 [N,K], scale fp32 [N])."""
    with torch.no_grad():  # w is usually a requires-grad Parameter — the
        # fp32 temps would otherwise be RETAINED by autograd (141 GB of
        # them for llama3-70b)
        wf = w.detach().float()
        scale = wf.abs().amax(dim=1).clamp_min(1e-12) / 448.0
        q = (wf / scale.unsqueeze(1)).to(torch.float8_e4m3fn)
        return q.view(torch.uint8).contiguous(), scale.contiguous()


def dequantize_fp8_rows(w8: torch.Tensor, scale: torch.Tensor,
                        dtype=torch.bfloat16) -> torch.Tensor:
    return (w8.view(torch.float8_e4m3fn).float()
            * scale.unsqueeze(1)).to(dtype)


def skinny_linear_q8(x: torch.Tensor, w8: torch.Tensor,
                     scale: torch.Tensor,
                     residual: Optional[torch.Tensor] = None,
                     rms=None, fuse_swiglu: bool = False) -> torch.Tensor:
    """Quantized decode linear y = x @ dequant(W8)^T (+ residual): fp8
    weights halve the HBM traffic of the weight-BW-bound decode GEMMs
    (skinny_gemm.hip kf_skinny_q8_kernel); activations stay bf16 and the
    MFMA runs bf16 (W8A16). Serving-only, behind KF_SERVE_QUANT=fp8."""
    shape = x.shape
    M = 1
    for d in shape[:-1]:
        M *= int(d)
    K = shape[-1] // 2 if fuse_swiglu else shape[-1]
    N = w8.shape[0]
    if (not _use_native(x) or M > 32 or K % 1024 or N % 16
            or x.dtype != torch.bfloat16):
        if rms is not None:
            x = rms[0](x)
        if fuse_swiglu:
            x = swiglu(x)
        y = torch.nn.functional.linear(
            x, dequantize_fp8_rows(w8, scale, x.dtype))
        return y + residual if residual is not None else y
    lib = _backend.require()
    x2 = x.reshape(M, shape[-1])
    if not x2.is_contiguous():
        x2 = x2.contiguous()
    r2 = None
    if residual is not None:
        r2 = residual.reshape(M, N)
        if not r2.is_contiguous():
            r2 = r2.contiguous()
    g32, eps = _rms_args(x, rms)
    y = torch.empty(M, N, dtype=x.dtype, device=x.device)
    _backend.check(
        lib.kf_skinny_gemm_q8(_p(y), _p(x2), _p(w8), _fp(scale), _p(r2),
                              _fp(g32), eps, int(fuse_swiglu), M, N, K,
                              shape[-1], 0, 0, _stream()),
        "skinny_gemm_q8")
    return y.view(*shape[:-1], N)


def decode_rope_store(qkv: torch.Tensor, cache_k: torch.Tensor,
                      cache_v: torch.Tensor, cos: torch.Tensor,
                      sin: torch.Tensor, slots: torch.Tensor,
                      positions: torch.Tensor, n_heads: int,
                      n_kv_heads: int) -> torch.Tensor:
    """Fused decode-step RoPE + cache scatter (decode_fused.hip): consume
    the fused QKV projection [N, 1, (Hq+2*Hkv)*D] directly — rotate q and
    k, write q to a fresh [N, Hq, D] tensor, scatter rotated k and copied
    v into cache row [slot[i], position[i]]. Replaces split + rope clones
    + kv_store (~4 launches and two D2D copies per layer). Serving only
    (no autograd); capture-safe."""
    D = cache_k.shape[3]
    n = qkv.shape[0]
    q = torch.empty(n, n_heads, D, dtype=qkv.dtype, device=qkv.device)
    if (_use_native(qkv) and qkv.dtype == torch.bfloat16 and D == 128):
        lib = _backend.require()
        q3 = qkv.reshape(n, -1)
        if not q3.is_contiguous():
            q3 = q3.contiguous()
        _backend.check(
            lib.kf_decode_rope_store(_p(q), _p(cache_k), _p(cache_v),
                                     _p(q3), _fp(cos), _fp(sin),
                                     _i32p(slots), _p(positions), n,
                                     n_heads, n_kv_heads, D,
                                     cache_k.shape[1], _stream()),
            "decode_rope_store")
        return q
    # reference path: split + rope + indexed store
    qq, kk, vv = qkv.reshape(n, 1, -1).split(
        [n_heads * D, n_kv_heads * D, n_kv_heads * D], dim=-1)
    qq = qq.view(n, 1, n_heads, D)
    kk = kk.view(n, 1, n_kv_heads, D)
    vv = vv.view(n, 1, n_kv_heads, D)
    qq, kk = rope(qq, kk, cos, sin, positions=positions)
    kv_store(cache_k, cache_v, kk, vv, slots, positions)
    return qq.reshape(n, n_heads, D)


# ------------------------------------------------------- Flash attention --

class _FlashAttnHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        lib = _backend.require()
        B, S, Hq, D = q.shape
        Hkv = k.shape[2]
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o = torch.empty_like(q)
        lse = torch.empty(B, Hq, S, dtype=torch.float32, device=q.device)
        _backend.check(
            lib.kf_attn_fwd(_p(o), _fp(lse), _p(q), _p(k), _p(v), B, S, Hq,
                            Hkv, D, 0, 0, float(scale), int(causal),
                            _stream()), "attn_fwd")
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.meta = (causal, scale)
        return o

    @staticmethod
    def backward(ctx, dout):
        lib = _backend.require()
        q, k, v, o, lse = ctx.saved_tensors
        causal, scale = ctx.meta
        B, S, Hq, D = q.shape
        Hkv = k.shape[2]
        dout = dout.contiguous()
        dq = torch.empty_like(q)
        dk = torch.empty_like(k)
        dv = torch.empty_like(v)
        delta = torch.empty(B, Hq, S, dtype=torch.float32, device=q.device)
        _backend.check(
            lib.kf_attn_bwd(_p(dq), _p(dk), _p(dv), _p(dout), _p(q), _p(k),
                            _p(v), _p(o), _fp(lse), _fp(delta), B, S, Hq, Hkv,
                            D, 0, 0, 0, 0, float(scale), int(causal),
                            _stream()), "attn_bwd")
        return dq, dk, dv, None, None


def flash_attention_rect(q: torch.Tensor, k: torch.Tensor,
                         v: torch.Tensor, q_offset: int,
                         scale: Optional[float] = None) -> torch.Tensor:
    """Rectangular-causal attention for chunked prefill (inference only,
    no autograd): q [B, C, Hq, D] holds C query rows at global offset
    q_offset; k/v [B, Skv, Hkv, D] hold the KV prefix INCLUDING the chunk
    (Skv >= q_offset + C). Row i attends kv <= q_offset + i.

    Native path: kf_attn_fwd4_rect (attention_fwd4.hip) — q is padded to a
    256-row multiple and the kv buffers must have row capacity up to the
    next 64 multiple of Skv (the serving KV-cache slabs do; this wrapper
    re-pads otherwise)."""
    B, C, Hq, D = q.shape
    Skv = k.shape[1]
    Hkv = k.shape[2]
    if scale is None:
        scale = D ** -0.5
    assert Skv >= q_offset + C, (Skv, q_offset, C)
    if _use_native(q):
        lib = _backend.require()
        pad = (256 - C % 256) % 256
        if pad:
            q = torch.cat([q, q.new_zeros(B, pad, Hq, D)], dim=1)
        o = torch.empty_like(q)
        lse = torch.empty(B, Hq, q.shape[1], dtype=torch.float32,
                          device=q.device)
        q = q.contiguous()
        # rows must be contiguous [.., Hkv, D] and the buffer must extend
        # to the next KV-TILE multiple of Skv (A4_KT=64 in
        # attention_fwd4.hip; 128 kept for headroom if the tile grows)
        need = (Skv + 127) // 128 * 128
        # the kernel's kv batch stride is b * Sq_padded rows, so the
        # zero-copy fast path is B == 1 only (the serving slabs)
        def _rows_ok(t):
            return (B == 1 and t.stride(1) == Hkv * D and t.stride(2) == D
                    and t.stride(3) == 1
                    and t.stride(0) // (Hkv * D) >= need)
        if not (_rows_ok(k) and _rows_ok(v)):
            rows = max(need, q.shape[1])
            if B > 1 and rows != q.shape[1]:
                raise ValueError(
                    "batched rectangular attention requires Skv <= padded "
                    f"Sq (kernel batch stride); got Skv={Skv} B={B}")
            kp = torch.zeros(B, rows, Hkv, D, dtype=k.dtype, device=k.device)
            vp = torch.zeros_like(kp)
            kp[:, :Skv] = k[:, :Skv]
            vp[:, :Skv] = v[:, :Skv]
            k, v = kp, vp
        _backend.check(
            lib.kf_attn_fwd4_rect(_p(o), _fp(lse), _p(q), _p(k), _p(v),
                                  B, q.shape[1], Skv, Hq, Hkv, D,
                                  0, Hkv * D,
                                  ctypes.c_float(float(scale)), q_offset,
                                  _stream()), "attn_fwd4_rect")
        return o[:, :C]
    # reference: sdpa's tril(Sk - Sq) IS the rectangular-causal mask when
    # the chunk sits at the END of the kv prefix
    kv_end = q_offset + C
    o = reference.sdpa(q.float().transpose(1, 2),
                       k[:, :kv_end].float().transpose(1, 2),
                       v[:, :kv_end].float().transpose(1, 2),
                       causal=True, scale=scale).transpose(1, 2)
    return o.to(q.dtype)


def masked_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     kv_len: int, scale: Optional[float] = None
                     ) -> torch.Tensor:
    """Bidirectional (non-causal) attention over only the first `kv_len`
    kv rows — the padded-batch classifier case (BERT serving): rows
    beyond kv_len are padding and must not be attended. Uses the
    rectangular-causal kernel with q_offset = kv_len, which makes every
    query's limit min(row + kv_len, kv_len-1) = kv_len-1 (inference
    only, no autograd; D == 128 on the native path)."""
    B, S, Hq, D = q.shape
    if scale is None:
        scale = D ** -0.5
    if _use_native(q) and D == 128:
        lib = _backend.require()
        Hkv = k.shape[2]
        pad = (256 - S % 256) % 256
        if pad:
            q = torch.cat([q, q.new_zeros(B, pad, Hq, D)], dim=1)
        o = torch.empty_like(q)
        lse = torch.empty(B, Hq, q.shape[1], dtype=torch.float32,
                          device=q.device)
        q = q.contiguous()
        # the kernel's kv batch stride is b * Sq_padded rows (training
        # layout) — repack kv into that shape, zeroed past kv_len
        Sp = q.shape[1]
        kp = torch.zeros(B, Sp, Hkv, D, dtype=k.dtype, device=k.device)
        vp = torch.zeros_like(kp)
        kp[:, :kv_len] = k[:, :kv_len]
        vp[:, :kv_len] = v[:, :kv_len]
        _backend.check(
            lib.kf_attn_fwd4_rect(_p(o), _fp(lse), _p(q), _p(kp), _p(vp),
                                  B, Sp, kv_len, Hq, Hkv, D,
                                  0, Hkv * D,
                                  ctypes.c_float(float(scale)), kv_len,
                                  _stream()), "attn_fwd4_rect")
        return o[:, :S]
    o = reference.sdpa(q.float().transpose(1, 2),
                       k[:, :kv_len].float().transpose(1, 2),
                       v[:, :kv_len].float().transpose(1, 2),
                       causal=False, scale=scale).transpose(1, 2)
    return o.to(q.dtype)


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    causal: bool = True, scale: Optional[float] = None):
    """bshd layout: q [B,S,Hq,D], k/v [B,S,Hkv,D] -> o [B,S,Hq,D].

    The CDNA4 kernel tiles q in 128-row blocks; for causal attention,
    sequences are zero-padded to a 128 multiple here (padded key rows are
    causally masked for every real query, so results are exact) and the
    output is sliced back. Non-causal input must already satisfy
    seq_len % 128 == 0.
    """
    if scale is None:
        scale = q.shape[-1] ** -0.5
    if _use_native(q):
        S = q.shape[1]
        pad = (128 - S % 128) % 128
        if pad:
            if not causal:
                raise ValueError(
                    "non-causal flash_attention requires seq_len % 128 == 0 "
                    f"(got {S}); pad inputs with an attention mask upstream")
            zq = q.new_zeros(q.shape[0], pad, q.shape[2], q.shape[3])
            zk = k.new_zeros(k.shape[0], pad, k.shape[2], k.shape[3])
            q = torch.cat([q, zq], dim=1)
            k = torch.cat([k, zk], dim=1)
            v = torch.cat([v, zk], dim=1)
            return _FlashAttnHip.apply(q, k, v, causal, scale)[:, :S]
        return _FlashAttnHip.apply(q, k, v, causal, scale)
    # reference path works in bhsd
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
    o = reference.sdpa(qt, kt, vt, causal=causal, scale=scale)
    return o.transpose(1, 2)


def _p_off(t: torch.Tensor, elem_off: int) -> ctypes.c_void_p:
    return ctypes.c_void_p(t.data_ptr() + elem_off * t.element_size())


# ------------------------------------------------------------------ SwiGLU --

class _SwigluHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h):
        lib = _backend.require()
        shape = h.shape
        F2 = shape[-1]
        h2 = h.contiguous().view(-1, F2)
        T = h2.shape[0]
        y = torch.empty(T, F2 // 2, dtype=h.dtype, device=h.device)
        _backend.check(lib.kf_swiglu_fwd(_p(y), _p(h2), T, F2 // 2,
                                         _stream()), "swiglu_fwd")
        ctx.save_for_backward(h2)
        ctx.shape = shape
        return y.view(*shape[:-1], F2 // 2)

    @staticmethod
    def backward(ctx, dy):
        lib = _backend.require()
        (h2,) = ctx.saved_tensors
        T, F2 = h2.shape
        dy2 = dy.contiguous().view(T, F2 // 2)
        dh = torch.empty_like(h2)
        _backend.check(lib.kf_swiglu_bwd(_p(dh), _p(dy2), _p(h2), T, F2 // 2,
                                         _stream()), "swiglu_bwd")
        return dh.view(ctx.shape)


def swiglu(h: torch.Tensor) -> torch.Tensor:
    """h = [gate ++ up] on the last dim -> silu(gate) * up."""
    if _use_native(h):
        return _SwigluHip.apply(h)
    g, u = h.chunk(2, dim=-1)
    return torch.nn.functional.silu(g) * u


# -------------------------------------------- Fused QKV -> RoPE -> attention

class _FusedQkvAttentionHip(torch.autograd.Function):
    """RoPE applied IN PLACE on the q/k regions of the fused qkv projection,
    then flash attention reading q/k/v as strided views — zero copies, and
    backward writes dq/dk/dv directly into one dqkv buffer (no torch.cat).

    In-place note: qkv is the wqkv GEMM output; torch.linear's backward
    needs its input and weight, never its output, so rotating the buffer in
    place is safe. No other consumer may read qkv afterwards.
    """

    @staticmethod
    def forward(ctx, qkv, cos, sin, Hq, Hkv, D, causal, scale):
        lib = _backend.require()
        B, S, _ = qkv.shape
        ts = (Hq + 2 * Hkv) * D  # token stride in elements
        qkv = qkv.contiguous()
        _backend.check(
            lib.kf_rope(_p(qkv), _p_off(qkv, Hq * D), _fp(cos), _fp(sin),
                        _ip_or_null(None), B, S, Hq, Hkv, D, ts, ts, 0, 0,
                        _stream()), "rope_fwd")
        o = torch.empty(B, S, Hq * D, dtype=qkv.dtype, device=qkv.device)
        lse = torch.empty(B, Hq, S, dtype=torch.float32, device=qkv.device)
        _backend.check(
            lib.kf_attn_fwd(_p(o), _fp(lse), _p(qkv), _p_off(qkv, Hq * D),
                            _p_off(qkv, (Hq + Hkv) * D), B, S, Hq, Hkv, D,
                            ts, ts, float(scale), int(causal), _stream()),
            "attn_fwd")
        ctx.save_for_backward(qkv, cos, sin, o, lse)
        ctx.meta = (Hq, Hkv, D, causal, scale)
        return o

    @staticmethod
    def backward(ctx, dout):
        lib = _backend.require()
        qkv, cos, sin, o, lse = ctx.saved_tensors
        Hq, Hkv, D, causal, scale = ctx.meta
        B, S, _ = qkv.shape
        ts = (Hq + 2 * Hkv) * D
        dout = dout.contiguous()
        dqkv = torch.empty_like(qkv)
        delta = torch.empty(B, Hq, S, dtype=torch.float32, device=qkv.device)
        _backend.check(
            lib.kf_attn_bwd(_p(dqkv), _p_off(dqkv, Hq * D),
                            _p_off(dqkv, (Hq + Hkv) * D), _p(dout), _p(qkv),
                            _p_off(qkv, Hq * D), _p_off(qkv, (Hq + Hkv) * D),
                            _p(o), _fp(lse), _fp(delta), B, S, Hq, Hkv, D,
                            ts, ts, ts, ts, float(scale), int(causal),
                            _stream()), "attn_bwd")
        # adjoint of the in-place rotation on the dq/dk regions
        _backend.check(
            lib.kf_rope(_p(dqkv), _p_off(dqkv, Hq * D), _fp(cos), _fp(sin),
                        _ip_or_null(None), B, S, Hq, Hkv, D, ts, ts, 0, 1,
                        _stream()), "rope_bwd")
        return dqkv, None, None, None, None, None, None, None


def fused_qkv_attention(qkv: torch.Tensor, cos: torch.Tensor,
                        sin: torch.Tensor, Hq: int, Hkv: int, D: int,
                        causal: bool = True,
                        scale: Optional[float] = None) -> torch.Tensor:
    """qkv [B,S,(Hq+2Hkv)*D] -> attention output [B,S,Hq*D].

    Native path requires S % 128 == 0 (training shapes); otherwise (and on
    CPU) falls back to the composed split->rope->attention ops.
    """
    if scale is None:
        scale = D ** -0.5
    B, S, _ = qkv.shape
    if _use_native(qkv) and S % 128 == 0:
        return _FusedQkvAttentionHip.apply(qkv, cos, sin, Hq, Hkv, D, causal,
                                           scale)
    q, k, v = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    q = q.reshape(B, S, Hq, D)
    k = k.reshape(B, S, Hkv, D)
    v = v.reshape(B, S, Hkv, D)
    q, k = rope(q, k, cos, sin)
    o = flash_attention(q, k, v, causal=causal, scale=scale)
    return o.reshape(B, S, Hq * D)


# --------------------------------------------------------------- LayerNorm --

class _LayerNormHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        lib = _backend.require()
        shape = x.shape
        x2 = x.contiguous().view(-1, shape[-1])
        rows, cols = x2.shape
        y = torch.empty_like(x2)
        mu = torch.empty(rows, dtype=torch.float32, device=x.device)
        rstd = torch.empty(rows, dtype=torch.float32, device=x.device)
        _backend.check(
            lib.kf_layernorm_fwd(_p(y), _fp(mu), _fp(rstd), _p(x2), _p(weight),
                                 _p(bias), rows, cols, float(eps), _stream()),
            "layernorm_fwd")
        ctx.save_for_backward(x2, weight, mu, rstd)
        ctx.shape = shape
        return y.view(shape)

    @staticmethod
    def backward(ctx, dy):
        lib = _backend.require()
        x2, weight, mu, rstd = ctx.saved_tensors
        rows, cols = x2.shape
        dy2 = dy.contiguous().view(rows, cols)
        dx = torch.empty_like(x2)
        dw = torch.empty_like(weight)
        db = torch.empty_like(weight)
        part = torch.zeros(2 * cols, dtype=torch.float32, device=x2.device)
        _backend.check(
            lib.kf_layernorm_bwd(_p(dx), _p(dw), _p(db), _fp(part), _p(dy2),
                                 _p(x2), _p(weight), _fp(mu), _fp(rstd), rows,
                                 cols, _stream()), "layernorm_bwd")
        return dx.view(ctx.shape), dw, db, None


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-5):
    if _use_native(x):
        return _LayerNormHip.apply(x, weight, bias, eps)
    return reference.layer_norm(x, weight, bias, eps)


# --------------------------------------------------------- Decode attention --

def attention_decode(q: torch.Tensor, kcache: torch.Tensor,
                     vcache: torch.Tensor, slots: torch.Tensor,
                     lens: torch.Tensor, scale: Optional[float] = None):
    """Single-token GQA attention over the KV cache (no autograd — serving).

    q [N,Hq,D] bf16; kcache/vcache [SLOTS,SMAX,Hkv,D] bf16;
    slots/lens int32 [N]. Returns o [N,Hq,D].
    """
    N, Hq, D = q.shape
    Hkv = kcache.shape[2]
    if scale is None:
        scale = D ** -0.5
    if _use_native(q):
        lib = _backend.require()
        q = q.contiguous()
        out = torch.empty_like(q)
        # flash-decoding split: N*Hkv blocks alone underfill the 256-CU
        # chip at serving batch sizes; split the sequence over grid.z and
        # merge partials (one extra tiny kernel). Static per (N, Hkv) so
        # hipGraph capture sees fixed shapes.
        splits = int(os.environ.get("KF_DECODE_SPLITS", "0")) or \
            min(8, max(1, 512 // max(1, N * Hkv)))
        po = pm = None
        if splits > 1:
            po = torch.empty(N, Hq, splits, D, dtype=torch.float32,
                             device=q.device)
            pm = torch.empty(N, Hq, splits, 2, dtype=torch.float32,
                             device=q.device)
        _backend.check(
            lib.kf_attn_decode(_p(out), _p(po), _p(pm), _p(q), _p(kcache),
                               _p(vcache), _i32p(slots), _i32p(lens), N,
                               kcache.shape[1], Hq, Hkv, D, splits,
                               float(scale), _stream()),
            "attn_decode")
        return out
    # reference path: per-sequence sdpa over the cached prefix
    outs = []
    for i in range(N):
        L = int(lens[i])
        s = int(slots[i])
        kk = kcache[s, :L].unsqueeze(0).transpose(1, 2)   # [1,Hkv,L,D]
        vv = vcache[s, :L].unsqueeze(0).transpose(1, 2)
        qq = q[i].view(1, 1, Hq, D).transpose(1, 2)       # [1,Hq,1,D]
        o = reference.sdpa(qq, kk, vv, causal=False, scale=scale)
        outs.append(o.transpose(1, 2).reshape(1, Hq, D))
    return torch.cat(outs)


# --------------------------------------------------------- Cross entropy --

class _CrossEntropyHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index, inplace_ok=False):
        lib = _backend.require()
        T, V = logits.shape
        logits = logits.contiguous()
        loss_sum = torch.zeros(1, dtype=torch.float32, device=logits.device)
        lse = torch.empty(T, dtype=torch.float32, device=logits.device)
        _backend.check(
            lib.kf_ce_fwd(_fp(loss_sum), _fp(lse), _p(logits), _ip(targets),
                          T, V, ignore_index, _stream()), "ce_fwd")
        valid = (targets != ignore_index).sum()
        inv_valid = torch.where(valid > 0, 1.0 / valid.float(),
                                torch.zeros((), device=logits.device))
        ctx.save_for_backward(logits, lse, targets, inv_valid)
        ctx.ignore_index = ignore_index
        ctx.inplace_ok = inplace_ok
        return (loss_sum * inv_valid).squeeze(0)

    @staticmethod
    def backward(ctx, grad_out):
        lib = _backend.require()
        logits, lse, targets, inv_valid = ctx.saved_tensors
        T, V = logits.shape
        scale = (grad_out.float() * inv_valid).contiguous()
        # dlogits is written INTO the saved logits buffer when the caller
        # passed a NON-LEAF (the model path: logits = lm_head(x), whose
        # backward uses x and W, never this output): the kernel is
        # elementwise per position (read lv -> write ov at the same
        # address), and at mb6-7 x seq 4096 x 128256-vocab this buffer is
        # ~7 GB — materializing a second one was the OOM that blocked
        # micro-batch 7. Leaf inputs (user holds the tensor / wants
        # .grad) always get a fresh buffer. KF_CE_INPLACE=0 disables.
        if ctx.inplace_ok and os.environ.get("KF_CE_INPLACE", "1") == "1":
            dlogits = logits
        else:
            dlogits = torch.empty_like(logits)
        _backend.check(
            lib.kf_ce_bwd(_p(dlogits), _p(logits), _fp(lse), _ip(targets),
                          _fp(scale), T, V, ctx.ignore_index, _stream()),
            "ce_bwd")
        return dlogits, None, None, None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                  ignore_index: int = -100):
    """Mean CE over non-ignored rows. logits [T,V] bf16, targets [T] int64."""
    if _use_native(logits):
        inplace_ok = not logits.is_leaf and logits.is_contiguous()
        return _CrossEntropyHip.apply(logits, targets, ignore_index,
                                      inplace_ok)
    return reference.softmax_cross_entropy(logits, targets, ignore_index)


# ----------------------------------------------------------- Fused AdamW --

def fused_adamw(p16: torch.Tensor, p32: torch.Tensor, grad: torch.Tensor,
                m: torch.Tensor, v: torch.Tensor,
                wd_mask: Optional[torch.Tensor], lr: float, beta1: float,
                beta2: float, eps: float, weight_decay: float, step: int):
    """One fused update over a flat bf16 parameter shard + fp32 state.

    p16/grad bf16 [N]; p32/m/v fp32 [N]; wd_mask fp32 {0,1} [N] or None.
    """
    if p16.is_cuda and _backend.native_enabled():
        lib = _backend.require()
        _backend.check(
            lib.kf_adamw(_p(p16), _fp(p32), _p(grad), _fp(m), _fp(v),
                         _fp(wd_mask), p16.numel(), lr, beta1, beta2, eps,
                         weight_decay, step, _stream()), "adamw")
        return
    # reference path (CPU tests)
    g32 = grad.float()
    m.mul_(beta1).add_(g32, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g32, g32, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (v / bc2).sqrt().add_(eps)
    decay = weight_decay if wd_mask is None else wd_mask * weight_decay
    p32.mul_(1 - lr * decay)
    p32.addcdiv_(m / bc1, denom, value=-lr)
    p16.copy_(p32.to(torch.bfloat16))
