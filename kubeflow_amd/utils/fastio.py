"""Threaded flat-tensor file I/O over libkfio (ops/csrc_cpp/kfio.cpp).

Used by runtime/checkpoint.py for the big flat buffers (bf16 weights, fp32
optimizer state): raw bytes + threaded pwrite instead of torch.save's
single-threaded pickle stream. Falls back to plain Python file I/O if the
library is missing (e.g. source checkout without build()) — same on-disk
format either way: the tensor's raw bytes, nothing else.
"""
from __future__ import annotations

import ctypes
import os
from pathlib import Path
from typing import Optional

import torch

_LIB_PATH = Path(__file__).resolve().parent.parent / "ops" / "libkfio.so"
_lib = None
_tried = False


def _load():
    global _lib, _tried
    if _tried:
        return _lib
    _tried = True
    try:
        if not _LIB_PATH.exists():
            from kubeflow_amd.ops import build_ext
            build_ext.build_io(verbose=False)
        lib = ctypes.CDLL(str(_LIB_PATH))
        lib.kf_write_file.restype = ctypes.c_int
        lib.kf_write_file.argtypes = [ctypes.c_char_p, ctypes.c_void_p,
                                      ctypes.c_int64, ctypes.c_int]
        lib.kf_read_file.restype = ctypes.c_int
        lib.kf_read_file.argtypes = [ctypes.c_char_p, ctypes.c_void_p,
                                     ctypes.c_int64, ctypes.c_int]
        lib.kf_file_size.restype = ctypes.c_int64
        lib.kf_file_size.argtypes = [ctypes.c_char_p]
        _lib = lib
    except Exception:
        _lib = None
    return _lib


def _nthreads() -> int:
    return min(16, os.cpu_count() or 4)


def write_tensor(path: str, t: torch.Tensor):
    """Write a tensor's raw bytes (host-staged if on GPU). Atomic: bytes go
    to a .tmp sibling and are renamed in — ftruncate preallocates the full
    size, so a crash mid-write would otherwise leave a full-size file of
    partial content that passes read_into's size check."""
    t = t.detach()
    if t.device.type != "cpu":
        t = t.cpu()
    t = t.contiguous()
    n = t.numel() * t.element_size()
    tmp = path + ".tmp"
    lib = _load()
    if lib is not None:
        rc = lib.kf_write_file(tmp.encode(), ctypes.c_void_p(t.data_ptr()),
                               n, _nthreads())
        if rc != 0:
            raise OSError(rc, f"kf_write_file({tmp}): {os.strerror(rc)}")
    else:
        with open(tmp, "wb") as f:  # fallback: plain write, same format
            f.write(t.numpy().tobytes() if t.dtype != torch.bfloat16
                    else t.view(torch.uint8).numpy().tobytes())
    os.replace(tmp, path)


def file_size(path: str) -> int:
    lib = _load()
    if lib is not None:
        return int(lib.kf_file_size(path.encode()))
    try:
        return os.path.getsize(path)
    except OSError:
        return -1


def read_into(path: str, out: torch.Tensor):
    """Read raw bytes into an existing tensor (via host staging if out is
    on GPU). The file must be EXACTLY the tensor's byte size — a mismatch
    means the checkpoint was written by a different layout (e.g. another
    world size for sharded state)."""
    n = out.numel() * out.element_size()
    sz = file_size(path)
    if sz != n:
        raise OSError(f"{path}: {sz} bytes on disk, expected exactly {n} "
                      "(checkpoint layout/world-size mismatch?)")
    host = out if out.device.type == "cpu" else torch.empty(
        out.shape, dtype=out.dtype, device="cpu")
    host = host.contiguous()
    lib = _load()
    if lib is not None:
        rc = lib.kf_read_file(path.encode(),
                              ctypes.c_void_p(host.data_ptr()), n,
                              _nthreads())
        if rc == -2:
            raise OSError(f"{path}: file smaller than expected {n} bytes")
        if rc != 0:
            raise OSError(rc, f"kf_read_file({path}): {os.strerror(rc)}")
    else:
        with open(path, "rb") as f:
            raw = f.read(n)
        if len(raw) < n:
            raise OSError(f"{path}: file smaller than expected {n} bytes")
        host.view(torch.uint8).copy_(
            torch.frombuffer(bytearray(raw), dtype=torch.uint8)
            .view(host.view(torch.uint8).shape))
    if host is not out:
        out.copy_(host.to(out.device))
    return out


def native_available() -> bool:
    return _load() is not None
