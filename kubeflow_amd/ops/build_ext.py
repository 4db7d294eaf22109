"""Build driver for the CDNA4 kernel library `libkfops.so`.

Invokes hipcc directly (no hipify, no CUDA shims — the sources are written
HIP/gfx950-native). The library has a C ABI (raw pointers + hipStream_t) and
is loaded via ctypes by `kubeflow_amd.ops._backend`; it deliberately does NOT
link libtorch, so there are no ABI/version couplings — tensors cross the
boundary as data_ptr() integers on torch's current HIP stream.

Built IN-TREE (kubeflow_amd/ops/libkfops.so) so the .so travels with the
repo snapshot to GPU boxes. Rebuilds are incremental on source mtimes.
"""
from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

CSRC = Path(__file__).resolve().parent / "csrc"
OUT = Path(__file__).resolve().parent / "libkfops.so"
BUILD = CSRC / ".build"

HIPCC = os.environ.get("KF_HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("KF_GFX_ARCH", "gfx950")

CFLAGS = [
    f"--offload-arch={ARCH}",
    "-O3",
    "-std=c++17",
    "-fPIC",
    "-fvisibility=hidden",
    "-Wall",
]


def _sources():
    return sorted(CSRC.glob("*.hip"))


def needs_build() -> bool:
    if not OUT.exists():
        return True
    out_mtime = OUT.stat().st_mtime
    deps = list(_sources()) + list(CSRC.glob("*.h")) + [Path(__file__)]
    return any(p.stat().st_mtime > out_mtime for p in deps)


CSRC_CPP = Path(__file__).resolve().parent / "csrc_cpp"
OUT_IO = Path(__file__).resolve().parent / "libkfio.so"
CXX = os.environ.get("KF_CXX", "g++")


def build_io(verbose: bool = True, force: bool = False) -> Path:
    """Plain-C++ IO library (no HIP): threaded checkpoint reads/writes."""
    src = CSRC_CPP / "kfio.cpp"
    if (OUT_IO.exists() and not force
            and OUT_IO.stat().st_mtime > src.stat().st_mtime):
        return OUT_IO
    cmd = [CXX, "-O2", "-std=c++17", "-shared", "-fPIC",
           "-fvisibility=hidden", "-pthread", str(src), "-o", str(OUT_IO)]
    if verbose:
        print("[kfio]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return OUT_IO


def build(verbose: bool = True, force: bool = False) -> Path:
    build_io(verbose=verbose, force=force)
    if not force and not needs_build():
        return OUT
    BUILD.mkdir(exist_ok=True)
    objs = []
    procs = []
    for src in _sources():
        obj = BUILD / (src.stem + ".o")
        objs.append(obj)
        hdr_mtime = max(p.stat().st_mtime for p in CSRC.glob("*.h"))
        if (obj.exists() and not force
                and obj.stat().st_mtime > max(src.stat().st_mtime, hdr_mtime)):
            continue
        cmd = [HIPCC, "-c", *CFLAGS, str(src), "-o", str(obj)]
        if verbose:
            print("[kfops]", " ".join(cmd), file=sys.stderr)
        procs.append((src, subprocess.Popen(cmd, stderr=subprocess.PIPE)))
    failed = False
    for src, p in procs:
        _, err = p.communicate()
        if p.returncode != 0:
            failed = True
            print(f"[kfops] FAILED {src.name}:\n{err.decode()}", file=sys.stderr)
        elif err.strip() and verbose:
            print(err.decode(), file=sys.stderr)
    if failed:
        raise RuntimeError("hipcc compilation failed")
    cmd = [HIPCC, "-shared", *CFLAGS, *map(str, objs), "-o", str(OUT)]
    if verbose:
        print("[kfops]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(OUT)
