"""Pre-tuned hipBLASLt/rocBLAS GEMM algorithm table (PyTorch TunableOp).

`tuned/tunableop_gfx950.csv` was produced by exhaustive TunableOp search on
an MI355X for the Llama-3-8B training shapes (e.g. the 6144x16384x4096 qkv
GEMM runs at ~1.6 PF with the tuned pick vs ~1.3 PF default heuristic).
enable() must run BEFORE the first GEMM; call it at process entry
(bench.py, runtime/worker.py, runtime/serving_server.py do).
"""
import os
from pathlib import Path

# TunableOp expands FILENAME per device: "tunableop.csv" -> "tunableop0.csv"
BASE = Path(__file__).resolve().parent / "tuned" / "tunableop.csv"
CSV = Path(__file__).resolve().parent / "tuned" / "tunableop0.csv"


def enable() -> bool:
    if not CSV.exists():
        return False
    if os.environ.get("PYTORCH_TUNABLEOP_ENABLED") is not None:
        return True  # caller controls it (e.g. a re-tuning run)
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"   # use table, never re-tune
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = str(BASE)
    return True
