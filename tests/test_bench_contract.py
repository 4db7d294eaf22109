"""The driver depends on bench.py's exact CLI + JSON contract — test it on
CPU with a tiny model so contract breaks are caught before round end."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    env = dict(os.environ, KF_BENCH_MODEL="llama-tiny", KF_BENCH_SEQ="64",
               KF_BENCH_MB="2", PYTHONPATH=REPO)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "2",
         "--warmup", "1"], env=env, capture_output=True, text=True,
        timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines()
             if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0
    cfg = d["config"]
    assert cfg["seq_len"] == 64 and cfg["global_batch"] == 2
    assert "model" in cfg and "parallelism" in cfg
