"""Warm worker — a pre-forked PyTorchJob rank waiting for its assignment.

SURVEY.md §7 hard part 3: "gang-start latency (HIP context creation per
process is slow; pre-fork a warm worker pool to hit a good job-start p50)".
The ~1.5 s of `import torch` + interpreter start dominates cold job-start;
this process pays it ahead of time, then blocks on stdin until the launcher
hands it a rank assignment as one JSON line:

    {"spec_path": ..., "rank": 0, "world_size": 1, "master_port": 12345,
     "workdir": ..., "job_uid": ..., "gpu": 3 | null,
     "env": {...extra env (PodDefaults)...}}

CUDA/HIP is NOT touched before the assignment arrives, so HIP_VISIBLE_DEVICES
set at assignment time still controls device visibility (torch initializes
the HIP runtime lazily on first use).
"""
from __future__ import annotations

import json
import os
import sys


def main() -> int:
    # Pay the import cost up-front; do NOT initialize CUDA here.
    from kubeflow_amd.ops import tunable as _t
    _t.enable()
    import torch  # noqa: F401  (the expensive import)
    import kubeflow_amd.models  # noqa: F401
    import kubeflow_amd.runtime.worker as worker

    sys.stderr.write("[warm-worker] ready\n")
    sys.stderr.flush()
    line = sys.stdin.readline()
    if not line.strip():
        return 0  # pool shutdown
    asg = json.loads(line)

    os.environ["RANK"] = str(asg["rank"])
    os.environ["WORLD_SIZE"] = str(asg["world_size"])
    os.environ["LOCAL_RANK"] = "0" if asg.get("gpu") is not None else str(
        asg["rank"])
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(asg["master_port"])
    os.environ["KF_JOB_UID"] = asg.get("job_uid", "")
    os.environ["KF_JOB_WORKDIR"] = asg["workdir"]
    if asg.get("gpu") is not None:
        os.environ["HIP_VISIBLE_DEVICES"] = str(asg["gpu"])
    for k, v in (asg.get("env") or {}).items():
        os.environ[k] = str(v)
    rank_dir = os.path.join(asg["workdir"], f"rank-{asg['rank']}")
    os.makedirs(rank_dir, exist_ok=True)
    os.chdir(rank_dir)
    return worker.main(["--spec", asg["spec_path"]])


if __name__ == "__main__":
    sys.exit(main())
