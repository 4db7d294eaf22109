#!/usr/bin/env python3
"""Job-start latency benchmark: PyTorchJob submit -> all ranks in the train
loop (the metric the gang scheduler owns — BASELINE.md row 'job-start p50').

Runs N iterations of a 1-step job through the FULL control plane (store ->
reconcile -> gang launch -> worker init -> first status heartbeat) and
reports p50/p90. Works on CPU (mnist-mlp) and GPU (llama-tiny).
"""
import json
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeflow_amd.api import new_object
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.platform import Platform


def run(n=5, model="mnist-mlp", gpus=0):
    lat_start, lat_done = [], []
    with Platform(root_dir=tempfile.mkdtemp(prefix="jobstart-")) as plat:
        if plat.warm_pool is not None:
            plat.warm_pool.wait_ready(1, timeout=60)
        for i in range(n):
            name = f"js-{i}"
            spec = {"pytorchReplicaSpecs": {"Worker": {
                "replicas": 1, "restartPolicy": "Never",
                "template": {"model": model, "steps": 1, "micro_batch": 2,
                             "seq_len": 128, "gpus_per_replica": gpus,
                             "status_every": 1, "save_final": False}}}}
            t0 = time.time()
            plat.store.create(new_object("PyTorchJob", name, "default",
                                         spec=spec))
            started = done = None
            deadline = time.time() + 180
            while time.time() < deadline:
                job = plat.store.get("PyTorchJob", name, "default")
                st = job.get("status", {}).get("replicaStatuses", {})
                if started is None and (st.get("Worker", {}).get("active")
                                        or st.get("Worker", {}).get("succeeded")):
                    started = time.time() - t0
                if has_condition(job, "Succeeded"):
                    done = time.time() - t0
                    break
                if has_condition(job, "Failed"):
                    raise RuntimeError(job["status"])
                time.sleep(0.05)
            lat_start.append(started)
            lat_done.append(done)
            if plat.warm_pool is not None:  # measure the warm-hit path
                plat.warm_pool.wait_ready(1, timeout=60)
    lat_start.sort()
    lat_done.sort()
    out = {
        "metric": "pytorchjob_job_start_p50_s",
        "model": model,
        "n": n,
        "start_p50_s": round(lat_start[len(lat_start) // 2], 3),
        "start_p90_s": round(lat_start[int(len(lat_start) * 0.9)], 3),
        "complete_p50_s": round(lat_done[len(lat_done) // 2], 3),
    }
    print(json.dumps(out), flush=True)
    return out


if __name__ == "__main__":
    model = sys.argv[1] if len(sys.argv) > 1 else "mnist-mlp"
    gpus = int(sys.argv[2]) if len(sys.argv) > 2 else 0
    run(model=model, gpus=gpus)
