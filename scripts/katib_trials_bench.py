"""Measured Katib trial throughput (VERDICT round-1 item 8): N concurrent
BERT fine-tune trials sharing the available GPUs via gpu_shared
co-scheduling. Prints one JSON line with measured trials/hour."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeflow_amd.api import new_object
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.platform import Platform

PARAMS = [
    {"name": "lr", "parameterType": "double",
     "feasibleSpace": {"min": "1e-5", "max": "1e-3"}},
]


def main():
    parallel = int(sys.argv[1]) if len(sys.argv) > 1 else 2
    total = int(sys.argv[2]) if len(sys.argv) > 2 else 6
    steps = int(sys.argv[3]) if len(sys.argv) > 3 else 50
    root = os.path.join("/tmp", f"katib-bench-{int(time.time())}")
    with Platform(root_dir=root) as plat:
        exp = new_object("Experiment", "bert-hpo", "default", spec={
            "objective": {"type": "minimize",
                          "objectiveMetricName": "loss"},
            "algorithm": {"algorithmName": "random"},
            "parallelTrialCount": parallel,
            "maxTrialCount": total,
            "maxFailedTrialCount": 2,
            "parameters": PARAMS,
            "trialTemplate": {"model": "bert-base-hd128", "steps": steps,
                              "micro_batch": 32, "seq_len": 512,
                              "gpus_per_replica": 1, "gpu_shared": True,
                              "gpu_memory": "48Gi", "status_every": 10,
                              "save_final": False, "replicas": 1},
        })
        t0 = time.time()
        plat.store.create(exp)
        deadline = time.time() + 3600
        while time.time() < deadline:
            obj = plat.store.get("Experiment", "bert-hpo", "default")
            if has_condition(obj, "Succeeded") or has_condition(obj, "Failed"):
                break
            time.sleep(1)
        wall = time.time() - t0
        st = obj["status"]
        done = st.get("trialsSucceeded", st.get("trials", 0))
        print(json.dumps({
            "metric": "katib_trials_per_hour",
            "value": round(done * 3600 / wall, 1),
            "trials_succeeded": done,
            "parallel": parallel,
            "wall_s": round(wall, 1),
            "steps_per_trial": steps,
            "model": "bert-base-hd128",
            "gpus": plat.scheduler.inv.n_gpus,
            "measured": True,
        }), flush=True)


if __name__ == "__main__":
    main()
