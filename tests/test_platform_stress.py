"""Mixed-workload platform stress: concurrent jobs + notebook + experiment +
pipeline + serving on one control plane — shakes out cross-controller races
(the reference's kf_is_ready-style 'everything at once' tier)."""
import json
import time
import urllib.request

import pytest

from kubeflow_amd.api import new_object
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.platform import Platform


def test_everything_at_once(tmp_path):
    with Platform(root_dir=str(tmp_path)) as plat:
        store = plat.store
        # profile -> namespace provisioning
        store.create(new_object("Profile", "team", None, spec={
            "owner": {"kind": "User", "name": "lead@example.com"}}))
        # two training jobs
        for i in range(2):
            store.create(new_object("PyTorchJob", f"train-{i}", "team", spec={
                "pytorchReplicaSpecs": {"Worker": {
                    "replicas": 1, "restartPolicy": "Never",
                    "template": {"model": "mnist-mlp", "steps": 4,
                                 "micro_batch": 8, "gpus_per_replica": 0,
                                 "status_every": 2, "save_final": False}}}}))
        # a notebook session
        store.create(new_object("Notebook", "nb", "team", spec={}))
        # an HPO experiment (2 trials)
        store.create(new_object("Experiment", "hpo", "team", spec={
            "objective": {"type": "minimize", "objectiveMetricName": "loss"},
            "algorithm": {"algorithmName": "random"},
            "parallelTrialCount": 2, "maxTrialCount": 2,
            "parameters": [{"name": "lr", "parameterType": "double",
                            "feasibleSpace": {"min": "0.001", "max": "0.1"}}],
            "trialTemplate": {"model": "mnist-mlp", "steps": 2,
                              "gpus_per_replica": 0, "save_final": False,
                              "replicas": 1}}))
        # a 2-step pipeline
        store.create(new_object("PipelineRun", "pipe", "team", spec={
            "tasks": [
                {"name": "a", "dependencies": [],
                 "template": {"model": "mnist-mlp", "steps": 2,
                              "gpus_per_replica": 0, "save_final": False}},
                {"name": "b", "dependencies": ["a"],
                 "template": {"model": "mnist-mlp", "steps": 2,
                              "gpus_per_replica": 0, "save_final": False}}]}))
        # a served model
        store.create(new_object("InferenceService", "svc", "team", spec={
            "predictor": {"model": "llama-tiny", "gpus": 0, "maxSlots": 2,
                          "maxSeqLen": 128, "maxBatch": 2}}))

        deadline = time.time() + 300

        def all_done():
            checks = {
                "train-0": has_condition(store.get("PyTorchJob", "train-0",
                                                   "team"), "Succeeded"),
                "train-1": has_condition(store.get("PyTorchJob", "train-1",
                                                   "team"), "Succeeded"),
                "nb": store.get("Notebook", "nb",
                                "team")["status"].get("readyReplicas") == 1,
                "hpo": has_condition(store.get("Experiment", "hpo", "team"),
                                     "Succeeded"),
                "pipe": has_condition(store.get("PipelineRun", "pipe",
                                                "team"), "Succeeded"),
                "svc": has_condition(store.get("InferenceService", "svc",
                                               "team"), "Ready"),
                "ns": True,
            }
            return checks

        while time.time() < deadline:
            checks = all_done()
            if all(checks.values()):
                break
            # nothing may have Failed
            for kind, name in (("PyTorchJob", "train-0"),
                               ("PyTorchJob", "train-1"),
                               ("Experiment", "hpo"), ("PipelineRun", "pipe")):
                obj = store.get(kind, name, "team")
                assert not has_condition(obj, "Failed"), (kind, obj["status"])
            time.sleep(0.5)
        assert all(checks.values()), checks

        # profile provisioned the namespace artifacts meanwhile
        assert store.get("Namespace", "team", None)
        assert store.get("RoleBinding", "namespaceAdmin", "team")

        # serving answers
        url = store.get("InferenceService", "svc", "team")["status"]["url"]
        body = json.dumps({"prompt_tokens": [1, 2], "max_new_tokens": 3}).encode()
        req = urllib.request.Request(f"{url}/v2/generate", data=body,
                                     headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=60) as r:
            assert len(json.loads(r.read())["tokens"]) == 3

        # teardown: deleting everything leaves no running processes
        for kind, name in (("PyTorchJob", "train-0"), ("PyTorchJob", "train-1"),
                           ("Notebook", "nb"), ("Experiment", "hpo"),
                           ("PipelineRun", "pipe"),
                           ("InferenceService", "svc")):
            store.delete(kind, name, "team")
        time.sleep(2)
        assert not plat.pytorchjob.gangs
        assert not plat.inference.gangs
        assert not plat.notebook.sessions
