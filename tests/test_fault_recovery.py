"""Fault injection: kill a running worker -> gang dies -> OnFailure restart
-> job resumes from checkpoint and succeeds.

The reference's recovery model is reconcile-until-converged + restart
policies delegated to StatefulSet/Deployment controllers;
here the gang supervisor owns it: kill-on-peer-death + requeue with
restartPolicy semantics + checkpoint resume.
"""
import os
import signal
import time

import pytest

from kubeflow_amd.api import new_object
from kubeflow_amd.api.objects import has_condition, get_condition
from kubeflow_amd.platform import Platform


def test_worker_kill_restart_and_resume(tmp_path):
    with Platform(root_dir=str(tmp_path)) as plat:
        job = new_object("PyTorchJob", "crashy", "default", spec={
            "pytorchReplicaSpecs": {"Worker": {
                "replicas": 1,
                "restartPolicy": "OnFailure",
                "template": {
                    "model": "mnist-mlp", "steps": 30, "micro_batch": 8,
                    "gpus_per_replica": 0, "status_every": 1,
                    "save_every": 5, "resume": True,
                },
            }},
            "backoffLimit": 3,
        })
        plat.store.create(job)

        # wait until the worker made some progress (a checkpoint exists)
        deadline = time.time() + 120
        uid = plat.store.get("PyTorchJob", "crashy", "default")["metadata"]["uid"]
        ckdir = os.path.join(str(tmp_path), "jobs", "default",
                             f"crashy-{uid[:8]}", "checkpoints")
        while time.time() < deadline:
            if os.path.exists(os.path.join(ckdir, "latest")):
                break
            time.sleep(0.2)
        assert os.path.exists(os.path.join(ckdir, "latest")), "no checkpoint"

        # fault injection: SIGKILL the rank-0 process of the live gang
        gang = plat.pytorchjob.gangs.get(uid)
        assert gang is not None
        victim_pid = gang.ranks[0].proc.pid
        os.kill(victim_pid, signal.SIGKILL)

        # the controller must mark Restarting, relaunch, and finish.
        # Restarting flips back to False once Running resumes (the
        # training-operator transition), so presence of the condition —
        # not its current status — is the restart witness.
        deadline = time.time() + 240
        saw_restart = False
        while time.time() < deadline:
            obj = plat.store.get("PyTorchJob", "crashy", "default")
            if get_condition(obj, "Restarting") is not None:
                saw_restart = True
            if has_condition(obj, "Succeeded"):
                break
            if has_condition(obj, "Failed"):
                raise AssertionError(obj["status"])
            time.sleep(0.3)
        assert saw_restart, "never observed Restarting condition"
        assert has_condition(obj, "Succeeded"), obj["status"]
        # and it is no longer marked Restarting after recovery
        assert not has_condition(obj, "Restarting"), obj["status"]
        # resumed (not restarted from scratch): final checkpoint is step-30
        with open(os.path.join(ckdir, "latest")) as f:
            assert f.read().strip() == "step-30"
        # and events tell the story
        reasons = [e["reason"] for e in plat.store.events_for(obj)]
        assert "JobRestarting" in reasons and "JobSucceeded" in reasons


def test_tfjob_parity(tmp_path):
    """TFJob uses tfReplicaSpecs but the same machinery."""
    with Platform(root_dir=str(tmp_path)) as plat:
        job = new_object("TFJob", "tf1", "default", spec={
            "tfReplicaSpecs": {"Worker": {
                "replicas": 1, "restartPolicy": "Never",
                "template": {"model": "mnist-mlp", "steps": 3,
                             "micro_batch": 8, "gpus_per_replica": 0,
                             "save_final": False}}}},
            api_version="kubeflow.org/v1")
        plat.store.create(job)
        deadline = time.time() + 120
        while time.time() < deadline:
            obj = plat.store.get("TFJob", "tf1", "default")
            if has_condition(obj, "Succeeded"):
                return
            if has_condition(obj, "Failed"):
                raise AssertionError(obj["status"])
            time.sleep(0.3)
        raise AssertionError(f"timeout: {obj['status']}")
