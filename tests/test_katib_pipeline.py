"""Katib Experiment + Pipeline DAG e2e on CPU (tiny models, world_size=1).

Mirrors the reference's katib_studyjob_test (submit CR, poll conditions)
and BASELINE config 5's DAG chaining.
"""
import os
import time

import pytest

from kubeflow_amd.api import new_object
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.katib import make_suggestion
from kubeflow_amd.platform import Platform


def _wait(store, kind, name, ns, types=("Succeeded", "Failed"), timeout=180):
    deadline = time.time() + timeout
    while time.time() < deadline:
        obj = store.get(kind, name, ns)
        for t in types:
            if has_condition(obj, t):
                return t, obj
        time.sleep(0.3)
    raise AssertionError(f"{kind}/{name} not terminal: {obj['status']}")


PARAMS = [
    {"name": "lr", "parameterType": "double",
     "feasibleSpace": {"min": "0.001", "max": "0.1", "logScale": True}},
    {"name": "micro_batch", "parameterType": "int",
     "feasibleSpace": {"min": "8", "max": "32"}},
]


def test_suggestion_algorithms():
    for alg in ("random", "grid", "bayesianoptimization", "tpe"):
        sug = make_suggestion(alg, PARAMS, seed=1)
        props = sug.suggest([], 4)
        assert len(props) == 4
        for p in props:
            assert 0.001 <= p["lr"] <= 0.1
            assert 8 <= p["micro_batch"] <= 32
    # bayesopt with observations proposes valid points
    sug = make_suggestion("bayesianoptimization", PARAMS, seed=1)
    obs = [({"lr": 0.01, "micro_batch": 16}, 0.5),
           ({"lr": 0.05, "micro_batch": 8}, 0.9),
           ({"lr": 0.002, "micro_batch": 32}, 0.3),
           ({"lr": 0.09, "micro_batch": 12}, 1.1)]
    props = sug.suggest(obs, 2)
    assert len(props) == 2 and all(0.001 <= p["lr"] <= 0.1 for p in props)


def test_tpe_concentrates_on_good_region():
    """On a known quadratic objective, TPE proposals after observations
    should sit closer to the optimum than uniform-random ones."""
    import math as _m
    params = [{"name": "x", "parameterType": "double",
               "feasibleSpace": {"min": "0.0", "max": "1.0"}}]
    opt = 0.3
    obs = []
    rng = __import__("random").Random(7)
    for _ in range(16):
        x = rng.random()
        obs.append(({"x": x}, (x - opt) ** 2))
    tpe = make_suggestion("tpe", params, seed=2)
    props = tpe.suggest(obs, 20)
    assert all(0.0 <= p["x"] <= 1.0 for p in props)
    mean_dist = sum(abs(p["x"] - opt) for p in props) / len(props)
    # uniform random would average |x-0.3| ~ 0.29; demand clearly better
    assert mean_dist < 0.15, mean_dist


def test_experiment_e2e(tmp_path):
    with Platform(root_dir=str(tmp_path)) as plat:
        exp = new_object("Experiment", "hpo-1", "default", spec={
            "objective": {"type": "minimize",
                          "objectiveMetricName": "loss"},
            "algorithm": {"algorithmName": "random"},
            "parallelTrialCount": 2,
            "maxTrialCount": 4,
            "maxFailedTrialCount": 2,
            "parameters": PARAMS,
            "trialTemplate": {
                "model": "mnist-mlp", "steps": 4, "gpus_per_replica": 0,
                "status_every": 2, "save_final": False, "replicas": 1,
            },
        }, api_version="kubeflow.org/v1beta1")
        plat.store.create(exp)
        state, obj = _wait(plat.store, "Experiment", "hpo-1", "default",
                           timeout=300)
        assert state == "Succeeded", obj["status"]
        assert obj["status"]["trials"] == 4
        assert obj["status"]["trialsSucceeded"] == 4
        opt = obj["status"]["currentOptimalTrial"]
        assert opt["observation"]["metrics"][0]["latest"] is not None
        # trials carry assignments within the space
        trials = plat.store.list("Trial", "default", {"experiment": "hpo-1"})
        assert len(trials) == 4
        for t in trials:
            a = {x["name"]: x["value"]
                 for x in t["spec"]["parameterAssignments"]}
            assert 0.001 <= a["lr"] <= 0.1


def test_pipeline_dag_e2e(tmp_path):
    with Platform(root_dir=str(tmp_path)) as plat:
        run = new_object("PipelineRun", "pipe-1", "default", spec={
            "tasks": [
                {"name": "preprocess", "dependencies": [],
                 "template": {"model": "mnist-mlp", "steps": 2,
                              "gpus_per_replica": 0, "save_final": False}},
                {"name": "train", "dependencies": ["preprocess"],
                 "template": {"model": "mnist-mlp", "steps": 3,
                              "gpus_per_replica": 0, "save_final": False}},
                {"name": "eval", "dependencies": ["train"],
                 "template": {"model": "mnist-mlp", "steps": 2,
                              "gpus_per_replica": 0, "save_final": False}},
                {"name": "deploy", "dependencies": ["eval"],
                 "kind": "InferenceService",
                 "template": {"model": "llama-tiny", "gpus": 0,
                              "maxSlots": 2, "maxSeqLen": 256}},
            ],
        }, api_version="pipelines.kubeflow.org/v1")
        plat.store.create(run)
        state, obj = _wait(plat.store, "PipelineRun", "pipe-1", "default",
                           timeout=300)
        assert state == "Succeeded", obj["status"]
        ts = obj["status"]["taskStates"]
        assert all(ts[k] == "Succeeded"
                   for k in ("preprocess", "train", "eval", "deploy"))
        # ordering respected: train started only after preprocess succeeded
        evs = plat.store.events_for(obj)
        started = [e["message"] for e in evs if e["reason"] == "TaskStarted"]
        assert started.index(
            next(m for m in started if m.startswith("preprocess"))) < \
            started.index(next(m for m in started if m.startswith("train")))


def test_pipeline_invalid_dag(tmp_path):
    with Platform(root_dir=str(tmp_path)) as plat:
        run = new_object("PipelineRun", "bad", "default", spec={
            "tasks": [{"name": "a", "dependencies": ["ghost"],
                       "template": {}}]})
        plat.store.create(run)
        state, obj = _wait(plat.store, "PipelineRun", "bad", "default",
                           timeout=30)
        assert state == "Failed"


def test_experiment_goal_early_stop(tmp_path):
    """Objective goal reached -> Succeeded with GoalReached, no extra trials
    beyond the running ones."""
    with Platform(root_dir=str(tmp_path)) as plat:
        exp = new_object("Experiment", "goal-exp", "default", spec={
            "objective": {"type": "minimize", "objectiveMetricName": "loss",
                          "goal": 1e9},  # any finite loss satisfies it
            "algorithm": {"algorithmName": "random"},
            "parallelTrialCount": 1,
            "maxTrialCount": 10,
            "parameters": PARAMS,
            "trialTemplate": {"model": "mnist-mlp", "steps": 3,
                              "gpus_per_replica": 0, "status_every": 1,
                              "save_final": False, "replicas": 1},
        })
        plat.store.create(exp)
        state, obj = _wait(plat.store, "Experiment", "goal-exp", "default",
                           timeout=180)
        assert state == "Succeeded"
        conds = {c["type"]: c for c in obj["status"]["conditions"]}
        assert conds["Succeeded"]["reason"] == "GoalReached"
        assert obj["status"]["trials"] <= 2  # stopped early, not 10


def test_experiment_failure_budget(tmp_path):
    """Trials that keep failing exhaust maxFailedTrialCount -> Experiment
    Failed (katib failure-budget semantics)."""
    with Platform(root_dir=str(tmp_path)) as plat:
        exp = new_object("Experiment", "doomed", "default", spec={
            "objective": {"type": "minimize", "objectiveMetricName": "loss"},
            "algorithm": {"algorithmName": "random"},
            "parallelTrialCount": 2,
            "maxTrialCount": 8,
            "maxFailedTrialCount": 2,
            "parameters": PARAMS,
            # nonexistent model -> every trial's worker exits 1
            "trialTemplate": {"model": "no-such-model", "steps": 2,
                              "gpus_per_replica": 0, "save_final": False,
                              "replicas": 1},
        })
        plat.store.create(exp)
        state, obj = _wait(plat.store, "Experiment", "doomed", "default",
                           timeout=120)
        assert state == "Failed", obj["status"]
        conds = {c["type"]: c for c in obj["status"]["conditions"]}
        assert conds["Failed"]["reason"] == "TooManyFailedTrials"
        assert obj["status"]["trialsFailed"] >= 2


def test_space_unit_roundtrip_property():
    """to_unit/from_unit roundtrip across the parameter space (hypothesis)."""
    from hypothesis import given, settings, strategies as st
    from kubeflow_amd.katib.suggestion import _Space

    params = [
        {"name": "lr", "parameterType": "double",
         "feasibleSpace": {"min": "0.001", "max": "0.1", "logScale": True}},
        {"name": "bs", "parameterType": "int",
         "feasibleSpace": {"min": "8", "max": "64"}},
        {"name": "opt", "parameterType": "categorical",
         "feasibleSpace": {"list": ["adamw", "sgd", "lion"]}},
    ]
    space = _Space(params)

    @settings(max_examples=200, deadline=None)
    @given(st.floats(0, 1), st.floats(0, 1), st.floats(0, 1))
    def check(u1, u2, u3):
        a = space.from_unit([u1, u2, u3])
        assert 0.001 <= a["lr"] <= 0.1
        assert 8 <= a["bs"] <= 64 and isinstance(a["bs"], int)
        assert a["opt"] in ("adamw", "sgd", "lion")
        # encode-decode is a projection: applying it twice is stable
        b = space.from_unit(space.to_unit(a))
        assert abs(b["lr"] - a["lr"]) < 1e-9 * max(1, abs(a["lr"]))
        assert b["bs"] == a["bs"] and b["opt"] == a["opt"]

    check()


def test_pipeline_train_then_deploy_trained_weights(tmp_path):
    """BASELINE config 5's train->deploy seam with real weight flow: the
    train task checkpoints to a PVC; the deploy task's InferenceService
    uses storageUri=pvc://... and must serve the TRAINED weights (engine
    reports the checkpoint step in its model metadata)."""
    import json
    import urllib.request

    from kubeflow_amd.api.objects import get_condition

    with Platform(root_dir=str(tmp_path)) as plat:
        plat.store.create(new_object(
            "PersistentVolumeClaim", "model-store", "default",
            spec={"resources": {"requests": {"storage": "1Gi"}}}))
        ck = os.path.join(str(tmp_path), "volumes", "default", "model-store",
                          "llama")
        run = new_object("PipelineRun", "pipe-deploy", "default", spec={
            "tasks": [
                {"name": "train", "dependencies": [],
                 "template": {"model": "llama-tiny", "steps": 2,
                              "gpus_per_replica": 0, "micro_batch": 2, "seq_len": 32,
                              "checkpoint_dir": ck}},
                {"name": "deploy", "dependencies": ["train"],
                 "kind": "InferenceService",
                 "template": {"model": "llama-tiny", "gpus": 0,
                              "maxSlots": 2, "maxSeqLen": 128,
                              "storageUri": "pvc://model-store/llama"}},
            ],
        }, api_version="pipelines.kubeflow.org/v1")
        plat.store.create(run)
        state, obj = _wait(plat.store, "PipelineRun", "pipe-deploy",
                           "default", timeout=300)
        assert state == "Succeeded", obj["status"]
        # the deployed engine loaded the trained checkpoint
        svc = plat.store.get("InferenceService", "pipe-deploy-deploy",
                             "default")
        ready = get_condition(svc, "Ready")
        url = ready["message"]
        with urllib.request.urlopen(
                f"{url}/v1/models/pipe-deploy-deploy", timeout=10) as r:
            meta = json.load(r)
        assert meta["loaded_step"] == 2, meta
        assert meta["storage_uri"] == "pvc://model-store/llama"
