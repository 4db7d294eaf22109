"""Full platform on a real MI355X: gang-scheduled GPU PyTorchJob + served
InferenceService over HTTP — the reference's cluster-E2E tier
(kf_is_ready + tf_serving tests) on actual hardware."""
import json
import time
import urllib.request

import pytest
import torch

from kubeflow_amd.api import new_object
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.platform import Platform

pytestmark = pytest.mark.gpu


def _wait(pred, timeout, period=0.5):
    deadline = time.time() + timeout
    while time.time() < deadline:
        v = pred()
        if v:
            return v
        time.sleep(period)
    raise AssertionError("condition not reached")


def test_gpu_pytorchjob_and_inference(tmp_path):
    assert torch.cuda.is_available()
    with Platform(root_dir=str(tmp_path)) as plat:
        assert plat.inventory.n_gpus >= 1
        # --- training job on a real GPU through the whole control plane ---
        job = new_object("PyTorchJob", "gpu-train", "default", spec={
            "pytorchReplicaSpecs": {"Worker": {
                "replicas": 1, "restartPolicy": "Never",
                "template": {"model": "llama-tiny", "steps": 3,
                             "micro_batch": 2, "seq_len": 256,
                             "gpus_per_replica": 1, "status_every": 1,
                             "save_final": False}}}})
        plat.store.create(job)

        def trained():
            o = plat.store.get("PyTorchJob", "gpu-train", "default")
            if has_condition(o, "Failed"):
                raise AssertionError(str(o["status"]))
            return o if has_condition(o, "Succeeded") else None
        job = _wait(trained, timeout=240)
        assert job["status"]["trainingMetrics"]["loss"] is not None

        # --- serving on the same GPU, queried over HTTP ---
        svc = new_object("InferenceService", "gpu-serve", "default", spec={
            "predictor": {"model": "llama-tiny", "gpus": 1,
                          "maxSlots": 4, "maxSeqLen": 512, "maxBatch": 4}},
            api_version="serving.kserve.io/v1beta1")
        plat.store.create(svc)

        def ready():
            o = plat.store.get("InferenceService", "gpu-serve", "default")
            return o if has_condition(o, "Ready") else None
        svc = _wait(ready, timeout=240)
        url = svc["status"]["url"]
        body = json.dumps({"instances": [
            {"prompt_tokens": [1, 2, 3, 4], "max_new_tokens": 6}]}).encode()
        req = urllib.request.Request(
            f"{url}/v1/models/gpu-serve:predict", data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=120) as resp:
            out = json.loads(resp.read())
        pred = out["predictions"][0]
        assert pred["error"] == "" and len(pred["tokens"]) == 6


@pytest.mark.gpu
def test_chunked_prefill_gpu_equality():
    """On hardware, chunked prefill (rect kernel over the KV cache) must
    generate the same greedy tokens as inline full prefill."""
    import torch
    from kubeflow_amd.runtime.serving import InferenceEngine

    torch.manual_seed(5)
    a = InferenceEngine("llama-tiny", max_slots=4, smax=1024, max_batch=4)
    torch.manual_seed(5)
    b = InferenceEngine("llama-tiny", max_slots=4, smax=1024, max_batch=4)
    b.PREFILL_CHUNK = 256
    a.start(precapture=False)
    b.start(precapture=False)
    try:
        prompt = [(i * 13) % a.model.cfg.vocab_size for i in range(1, 700)]
        # keep one stream active on b so the chunked path engages
        a.generate([1, 2, 3], max_new_tokens=20, timeout=120)
        b.generate([1, 2, 3], max_new_tokens=20, timeout=120)
        ra = a.generate(prompt, max_new_tokens=16, timeout=120)
        rb = b.generate(prompt, max_new_tokens=16, timeout=120)
        assert not ra.error and not rb.error, (ra.error, rb.error)
        assert ra.generated == rb.generated, (ra.generated, rb.generated)
    finally:
        a.stop()
        b.stop()
