"""kubeflow-amd CLI — kubectl/kfctl-style control of the platform.

    python -m kubeflow_amd.cli serve             # start platform + REST API
    python -m kubeflow_amd.cli get pytorchjobs -n default
    python -m kubeflow_amd.cli apply -f job.yaml
    python -m kubeflow_amd.cli delete notebooks my-nb -n alice
    python -m kubeflow_amd.cli submit-train --model llama3-8b --gpus 8
    python -m kubeflow_amd.cli logs my-job -n default
    python -m kubeflow_amd.cli gpus

`serve` owns the in-process control plane; every other command talks to it
over the REST API (KF_API, default http://127.0.0.1:8099) with the same
envelope the web apps use.
"""
from __future__ import annotations

import json
import os
import time
from typing import Optional

import typer
import urllib.request

app = typer.Typer(help="MI355X-native Kubeflow platform CLI",
                  no_args_is_help=True, add_completion=False)

API = os.environ.get("KF_API", "http://127.0.0.1:8099")
USER = os.environ.get("KF_USER", "admin@kubeflow.org")


def _req(method: str, path: str, body: Optional[dict] = None):
    data = json.dumps(body).encode() if body is not None else None
    req = urllib.request.Request(API + path, data=data, method=method,
                                 headers={"Content-Type": "application/json",
                                          "kubeflow-userid": USER})
    try:
        with urllib.request.urlopen(req, timeout=30) as r:
            return json.loads(r.read())
    except urllib.error.HTTPError as e:
        payload = json.loads(e.read())
        typer.secho(f"error {payload.get('status')}: {payload.get('log')}",
                    fg="red", err=True)
        raise typer.Exit(1)
    except urllib.error.URLError:
        typer.secho(f"cannot reach platform API at {API} — run "
                    "`python -m kubeflow_amd.cli serve` first", fg="red",
                    err=True)
        raise typer.Exit(1)


@app.command()
def serve(root: str = typer.Option("/tmp/kubeflow-amd", help="state dir"),
          port: int = typer.Option(8099),
          persist: bool = typer.Option(True, help="persist the object store")):
    """Start the platform: controllers + gang scheduler + REST API."""
    import uvicorn
    from kubeflow_amd.platform import Platform
    from kubeflow_amd.api.server import build_app
    plat = Platform(root_dir=root, persist=persist).start()
    api = build_app(plat.store, scheduler=plat.scheduler, root_dir=root,
                    volume_controller=plat.volume)
    typer.secho(f"kubeflow-amd platform on http://127.0.0.1:{port} "
                f"(state: {root}, GPUs: {plat.inventory.n_gpus})", fg="green")
    try:
        uvicorn.run(api, host="127.0.0.1", port=port, log_level="warning")
    finally:
        plat.stop()


@app.command()
def get(kind: str, name: Optional[str] = typer.Argument(None),
        namespace: str = typer.Option("default", "-n", "--namespace"),
        output: str = typer.Option("table", "-o")):
    """List or get resources (plural kind, e.g. pytorchjobs, notebooks)."""
    if name:
        body = _req("GET", f"/api/namespaces/{namespace}/{kind}/{name}")
        field = next(k for k in body if k not in
                     ("status", "success", "user", "events"))
        print(json.dumps(body[field], indent=2))
        return
    body = _req("GET", f"/api/namespaces/{namespace}/{kind}")
    items = body.get(kind, [])
    if output == "json":
        print(json.dumps(items, indent=2))
        return
    rows = []
    for o in items:
        conds = [c["type"] for c in o.get("status", {}).get("conditions", [])
                 if c.get("status") == "True"]
        rows.append((o["metadata"]["name"],
                     ",".join(conds) or "-",
                     o["metadata"].get("creationTimestamp", "")))
    w = max([len(r[0]) for r in rows] + [4])
    print(f"{'NAME':<{w}}  {'CONDITIONS':<24}  CREATED")
    for r in rows:
        print(f"{r[0]:<{w}}  {r[1]:<24}  {r[2]}")


@app.command()
def apply(file: str = typer.Option(..., "-f", "--file")):
    """Create resources from a YAML/JSON manifest (kubectl apply style)."""
    import yaml
    with open(file) as f:
        docs = list(yaml.safe_load_all(f))
    plural = {"PyTorchJob": "pytorchjobs", "TFJob": "tfjobs",
              "Notebook": "notebooks", "Tensorboard": "tensorboards",
              "PersistentVolumeClaim": "pvcs", "PodDefault": "poddefaults",
              "ConfigMap": "configmaps", "ResourceQuota": "resourcequotas",
              "InferenceService": "inferenceservices",
              "Experiment": "experiments", "PipelineRun": "pipelineruns"}
    for doc in docs:
        if not doc:
            continue
        kind = doc["kind"]
        ns = doc.get("metadata", {}).get("namespace", "default")
        if kind == "Profile":
            _req("POST", "/kfam/v1/profiles", doc)
            typer.secho(f"profile/{doc['metadata']['name']} created",
                        fg="green")
            continue
        body = _req("POST", f"/api/namespaces/{ns}/{plural[kind]}", doc)
        typer.secho(f"{plural[kind][:-1]}/{doc['metadata']['name']} created",
                    fg="green")


@app.command()
def delete(kind: str, name: str,
           namespace: str = typer.Option("default", "-n", "--namespace")):
    _req("DELETE", f"/api/namespaces/{namespace}/{kind}/{name}")
    typer.secho(f"{kind}/{name} deleted", fg="green")


@app.command("submit-train")
def submit_train(model: str = typer.Option("llama3-8b"),
                 name: Optional[str] = typer.Option(None),
                 gpus: int = typer.Option(1, help="replicas (1 GPU each)"),
                 steps: int = typer.Option(100),
                 micro_batch: int = typer.Option(4),
                 seq_len: int = typer.Option(4096),
                 lr: float = typer.Option(3e-4),
                 strategy: str = typer.Option(
                     "ddp", help="ddp | tp | pp | ulysses | ep"),
                 degree: int = typer.Option(
                     0, help="tp: degree<replicas builds a TPxDP mesh"),
                 zero: bool = typer.Option(
                     False, help="ZeRO-1 sharded optimizer (ddp only)"),
                 namespace: str = typer.Option("default", "-n")):
    """Submit a PyTorchJob (one process per GPU over RCCL/xGMI)."""
    name = name or f"train-{model.replace('.', '-')}-{int(time.time()) % 100000}"
    template = {"model": model, "steps": steps,
                "micro_batch": micro_batch, "seq_len": seq_len,
                "lr": lr, "gpus_per_replica": 1 if gpus else 0}
    if strategy != "ddp":
        template["parallelism"] = {"strategy": strategy,
                                   "degree": degree or gpus}
    if zero:
        template["zero"] = True
    spec = {"pytorchReplicaSpecs": {"Worker": {
        "replicas": gpus, "restartPolicy": "OnFailure",
        "template": template}}}
    _req("POST", f"/api/namespaces/{namespace}/pytorchjobs",
         {"name": name, "spec": spec})
    typer.secho(f"pytorchjob/{name} submitted", fg="green")


@app.command()
def logs(name: str,
         namespace: str = typer.Option("default", "-n", "--namespace"),
         rank: int = typer.Option(0), tail: int = typer.Option(200)):
    """Tail a PyTorchJob rank's worker log."""
    body = _req("GET", f"/api/namespaces/{namespace}/pytorchjobs/{name}/logs"
                f"?rank={rank}&tail={tail}")
    print(body.get("logs", ""), end="")


@app.command()
def status(name: str,
           namespace: str = typer.Option("default", "-n", "--namespace")):
    """Show a PyTorchJob's status block (conditions, metrics, replicas)."""
    body = _req("GET", f"/api/namespaces/{namespace}/pytorchjobs/{name}")
    job = next(v for k, v in body.items()
               if isinstance(v, dict) and "metadata" in v)
    print(json.dumps(job.get("status", {}), indent=2))


@app.command()
def gpus():
    """Node GPU inventory + scheduler utilization."""
    body = _req("GET", "/api/metrics/node")
    print(json.dumps({k: body[k] for k in ("scheduler", "gpus")
                      if k in body}, indent=2))


@app.command()
def events(namespace: str = typer.Option("default", "-n", "--namespace")):
    body = _req("GET", f"/api/activities/{namespace}")
    for e in body.get("activities", [])[:30]:
        inv = e.get("involvedObject", {})
        print(f"{e.get('lastTimestamp','')}  {e.get('type','')[:7]:<8}"
              f"{e.get('reason',''):<24} {inv.get('kind','')}/"
              f"{inv.get('name','')}: {e.get('message','')[:60]}")


if __name__ == "__main__":
    app()
