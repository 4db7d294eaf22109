"""Round-robin reverse proxy fronting InferenceService replicas.

KServe scales predictors with `spec.predictor.replicas` and lets the
Kubernetes Service spread requests; in the process model this proxy IS
that Service: one process listening on the published port, forwarding
each request round-robin to the replica engines (health-aware — a dead
backend is skipped and retried on the next replica).

Spec (JSON file, --spec): {"port": P, "backends": ["http://127.0.0.1:p1",
...], "name": "..."}
"""
from __future__ import annotations

import argparse
import itertools
import json
import threading
import urllib.error
import urllib.request

from fastapi import FastAPI, Request, Response
import uvicorn


def build_app(spec: dict) -> FastAPI:
    app = FastAPI(title=f"kubeflow-amd serving proxy: {spec.get('name')}")
    backends = list(spec["backends"])
    rr = itertools.cycle(range(len(backends)))
    lock = threading.Lock()

    def next_order():
        with lock:
            start = next(rr)
        return [backends[(start + i) % len(backends)]
                for i in range(len(backends))]

    @app.get("/healthz")
    def healthz():
        up = []
        for b in backends:
            try:
                with urllib.request.urlopen(f"{b}/healthz", timeout=2) as r:
                    up.append(r.status == 200)
            except Exception:
                up.append(False)
        # ready when at least one replica serves (KServe marks the service
        # Ready on available replicas, not all)
        code = 200 if any(up) else 503
        return Response(json.dumps({"status": "ok" if code == 200 else
                                    "unavailable",
                                    "replicas": len(backends),
                                    "ready": sum(up)}),
                        status_code=code, media_type="application/json")

    @app.api_route("/{path:path}", methods=["GET", "POST"])
    async def forward(path: str, request: Request):
        body = await request.body()
        last_err = None
        for backend in next_order():
            req = urllib.request.Request(
                f"{backend}/{path}", data=body if body else None,
                headers={"Content-Type":
                         request.headers.get("content-type",
                                             "application/json")},
                method=request.method)
            try:
                import asyncio
                loop = asyncio.get_event_loop()

                def _do(req=req):
                    with urllib.request.urlopen(req, timeout=600) as r:
                        return r.status, r.read()

                status, data = await loop.run_in_executor(None, _do)
                return Response(data, status_code=status,
                                media_type="application/json")
            except urllib.error.HTTPError as e:
                # backend answered: pass its error through (no failover —
                # it is an application error, not a dead replica)
                return Response(e.read(), status_code=e.code,
                                media_type="application/json")
            except Exception as e:
                last_err = e  # dead/unreachable replica: try the next
        return Response(json.dumps({"error": f"no replica reachable: "
                                             f"{last_err}"}),
                        status_code=503, media_type="application/json")

    return app


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--spec", required=True)
    args = ap.parse_args()
    with open(args.spec) as f:
        spec = json.load(f)
    app = build_app(spec)
    uvicorn.run(app, host="127.0.0.1", port=int(spec["port"]),
                log_level="warning")


if __name__ == "__main__":
    main()
