"""Notebook session server — the workload-image contract, as a process.

Replaces the reference's jupyter container images
(example-notebook-servers/jupyter: boots via s6, serves on a port, honors
NB_PREFIX, exposes GET <prefix>/api/status with last_activity —
notebook-controller/pkg/culler/culler.go:39-44,138-169 depends on exactly
that endpoint). This process keeps the same contract so the culler works
unchanged; the "kernel" is a Python exec sandbox instead of Jupyter (no
Jupyter in the offline image).

    GET  <prefix>/api/status       {started, last_activity, connections}
    GET  <prefix>/                 session info page (JSON)
    POST <prefix>/api/execute      {"code": "..."} -> {"output": ...}
    GET  /healthz
"""
from __future__ import annotations

import argparse
import contextlib
import io
import json
import os
import time

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse
import uvicorn


def build_app(prefix: str, workdir: str) -> FastAPI:
    app = FastAPI(title="kubeflow-amd notebook session")
    state = {
        "started": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
        "last_activity": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
        "connections": 0,
        "ns": {},  # exec namespace
    }

    def touch():
        state["last_activity"] = time.strftime("%Y-%m-%dT%H:%M:%SZ",
                                               time.gmtime())

    @app.get("/healthz")
    def healthz():
        return {"status": "ok"}

    @app.get(f"{prefix}/api/status")
    def status():
        # same shape the culler parses (culler.go:138-169): last_activity
        return {"started": state["started"],
                "last_activity": state["last_activity"],
                "connections": state["connections"],
                "kernels": []}

    @app.get(f"{prefix}/")
    def index():
        touch()
        return {"notebook": prefix, "workdir": workdir,
                "last_activity": state["last_activity"]}

    @app.post(f"{prefix}/api/execute")
    async def execute(req: Request):
        touch()
        body = await req.json()
        code = body.get("code", "")
        buf = io.StringIO()
        try:
            with contextlib.redirect_stdout(buf):
                exec(compile(code, "<cell>", "exec"), state["ns"])
            return {"status": "ok", "output": buf.getvalue()}
        except Exception as e:
            return JSONResponse({"status": "error",
                                 "ename": type(e).__name__,
                                 "evalue": str(e),
                                 "output": buf.getvalue()}, status_code=400)

    return app


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--spec", required=True)
    args = ap.parse_args(argv)
    with open(args.spec) as f:
        spec = json.load(f)
    prefix = spec.get("nb_prefix") or os.environ.get("NB_PREFIX", "")
    app = build_app(prefix, os.getcwd())
    uvicorn.run(app, host="127.0.0.1", port=int(spec.get("port", 8888)),
                log_level="warning")


if __name__ == "__main__":
    main()
