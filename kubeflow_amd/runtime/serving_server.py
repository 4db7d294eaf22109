"""InferenceService server process — KServe-compatible HTTP front end.

Launched by the InferenceService controller as
    python -m kubeflow_amd.runtime.serving_server --spec <dir>/spec.json

Endpoints (KServe data-plane shapes, which the reference's E2E suite probes
— testing/test_tf_serving.py hits a served model over HTTP):
    GET  /healthz                     liveness/readiness
    GET  /v1/models/<name>            model metadata + ready flag
    POST /v1/models/<name>:predict    {"instances": [{"prompt_tokens": [...],
                                       "max_new_tokens": N}, ...]}
    POST /v2/generate                 single generate request
    POST /v1/models/<name>:generate_stream   SSE token stream
    GET  /metrics                     Prometheus text format
"""
from __future__ import annotations

import argparse
import json
import time

from kubeflow_amd.ops import tunable as _kf_tunable
_kf_tunable.enable()
from fastapi import FastAPI, Request as HttpRequest
from fastapi.responses import PlainTextResponse
import uvicorn

from kubeflow_amd.runtime.serving import InferenceEngine


def build_app(spec: dict) -> FastAPI:
    name = spec.get("name", "model")
    engine = InferenceEngine(
        spec.get("model", "llama-tiny"),
        max_slots=int(spec.get("max_slots", 16)),
        smax=int(spec.get("max_seq_len", 2048)),
        max_batch=int(spec.get("max_batch", 16)),
        storage_uri=spec.get("ckpt_dir") or None,
        quant=spec.get("quantization") or None,
    ).start()
    app = FastAPI(title=f"kubeflow-amd inference: {name}")
    app.state.engine = engine
    t0 = time.time()

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "uptime_s": time.time() - t0}

    @app.get("/v1/models")
    def models_list():  # KServe model-list surface
        return {"models": [name]}

    @app.get(f"/v1/models/{name}")
    def model_meta():
        return {"name": name, "ready": True,
                "model": spec.get("model"),
                "loaded_step": engine.loaded_step,
                "storage_uri": spec.get("storage_uri"),
                "stats": engine.stats}

    @app.post(f"/v1/models/{name}:predict")
    async def predict(req: HttpRequest):
        body = await req.json()
        instances = body.get("instances", [])
        # submit ALL instances first so they decode as one continuous
        # batch; waiting per-instance would serialize the request
        from kubeflow_amd.runtime.serving import Request
        reqs = []
        for inst in instances:
            r = Request(rid=f"p{time.monotonic_ns()}-{len(reqs)}",
                        prompt=list(inst.get("prompt_tokens", [1])),
                        max_new_tokens=int(inst.get("max_new_tokens", 16)),
                        temperature=float(inst.get("temperature", 0.0)))
            engine.submit(r)
            reqs.append(r)
        preds = []
        for r in reqs:
            if not r.done.wait(120.0):
                r.error = r.error or "timeout"
            preds.append({
                "tokens": r.generated,
                **({"class": r.generated[0] if r.generated else None,
                    "scores": r.scores} if engine.classify else {}),
                "error": r.error,
                "ttft_ms": (None if r.first_token_at is None else
                            round(1000 * (r.first_token_at - r.submitted), 2)),
                "latency_ms": (None if r.finished_at is None else
                               round(1000 * (r.finished_at - r.submitted), 2)),
            })
        return {"predictions": preds}

    @app.post("/v2/generate")
    async def generate(req: HttpRequest):
        body = await req.json()
        r = engine.generate(body.get("prompt_tokens", [1]),
                            int(body.get("max_new_tokens", 16)),
                            temperature=float(body.get("temperature", 0.0)))
        return {"tokens": r.generated, "error": r.error}

    @app.post(f"/v1/models/{name}:generate_stream")
    async def generate_stream(req: HttpRequest):
        """Server-sent events: one `data: {"token": t}` line per decoded
        token, then `data: [DONE]` (the KServe/OpenAI streaming shape)."""
        import anyio
        from fastapi.responses import StreamingResponse

        body = await req.json()
        gen = engine.generate_stream(
            body.get("prompt_tokens", [1]),
            int(body.get("max_new_tokens", 16)),
            temperature=float(body.get("temperature", 0.0)))

        async def _events():
            while True:
                tok = await anyio.to_thread.run_sync(
                    lambda: next(gen, None))
                if tok is None:
                    yield "data: [DONE]\n\n"
                    return
                yield f'data: {{"token": {tok}}}\n\n'

        return StreamingResponse(_events(), media_type="text/event-stream")

    @app.get("/metrics")
    def metrics():
        s = engine.stats
        lines = [
            "# TYPE kf_serving_requests_total counter",
            f'kf_serving_requests_total{{model="{name}"}} {s["requests"]}',
            "# TYPE kf_serving_completed_total counter",
            f'kf_serving_completed_total{{model="{name}"}} {s["completed"]}',
            "# TYPE kf_serving_tokens_out_total counter",
            f'kf_serving_tokens_out_total{{model="{name}"}} {s["tokens_out"]}',
            "# TYPE kf_serving_prefill_tokens_total counter",
            f'kf_serving_prefill_tokens_total{{model="{name}"}} '
            f'{s["prefill_tokens"]}',
            "# TYPE kf_serving_graph_replays_total counter",
            f'kf_serving_graph_replays_total{{model="{name}"}} '
            f'{s["graph_replays"]}',
            "# TYPE kf_serving_active_streams gauge",
            f'kf_serving_active_streams{{model="{name}"}} '
            f'{len(engine.active)}',
            "# TYPE kf_serving_quantized gauge",
            f'kf_serving_quantized{{model="{name}"}} '
            f'{1 if engine.quant else 0}',
        ]
        return PlainTextResponse("\n".join(lines) + "\n")

    return app


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--spec", required=True)
    args = ap.parse_args(argv)
    with open(args.spec) as f:
        spec = json.load(f)
    app = build_app(spec)
    uvicorn.run(app, host="127.0.0.1", port=int(spec.get("port", 8085)),
                log_level="warning")


if __name__ == "__main__":
    main()
