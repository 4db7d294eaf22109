"""Platform REST API — CRUD web apps + kfam + central-dashboard BFF in one.

Endpoint surface and semantics ported from the reference (SURVEY.md
Appendix A):

  * response envelope: success -> {"status": 200, "success": true,
    "user": <user>, "<field>": <data>} ; failure -> {"success": false,
    "log": msg, "status": code, "user": <user>}
    (crud_backend/api/utils.py:6-25, error handlers handlers.py:13-41);
  * identity: trusted `kubeflow-userid` header (env USERID_HEADER /
    USERID_PREFIX), defaulting to anonymous@kubeflow.org like the dashboard
    middleware (attach_user_middleware.ts:8-24); APP_DISABLE_AUTH skips
    checks (crud_backend/settings.py:3-6);
  * authz: SubjectAccessReview analog — the user needs a kfam role in the
    namespace (view for GET, edit for mutations, admin for bindings); a
    namespace with no bindings at all is open (single-user bootstrap);
  * CSRF: double-submit cookie XSRF-TOKEN vs X-XSRF-TOKEN header on
    mutations (crud_backend/csrf.py:57-112), enabled with KF_CSRF=1;
  * generic resource routes /api/namespaces/<ns>/<plural>[...] over the
    object store (the custom-objects API shape the web apps use);
  * notebook stop/start via PATCH {"stopped": bool} -> the
    `kubeflow-resource-stopped` annotation (apps/common/routes/patch.py);
  * kfam REST /kfam/v1/{profiles,bindings,role/clusteradmin}
    (access-management/kfam/routers.go:32-100);
  * dashboard BFF /api/{workgroup/*,activities,namespaces,dashboard-links,
    metrics/*} (centraldashboard/app/api.ts, api_workgroup.ts);
  * Prometheus /metrics with the reference's metric names
    (notebook-controller/pkg/metrics/metrics.go, monitoring.go:25-60).
"""
from __future__ import annotations

import os
import secrets
import time
from typing import Optional

from fastapi import FastAPI, Request
from fastapi.responses import (FileResponse, JSONResponse,
                               PlainTextResponse)

from kubeflow_amd.api import (ObjectStore, new_object, NotFoundError,
                              AlreadyExistsError, ConflictError)
from kubeflow_amd.kfam import BindingClient

PLURALS = {
    "notebooks": ("Notebook", "kubeflow.org/v1beta1"),
    "tensorboards": ("Tensorboard", "tensorboard.kubeflow.org/v1alpha1"),
    "pvcs": ("PersistentVolumeClaim", "v1"),
    "persistentvolumeclaims": ("PersistentVolumeClaim", "v1"),
    "poddefaults": ("PodDefault", "kubeflow.org/v1alpha1"),
    "pytorchjobs": ("PyTorchJob", "kubeflow.org/v1"),
    "tfjobs": ("TFJob", "kubeflow.org/v1"),
    "inferenceservices": ("InferenceService", "serving.kserve.io/v1beta1"),
    "experiments": ("Experiment", "kubeflow.org/v1beta1"),
    "trials": ("Trial", "kubeflow.org/v1beta1"),
    "pipelineruns": ("PipelineRun", "pipelines.kubeflow.org/v1"),
    "events": ("Event", "v1"),
    "configmaps": ("ConfigMap", "v1"),
    "resourcequotas": ("ResourceQuota", "v1"),
}

STOP_ANNOTATION = "kubeflow-resource-stopped"


class ApiError(Exception):
    def __init__(self, status: int, msg: str):
        self.status = status
        self.msg = msg


def build_app(store: ObjectStore, scheduler=None,
              root_dir: Optional[str] = None,
              volume_controller=None) -> FastAPI:
    app = FastAPI(title="kubeflow-amd platform API")
    bindings = BindingClient(store)
    userid_header = os.environ.get("USERID_HEADER", "kubeflow-userid")
    userid_prefix = os.environ.get("USERID_PREFIX", "")
    disable_auth = os.environ.get("APP_DISABLE_AUTH", "").lower() in (
        "true", "1")
    csrf_on = os.environ.get("KF_CSRF", "0") == "1"
    cluster_admins = set(filter(None, os.environ.get(
        "KF_CLUSTER_ADMINS", "").split(",")))
    started = time.time()

    # ---- reference-named prometheus metrics -----------------------------
    try:
        from prometheus_client import Counter, Gauge, CollectorRegistry
        from prometheus_client import generate_latest
        registry = CollectorRegistry()
        m_req = Counter("request_kf", "requests to kubeflow-amd API",
                        ["path_prefix"], registry=registry)
        m_fail = Counter("request_kf_failure", "failed requests",
                         ["severity"], registry=registry)
        m_heartbeat = Counter("service_heartbeat", "heartbeat",
                              registry=registry)
        g_nb_running = Gauge("notebook_running_total", "running notebooks",
                             registry=registry)
        m_nb_created = Counter("notebook_create_total", "created notebooks",
                               registry=registry)
        m_nb_deleted = Counter("notebook_delete_total", "deleted notebooks",
                               registry=registry)
        g_nb_failed = Gauge("notebook_failed_total", "failed notebooks",
                            registry=registry)
        g_nb_culled = Gauge("notebook_culled_total", "culled notebooks",
                            registry=registry)
    except ImportError:  # pragma: no cover
        registry = None

    # ------------------------------------------------------------ helpers
    def user_of(req: Request) -> str:
        raw = req.headers.get(userid_header)
        if raw is None:
            return "anonymous@kubeflow.org"
        if userid_prefix and raw.startswith(userid_prefix):
            raw = raw[len(userid_prefix):]
        return raw

    def ok(user: str, **fields) -> JSONResponse:
        return JSONResponse({"status": 200, "success": True, "user": user,
                             **fields})

    def fail(user: str, status: int, msg: str) -> JSONResponse:
        if registry:
            m_fail.labels(severity="4xx" if status < 500 else "5xx").inc()
        return JSONResponse({"success": False, "log": msg, "status": status,
                             "user": user}, status_code=status)

    def is_cluster_admin(user: str) -> bool:
        return user in cluster_admins or disable_auth

    strict_authz = os.environ.get("KF_STRICT_AUTHZ", "0") == "1"

    def authz(user: str, namespace: Optional[str], verb: str):
        """SubjectAccessReview analog (crud_backend/authz.py:25-132).

        A namespace with NO bindings at all is open by default (single-user
        bootstrap — otherwise a fresh install locks everyone out); under
        KF_STRICT_AUTHZ=1 the bootstrap hole closes: only profile owners
        and cluster admins may touch binding-less namespaces, so deleting
        the last binding no longer silently opens the namespace."""
        if disable_auth or namespace is None:
            return
        if is_cluster_admin(user):
            return
        role = bindings.role_for(user, namespace)
        ns_bindings = [rb for rb in store.list("RoleBinding", namespace)
                       if "user" in rb["metadata"].get("annotations", {})]
        if not ns_bindings:
            if not strict_authz:
                return  # unmanaged namespace: open (single-user bootstrap)
            for p in store.list("Profile"):
                if p["metadata"]["name"] == namespace and                         _owner_name(p) == user:
                    return
            raise ApiError(
                403, f"user {user} has no access to {namespace} "
                "(strict authz: binding-less namespaces are owner-only)")
        if role is None:
            raise ApiError(403, f"user {user} has no access to {namespace}")
        if verb != "get" and role == "view":
            raise ApiError(403, f"user {user} is view-only in {namespace}")

    def csrf_check(req: Request):
        if not csrf_on or req.method in ("GET", "HEAD", "OPTIONS"):
            return
        cookie = req.cookies.get("XSRF-TOKEN")
        header = req.headers.get("X-XSRF-TOKEN")
        if not cookie or cookie != header:
            raise ApiError(403, "CSRF token missing or invalid")

    @app.middleware("http")
    async def envelope_errors(req: Request, call_next):
        if registry:
            parts = req.url.path.strip("/").split("/")
            m_req.labels(path_prefix=parts[0] if parts else "").inc()
        try:
            resp = await call_next(req)
        except ApiError as e:
            return fail(user_of(req), e.status, e.msg)
        except NotFoundError as e:
            return fail(user_of(req), 404, str(e))
        except AlreadyExistsError as e:
            return fail(user_of(req), 409, str(e))
        except ConflictError as e:
            return fail(user_of(req), 409, str(e))
        except Exception as e:  # catch-all 500 (handlers.py:35-41)
            import traceback
            traceback.print_exc()
            return fail(user_of(req), 500, f"{type(e).__name__}: {e}")
        if csrf_on and "XSRF-TOKEN" not in req.cookies:
            resp.set_cookie("XSRF-TOKEN", secrets.token_urlsafe(16),
                            httponly=False)
        return resp

    # ---------------------------------------------------------- app config
    from kubeflow_amd import config as kfconfig
    cfg = kfconfig.load()

    @app.get("/api/config")
    def get_config(request: Request):
        me = user_of(request)
        return ok(me, config=cfg["spawner"])

    # -------------------------------------------------------------- probes
    @app.get("/healthz")
    @app.get("/api/status/health")
    def healthz():
        return {"status": "ok", "uptime_s": time.time() - started}

    # ------------------------------------------------- generic CRUD routes
    def _kind_of(plural: str):
        if plural not in PLURALS:
            raise ApiError(404, f"unknown resource kind {plural!r}")
        return PLURALS[plural]

    @app.get("/api/namespaces/{ns}/{plural}")
    def list_resources(ns: str, plural: str, request: Request):
        user = user_of(request)
        kind, _ = _kind_of(plural)
        authz(user, ns, "get")
        return ok(user, **{plural: store.list(kind, ns)})

    @app.post("/api/namespaces/{ns}/{plural}")
    async def create_resource(ns: str, plural: str, request: Request):
        user = user_of(request)
        csrf_check(request)
        kind, api_version = _kind_of(plural)
        authz(user, ns, "create")
        body = await request.json()
        if "metadata" in body:  # full object
            obj = body
            obj.setdefault("apiVersion", api_version)
            obj.setdefault("kind", kind)
            obj["metadata"].setdefault("namespace", ns)
            obj.setdefault("status", {}).setdefault("conditions", [])
            for key in ("uid", "resourceVersion"):
                obj["metadata"].pop(key, None)
            tpl = new_object(kind, obj["metadata"]["name"], ns)
            tpl.update({k: v for k, v in obj.items() if k != "metadata"})
            tpl["metadata"].update({k: v for k, v in obj["metadata"].items()
                                    if k not in ("uid", "resourceVersion")})
            obj = tpl
        else:  # short form {"name": ..., "spec"/fields...}
            name = body.pop("name")
            spec = body.get("spec", body)
            if kind == "Notebook":
                # admin value/readOnly enforcement (spawner config parity)
                spec = dict(spec)
                spec.update(kfconfig.enforce_spawner(cfg, spec))
            obj = new_object(kind, name, ns, spec=spec,
                             api_version=api_version)
        created = store.create(obj)
        if registry and kind == "Notebook":
            m_nb_created.inc()
        return ok(user, **{kind.lower(): created,
                           "message": f"{kind} {created['metadata']['name']} created"})

    @app.get("/api/namespaces/{ns}/{plural}/{name}")
    def get_resource(ns: str, plural: str, name: str, request: Request):
        user = user_of(request)
        kind, _ = _kind_of(plural)
        authz(user, ns, "get")
        obj = store.get(kind, name, ns)
        field = kind[0].lower() + kind[1:]
        payload = {field: obj}
        if kind == "Notebook":  # events for status derivation parity
            payload["events"] = store.events_for(obj)
        return ok(user, **payload)

    @app.patch("/api/namespaces/{ns}/{plural}/{name}")
    async def patch_resource(ns: str, plural: str, name: str,
                             request: Request):
        user = user_of(request)
        csrf_check(request)
        kind, _ = _kind_of(plural)
        authz(user, ns, "update")
        body = await request.json()
        if "stopped" in body:  # stop/start semantics (patch.py:22-76)
            if body["stopped"]:
                patch = {"metadata": {"annotations": {
                    STOP_ANNOTATION: time.strftime("%Y-%m-%dT%H:%M:%SZ",
                                                   time.gmtime())}}}
            else:
                patch = {"metadata": {"annotations": {STOP_ANNOTATION: None}}}
        else:
            patch = body
        obj = store.patch(kind, name, ns, patch)
        return ok(user, **{kind.lower(): obj})

    @app.delete("/api/namespaces/{ns}/{plural}/{name}")
    def delete_resource(ns: str, plural: str, name: str, request: Request):
        user = user_of(request)
        csrf_check(request)
        kind, _ = _kind_of(plural)
        authz(user, ns, "delete")
        store.delete(kind, name, ns)
        if registry and kind == "Notebook":
            m_nb_deleted.inc()
        return ok(user, message=f"{kind} {name} deleted")

    # ------------------------------------------------------------- kfam
    def _owner_or_admin(me: str, ns: str) -> bool:
        """isOwnerOrAdmin (access-management/kfam/api_default.go:116-118):
        cluster admin, the namespace profile's owner, or a kfam admin."""
        if is_cluster_admin(me):
            return True
        for p in store.list("Profile"):
            if p["metadata"]["name"] == ns and _owner_name(p) == me:
                return True
        return bindings.role_for(me, ns) == "admin"

    @app.get("/kfam/v1/bindings")
    def kfam_list_bindings(request: Request, user: Optional[str] = None,
                           namespace: Optional[str] = None,
                           role: Optional[str] = None):
        me = user_of(request)
        out = bindings.list(user, namespace, role)
        if not disable_auth and not is_cluster_admin(me):
            # reads are permission-checked like writes: a non-admin sees
            # only their own bindings plus namespaces they own/administer
            out = [b for b in out
                   if b.get("user", {}).get("name") == me
                   or _owner_or_admin(me, b.get("referredNamespace", ""))]
        return ok(me, bindings=out)

    @app.post("/kfam/v1/bindings")
    async def kfam_create_binding(request: Request):
        me = user_of(request)
        csrf_check(request)
        body = await request.json()
        ns = body["referredNamespace"]
        if not _owner_or_admin(me, ns):
            raise ApiError(403, f"{me} is not owner or admin of {ns}")
        b = bindings.create(body["user"]["name"], ns,
                            body.get("roleRef", {}).get("name", "edit")
                            .replace("kubeflow-", ""))
        return ok(me, binding=b["metadata"]["name"])

    @app.delete("/kfam/v1/bindings")
    async def kfam_delete_binding(request: Request):
        me = user_of(request)
        csrf_check(request)
        body = await request.json()
        ns = body["referredNamespace"]
        if not _owner_or_admin(me, ns):
            raise ApiError(403, f"{me} is not owner or admin of {ns}")
        bindings.delete(body["user"]["name"], ns,
                        body.get("roleRef", {}).get("name", "edit")
                        .replace("kubeflow-", ""))
        return ok(me, message="deleted")

    @app.get("/kfam/v1/role/clusteradmin")
    def kfam_cluster_admin(request: Request):
        me = user_of(request)
        return ok(me, clusterAdmin=is_cluster_admin(me))

    @app.get("/kfam/v1/profiles")
    def kfam_list_profiles(request: Request):
        me = user_of(request)
        return ok(me, profiles=store.list("Profile"))

    @app.post("/kfam/v1/profiles")
    async def kfam_create_profile(request: Request):
        me = user_of(request)
        csrf_check(request)
        body = await request.json()
        name = body["metadata"]["name"] if "metadata" in body else body["name"]
        owner = (body.get("spec", {}).get("owner")
                 or {"kind": "User", "name": me})
        prof = new_object("Profile", name, None,
                          spec={"owner": owner},
                          api_version="kubeflow.org/v1")
        store.create(prof)
        return ok(me, profile=name)

    # -------------------------------------------------- dashboard BFF
    @app.get("/api/namespaces")
    def list_namespaces(request: Request):
        me = user_of(request)
        return ok(me, namespaces=[n["metadata"]["name"]
                                  for n in store.list("Namespace")])

    @app.get("/api/workgroup/exists")
    def workgroup_exists(request: Request):
        me = user_of(request)
        owned = [p for p in store.list("Profile")
                 if _owner_name(p) == me]
        shared = {b["referredNamespace"] for b in bindings.list(user=me)}
        return ok(me, hasWorkgroup=bool(owned),
                  hasAuth=me != "anonymous@kubeflow.org",
                  namespaces=sorted({p["metadata"]["name"] for p in owned}
                                    | shared))

    @app.post("/api/workgroup/create")
    async def workgroup_create(request: Request):
        me = user_of(request)
        csrf_check(request)
        body = {}
        try:
            body = await request.json()
        except Exception:
            pass
        name = body.get("namespace") or me.split("@")[0].replace(".", "-")
        prof = new_object("Profile", name, None,
                          spec={"owner": {"kind": "User", "name": me}},
                          api_version="kubeflow.org/v1")
        store.create(prof)
        return ok(me, message=f"profile {name} created", namespace=name)

    @app.get("/api/workgroup/env-info")
    def env_info(request: Request):
        me = user_of(request)
        profiles = store.list("Profile")
        owned = [p["metadata"]["name"] for p in profiles
                 if _owner_name(p) == me]
        return ok(me, isClusterAdmin=is_cluster_admin(me),
                  namespaces=[{
                      "namespace": p["metadata"]["name"],
                      "role": ("owner" if _owner_name(p) == me else
                               bindings.role_for(me, p["metadata"]["name"])
                               or "none"),
                      "user": me} for p in profiles],
                  platform={"provider": "amd-mi355x",
                            "providerName": "kubeflow-amd",
                            "kubeflowVersion": "0.1.0"})

    @app.get("/api/activities/{ns}")
    def activities(ns: str, request: Request):
        me = user_of(request)
        authz(me, ns, "get")
        evs = store.list("Event", ns)
        evs.sort(key=lambda e: e.get("lastTimestamp", ""), reverse=True)
        return ok(me, activities=evs[:100])

    @app.get("/api/dashboard-links")
    def dashboard_links(request: Request):
        me = user_of(request)
        return ok(me, menuLinks=[
            {"type": "item", "link": "/jupyter/", "text": "Notebooks"},
            {"type": "item", "link": "/tensorboards/", "text": "Tensorboards"},
            {"type": "item", "link": "/volumes/", "text": "Volumes"},
            {"type": "item", "link": "/katib/", "text": "Experiments (AutoML)"},
            {"type": "item", "link": "/pipeline/", "text": "Pipelines"},
            {"type": "item", "link": "/models/", "text": "Models"},
        ], externalLinks=[], quickLinks=[
            {"text": "Create a new Notebook server",
             "link": "/jupyter/new"},
            {"text": "Submit a PyTorchJob",
             "link": "/api/namespaces/default/pytorchjobs"},
        ], documentationItems=[])

    @app.get("/api/metrics/{which}")
    def node_metrics(which: str, request: Request):
        me = user_of(request)
        util = scheduler.utilization() if scheduler else {}
        gpus = []
        try:
            import torch
            if torch.cuda.is_available():
                for i in range(torch.cuda.device_count()):
                    free, total = torch.cuda.mem_get_info(i)
                    gpus.append({"index": i, "hbm_used": total - free,
                                 "hbm_total": total})
        except Exception:
            pass
        return ok(me, metric=which, scheduler=util, gpus=gpus)

    # ---------------------------------------------------------------- logs
    def _tail_log(user, subdir, kind, ns, name, rank, tail):
        if root_dir is None:
            raise ApiError(404, "no logs root configured")
        obj = store.get(kind, name, ns)
        uid = obj["metadata"]["uid"]
        path = os.path.join(root_dir, subdir, ns, f"{name}-{uid[:8]}",
                            f"rank-{rank}", "worker.log")
        if not os.path.exists(path):
            raise ApiError(404, f"no log for rank {rank}")
        with open(path, errors="replace") as f:
            lines = f.readlines()[-tail:]
        return ok(user, logs="".join(lines))

    @app.get("/api/namespaces/{ns}/pytorchjobs/{name}/logs")
    def job_logs(ns: str, name: str, request: Request, rank: int = 0,
                 tail: int = 200):
        user = user_of(request)
        authz(user, ns, "get")
        return _tail_log(user, "jobs", "PyTorchJob", ns, name, rank, tail)

    @app.get("/api/namespaces/{ns}/inferenceservices/{name}/logs")
    def svc_logs(ns: str, name: str, request: Request, rank: int = 0,
                 tail: int = 200):
        user = user_of(request)
        authz(user, ns, "get")
        return _tail_log(user, "serving", "InferenceService", ns, name,
                         rank, tail)

    # ------------------------------------------------ central dashboard UI
    # ----------------------------------------------------------- dashboard
    # SPA (kubeflow_amd/dashboard/static): namespace selector + routed
    # resource views with exponential-backoff polling over this BFF —
    # the central-dashboard + crud-web-app frontend layer rebuilt as
    # hand-rolled ES modules (no build step). /classic keeps the
    # server-rendered fallback table.
    _static_dir = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "dashboard", "static")

    @app.get("/")
    def dashboard_spa():
        return FileResponse(os.path.join(_static_dir, "index.html"))

    @app.get("/static/{fname}")
    def dashboard_static(fname: str):
        safe = os.path.basename(fname)
        path = os.path.join(_static_dir, safe)
        if not os.path.exists(path):
            raise ApiError(404, f"no static asset {safe}")
        media = {"js": "text/javascript", "css": "text/css",
                 "html": "text/html"}.get(safe.rsplit(".", 1)[-1])
        return FileResponse(path, media_type=media)

    @app.get("/classic")
    def dashboard(request: Request):
        """Minimal server-rendered dashboard shell — the central-dashboard
        equivalent (namespaces, workloads, activities, GPU utilization)."""
        import html as _html
        me = _html.escape(user_of(request))
        util = scheduler.utilization() if scheduler else {}
        sections = []
        for plural, (kind, _) in PLURALS.items():
            if plural in ("persistentvolumeclaims", "events"):
                continue
            objs = store.list(kind)
            if not objs:
                continue
            rows = "".join(
                f"<tr><td>{_html.escape(o['metadata'].get('namespace') or '')}</td>"
                f"<td>{_html.escape(o['metadata']['name'])}</td>"
                f"<td>{', '.join(c['type'] for c in o.get('status', {}).get('conditions', []) if c.get('status') == 'True') or '-'}</td></tr>"
                for o in objs[:50])
            sections.append(
                f"<h3>{kind}s ({len(objs)})</h3>"
                f"<table border=1 cellpadding=4><tr><th>namespace</th>"
                f"<th>name</th><th>conditions</th></tr>{rows}</table>")
        html = (
            "<html><head><title>kubeflow-amd</title></head><body>"
            f"<h1>kubeflow-amd — MI355X platform</h1>"
            f"<p>user: {me} · GPUs: {util.get('total_gpus', 0)} "
            f"(busy: {util.get('exclusive_busy', 0)}) · jobs: "
            f"{util.get('jobs', 0)}</p>"
            + "".join(sections) +
            "<p><a href='/docs'>REST API docs (OpenAPI)</a> · "
            "<a href='/metrics'>metrics</a></p></body></html>")
        from fastapi.responses import HTMLResponse
        return HTMLResponse(html)

    # ------------------------------------------------------------ metrics
    @app.get("/metrics")
    def metrics():
        if registry is None:
            return PlainTextResponse("")
        m_heartbeat.inc()
        nbs = store.list("Notebook")
        g_nb_running.set(sum(1 for nb in nbs
                             if nb.get("status", {}).get("readyReplicas")))
        # scraped on pull like the reference's custom Collector
        # (notebook-controller/pkg/metrics/metrics.go:82-99)
        g_nb_failed.set(sum(
            1 for nb in nbs
            if any(c.get("type") == "Failed" and c.get("status") == "True"
                   for c in nb.get("status", {}).get("conditions", []))
            or (nb.get("status", {}).get("containerState", {})
                .get("waiting", {}).get("reason") in ("Error",
                                                      "CrashLoopBackOff"))))
        g_nb_culled.set(sum(
            1 for nb in nbs
            if nb["metadata"].get("annotations", {}).get(
                "notebooks.kubeflow.org/culled") == "true"))
        return PlainTextResponse(generate_latest(registry).decode())

    return app


def _owner_name(profile: dict) -> str:
    owner = profile.get("spec", {}).get("owner", {})
    return owner.get("name") if isinstance(owner, dict) else str(owner)
