"""HTTP API surface over the Store (fastapi) — the kube-apiserver stand-in.

Gives the control plane a real wire interface: the initc waiter (grove_amd/initc.py),
external tooling and multi-process deployments talk to the store over HTTP exactly like
the reference's components talk to the apiserver. Supports CRUD, the status
subresource, label selectors, and ndjson watch streams.

Paths (kube-style):
  GET/POST   /apis/{group}/{version}/namespaces/{ns}/{plural}
  GET/PUT/DELETE /apis/{group}/{version}/namespaces/{ns}/{plural}/{name}
  PUT        .../{name}/status
  GET        ...?labelSelector=k=v,k2=v2
  GET        ...?watch=true   (ndjson stream of {"type", "object"})
Core v1 kinds use /api/v1/... ; cluster-scoped kinds omit the namespaces segment.
"""
import json
import queue
import threading
from typing import Dict, Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, PlainTextResponse, StreamingResponse

from .store import Store, ApiError
from .identity import as_user, ANONYMOUS_USER

# plural -> kind for everything the stack serves
PLURALS: Dict[str, str] = {
    "podcliquesets": "PodCliqueSet",
    "podcliques": "PodClique",
    "podcliquescalinggroups": "PodCliqueScalingGroup",
    "podgangs": "PodGang",
    "clustertopologybindings": "ClusterTopologyBinding",
    "schedulertopologies": "SchedulerTopology",
    "pods": "Pod",
    "services": "Service",
    "secrets": "Secret",
    "serviceaccounts": "ServiceAccount",
    "roles": "Role",
    "rolebindings": "RoleBinding",
    "horizontalpodautoscalers": "HorizontalPodAutoscaler",
    "resourceclaims": "ResourceClaim",
    "nodes": "Node",
}
CLUSTER_SCOPED_PLURALS = {"clustertopologybindings", "schedulertopologies", "nodes"}


def parse_selector(sel: Optional[str]) -> Optional[Dict[str, str]]:
    if not sel:
        return None
    out = {}
    for part in sel.split(","):
        if "=" in part:
            k, v = part.split("=", 1)
            out[k.strip()] = v.strip()
    return out


def build_app(store: Store, metrics_fn=None, auth_tokens: Optional[Dict[str, str]] = None):
    """auth_tokens maps bearer token -> user identity. HTTP callers authenticate
    per request: a bearer token resolves through the static map, then through
    SA-token Secrets in the store (kubernetes.io/service-account-token type, the
    <pcs>-ic-sat contract); anything else runs as system:anonymous, which the
    Authorizer treats as a non-operator identity — unauthenticated peers can no
    longer mutate grove-managed resources over the wire."""
    app = FastAPI(title="grove-amd apiserver")

    def user_of(request: Request) -> str:
        auth = request.headers.get("authorization", "")
        if auth.lower().startswith("bearer "):
            token = auth[7:].strip()
            if auth_tokens and token in auth_tokens:
                return auth_tokens[token]
            for s in store.list("Secret", None):
                if s.get("type") != "kubernetes.io/service-account-token":
                    continue
                tok = (s.get("stringData") or {}).get("token") or \
                      (s.get("data") or {}).get("token")
                if tok and tok == token:
                    md = s.get("metadata", {})
                    sa = (md.get("annotations") or {}).get(
                        "kubernetes.io/service-account.name", "")
                    ns = md.get("namespace", "default")
                    return f"system:serviceaccount:{ns}:{sa}"
        return ANONYMOUS_USER

    def err(e: ApiError):
        return JSONResponse(status_code=e.code, content={
            "kind": "Status", "status": "Failure", "reason": e.reason,
            "message": e.message, "code": e.code})

    def kind_of(plural: str) -> str:
        kind = PLURALS.get(plural)
        if kind is None:
            raise ApiError(404, "NotFound", f"unknown resource {plural!r}")
        return kind

    async def handle_list_or_watch(request: Request, plural: str,
                                   ns: Optional[str]):
        kind = kind_of(plural)
        params = request.query_params
        if params.get("watch") in ("true", "1"):
            since_rv = params.get("resourceVersion") or None
            bookmarks = params.get("allowWatchBookmarks") in ("true", "1")
            w = store.watch(kind,
                            seed=(since_rv is None
                                  and params.get("seed", "true") in ("true", "1")),
                            since_rv=since_rv)

            def stream():
                try:
                    while True:
                        try:
                            ev, obj = w.queue.get(timeout=1.0)
                        except queue.Empty:
                            if bookmarks:
                                # kube BOOKMARK: progress marker carrying only the
                                # current resourceVersion, so clients can resume
                                # without replaying history
                                yield json.dumps({"type": "BOOKMARK", "object": {
                                    "kind": kind, "metadata": {
                                        "resourceVersion": store.current_rv()}}}) \
                                    + "\n"
                            else:
                                yield "\n"  # keepalive
                            continue
                        yield json.dumps({"type": ev, "object": obj}) + "\n"
                finally:
                    w.stop()
            return StreamingResponse(stream(), media_type="application/x-ndjson")
        selector = parse_selector(params.get("labelSelector"))
        limit = int(params["limit"]) if params.get("limit") else None
        cont = params.get("continue") or None
        items, next_cont, rv = store.list_page(kind, ns, selector, limit, cont)
        list_meta = {"resourceVersion": rv}
        if next_cont:
            list_meta["continue"] = next_cont
        return JSONResponse({"kind": f"{kind}List", "apiVersion": "v1",
                             "metadata": list_meta, "items": items})

    # ---- namespaced ----
    @app.get("/apis/{group}/{version}/namespaces/{ns}/{plural}")
    @app.get("/api/{version}/namespaces/{ns}/{plural}")
    async def list_ns(request: Request, plural: str, ns: str,
                      group: str = "", version: str = "v1"):
        try:
            return await handle_list_or_watch(request, plural, ns)
        except ApiError as e:
            return err(e)

    @app.post("/apis/{group}/{version}/namespaces/{ns}/{plural}")
    @app.post("/api/{version}/namespaces/{ns}/{plural}")
    async def create_ns(request: Request, plural: str, ns: str,
                        group: str = "", version: str = "v1"):
        try:
            obj = await request.json()
            obj.setdefault("kind", kind_of(plural))
            obj.setdefault("metadata", {})["namespace"] = ns
            with as_user(user_of(request)):
                return JSONResponse(store.create(obj), status_code=201)
        except ApiError as e:
            return err(e)

    @app.get("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}")
    @app.get("/api/{version}/namespaces/{ns}/{plural}/{name}")
    async def get_ns(plural: str, ns: str, name: str,
                     group: str = "", version: str = "v1"):
        try:
            return JSONResponse(store.get(kind_of(plural), ns, name))
        except ApiError as e:
            return err(e)

    @app.put("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}")
    @app.put("/api/{version}/namespaces/{ns}/{plural}/{name}")
    async def update_ns(request: Request, plural: str, ns: str, name: str,
                        group: str = "", version: str = "v1"):
        try:
            obj = await request.json()
            obj.setdefault("kind", kind_of(plural))
            obj.setdefault("metadata", {})["namespace"] = ns
            obj["metadata"]["name"] = name
            with as_user(user_of(request)):
                return JSONResponse(store.update(obj))
        except ApiError as e:
            return err(e)

    @app.put("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}/status")
    @app.put("/api/{version}/namespaces/{ns}/{plural}/{name}/status")
    async def update_status_ns(request: Request, plural: str, ns: str, name: str,
                               group: str = "", version: str = "v1"):
        try:
            obj = await request.json()
            obj.setdefault("kind", kind_of(plural))
            obj.setdefault("metadata", {})["namespace"] = ns
            obj["metadata"]["name"] = name
            with as_user(user_of(request)):
                return JSONResponse(store.update_status(obj))
        except ApiError as e:
            return err(e)

    @app.delete("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}")
    @app.delete("/api/{version}/namespaces/{ns}/{plural}/{name}")
    async def delete_ns(request: Request, plural: str, ns: str, name: str,
                        group: str = "", version: str = "v1"):
        try:
            with as_user(user_of(request)):
                store.delete(kind_of(plural), ns, name)
            return JSONResponse({"kind": "Status", "status": "Success"})
        except ApiError as e:
            return err(e)

    # kubectl `patch --type=merge|strategic` analog: strategy picked by the
    # request Content-Type, exactly like the apiserver (kubecore/patching.py).
    from .patching import json_merge_patch, strategic_merge_patch

    def _patch_fn(request: Request):
        ctype = request.headers.get("content-type", "")
        if "strategic-merge-patch" in ctype:
            return strategic_merge_patch
        return json_merge_patch

    @app.patch("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}")
    @app.patch("/api/{version}/namespaces/{ns}/{plural}/{name}")
    async def patch_ns(request: Request, plural: str, ns: str, name: str,
                       group: str = "", version: str = "v1"):
        try:
            body = await request.json()
            merge = _patch_fn(request)

            def apply(o):
                merge(o, body)
            with as_user(user_of(request)):
                return JSONResponse(store.patch(kind_of(plural), ns, name, apply))
        except ApiError as e:
            return err(e)

    # merge-patch on the status subresource (client-go Status().Patch analog)
    @app.patch("/apis/{group}/{version}/namespaces/{ns}/{plural}/{name}/status")
    @app.patch("/api/{version}/namespaces/{ns}/{plural}/{name}/status")
    async def patch_status_ns(request: Request, plural: str, ns: str, name: str,
                              group: str = "", version: str = "v1"):
        try:
            body = await request.json()
            merge = _patch_fn(request)

            def apply(o):
                merge(o, body)
            with as_user(user_of(request)):
                return JSONResponse(
                    store.patch(kind_of(plural), ns, name, apply, status=True))
        except ApiError as e:
            return err(e)

    @app.patch("/apis/{group}/{version}/{plural}/{name}/status")
    @app.patch("/api/{version}/{plural}/{name}/status")
    async def patch_status_cluster(request: Request, plural: str, name: str,
                                   group: str = "", version: str = "v1"):
        try:
            body = await request.json()
            merge = _patch_fn(request)

            def apply(o):
                merge(o, body)
            with as_user(user_of(request)):
                return JSONResponse(
                    store.patch(kind_of(plural), None, name, apply, status=True))
        except ApiError as e:
            return err(e)

    @app.patch("/apis/{group}/{version}/{plural}/{name}")
    @app.patch("/api/{version}/{plural}/{name}")
    async def patch_cluster(request: Request, plural: str, name: str,
                            group: str = "", version: str = "v1"):
        try:
            body = await request.json()
            merge = _patch_fn(request)

            def apply(o):
                merge(o, body)
            with as_user(user_of(request)):
                return JSONResponse(store.patch(kind_of(plural), None, name, apply))
        except ApiError as e:
            return err(e)

    # ---- cluster-scoped ----
    @app.get("/apis/{group}/{version}/{plural}")
    @app.get("/api/{version}/{plural}")
    async def list_cluster(request: Request, plural: str,
                           group: str = "", version: str = "v1"):
        try:
            return await handle_list_or_watch(request, plural, None)
        except ApiError as e:
            return err(e)

    @app.post("/apis/{group}/{version}/{plural}")
    @app.post("/api/{version}/{plural}")
    async def create_cluster(request: Request, plural: str,
                             group: str = "", version: str = "v1"):
        try:
            obj = await request.json()
            obj.setdefault("kind", kind_of(plural))
            with as_user(user_of(request)):
                return JSONResponse(store.create(obj), status_code=201)
        except ApiError as e:
            return err(e)

    @app.put("/apis/{group}/{version}/{plural}/{name}")
    @app.put("/api/{version}/{plural}/{name}")
    async def update_cluster(request: Request, plural: str, name: str,
                             group: str = "", version: str = "v1"):
        try:
            obj = await request.json()
            obj.setdefault("kind", kind_of(plural))
            obj.setdefault("metadata", {})["name"] = name
            with as_user(user_of(request)):
                return JSONResponse(store.update(obj))
        except ApiError as e:
            return err(e)

    @app.put("/apis/{group}/{version}/{plural}/{name}/status")
    @app.put("/api/{version}/{plural}/{name}/status")
    async def update_cluster_status(request: Request, plural: str, name: str,
                                    group: str = "", version: str = "v1"):
        try:
            obj = await request.json()
            obj.setdefault("kind", kind_of(plural))
            obj.setdefault("metadata", {})["name"] = name
            with as_user(user_of(request)):
                return JSONResponse(store.update_status(obj))
        except ApiError as e:
            return err(e)

    @app.delete("/apis/{group}/{version}/{plural}/{name}")
    @app.delete("/api/{version}/{plural}/{name}")
    async def delete_cluster(request: Request, plural: str, name: str,
                             group: str = "", version: str = "v1"):
        try:
            with as_user(user_of(request)):
                store.delete(kind_of(plural), None, name)
            return JSONResponse({"kind": "Status", "status": "Success"})
        except ApiError as e:
            return err(e)

    @app.get("/apis/{group}/{version}/{plural}/{name}")
    @app.get("/api/{version}/{plural}/{name}")
    async def get_cluster(plural: str, name: str,
                          group: str = "", version: str = "v1"):
        try:
            return JSONResponse(store.get(kind_of(plural), None, name))
        except ApiError as e:
            return err(e)

    # --- kubectl-style API discovery (client-go discovery analog) ---
    @app.get("/apis")
    async def discovery_groups():
        def g(name, version):
            return {"name": name,
                    "versions": [{"groupVersion": f"{name}/{version}",
                                  "version": version}],
                    "preferredVersion": {"groupVersion": f"{name}/{version}",
                                         "version": version}}
        return JSONResponse({"kind": "APIGroupList", "apiVersion": "v1",
                             "groups": [g("grove.io", "v1alpha1"),
                                        g("scheduler.grove.io", "v1alpha1"),
                                        g("autoscaling", "v2")]})

    @app.get("/apis/{group}/{version}")
    async def discovery_resources(group: str, version: str):
        gv = f"{group}/{version}"
        res = []
        for plural, kind in sorted(PLURALS.items()):
            k_group = ("grove.io/v1alpha1" if kind.startswith("PodClique")
                       or kind == "ClusterTopologyBinding" else
                       "scheduler.grove.io/v1alpha1" if kind == "PodGang" else None)
            if k_group != gv:
                continue
            res.append({"name": plural, "singularName": kind.lower(), "kind": kind,
                        "namespaced": plural not in CLUSTER_SCOPED_PLURALS,
                        "verbs": ["create", "delete", "get", "list", "patch",
                                  "update", "watch"]})
            res.append({"name": f"{plural}/status", "singularName": "", "kind": kind,
                        "namespaced": plural not in CLUSTER_SCOPED_PLURALS,
                        "verbs": ["get", "patch", "update"]})
        return JSONResponse({"kind": "APIResourceList", "apiVersion": "v1",
                             "groupVersion": gv, "resources": res})

    @app.get("/debug/profile")
    async def debug_profile(seconds: float = 2.0, interval_ms: float = 10.0):
        """Sampling profiler (the Pyroscope/pprof analog): samples every thread's
        stack and returns collapsed stacks (flamegraph.pl / speedscope format)."""
        import asyncio
        import collections
        import sys
        import traceback
        counts: "collections.Counter[str]" = collections.Counter()
        deadline = asyncio.get_event_loop().time() + min(seconds, 60.0)
        while asyncio.get_event_loop().time() < deadline:
            for tid, frame in sys._current_frames().items():
                stack = traceback.extract_stack(frame)
                key = ";".join(f"{f.name} ({f.filename.rsplit('/', 1)[-1]}:{f.lineno})"
                               for f in stack[-25:])
                counts[key] += 1
            await asyncio.sleep(max(interval_ms, 1.0) / 1000.0)
        lines = [f"{k} {v}" for k, v in counts.most_common()]
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.get("/debug/stacks")
    async def debug_stacks():
        import sys
        import traceback
        out = []
        for tid, frame in sys._current_frames().items():
            out.append(f"--- thread {tid} ---")
            out.extend(l.rstrip() for l in traceback.format_stack(frame))
        return PlainTextResponse("\n".join(out) + "\n")

    @app.get("/debug/events")
    async def debug_events():
        return JSONResponse(store.events[-200:])

    @app.get("/healthz")
    async def healthz():
        return {"status": "ok"}

    @app.get("/readyz")
    async def readyz():
        return {"status": "ok"}

    @app.get("/metrics")
    async def metrics():
        lines = ["# TYPE grove_store_objects gauge"]
        for kind, n in store.stats().items():
            lines.append(f'grove_store_objects{{kind="{kind}"}} {n}')
        if metrics_fn is not None:
            lines.extend(metrics_fn())
        return PlainTextResponse("\n".join(lines) + "\n")

    return app


class ApiServer:
    """uvicorn in a background thread."""

    def __init__(self, store: Store, host: str = "127.0.0.1", port: int = 8081,
                 metrics_fn=None, ssl_certfile: Optional[str] = None,
                 ssl_keyfile: Optional[str] = None,
                 auth_tokens: Optional[Dict[str, str]] = None):
        self.store = store
        self.host = host
        self.port = port
        self.metrics_fn = metrics_fn
        self.ssl_certfile = ssl_certfile
        self.ssl_keyfile = ssl_keyfile
        self.auth_tokens = auth_tokens
        self._server = None
        self._thread: Optional[threading.Thread] = None

    def start(self) -> "ApiServer":
        import uvicorn
        app = build_app(self.store, self.metrics_fn, self.auth_tokens)
        config = uvicorn.Config(app, host=self.host, port=self.port,
                                log_level="warning", lifespan="off",
                                ssl_certfile=self.ssl_certfile,
                                ssl_keyfile=self.ssl_keyfile)
        self._server = uvicorn.Server(config)
        self._thread = threading.Thread(target=self._server.run, daemon=True)
        self._thread.start()
        import time
        import urllib.request
        scheme = "https" if self.ssl_certfile else "http"
        import ssl as _ssl
        ctx = _ssl._create_unverified_context() if self.ssl_certfile else None
        for _ in range(200):
            try:
                urllib.request.urlopen(
                    f"{scheme}://{self.host}:{self.port}/healthz", timeout=0.2,
                    context=ctx)
                return self
            except Exception:
                time.sleep(0.05)
        raise RuntimeError("apiserver failed to start")

    @property
    def url(self) -> str:
        scheme = "https" if self.ssl_certfile else "http"
        return f"{scheme}://{self.host}:{self.port}"

    def stop(self) -> None:
        if self._server is not None:
            self._server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout=5)
