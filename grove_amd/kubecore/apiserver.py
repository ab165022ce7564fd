"""HTTP API surface over the Store — the kube-apiserver stand-in.

Gives the control plane a real wire interface: the initc waiter (grove_amd/initc.py),
external tooling and multi-process deployments talk to the store over HTTP exactly like
the reference's components talk to the apiserver. Supports CRUD, the status
subresource, label selectors + chunked lists, ndjson watch streams with
resourceVersion resume + BOOKMARKs, merge/strategic-merge PATCH, and bearer-token
authentication (static map or SA-token Secrets; unauthenticated = system:anonymous).

Paths (kube-style):
  GET/POST   /apis/{group}/{version}/namespaces/{ns}/{plural}
  GET/PUT/PATCH/DELETE .../{plural}/{name}   (+ /status subresource)
  GET        ...?labelSelector=k=v&limit=N&continue=TOK
  GET        ...?watch=true[&resourceVersion=RV][&allowWatchBookmarks=true]
Core v1 kinds use /api/v1/... ; cluster-scoped kinds omit the namespaces segment.

Architecture: the regular data-plane grammar is served by a raw ASGI dispatcher
(kubecore/dataplane.py — the framework-routed path cost ~670 µs/request in routing
machinery, the raw path ~5x less); FastAPI serves only the irregular surface
(discovery, /debug, health, metrics).
"""
import json
import threading
from typing import Dict, Optional

from fastapi import FastAPI
from fastapi.responses import JSONResponse, PlainTextResponse

from .store import Store
from .dataplane import DataPlane, parse_selector  # noqa: F401 (re-export)

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
    # apiextensions surface: lets the install-crds initc target THIS apiserver
    "customresourcedefinitions": "CustomResourceDefinition",
}
CLUSTER_SCOPED_PLURALS = {"clustertopologybindings", "schedulertopologies", "nodes",
                          "customresourcedefinitions"}


def build_app(store: Store, metrics_fn=None,
              auth_tokens: Optional[Dict[str, str]] = None):
    """ASGI app: raw data plane in front, FastAPI behind for the irregular routes."""
    plane = DataPlane(store, PLURALS, CLUSTER_SCOPED_PLURALS, auth_tokens)
    aux = _build_aux_app(store, metrics_fn)

    async def app(scope, receive, send):
        if scope["type"] == "http":
            parsed = plane.parse(scope.get("path", ""))
            if parsed is not None:
                await plane(scope, receive, send, parsed)
                return
            # debug introspection carries cluster state (events, stacks) and can
            # burn CPU (profile): when authentication is configured, require it
            # there too — anonymous callers get 403
            if auth_tokens and scope.get("path", "").startswith("/debug"):
                from .identity import ANONYMOUS_USER
                headers = {k.lower(): v for k, v in scope.get("headers") or []}
                if plane._user(headers) == ANONYMOUS_USER:
                    body = (b'{"kind":"Status","status":"Failure",'
                            b'"reason":"Forbidden","code":403}')
                    await send({"type": "http.response.start", "status": 403,
                                "headers": [(b"content-type",
                                             b"application/json")]})
                    await send({"type": "http.response.body", "body": body})
                    return
        await aux(scope, receive, send)

    app.dataplane = plane  # ApiServer.stop() signals open watch streams
    return app


def _build_aux_app(store: Store, metrics_fn=None) -> FastAPI:
    app = FastAPI(title="grove-amd apiserver")

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
        self._app = app
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
        plane = getattr(getattr(self, "_app", None), "dataplane", None)
        if plane is not None:
            plane.shutdown.set()  # end open watch streams promptly
        if self._server is not None:
            self._server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout=5)
