"""Raw ASGI data plane for the kube-style API — the apiserver's hot path.

The framework-routed (FastAPI) implementation costs ~670 µs per request in routing/
validation machinery alone; this hand-rolled dispatcher serves the regular kube path
grammar directly from the ASGI scope and keeps the store call inline, cutting the
wire hop that dominates the deployable shape's gang latency. The FastAPI app remains
mounted behind it for the irregular surface (discovery, /debug, health, metrics).

Grammar served (exactly the previous FastAPI routes):
  /api/{version}/...  |  /apis/{group}/{version}/...
  .../namespaces/{ns}/{plural}[/{name}[/status]]     (namespaced)
  .../{plural}[/{name}[/status]]                     (cluster-scoped)
  GET collection (+labelSelector/limit/continue, ?watch=true with
  resourceVersion/allowWatchBookmarks), POST create, GET/PUT/PATCH/DELETE object,
  PUT/PATCH status. PATCH strategy by Content-Type (merge vs strategic).
"""
from __future__ import annotations

import asyncio
import json
import queue
from typing import Any, Dict, List, Optional, Tuple
from urllib.parse import parse_qs, unquote

from .identity import as_user, ANONYMOUS_USER
from .patching import json_merge_patch, strategic_merge_patch
from .store import Store, ApiError

Obj = Dict[str, Any]


def parse_field_selector(sel: Optional[str]) -> Optional[Dict[str, str]]:
    if not sel:
        return None
    out = {}
    for part in sel.split(","):
        if "=" in part:
            k, v = part.split("=", 1)
            out[k.strip().lstrip("=")] = v.strip()
    return out


def parse_selector(sel: Optional[str]) -> Optional[Dict[str, str]]:
    if not sel:
        return None
    out = {}
    for part in sel.split(","):
        if "=" in part:
            k, v = part.split("=", 1)
            out[k.strip()] = v.strip()
    return out


class DataPlane:
    def __init__(self, store: Store, plurals: Dict[str, str],
                 cluster_scoped: set, auth_tokens: Optional[Dict[str, str]]):
        self.store = store
        self.plurals = plurals
        self.cluster_scoped = cluster_scoped
        self.auth_tokens = auth_tokens or {}
        # set by ApiServer.stop(): open watch streams end their responses so
        # uvicorn's graceful shutdown does not wait out its timeout
        import threading
        self.shutdown = threading.Event()

    # ------------------------------------------------------------------ routing
    def parse(self, path: str) -> Optional[Tuple[str, Optional[str],
                                                 Optional[str], bool]]:
        """-> (plural, ns, name, status) or None when the path is not data-plane."""
        parts = [unquote(p) for p in path.strip("/").split("/") if p]
        if not parts:
            return None
        if parts[0] == "api":
            rest = parts[2:]
        elif parts[0] == "apis":
            rest = parts[3:]
        else:
            return None
        if not rest:
            return None  # /apis or /apis/{g}/{v} -> discovery (FastAPI)
        ns: Optional[str] = None
        if rest[0] == "namespaces":
            if len(rest) < 3:
                return None
            ns = rest[1]
            rest = rest[2:]
        plural = rest[0]
        name = rest[1] if len(rest) > 1 else None
        status = len(rest) > 2 and rest[2] == "status"
        if len(rest) > 3 or (len(rest) == 3 and not status):
            return None
        return plural, ns, name, status

    def _kind(self, plural: str) -> str:
        kind = self.plurals.get(plural)
        if kind is None:
            raise ApiError(404, "NotFound", f"unknown resource {plural!r}")
        return kind

    def _user(self, headers: Dict[bytes, bytes]) -> str:
        auth = headers.get(b"authorization", b"").decode()
        if auth.lower().startswith("bearer "):
            token = auth[7:].strip()
            if token in self.auth_tokens:
                return self.auth_tokens[token]
            for s in self.store.list("Secret", None, copy_objects=False):
                if s.get("type") != "kubernetes.io/service-account-token":
                    continue
                tok = (s.get("stringData") or {}).get("token") or \
                      (s.get("data") or {}).get("token")
                if tok and tok == token:
                    md = s.get("metadata", {})
                    sa = (md.get("annotations") or {}).get(
                        "kubernetes.io/service-account.name", "")
                    return f"system:serviceaccount:" \
                           f"{md.get('namespace', 'default')}:{sa}"
        return ANONYMOUS_USER

    # ------------------------------------------------------------------ ASGI
    async def __call__(self, scope, receive, send, parsed) -> None:
        plural, ns, name, status_sub = parsed
        method = scope["method"]
        q = {k: v[0] for k, v in parse_qs(scope.get("query_string",
                                                    b"").decode()).items()}
        headers = dict(scope.get("headers") or [])
        try:
            kind = self._kind(plural)
            if status_sub and method in ("POST", "DELETE"):
                # the status subresource only supports read/update/patch —
                # a DELETE here must never delete the parent object
                raise ApiError(405, "MethodNotAllowed",
                               f"{method} is not supported on the status "
                               "subresource")
            if method == "GET" and name is None:
                if q.get("watch") in ("true", "1"):
                    await self._watch(send, kind, q)
                    return
                await self._list(send, kind, ns, q)
                return
            body = None
            if method in ("POST", "PUT", "PATCH"):
                body = await self._read_body(receive)
            if method == "GET":
                # zero-copy read: the object is serialized immediately and store
                # objects are immutable after insert, so no defensive copy needed
                await _json(send, 200, self.store.get(kind, ns, name, copy=False))
                return
            user = self._user(headers)
            if method == "POST":
                obj = body
                obj.setdefault("kind", kind)
                if ns is not None:
                    obj.setdefault("metadata", {})["namespace"] = ns
                with as_user(user):
                    await _json(send, 201, self.store.create(obj))
                return
            if method == "PUT":
                obj = body
                obj.setdefault("kind", kind)
                md = obj.setdefault("metadata", {})
                if ns is not None:
                    md["namespace"] = ns
                md["name"] = name
                with as_user(user):
                    out = self.store.update_status(obj) if status_sub \
                        else self.store.update(obj)
                await _json(send, 200, out)
                return
            if method == "PATCH":
                ctype = headers.get(b"content-type", b"").decode()
                merge = strategic_merge_patch if "strategic-merge-patch" in ctype \
                    else json_merge_patch

                def apply(o):
                    merge(o, body)
                with as_user(user):
                    out = self.store.patch(kind, ns, name, apply,
                                           status=status_sub)
                await _json(send, 200, out)
                return
            if method == "DELETE":
                with as_user(user):
                    self.store.delete(kind, ns, name)
                await _json(send, 200, {"kind": "Status", "status": "Success"})
                return
            await _json(send, 405, {"kind": "Status", "status": "Failure",
                                    "reason": "MethodNotAllowed", "code": 405})
        except ApiError as e:
            await _json(send, e.code, {
                "kind": "Status", "status": "Failure", "reason": e.reason,
                "message": e.message, "code": e.code})
        except Exception as e:  # internal fault -> kube-style 500, not a dead socket
            try:
                await _json(send, 500, {
                    "kind": "Status", "status": "Failure",
                    "reason": "InternalError",
                    "message": f"{type(e).__name__}: {e}", "code": 500})
            except Exception:
                pass  # response already started (watch stream teardown)

    async def _read_body(self, receive) -> Obj:
        chunks: List[bytes] = []
        while True:
            msg = await receive()
            if msg["type"] == "http.disconnect":
                raise ApiError(400, "BadRequest", "client disconnected")
            chunks.append(msg.get("body", b""))
            if not msg.get("more_body"):
                break
        try:
            return json.loads(b"".join(chunks) or b"{}")
        except json.JSONDecodeError as e:
            raise ApiError(400, "BadRequest", f"invalid JSON body: {e}")

    async def _list(self, send, kind: str, ns: Optional[str],
                    q: Dict[str, str]) -> None:
        selector = parse_selector(q.get("labelSelector"))
        fields = parse_field_selector(q.get("fieldSelector"))
        limit = int(q["limit"]) if q.get("limit") else None
        if fields and not limit and not q.get("continue"):
            items = self.store.list(kind, ns, selector, field_selector=fields,
                                    copy_objects=False)  # serialized immediately
            await _json(send, 200, {
                "kind": f"{kind}List", "apiVersion": "v1",
                "metadata": {"resourceVersion": self.store.current_rv()},
                "items": items})
            return
        items, next_cont, rv = self.store.list_page(
            kind, ns, selector, limit, q.get("continue") or None,
            copy_objects=False)  # serialized immediately
        meta: Dict[str, Any] = {"resourceVersion": rv}
        if next_cont:
            meta["continue"] = next_cont
        await _json(send, 200, {"kind": f"{kind}List", "apiVersion": "v1",
                                "metadata": meta, "items": items})

    async def _watch(self, send, kind: str, q: Dict[str, str]) -> None:
        since_rv = q.get("resourceVersion") or None
        bookmarks = q.get("allowWatchBookmarks") in ("true", "1")
        w = self.store.watch(
            kind,
            seed=(since_rv is None and q.get("seed", "true") in ("true", "1")),
            since_rv=since_rv,
            field_selector=parse_field_selector(q.get("fieldSelector")))
        await send({"type": "http.response.start", "status": 200,
                    "headers": [(b"content-type",
                                 b"application/x-ndjson")]})
        loop = asyncio.get_event_loop()
        idle = 0
        try:
            while not self.shutdown.is_set():
                try:
                    ev, obj = await loop.run_in_executor(
                        None, w.queue.get, True, 0.2)
                except RuntimeError:
                    return  # event loop / executor shutting down
                except queue.Empty:
                    idle += 1
                    if idle < 5:
                        continue  # keepalive cadence stays ~1 s
                    idle = 0
                    if bookmarks:
                        payload = json.dumps({"type": "BOOKMARK", "object": {
                            "kind": kind, "metadata": {
                                "resourceVersion":
                                    self.store.current_rv()}}}) + "\n"
                    else:
                        payload = "\n"  # keepalive
                    await send({"type": "http.response.body",
                                "body": payload.encode(), "more_body": True})
                    continue
                idle = 0
                await send({"type": "http.response.body",
                            "body": (json.dumps({"type": ev, "object": obj})
                                     + "\n").encode(),
                            "more_body": True})
            # graceful end-of-stream on shutdown
            await send({"type": "http.response.body", "body": b"",
                        "more_body": False})
        finally:
            w.stop()


async def _json(send, status: int, payload: Obj) -> None:
    body = json.dumps(payload).encode()
    await send({"type": "http.response.start", "status": status,
                "headers": [(b"content-type", b"application/json"),
                            (b"content-length",
                             str(len(body)).encode())]})
    await send({"type": "http.response.body", "body": body})
