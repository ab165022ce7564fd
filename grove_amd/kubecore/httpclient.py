"""HTTP store client — the client-go analog for remote components.

Speaks the apiserver's kube-style paths (kubecore/apiserver.py) and duck-types the
subset of the Store interface that node-side components use (get/try_get/list/patch/
update/delete + ndjson watch), so a ProcessKubelet or topology agent can run on a GPU
node against a remote operator exactly like the reference's kubelet/initc talk to the
apiserver via client-go.
"""
from __future__ import annotations

import json
import ssl
import time
import urllib.error
import urllib.parse
import urllib.request
from typing import Any, Callable, Dict, Iterator, List, Optional, Tuple

from .apiserver import PLURALS, CLUSTER_SCOPED_PLURALS
from .store import ApiError

Obj = Dict[str, Any]

KIND_TO_PLURAL = {v: k for k, v in PLURALS.items()}
GROVE_KINDS = {"PodCliqueSet", "PodClique", "PodCliqueScalingGroup",
               "ClusterTopologyBinding"}


def _group_of(kind: str) -> str:
    if kind in GROVE_KINDS:
        return "apis/grove.io/v1alpha1"
    if kind == "PodGang":
        return "apis/scheduler.grove.io/v1alpha1"
    if kind == "SchedulerTopology":
        return "apis/scheduler.amd.com/v1alpha1"
    return "api/v1"


class HttpStoreClient:
    def __init__(self, base_url: str, timeout: float = 10.0,
                 cafile: Optional[str] = None, token: Optional[str] = None):
        self.base_url = base_url.rstrip("/")
        self.timeout = timeout
        self.token = token
        self._ctx = ssl.create_default_context(cafile=cafile) if cafile else None
        # keep-alive connection per thread (urllib opens a fresh TCP connection
        # per request — measured as the dominant wire-path overhead); watch
        # streams still use urllib (dedicated long-lived connection)
        import threading as _threading
        import urllib.parse as _parse
        u = _parse.urlsplit(self.base_url)
        self._scheme = u.scheme
        self._netloc = u.netloc
        self._prefix = u.path.rstrip("/")
        self._local = _threading.local()

    def _conn(self):
        import http.client
        conn = getattr(self._local, "conn", None)
        if conn is None:
            if self._scheme == "https":
                conn = http.client.HTTPSConnection(
                    self._netloc, timeout=self.timeout, context=self._ctx)
            else:
                conn = http.client.HTTPConnection(self._netloc,
                                                  timeout=self.timeout)
            self._local.conn = conn
        return conn

    def _drop_conn(self):
        conn = getattr(self._local, "conn", None)
        if conn is not None:
            try:
                conn.close()
            except Exception:
                pass
            self._local.conn = None

    def _headers(self, content: bool = False) -> Dict[str, str]:
        h: Dict[str, str] = {}
        if content:
            h["Content-Type"] = "application/json"
        if self.token:
            h["Authorization"] = f"Bearer {self.token}"
        return h

    # ------------------------------------------------------------------ plumbing
    def _url(self, kind: str, namespace: Optional[str], name: Optional[str] = None,
             subresource: str = "", query: str = "") -> str:
        plural = KIND_TO_PLURAL[kind]
        base = f"{self.base_url}/{_group_of(kind)}"
        if plural in CLUSTER_SCOPED_PLURALS:
            url = f"{base}/{plural}"
        else:
            url = f"{base}/namespaces/{namespace or 'default'}/{plural}"
        if name:
            url += f"/{urllib.parse.quote(name)}"
        if subresource:
            url += f"/{subresource}"
        if query:
            url += f"?{query}"
        return url

    def _request(self, method: str, url: str, body: Optional[Obj] = None,
                 content_type: str = "application/json") -> Obj:
        data = json.dumps(body).encode() if body is not None else None
        path = url[len(f"{self._scheme}://{self._netloc}"):] \
            if url.startswith(f"{self._scheme}://{self._netloc}") else url
        headers = self._headers()
        headers["Content-Type"] = content_type
        for attempt in (0, 1):  # one retry on a stale keep-alive connection
            conn = self._conn()
            try:
                conn.request(method, path, body=data, headers=headers)
                resp = conn.getresponse()
                raw = resp.read()
            except ApiError:
                raise
            except Exception:
                self._drop_conn()
                if attempt == 1:
                    raise
                continue
            if resp.status >= 400:
                try:
                    payload = json.loads(raw)
                except Exception:
                    payload = {}
                raise ApiError(resp.status, payload.get("reason", "HTTPError"),
                               payload.get("message", f"HTTP {resp.status}"))
            return json.loads(raw)

    # ------------------------------------------------------------------ store API
    def get(self, kind: str, namespace: Optional[str], name: str,
            copy: bool = True) -> Obj:
        return self._request("GET", self._url(kind, namespace, name))

    def try_get(self, kind: str, namespace: Optional[str], name: str,
                copy: bool = True) -> Optional[Obj]:
        # `copy` is accepted for Store duck-type compatibility (wire objects are
        # always fresh copies).
        try:
            return self.get(kind, namespace, name)
        except ApiError:
            return None

    def list(self, kind: str, namespace: Optional[str] = None,
             label_selector: Optional[Dict[str, str]] = None,
             filter_fn: Optional[Callable[[Obj], bool]] = None,
             copy_objects: bool = True,
             field_selector: Optional[Dict[str, str]] = None) -> List[Obj]:
        parts = []
        if label_selector:
            sel = ",".join(f"{k}={v}" for k, v in label_selector.items())
            parts.append(f"labelSelector={urllib.parse.quote(sel)}")
        if field_selector:
            fs = ",".join(f"{k}={v}" for k, v in field_selector.items())
            parts.append(f"fieldSelector={urllib.parse.quote(fs)}")
        query = "&".join(parts)
        items = self._request("GET", self._url(kind, namespace, query=query))["items"]
        if filter_fn is not None:
            items = [o for o in items if filter_fn(o)]
        return items

    def create(self, obj: Obj) -> Obj:
        kind = obj["kind"]
        ns = obj.get("metadata", {}).get("namespace")
        return self._request("POST", self._url(kind, ns), obj)

    def update(self, obj: Obj) -> Obj:
        kind = obj["kind"]
        md = obj["metadata"]
        return self._request("PUT", self._url(kind, md.get("namespace"), md["name"]),
                             obj)

    def update_status(self, obj: Obj) -> Obj:
        kind = obj["kind"]
        md = obj["metadata"]
        return self._request(
            "PUT", self._url(kind, md.get("namespace"), md["name"], "status"), obj)

    def patch(self, kind: str, namespace: Optional[str], name: str,
              fn: Callable[[Obj], None], status: bool = False,
              retries: int = 10) -> Obj:
        last: Optional[ApiError] = None
        for _ in range(retries):
            obj = self.get(kind, namespace, name)
            fn(obj)
            try:
                return self.update_status(obj) if status else self.update(obj)
            except ApiError as e:
                if e.reason != "Conflict":
                    raise
                last = e
        raise last or ApiError(409, "Conflict", name)

    def merge_patch(self, kind: str, namespace: Optional[str], name: str,
                    patch: Obj, status: bool = False) -> Obj:
        """Server-side RFC 7386 merge patch (single round-trip, no read-modify-write
        conflict loop) — the client-go Patch(types.MergePatchType) analog.
        status=True targets the status subresource."""
        sub = "status" if status else ""
        return self._request("PATCH", self._url(kind, namespace, name, sub), patch)

    def strategic_merge_patch(self, kind: str, namespace: Optional[str], name: str,
                              patch: Obj) -> Obj:
        """client-go Patch(types.StrategicMergePatchType): merge-by-key list
        semantics for pod specs (containers by name etc.)."""
        return self._request(
            "PATCH", self._url(kind, namespace, name), patch,
            content_type="application/strategic-merge-patch+json")

    def list_page(self, kind: str, namespace: Optional[str] = None,
                  label_selector: Optional[Dict[str, str]] = None,
                  limit: Optional[int] = None,
                  continue_token: Optional[str] = None):
        """Chunked list: returns (items, next_continue, resourceVersion)."""
        import urllib.parse as _p
        parts = []
        if label_selector:
            sel = ",".join(f"{k}={v}" for k, v in label_selector.items())
            parts.append(f"labelSelector={_p.quote(sel)}")
        if limit:
            parts.append(f"limit={limit}")
        if continue_token:
            parts.append(f"continue={_p.quote(continue_token)}")
        out = self._request("GET", self._url(kind, namespace,
                                             query="&".join(parts)))
        md = out.get("metadata") or {}
        return out["items"], md.get("continue"), md.get("resourceVersion")

    def delete(self, kind: str, namespace: Optional[str], name: str,
               cascade: bool = True) -> None:
        self._request("DELETE", self._url(kind, namespace, name))

    def watch_events(self, kind: str, namespace: Optional[str] = None,
                     seed: bool = True, resource_version: Optional[str] = None,
                     bookmarks: bool = False,
                     field_selector: Optional[Dict[str, str]] = None
                     ) -> Iterator[Tuple[str, Obj]]:
        """ndjson watch stream; yields (event_type, object). resource_version
        resumes from that RV (replay + live); bookmarks=True interleaves BOOKMARK
        progress events carrying the current resourceVersion."""
        q = f"watch=true&seed={'true' if seed else 'false'}"
        if resource_version is not None:
            q += f"&resourceVersion={resource_version}"
        if bookmarks:
            q += "&allowWatchBookmarks=true"
        if field_selector:
            import urllib.parse as _p
            fs = ",".join(f"{k}={v}" for k, v in field_selector.items())
            q += f"&fieldSelector={_p.quote(fs)}"
        url = self._url(kind, namespace, query=q)
        req = urllib.request.Request(url, headers=self._headers())
        with urllib.request.urlopen(req, timeout=3600, context=self._ctx) as r:
            for raw in r:
                line = raw.decode().strip()
                if not line:
                    continue
                ev = json.loads(line)
                yield ev["type"], ev["object"]
