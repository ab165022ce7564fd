"""In-process Kubernetes-style API machinery.

The reference (ai-dynamo/grove) is a Go controller-runtime operator that talks to a real
kube-apiserver. This image has no cluster, no Go toolchain and no kubernetes client, so the
MI355X-native stack ships its own API machinery: a thread-safe object store with the
Kubernetes semantics the control plane depends on — resourceVersion optimistic concurrency,
metadata.generation bumps on spec changes, a status subresource, label selectors, watches,
finalizers + deletionTimestamp two-phase delete, ownerReference cascade GC, admission
(mutating + validating) chains, and Events.

Objects are plain dicts in the unstructured k8s shape:
    {"apiVersion", "kind", "metadata": {...}, "spec": {...}, "status": {...}}

Behavioral parity notes (not a port):
- two-phase delete w/ finalizers mirrors apiserver semantics the reference's finalizer flow
  relies on (operator/internal/controller/podcliqueset/reconciledelete.go).
- cascade GC replaces the kube garbage collector for ownerReferences.
"""
from __future__ import annotations

import itertools
import queue
import threading
import time
import uuid
from typing import Any, Callable, Dict, List, Optional, Tuple

Obj = Dict[str, Any]


def _py_json_copy(obj):
    t = type(obj)
    if t is dict:
        return {k: _py_json_copy(v) for k, v in obj.items()}
    if t is list:
        return [_py_json_copy(v) for v in obj]
    return obj


try:
    # native deep copy (raw CPython API in _sched.so): 6.5x the Python recursion on
    # a typical pod object — this is the store's hottest single function
    from ..scheduler._sched import json_copy  # type: ignore
except Exception:  # pragma: no cover - pre-build fallback
    json_copy = _py_json_copy

ADDED = "ADDED"
MODIFIED = "MODIFIED"
DELETED = "DELETED"


class ApiError(Exception):
    def __init__(self, code: int, reason: str, message: str = ""):
        super().__init__(f"{reason}: {message}" if message else reason)
        self.code = code
        self.reason = reason
        self.message = message


def not_found(kind: str, name: str) -> ApiError:
    return ApiError(404, "NotFound", f"{kind} {name!r} not found")


def conflict(kind: str, name: str, msg: str = "resourceVersion mismatch") -> ApiError:
    return ApiError(409, "Conflict", f"{kind} {name!r}: {msg}")


def already_exists(kind: str, name: str) -> ApiError:
    return ApiError(409, "AlreadyExists", f"{kind} {name!r} already exists")


def invalid(msg: str) -> ApiError:
    return ApiError(422, "Invalid", msg)


def forbidden(msg: str) -> ApiError:
    return ApiError(403, "Forbidden", msg)


def now_iso() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


def match_labels(labels: Optional[Dict[str, str]], selector: Optional[Dict[str, str]]) -> bool:
    if not selector:
        return True
    if not labels:
        return False
    return all(labels.get(k) == v for k, v in selector.items())


def meta(obj: Obj) -> Dict[str, Any]:
    return obj.setdefault("metadata", {})


def obj_key(obj: Obj) -> Tuple[str, str, str]:
    m = obj.get("metadata", {})
    return (obj.get("kind", ""), m.get("namespace", ""), m.get("name", ""))


# Kinds that are cluster-scoped (no namespace)
CLUSTER_SCOPED = {"ClusterTopologyBinding", "Node", "Namespace"}


class _KindTable:
    __slots__ = ("objects", "watchers", "label_indexes", "history")

    # events retained per kind for watch resume (watch cache depth; a client
    # resuming from an RV older than the window gets 410 Gone, kube semantics)
    HISTORY_DEPTH = 4096

    def __init__(self) -> None:
        # key: (namespace, name) -> obj ; cluster-scoped use namespace ""
        self.objects: Dict[Tuple[str, str], Obj] = {}
        self.watchers: List["Watch"] = []
        # label_key -> label_value -> set of object keys (inverted index for the
        # selector scans that dominate reconcile cost at 1000-pod scale)
        self.label_indexes: Dict[str, Dict[str, set]] = {}
        # bounded ring of (int_rv, event_type, obj) for resourceVersion resume
        self.history: List[Tuple[int, str, Obj]] = []

    def index_add(self, key: Tuple[str, str], obj: Obj) -> None:
        labels = obj.get("metadata", {}).get("labels") or {}
        for lk, idx in self.label_indexes.items():
            v = labels.get(lk)
            if v is not None:
                idx.setdefault(v, set()).add(key)

    def index_remove(self, key: Tuple[str, str], obj: Obj) -> None:
        labels = obj.get("metadata", {}).get("labels") or {}
        for lk, idx in self.label_indexes.items():
            v = labels.get(lk)
            if v is not None:
                s = idx.get(v)
                if s is not None:
                    s.discard(key)


def match_fields(obj: Obj, field_selector: Optional[Dict[str, str]]) -> bool:
    """kube fieldSelector subset: the fields real components select on —
    spec.nodeName (kubelet), metadata.name, metadata.namespace, status.phase."""
    if not field_selector:
        return True
    for path, want in field_selector.items():
        cur: Any = obj
        for part in path.split("."):
            if not isinstance(cur, dict):
                cur = None
                break
            cur = cur.get(part)
        if cur != want:
            return False
    return True


class Watch:
    """A subscription to one kind's events. Iterate or poll `.queue`."""

    def __init__(self, store: "Store", kind: str,
                 field_selector: Optional[Dict[str, str]] = None):
        self.store = store
        self.kind = kind
        self.field_selector = field_selector
        self.queue: "queue.Queue[Tuple[str, Obj]]" = queue.Queue()
        self._stopped = False

    def stop(self) -> None:
        self._stopped = True
        tbl = self.store._tables.get(self.kind)
        if tbl is not None:
            with self.store._lock:
                if self in tbl.watchers:
                    tbl.watchers.remove(self)

    def __iter__(self):
        while not self._stopped:
            try:
                yield self.queue.get(timeout=0.2)
            except queue.Empty:
                continue


class Store:
    """The in-process apiserver core."""

    def __init__(self) -> None:
        self._lock = threading.RLock()
        self._tables: Dict[str, _KindTable] = {}
        self._rv = itertools.count(1)
        self._uid_index: Dict[str, Tuple[str, str, str]] = {}  # uid -> (kind, ns, name)
        # owner uid -> set of (kind, ns, name) owned — makes cascade GC O(children)
        # instead of a full-store scan per delete (the 10k-pod delete path was
        # quadratic without it: 181 s for a 10k-pod tree)
        self._owned_by: Dict[str, set] = {}
        # admission chains keyed by kind
        self._mutators: Dict[str, List[Callable[[Obj, Optional[Obj]], None]]] = {}
        self._validators: Dict[str, List[Callable[[Obj, Optional[Obj]], None]]] = {}
        self._delete_validators: Dict[str, List[Callable[[Obj], None]]] = {}
        self.events: List[Obj] = []
        self._events_lock = threading.Lock()
        import collections as _collections
        self._pending_notify: "_collections.deque" = _collections.deque()

    # ------------------------------------------------------------------ admission
    def register_mutator(self, kind: str, fn: Callable[[Obj, Optional[Obj]], None]) -> None:
        self._mutators.setdefault(kind, []).append(fn)

    def register_validator(self, kind: str, fn: Callable[[Obj, Optional[Obj]], None]) -> None:
        self._validators.setdefault(kind, []).append(fn)

    def register_delete_validator(self, kind: str, fn: Callable[[Obj], None]) -> None:
        self._delete_validators.setdefault(kind, []).append(fn)

    # ------------------------------------------------------------------ internals
    def _table(self, kind: str) -> _KindTable:
        tbl = self._tables.get(kind)
        if tbl is None:
            tbl = self._tables.setdefault(kind, _KindTable())
        return tbl

    def _notify(self, tbl: _KindTable, ev: str, obj: Obj) -> None:
        # stored objects are immutable after insert; watchers share the reference
        # (read-only contract, same as list(copy_objects=False)).
        # Called UNDER the store lock: only the resume history (rv-ordered) is
        # appended here; the per-watcher queue fanout is deferred to
        # _drain_notifications OUTSIDE the lock — queue.put wakes waiter threads
        # and was a measured store-lock hold at 10k-pod scale.
        try:
            rv = int(obj["metadata"].get("resourceVersion", "0"))
        except (TypeError, ValueError):
            rv = 0
        tbl.history.append((rv, ev, obj))
        if len(tbl.history) > tbl.HISTORY_DEPTH:
            del tbl.history[: tbl.HISTORY_DEPTH // 2]
        self._pending_notify.append((tbl, ev, obj))

    def _drain_notifications(self) -> None:
        """Deliver queued watch events outside the store lock. Per-watcher FIFO
        order is preserved for events produced under the same lock hold; events
        from different writers may interleave (level-triggered consumers)."""
        while True:
            try:
                tbl, ev, obj = self._pending_notify.popleft()
            except IndexError:
                return
            for w in list(tbl.watchers):
                if w.field_selector is None or match_fields(obj, w.field_selector):
                    w.queue.put((ev, obj))

    def _next_rv(self) -> str:
        return str(next(self._rv))

    @staticmethod
    def _ns_of(kind: str, obj_meta: Dict[str, Any]) -> str:
        if kind in CLUSTER_SCOPED:
            return ""
        return obj_meta.get("namespace") or "default"

    # ------------------------------------------------------------------ CRUD
    def create(self, obj: Obj) -> Obj:
        obj = json_copy(obj)
        kind = obj.get("kind")
        if not kind:
            raise invalid("object has no kind")
        m = meta(obj)
        ns = self._ns_of(kind, m)
        if kind not in CLUSTER_SCOPED:
            m["namespace"] = ns
        if not m.get("name"):
            gen_name = m.get("generateName")
            if not gen_name:
                raise invalid(f"{kind}: metadata.name or generateName required")
            m["name"] = gen_name + uuid.uuid4().hex[:5]
        # admission: mutate then validate (old=None on create)
        for fn in self._mutators.get(kind, ()):
            fn(obj, None)
        for fn in self._validators.get(kind, ()):
            fn(obj, None)
        with self._lock:
            tbl = self._table(kind)
            key = (ns, m["name"])
            if key in tbl.objects:
                raise already_exists(kind, m["name"])
            m["uid"] = str(uuid.uuid4())
            m["resourceVersion"] = self._next_rv()
            m["generation"] = 1
            m["creationTimestamp"] = now_iso()
            tbl.objects[key] = obj
            tbl.index_add(key, obj)
            self._uid_index[m["uid"]] = (kind, ns, m["name"])
            for ref in m.get("ownerReferences") or []:
                if ref.get("uid"):
                    self._owned_by.setdefault(ref["uid"], set()).add(
                        (kind, ns, m["name"]))
            self._notify(tbl, ADDED, obj)
        self._drain_notifications()
        return json_copy(obj)  # caller gets a private copy; stored one is immutable

    def get(self, kind: str, namespace: Optional[str], name: str,
            copy: bool = True) -> Obj:
        ns = "" if kind in CLUSTER_SCOPED else (namespace or "default")
        with self._lock:
            tbl = self._table(kind)
            obj = tbl.objects.get((ns, name))
        if obj is None:
            raise not_found(kind, name)
        # stored objects are immutable after insert (updates replace wholesale), so
        # the defensive copy happens OUTSIDE the lock — the lock hold is a dict get.
        # copy=False returns the stored object itself for READ-ONLY consumers (the
        # immutability contract makes this safe as long as the caller never mutates).
        return json_copy(obj) if copy else obj

    def try_get(self, kind: str, namespace: Optional[str], name: str,
                copy: bool = True) -> Optional[Obj]:
        try:
            return self.get(kind, namespace, name, copy=copy)
        except ApiError:
            return None

    # Label keys kept in the inverted index (built lazily on first indexed list()).
    INDEXED_LABELS = frozenset((
        "grove.io/podclique", "grove.io/podgang", "grove.io/base-podgang",
        "grove.io/podcliquescalinggroup", "app.kubernetes.io/part-of",
    ))

    def list(self, kind: str, namespace: Optional[str] = None,
             label_selector: Optional[Dict[str, str]] = None,
             filter_fn: Optional[Callable[[Obj], bool]] = None,
             copy_objects: bool = True,
             field_selector: Optional[Dict[str, str]] = None) -> List[Obj]:
        """List objects. copy_objects=False returns direct references for read-only
        consumers (status rollups, the scheduler pass) — callers MUST NOT mutate."""
        with self._lock:
            tbl = self._table(kind)
            candidates = None
            if label_selector:
                for lk in label_selector:
                    if lk in self.INDEXED_LABELS:
                        idx = tbl.label_indexes.get(lk)
                        if idx is None:
                            idx = {}
                            for key, obj in tbl.objects.items():
                                v = (obj.get("metadata", {}).get("labels") or {}).get(lk)
                                if v is not None:
                                    idx.setdefault(v, set()).add(key)
                            tbl.label_indexes[lk] = idx
                        keys = idx.get(label_selector[lk], set())
                        candidates = [(k, tbl.objects[k]) for k in keys
                                      if k in tbl.objects]
                        break
            if candidates is None:
                # snapshot only (C-level dict copy); the python-level namespace/
                # label filter loop runs OUTSIDE the lock — full-table scans at
                # 10k objects were the dominant store-lock hold
                candidates = list(tbl.objects.items())
        refs = []
        for (ns, _name), obj in candidates:
            if namespace is not None and ns != namespace:
                continue
            if not match_labels(obj.get("metadata", {}).get("labels"), label_selector):
                continue
            if field_selector and not match_fields(obj, field_selector):
                continue
            refs.append(obj)
        # filtering callbacks + copies run OUTSIDE the lock (objects immutable)
        if filter_fn is not None:
            refs = [o for o in refs if filter_fn(o)]
        return [json_copy(o) for o in refs] if copy_objects else refs

    def _apply_update(self, obj: Obj, status_only: bool,
                      owned: bool = False, return_copy: bool = True) -> Obj:
        """Lock-split optimistic update: the expensive work (copies, admission,
        equality checks) runs OUTSIDE the lock against the immutable current object;
        insertion re-checks identity under the lock and retries on interleaving.
        owned=True promises obj is a private copy the store may keep (patch()
        fast-path — skips one full deep copy per write)."""
        if not owned:
            obj = json_copy(obj)
        kind = obj["kind"]
        m = meta(obj)
        ns = self._ns_of(kind, m)
        key = (ns, m["name"])
        for _attempt in range(16):
            with self._lock:
                tbl = self._table(kind)
                cur = tbl.objects.get(key)
            if cur is None:
                raise not_found(kind, m["name"])
            cur_m = cur["metadata"]
            if m.get("resourceVersion") and m["resourceVersion"] != cur_m["resourceVersion"]:
                raise conflict(kind, m["name"])
            if status_only:
                if cur.get("status", {}) == obj.get("status", {}):
                    return json_copy(cur) if return_copy else None  # no-op
                # structural sharing: spec (the object's bulk — podSpecs) is
                # immutable across a status write, so the new stored object
                # shares it; only metadata (rv bump) + status are fresh
                new = dict(cur)
                new["metadata"] = json_copy(cur["metadata"])
                new["status"] = json_copy(obj.get("status", {}))
            else:
                # admission on spec/metadata updates
                for fn in self._mutators.get(kind, ()):
                    fn(obj, cur)
                for fn in self._validators.get(kind, ()):
                    fn(obj, cur)
                new = obj
                # immutable fields
                for f in ("uid", "creationTimestamp", "generation"):
                    new["metadata"][f] = cur_m[f]
                new["metadata"]["namespace"] = cur_m.get("namespace", "")
                if cur.get("status") is not None and "status" not in new:
                    new["status"] = json_copy(cur["status"])
                else:
                    new["status"] = json_copy(cur.get("status", {}))
                if cur_m.get("deletionTimestamp"):
                    new["metadata"]["deletionTimestamp"] = cur_m["deletionTimestamp"]
                new["metadata"]["resourceVersion"] = cur_m["resourceVersion"]
                if new == cur:
                    return json_copy(cur) if return_copy else None  # no-op
                if new.get("spec") != cur.get("spec"):
                    new["metadata"]["generation"] = cur_m.get("generation", 1) + 1
            with self._lock:
                if tbl.objects.get(key) is not cur:
                    if m.get("resourceVersion"):
                        raise conflict(kind, m["name"])
                    continue  # interleaved writer; rebuild against the new current
                new["metadata"]["resourceVersion"] = self._next_rv()
                if not status_only and new["metadata"].get("labels") != cur_m.get("labels"):
                    tbl.index_remove(key, cur)
                    tbl.index_add(key, new)
                if not status_only and new["metadata"].get("ownerReferences") \
                        != cur_m.get("ownerReferences"):
                    loc = (kind, ns, m["name"])
                    for ref in cur_m.get("ownerReferences") or []:
                        if ref.get("uid"):
                            self._owned_by.get(ref["uid"], set()).discard(loc)
                    for ref in new["metadata"].get("ownerReferences") or []:
                        if ref.get("uid"):
                            self._owned_by.setdefault(ref["uid"], set()).add(loc)
                tbl.objects[key] = new
                self._notify(tbl, MODIFIED, new)
                # finalizer removal on a deleting object may allow actual deletion
                if new["metadata"].get("deletionTimestamp") \
                        and not new["metadata"].get("finalizers"):
                    self._finalize_delete(kind, ns, m["name"])
            self._drain_notifications()
            return json_copy(new) if return_copy else None
        raise conflict(kind, m["name"], "persistent write interleaving")

    def update(self, obj: Obj) -> Obj:
        return self._apply_update(obj, status_only=False)

    def update_status(self, obj: Obj) -> Obj:
        return self._apply_update(obj, status_only=True)

    def patch(self, kind: str, namespace: Optional[str], name: str,
              fn: Callable[[Obj], None], status: bool = False, retries: int = 10,
              return_copy: bool = True) -> Obj:
        """Optimistic-concurrency retry loop: get → fn(obj) → update.

        The working copy is handed to _apply_update as `owned` (it was copied from
        the immutable stored object here, and fn sees only the copy), saving one
        full deep copy per patch — the store's hottest write path at 1000-pod
        scale."""
        last: Optional[ApiError] = None
        for _ in range(retries):
            cur = self.get(kind, namespace, name, copy=False)
            if status:
                # status patches may only mutate obj["status"] (and read the
                # rest) — the working copy shares the immutable spec subtree,
                # skipping the podSpec deep copy that dominated status-write
                # cost at 10k-pod scale
                obj = dict(cur)
                obj["metadata"] = json_copy(cur["metadata"])
                obj["status"] = json_copy(cur.get("status") or {})
            else:
                obj = json_copy(cur)
            fn(obj)
            try:
                return self._apply_update(obj, status_only=status, owned=True,
                                          return_copy=return_copy)
            except ApiError as e:
                if e.reason != "Conflict":
                    raise
                last = e
        raise last or conflict(kind, name)

    # ------------------------------------------------------------------ delete + GC
    def delete(self, kind: str, namespace: Optional[str], name: str,
               cascade: bool = True) -> None:
        ns = "" if kind in CLUSTER_SCOPED else (namespace or "default")
        with self._lock:
            tbl = self._table(kind)
            obj = tbl.objects.get((ns, name))
            if obj is None:
                raise not_found(kind, name)
            for fn in self._delete_validators.get(kind, ()):
                fn(obj)
            m = obj["metadata"]
            if m.get("finalizers"):
                if not m.get("deletionTimestamp"):
                    marked = json_copy(obj)  # stored objects are immutable: replace
                    marked["metadata"]["deletionTimestamp"] = now_iso()
                    marked["metadata"]["resourceVersion"] = self._next_rv()
                    tbl.objects[(ns, name)] = marked
                    self._notify(tbl, MODIFIED, marked)
            else:
                self._finalize_delete(kind, ns, name, cascade=cascade)
        self._drain_notifications()

    def delete_collection(self, kind: str, namespace: Optional[str],
                          label_selector: Optional[Dict[str, str]] = None) -> int:
        """DeleteAllOf equivalent (used by gang termination)."""
        victims = self.list(kind, namespace, label_selector)
        n = 0
        for v in victims:
            try:
                self.delete(kind, v["metadata"].get("namespace"), v["metadata"]["name"])
                n += 1
            except ApiError:
                pass
        return n

    def _finalize_delete(self, kind: str, ns: str, name: str, cascade: bool = True) -> None:
        # caller holds lock (RLock re-entrant)
        with self._lock:
            tbl = self._table(kind)
            obj = tbl.objects.pop((ns, name), None)
            if obj is None:
                return
            tbl.index_remove((ns, name), obj)
            uid = obj["metadata"].get("uid")
            if uid:
                self._uid_index.pop(uid, None)
            loc = (kind, ns, name)
            for ref in obj["metadata"].get("ownerReferences") or []:
                if ref.get("uid"):
                    self._owned_by.get(ref["uid"], set()).discard(loc)
            self._notify(tbl, DELETED, obj)
            if cascade and uid:
                self._cascade(uid)

    def _cascade(self, owner_uid: str) -> None:
        """Delete all objects owned by owner_uid (kube GC stand-in) — indexed."""
        with self._lock:
            victims = list(self._owned_by.pop(owner_uid, ()) or ())
            for kind, ns, name in victims:
                try:
                    self.delete(kind, ns or None, name)
                except ApiError:
                    pass

    # ------------------------------------------------------------------ watches & events
    def watch(self, kind: str, seed: bool = False,
              since_rv: Optional[str] = None,
              field_selector: Optional[Dict[str, str]] = None) -> Watch:
        """since_rv: resume semantics — replay retained events with
        resourceVersion > since_rv, then stream live (client-go ListAndWatch /
        watch cache parity). Raises 410 Expired if since_rv predates the
        retained window."""
        w = Watch(self, kind, field_selector)
        with self._lock:
            tbl = self._table(kind)
            if since_rv is not None:
                try:
                    rv = int(since_rv)
                except (TypeError, ValueError):
                    raise invalid(f"malformed resourceVersion {since_rv!r}")
                if tbl.history and tbl.history[0][0] > rv + 1 \
                        and len(tbl.history) >= tbl.HISTORY_DEPTH // 2:
                    raise ApiError(410, "Expired",
                                   f"resourceVersion {since_rv} is too old")
                tbl.watchers.append(w)
                for (erv, ev, obj) in tbl.history:
                    if erv > rv and match_fields(obj, field_selector):
                        w.queue.put((ev, obj))
                return w
            tbl.watchers.append(w)
            if seed:
                for obj in tbl.objects.values():
                    if match_fields(obj, field_selector):
                        w.queue.put((ADDED, json_copy(obj)))
        return w

    def current_rv(self) -> str:
        """Most recently issued resourceVersion (list-consistency token)."""
        with self._lock:
            nxt = next(self._rv)
        return str(nxt - 1)  # peek costs one rv; monotonicity is all that matters

    def list_page(self, kind: str, namespace: Optional[str] = None,
                  label_selector: Optional[Dict[str, str]] = None,
                  limit: Optional[int] = None,
                  continue_token: Optional[str] = None,
                  copy_objects: bool = True
                  ) -> Tuple[List[Obj], Optional[str], str]:
        """Chunked list (apiserver limit/continue parity): deterministic
        (namespace, name) order; returns (items, next_continue, resourceVersion)."""
        import base64
        import json as _json
        if limit is None and continue_token is None:
            # unchunked: use the indexed list path (no full-table sort; the
            # inverted label indexes serve selector queries)
            return (self.list(kind, namespace, label_selector,
                              copy_objects=copy_objects), None,
                    self.current_rv())
        start_after: Optional[Tuple[str, str]] = None
        if continue_token:
            try:
                ns_name = _json.loads(base64.b64decode(continue_token))
                start_after = (ns_name[0], ns_name[1])
            except Exception:
                raise invalid("malformed continue token")
        with self._lock:
            tbl = self._table(kind)
            keys = sorted(tbl.objects.keys())
            rv = self.current_rv()
            refs = []
            for key in keys:
                if start_after is not None and key <= start_after:
                    continue
                ns, _name = key
                if namespace is not None and ns != namespace:
                    continue
                obj = tbl.objects[key]
                if not match_labels(obj.get("metadata", {}).get("labels"),
                                    label_selector):
                    continue
                refs.append((key, obj))
                if limit and len(refs) > limit:
                    break
        nxt = None
        if limit and len(refs) > limit:
            refs = refs[:limit]
            nxt = base64.b64encode(
                _json.dumps(list(refs[-1][0])).encode()).decode()
        return ([json_copy(o) for _k, o in refs] if copy_objects
                else [o for _k, o in refs]), nxt, rv

    def record_event(self, involved: Obj, etype: str, reason: str, message: str) -> None:
        ev = {
            "kind": "Event", "type": etype, "reason": reason, "message": message,
            "involvedObject": {"kind": involved.get("kind"),
                               "namespace": involved.get("metadata", {}).get("namespace"),
                               "name": involved.get("metadata", {}).get("name")},
            "timestamp": now_iso(),
        }
        with self._events_lock:
            self.events.append(ev)
            if len(self.events) > 10000:
                del self.events[:5000]

    # ------------------------------------------------------------------ helpers
    def owner_of(self, obj: Obj, kind: str) -> Optional[Obj]:
        for ref in obj.get("metadata", {}).get("ownerReferences", []) or []:
            if ref.get("kind") == kind:
                loc = self._uid_index.get(ref.get("uid", ""))
                if loc:
                    return self.try_get(loc[0], loc[1] or None, loc[2])
                return self.try_get(kind, obj["metadata"].get("namespace"), ref.get("name", ""))
        return None

    def stats(self) -> Dict[str, int]:
        with self._lock:
            return {k: len(t.objects) for k, t in self._tables.items()}


def owner_reference(obj: Obj, controller: bool = True) -> Dict[str, Any]:
    m = obj["metadata"]
    return {
        "apiVersion": obj.get("apiVersion", ""),
        "kind": obj["kind"],
        "name": m["name"],
        "uid": m["uid"],
        "controller": controller,
        "blockOwnerDeletion": True,
    }
