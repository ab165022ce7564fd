"""Typed clientset — the generated-client analog (client-go clientset/informers).

Parity role: the reference ships generated clientsets for both API groups
(operator/client/, scheduler/client/ — k8s code-gen output, SURVEY §2.1 "generated
clients"). Here the same ergonomic surface is a thin typed facade over either the
in-process Store or an HttpStoreClient (both duck-type get/list/create/update/
patch/delete/watch): `cs.podcliquesets("ns").get("x")`,
`cs.podgangs().watch(resource_version=rv)`.

Every resource accessor exposes: create, get, try_get, list, list_page (HTTP only),
update, update_status, patch, patch_status, delete, watch. Objects stay plain dicts
(the unstructured shape the whole stack uses) — typing here means kind/namespace
plumbing, not schemas.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Iterator, List, Optional, Tuple

Obj = Dict[str, Any]

# kind -> (accessor name, namespaced)
_RESOURCES = {
    "PodCliqueSet": ("podcliquesets", True),
    "PodClique": ("podcliques", True),
    "PodCliqueScalingGroup": ("podcliquescalinggroups", True),
    "PodGang": ("podgangs", True),
    "ClusterTopologyBinding": ("clustertopologybindings", False),
    "SchedulerTopology": ("schedulertopologies", False),
    "Pod": ("pods", True),
    "Service": ("services", True),
    "Secret": ("secrets", True),
    "ServiceAccount": ("serviceaccounts", True),
    "Role": ("roles", True),
    "RoleBinding": ("rolebindings", True),
    "HorizontalPodAutoscaler": ("horizontalpodautoscalers", True),
    "ResourceClaim": ("resourceclaims", True),
    "Node": ("nodes", False),
}


class ResourceClient:
    """Typed access to one kind in one namespace (or cluster-scoped)."""

    def __init__(self, backend, kind: str, namespace: Optional[str]):
        self._b = backend
        self.kind = kind
        self.namespace = namespace

    # ---- reads
    def get(self, name: str) -> Obj:
        return self._b.get(self.kind, self.namespace, name)

    def try_get(self, name: str) -> Optional[Obj]:
        return self._b.try_get(self.kind, self.namespace, name)

    def list(self, label_selector: Optional[Dict[str, str]] = None) -> List[Obj]:
        return self._b.list(self.kind, self.namespace, label_selector)

    def list_page(self, label_selector: Optional[Dict[str, str]] = None,
                  limit: Optional[int] = None,
                  continue_token: Optional[str] = None):
        if hasattr(self._b, "list_page"):
            return self._b.list_page(self.kind, self.namespace, label_selector,
                                     limit, continue_token)
        items = self.list(label_selector)
        return items, None, None

    # ---- writes
    def create(self, obj: Obj) -> Obj:
        obj = dict(obj)
        obj.setdefault("kind", self.kind)
        if self.namespace is not None:
            obj.setdefault("metadata", {}).setdefault("namespace", self.namespace)
        return self._b.create(obj)

    def update(self, obj: Obj) -> Obj:
        return self._b.update(obj)

    def update_status(self, obj: Obj) -> Obj:
        return self._b.update_status(obj)

    def patch(self, name: str, fn: Callable[[Obj], None]) -> Obj:
        return self._b.patch(self.kind, self.namespace, name, fn)

    def patch_status(self, name: str, fn: Callable[[Obj], None]) -> Obj:
        return self._b.patch(self.kind, self.namespace, name, fn, status=True)

    def delete(self, name: str) -> None:
        self._b.delete(self.kind, self.namespace, name)

    # ---- watch
    def watch(self, seed: bool = True,
              resource_version: Optional[str] = None
              ) -> Iterator[Tuple[str, Obj]]:
        if hasattr(self._b, "watch_events"):  # HTTP client
            return self._b.watch_events(self.kind, self.namespace, seed=seed,
                                        resource_version=resource_version)
        w = self._b.watch(self.kind, seed=seed, since_rv=resource_version)
        return iter(w)


class Clientset:
    """client-go Clientset analog over a Store or HttpStoreClient backend."""

    def __init__(self, backend):
        self._backend = backend

    def resource(self, kind: str, namespace: Optional[str] = None
                 ) -> ResourceClient:
        _accessor, namespaced = _RESOURCES[kind]
        ns = (namespace or "default") if namespaced else None
        return ResourceClient(self._backend, kind, ns)


def _make_accessor(kind: str, namespaced: bool):
    if namespaced:
        def accessor(self, namespace: str = "default") -> ResourceClient:
            return ResourceClient(self._backend, kind, namespace)
    else:
        def accessor(self) -> ResourceClient:  # type: ignore[misc]
            return ResourceClient(self._backend, kind, None)
    accessor.__name__ = _RESOURCES[kind][0]
    accessor.__doc__ = f"Typed client for {kind}."
    return accessor


for _kind, (_name, _namespaced) in _RESOURCES.items():
    setattr(Clientset, _name, _make_accessor(_kind, _namespaced))


def for_store(store) -> Clientset:
    return Clientset(store)


def for_server(base_url: str, token: Optional[str] = None,
               cafile: Optional[str] = None) -> Clientset:
    from .httpclient import HttpStoreClient
    return Clientset(HttpStoreClient(base_url, token=token, cafile=cafile))
