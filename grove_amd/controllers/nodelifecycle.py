"""Node lifecycle — eviction of pods from lost nodes.

Parity role: in the reference this is the kube node-lifecycle controller's job (pods on
a dead node get Failed/evicted, which drops PCLQ ScheduledReplicas and triggers the
MinAvailableBreached → gang-termination machinery). The in-process cluster
has no kubelet heartbeats, so node loss = Node deletion or Ready=False; this controller
evicts the node's pods so the recovery path downstream behaves exactly like the
reference's.
"""
from __future__ import annotations

import logging

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError
from ..utils.errors import report_api_error
from ..utils import conditions as cond
from .manager import Result

log = logging.getLogger("grove.nodelifecycle")


class NodeLifecycleReconciler:
    def __init__(self, store: Store):
        self.store = store

    def reconcile(self, _ns: str, name: str) -> Result:
        node = self.store.try_get("Node", None, name)
        lost = node is None or node["metadata"].get("deletionTimestamp") \
            or not self._ready(node)
        if not lost:
            return Result.DONE
        victims = self.store.list(
            "Pod", None, filter_fn=lambda p: p.get("spec", {}).get("nodeName") == name,
            copy_objects=False)
        for p in victims:
            if (p.get("status") or {}).get("phase") in ("Succeeded", "Failed"):
                continue
            try:
                self.store.delete("Pod", p["metadata"].get("namespace"),
                                  p["metadata"]["name"])
                log.info("evicted pod %s from lost node %s",
                         p["metadata"]["name"], name)
            except ApiError as e:
                report_api_error(self.store, "Pod",
                                 p["metadata"].get("namespace"),
                                 p["metadata"]["name"], "evict from lost node", e)
        return Result.DONE

    @staticmethod
    def _ready(node: Obj) -> bool:
        for cd in (node.get("status") or {}).get("conditions") or []:
            if cd.get("type") == "Ready":
                return cd.get("status") == "True"
        return True  # no conditions → assume ready (virtual nodes)
