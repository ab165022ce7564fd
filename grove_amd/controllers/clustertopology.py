"""ClusterTopology controller — CTB → backend topology resources + drift detection.

Parity source: operator/internal/controller/clustertopology/reconciler.go:57-148:
auto-managed backends get SyncTopology; externally-managed get CheckTopologyDrift; the
SchedulerTopologyDrift condition + events surface the result on the CTB.
"""
from __future__ import annotations

import logging
from typing import Optional

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError
from ..utils.errors import report_api_error
from ..scheduler.backends import Registry
from ..utils import conditions as cond
from .manager import Result

log = logging.getLogger("grove.clustertopology")


class ClusterTopologyReconciler:
    def __init__(self, store: Store, registry: Registry, auto_manage: bool = True):
        self.store = store
        self.registry = registry
        self.auto_manage = auto_manage

    def reconcile(self, _namespace: str, name: str) -> Result:
        ctb = self.store.try_get(c.KIND_CTB, None, name)
        if ctb is None or ctb["metadata"].get("deletionTimestamp"):
            return Result.DONE
        drift: Optional[str] = None
        refs = []
        for backend in self.registry.all():
            if backend.topology_resource_name(ctb) is None:
                continue  # not topology-aware
            if self.auto_manage:
                backend.sync_topology(ctb)
            d = backend.check_topology_drift(ctb)
            refs.append({"backend": backend.name,
                         "resourceName": backend.topology_resource_name(ctb),
                         "inSync": d is None})
            if d is not None:
                drift = f"{backend.name}: {d}"

        def upd(o: Obj) -> None:
            st = o.setdefault("status", {})
            st["schedulerTopologyReferences"] = refs
            st["observedGeneration"] = o["metadata"].get("generation")
            if drift is None:
                cond.set_condition(o, c.COND_SCHEDULER_TOPOLOGY_DRIFT, False,
                                   c.REASON_IN_SYNC)
            else:
                cond.set_condition(o, c.COND_SCHEDULER_TOPOLOGY_DRIFT, True,
                                   c.REASON_DRIFT, drift)
        try:
            self.store.patch(c.KIND_CTB, None, name, upd, status=True)
        except ApiError as e:
            report_api_error(self.store, c.KIND_CTB, None, name,
                             "write CTB status", e)
        if drift is not None:
            self.store.record_event(ctb, "Warning", c.REASON_DRIFT, drift)
        return Result.DONE
