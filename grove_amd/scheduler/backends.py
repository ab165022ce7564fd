"""Scheduler backend layer — the pluggable gang/topology dispatch.

Parity source: operator/internal/scheduler/types.go:35-91 (Backend +
TopologyAwareBackend + Registry) and registry.go:97-110. The reference ships kube/kai/
volcano/lpx backends that translate PodGangs into external schedulers' CRs; this build's
primary backend is the NATIVE amd-gang-scheduler (plugin.py consumes PodGangs directly —
no translation CR), with the no-gang `default-scheduler` backend kept for plumbing
parity. The AMD backend is topology-aware: it materializes a SchedulerTopology CR from
ClusterTopologyBinding levels (the KAI Topology CR analog, kai/topology.go:33) extended
with the xGMI-hive level.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError, invalid

log = logging.getLogger("grove.backends")

KIND_SCHEDULER_TOPOLOGY = "SchedulerTopology"


class Backend:
    """scheduler/types.go:35-53 equivalent."""

    name: str = ""

    def init(self, store: Store) -> None:
        self.store = store

    def sync_podgang(self, podgang: Obj) -> None:
        raise NotImplementedError

    def prepare_pod(self, pod: Obj) -> None:
        pod["spec"]["schedulerName"] = self.name

    def validate_podcliqueset(self, pcs: Obj) -> None:
        pass

    # --- TopologyAwareBackend (types.go:59-91); None return = unsupported ---
    def topology_resource_name(self, ctb: Obj) -> Optional[str]:
        return None

    def sync_topology(self, ctb: Obj) -> None:
        raise NotImplementedError

    def check_topology_drift(self, ctb: Obj) -> Optional[str]:
        raise NotImplementedError


class KubeBackend(Backend):
    """No-gang backend (scheduler/kube/backend.go:82): sets schedulerName only; the
    scheduling-gate machinery still serializes startup."""

    name = c.SCHEDULER_DEFAULT

    def sync_podgang(self, podgang: Obj) -> None:
        return  # default scheduler has no gang resource


class AmdGangBackend(Backend):
    """The native backend: PodGang IS the contract consumed by plugin.py, so
    sync_podgang only stamps discovery annotations; topology is materialized as a
    SchedulerTopology CR whose levels extend the CTB with the xGMI-hive level."""

    name = c.SCHEDULER_AMD_GANG

    def sync_podgang(self, podgang: Obj) -> None:
        ns = podgang["metadata"].get("namespace", "default")
        ctbs = self.store.list(c.KIND_CTB, copy_objects=False)
        if not ctbs:
            return
        ctb_name = ctbs[0]["metadata"]["name"]
        ann = podgang["metadata"].get("annotations") or {}
        if ann.get(c.ANNOTATION_TOPOLOGY_NAME) == ctb_name:
            return

        def stamp(o: Obj) -> None:
            o["metadata"].setdefault("annotations", {})[
                c.ANNOTATION_TOPOLOGY_NAME] = ctb_name
        try:
            self.store.patch(c.KIND_PODGANG, ns, podgang["metadata"]["name"], stamp)
        except ApiError as e:
            from ..utils.errors import report_api_error
            report_api_error(self.store, c.KIND_PODGANG, ns,
                             podgang["metadata"]["name"], "stamp topology name", e)

    def validate_podcliqueset(self, pcs: Obj) -> None:
        return  # native backend accepts all topology constraints

    def topology_resource_name(self, ctb: Obj) -> str:
        return ctb["metadata"]["name"]

    def _levels(self, ctb: Obj) -> List[Dict[str, str]]:
        levels = [{"domain": lv.get("domain"),
                   "nodeLabelKey": lv.get("key")}
                  for lv in (ctb.get("spec") or {}).get("levels") or []]
        # narrowest native level: the xGMI hive (below host/numa)
        if not any(lv["domain"] == "xgmi-hive" for lv in levels):
            levels.append({"domain": "xgmi-hive", "nodeLabelKey": c.NODE_LABEL_XGMI_HIVE})
        return levels

    def sync_topology(self, ctb: Obj) -> None:
        name = self.topology_resource_name(ctb)
        want_levels = self._levels(ctb)
        cur = self.store.try_get(KIND_SCHEDULER_TOPOLOGY, None, name)
        if cur is None:
            self.store.create({
                "apiVersion": "scheduler.amd.com/v1alpha1",
                "kind": KIND_SCHEDULER_TOPOLOGY,
                "metadata": {"name": name,
                             "labels": {c.LABEL_MANAGED_BY: c.LABEL_MANAGED_BY_VALUE}},
                "spec": {"levels": want_levels},
            })
            return
        if (cur.get("spec") or {}).get("levels") != want_levels:
            def upd(o: Obj) -> None:
                o["spec"]["levels"] = want_levels
            self.store.patch(KIND_SCHEDULER_TOPOLOGY, None, name, upd)

    def check_topology_drift(self, ctb: Obj) -> Optional[str]:
        name = self.topology_resource_name(ctb)
        cur = self.store.try_get(KIND_SCHEDULER_TOPOLOGY, None, name)
        if cur is None:
            return "topology resource not found"
        if (cur.get("spec") or {}).get("levels") != self._levels(ctb):
            return "levels drifted from ClusterTopologyBinding"
        return None


class LpxBackend(Backend):
    """Name-only backend (scheduler/lpx/backend.go:86): sets schedulerName and
    REJECTS grove topology constraints at admission (backend.go:68-86) — lpx has
    no topology support, so constrained workloads must not silently lose their
    packing guarantee."""

    name = "lpx-scheduler"

    def sync_podgang(self, podgang: Obj) -> None:
        return  # no gang resource; lpx consumes pods directly

    def validate_podcliqueset(self, pcs: Obj) -> None:
        tmpl = (pcs.get("spec") or {}).get("template") or {}
        sites = [("spec.template.topologyConstraint",
                  tmpl.get("topologyConstraint"))]
        sites += [(f"spec.template.cliques[{cl.get('name')}].topologyConstraint",
                   cl.get("topologyConstraint"))
                  for cl in tmpl.get("cliques") or []]
        sites += [(f"spec.template.podCliqueScalingGroups[{sg.get('name')}]"
                   ".topologyConstraint", sg.get("topologyConstraint"))
                  for sg in tmpl.get("podCliqueScalingGroups") or []]
        for path, tc in sites:
            if tc:
                raise invalid(
                    f"{path}: topology constraints are not supported by the "
                    f"{self.name} backend")


class Registry:
    """scheduler/registry/registry.go:45-115 equivalent."""

    def __init__(self, store: Store, default: str = c.SCHEDULER_AMD_GANG,
                 backends: Optional[List[Backend]] = None):
        self._by_name: Dict[str, Backend] = {}
        for b in backends or [AmdGangBackend(), KubeBackend(), LpxBackend()]:
            b.init(store)
            self._by_name[b.name] = b
        if default not in self._by_name:
            raise invalid(f"unknown default scheduler backend {default!r}")
        self.default_name = default

    def get(self, name: Optional[str]) -> Backend:
        return self._by_name.get(name or "", self._by_name[self.default_name])

    def all(self) -> List[Backend]:
        return list(self._by_name.values())

    def resolve_for_podgang(self, podgang: Obj) -> Backend:
        return self.get(podgang["metadata"].get("labels", {}).get(c.LABEL_SCHEDULER_NAME))
