"""Cluster facade: wires store + admission + controllers + scheduler + kubelets.

This is the in-process equivalent of the reference's deployment unit (operator manager +
webhooks + external scheduler + kubelet/KWOK cluster, cmd/main.go:44): one object that
tests, the benchmark harness, and the GPU node agent all drive the same way.
"""
from __future__ import annotations

import logging
import time
from typing import Callable, Dict, List, Optional

import yaml as _yaml

from .api import constants as c
from .api.defaulting import default_podcliqueset, default_podclique, default_pcsg
from .api.validation import (validate_podcliqueset, validate_clustertopologybinding,
                             validate_pcsg, validate_podclique)
from .controllers.manager import Controller, Manager, Result
from .controllers.podclique import PodCliqueReconciler
from .controllers.podcliqueset import PodCliqueSetReconciler
from .controllers.podcliquescalinggroup import PCSGReconciler
from .controllers.podgang import PodGangReconciler
from .controllers.clustertopology import ClusterTopologyReconciler
from .controllers.hpa import HPAReconciler
from .controllers.nodelifecycle import NodeLifecycleReconciler
from .scheduler.backends import Registry
from .kubecore.store import Store, Obj
from .kubelet.virtual import VirtualKubelet, make_virtual_node
from .scheduler.plugin import GangScheduler
from .utils import conditions as cond

log = logging.getLogger("grove.cluster")


class Cluster:
    def __init__(self,
                 scheduler_name: str = c.SCHEDULER_AMD_GANG,
                 concurrent_syncs: int = 4,
                 controller_workers: Optional[Dict[str, int]] = None,
                 startup_latency_s: float = 0.0,
                 ready_latency_s: float = 0.0,
                 pod_payload: Optional[Callable[[Obj], None]] = None,
                 use_native_scheduler: Optional[bool] = None,
                 enable_authorizer: bool = True,
                 auto_xgmi_domain: bool = False):
        self.store = Store()
        self.scheduler_name = scheduler_name

        # structural-schema admission first (the apiserver's CRD schema layer:
        # defaults + type/enum/bounds checks, api/schemavalidate.py), then the
        # webhook-parity semantic admission (defaulting before validation)
        from .api.schemavalidate import StructuralSchemaAdmission
        StructuralSchemaAdmission().register(self.store)
        self.store.register_mutator(c.KIND_PCS, default_podcliqueset)
        self.store.register_validator(c.KIND_PCS, validate_podcliqueset)
        from .api.validation import validate_xgmi_groups
        self.store.register_validator(c.KIND_PCS, validate_xgmi_groups)
        from .api.validation import TopologyConstraintValidator
        self.store.register_validator(c.KIND_PCS,
                                      TopologyConstraintValidator(self.store))
        self.store.register_mutator(c.KIND_PCLQ, default_podclique)
        self.store.register_validator(c.KIND_PCLQ, validate_podclique)
        self.store.register_mutator(c.KIND_PCSG, default_pcsg)
        self.store.register_validator(c.KIND_PCSG, validate_pcsg)
        self.store.register_validator(c.KIND_CTB, validate_clustertopologybinding)
        if enable_authorizer:
            from .api.authorization import Authorizer
            self.authorizer = Authorizer(self.store)
            self.authorizer.register()

        self.manager = Manager(self.store)
        self.registry = Registry(self.store, default=scheduler_name)
        self.pcs_rec = PodCliqueSetReconciler(self.store, scheduler_name,
                                              auto_xgmi_domain=auto_xgmi_domain)
        self.pclq_rec = PodCliqueReconciler(self.store, scheduler_name)
        self.pcsg_rec = PCSGReconciler(self.store, scheduler_name,
                                       auto_xgmi_domain=auto_xgmi_domain)
        self.podgang_rec = PodGangReconciler(self.store, self.registry)
        self.ctb_rec = ClusterTopologyReconciler(self.store, self.registry)
        self.hpa_rec = HPAReconciler(self.store)
        self.node_rec = NodeLifecycleReconciler(self.store)
        self.scheduler = GangScheduler(self.store, scheduler_name,
                                       use_native=use_native_scheduler)
        self.kubelet = VirtualKubelet(self.store,
                                      startup_latency_s=startup_latency_s,
                                      ready_latency_s=ready_latency_s,
                                      payload=pod_payload)

        m = self.manager
        workers = controller_workers or {}

        def w(name: str) -> int:
            return int(workers.get(name, concurrent_syncs))

        self.c_pcs = m.add_controller(Controller(
            "podcliqueset", self.pcs_rec.reconcile, workers=w("podCliqueSet")))
        self.c_pclq = m.add_controller(Controller(
            "podclique", self.pclq_rec.reconcile, workers=w("podClique")))
        self.c_pcsg = m.add_controller(Controller(
            "podcliquescalinggroup", self.pcsg_rec.reconcile,
            workers=w("podCliqueScalingGroup")))
        self.c_podgang = m.add_controller(Controller(
            "podgang", self.podgang_rec.reconcile, workers=2))
        self.c_ctb = m.add_controller(Controller(
            "clustertopology", self.ctb_rec.reconcile, workers=1))
        self.c_hpa = m.add_controller(Controller(
            "hpa", self.hpa_rec.reconcile, workers=1))
        self.c_node = m.add_controller(Controller(
            "node-lifecycle", self.node_rec.reconcile, workers=1))
        self.c_sched = m.add_controller(Controller(
            "gang-scheduler", lambda ns, n: self.scheduler.reconcile(ns, n) or Result.DONE,
            workers=1))
        self.c_kubelet = m.add_controller(Controller(
            "kubelet", self.kubelet.reconcile, workers=max(2, concurrent_syncs)))

        self._wire_watches()
        self._started = False

    # ------------------------------------------------------------------ watches
    def _wire_watches(self) -> None:
        m = self.manager

        def on_pcs(ev: str, obj: Obj, _old) -> None:
            md = obj["metadata"]
            ns = md.get("namespace", "default")
            self.c_pcs.enqueue(ns, md["name"])
            # spec changes flow to child PCSGs (template propagation — e.g. an
            # OnDelete template change must reach member PCLQs without any PCSG
            # spec change; reference podcliquescalinggroup watch predicates).
            # Rolling-update replica SELECTION is a status-only write (no
            # generation bump) but gates the PCSG member-template propagation -
            # without this edge a PCSG pass that ran just before selection
            # latches its fingerprint and the selected replica's members never
            # roll (observed stall).
            def _upd_sig(o):
                if o is None:
                    return None
                st = o.get("status") or {}
                prog = st.get("updateProgress") or {}
                return (st.get("currentGenerationHash"),
                        tuple(sorted(int(e.get("replicaIndex", -1))
                                     for e in prog.get("currentlyUpdating")
                                     or [])))
            if _old is not None and (
                    md.get("generation") != _old["metadata"].get("generation")
                    or _upd_sig(obj) != _upd_sig(_old)):
                for g in self.store.list(c.KIND_PCSG, ns,
                                         {c.LABEL_PART_OF: md["name"]},
                                         copy_objects=False):
                    self.c_pcsg.enqueue(ns, g["metadata"]["name"])

        def on_pclq(ev: str, obj: Obj, _old) -> None:
            md = obj["metadata"]
            ns = md.get("namespace", "default")
            self.c_pclq.enqueue(ns, md["name"])
            pcs = md.get("labels", {}).get(c.LABEL_PART_OF)
            if pcs:
                self.c_pcs.enqueue(ns, pcs)
            pcsg = md.get("labels", {}).get(c.LABEL_PCSG)
            if pcsg:
                self.c_pcsg.enqueue(ns, pcsg)
            # A base-gang member's status change (scheduledReplicas) can unblock gate
            # removal for every scaled-gang PCLQ anchored to that base gang
            # (pod/syncflow.go:386-424 watch-mapping parity).
            pg = md.get("labels", {}).get(c.LABEL_PODGANG)
            if pg and not md.get("labels", {}).get(c.LABEL_BASE_PODGANG):
                # only a scheduledReplicas change can unblock dependents — skip the
                # (indexed, but per-event) lookup for all other status churn
                new_sched = (obj.get("status") or {}).get("scheduledReplicas")
                old_sched = ((_old or {}).get("status") or {}).get("scheduledReplicas")
                if new_sched != old_sched or ev == "DELETED":
                    for q in self.store.list(c.KIND_PCLQ, ns,
                                             {c.LABEL_BASE_PODGANG: pg},
                                             copy_objects=False):
                        self.c_pclq.enqueue(ns, q["metadata"]["name"])

        def on_pcsg(ev: str, obj: Obj, _old) -> None:
            md = obj["metadata"]
            ns = md.get("namespace", "default")
            self.c_pcsg.enqueue(ns, md["name"])
            pcs = md.get("labels", {}).get(c.LABEL_PART_OF)
            if pcs:
                self.c_pcs.enqueue(ns, pcs)

        def on_pod(ev: str, obj: Obj, _old) -> None:
            md = obj["metadata"]
            ns = md.get("namespace", "default")
            pclq = md.get("labels", {}).get(c.LABEL_PODCLIQUE)
            if pclq:
                self.c_pclq.enqueue(ns, pclq)
            if ev in ("ADDED", "DELETED"):
                pcs_name = md.get("labels", {}).get(c.LABEL_PART_OF)
                if pcs_name:
                    # pod-set membership changed -> the PCS structural sync
                    # (PodGang podReferences etc.) must run again
                    self.pcs_rec.bump_pod_epoch(ns, pcs_name)
            self.scheduler.note_pod_event(ev, obj, _old)
            self.c_sched.enqueue("", "pass")
            if obj.get("spec", {}).get("nodeName"):
                self.c_kubelet.enqueue(ns, md["name"])

        def on_podgang(ev: str, obj: Obj, _old) -> None:
            md = obj["metadata"]
            ns = md.get("namespace", "default")
            self.scheduler.note_placement_event()
            self.c_sched.enqueue("", "pass")
            if ev != "DELETED":
                self.c_podgang.enqueue(ns, md["name"])
            # gate-removal re-check for every member clique
            for group in (obj.get("spec") or {}).get("podgroups") or []:
                self.c_pclq.enqueue(ns, group.get("name", ""))
            base = md.get("labels", {}).get(c.LABEL_BASE_PODGANG)
            if base:
                # scaled gangs unblock when base gets scheduled
                pass

        def on_node(ev: str, obj: Obj, _old) -> None:
            self.scheduler.note_placement_event()
            self.c_sched.enqueue("", "pass")
            if ev in ("DELETED", "MODIFIED"):
                self.c_node.enqueue("", obj["metadata"]["name"])

        def on_ctb(ev: str, obj: Obj, _old) -> None:
            self.c_ctb.enqueue("", obj["metadata"]["name"])
            # topology translation feeds PodGang specs: re-sync every PCS
            self.pcs_rec.invalidate_sync_fingerprints()
            for p in self.store.list(c.KIND_PCS):
                self.c_pcs.enqueue(p["metadata"].get("namespace", "default"),
                                   p["metadata"]["name"])

        def on_hpa(ev: str, obj: Obj, _old) -> None:
            if ev == "ADDED":
                md = obj["metadata"]
                self.c_hpa.enqueue(md.get("namespace", "default"), md["name"])

        m.watch("HorizontalPodAutoscaler", on_hpa)
        m.watch(c.KIND_CTB, on_ctb)
        m.watch(c.KIND_PCS, on_pcs)
        m.watch(c.KIND_PCLQ, on_pclq)
        m.watch(c.KIND_PCSG, on_pcsg)
        m.watch("Pod", on_pod)
        m.watch(c.KIND_PODGANG, on_podgang)
        m.watch("Node", on_node)

    def metrics_lines(self):
        """Prometheus-style controller metrics (served by the apiserver /metrics)."""
        lines = ["# TYPE grove_reconcile_total counter",
                 "# TYPE grove_reconcile_seconds_total counter"]
        for ctrl in self.manager.controllers:
            lines.append(
                f'grove_reconcile_total{{controller="{ctrl.name}"}} '
                f'{ctrl.reconcile_count}')
            lines.append(
                f'grove_reconcile_seconds_total{{controller="{ctrl.name}"}} '
                f'{ctrl.reconcile_seconds:.6f}')
        lines.append("# TYPE grove_events_total counter")
        lines.append(f"grove_events_total {len(self.store.events)}")
        return lines

    # ------------------------------------------------------------------ lifecycle
    def start(self) -> "Cluster":
        if not self._started:
            # pre-manager startup topology sync (internal/clustertopology/
            # clustertopology.go:71 parity): every CTB synced to every TAS backend
            # with a direct client before controllers run
            for ctb in self.store.list(c.KIND_CTB):
                try:
                    self.ctb_rec.reconcile("", ctb["metadata"]["name"])
                except Exception:
                    log.debug("startup topology sync failed", exc_info=True)
            self.manager.start()
            self._started = True
        return self

    def stop(self) -> None:
        if self._started:
            self.manager.stop()
            self._started = False

    def __enter__(self) -> "Cluster":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()

    # ------------------------------------------------------------------ node helpers
    def add_virtual_nodes(self, count: int, gpus: int = 0, prefix: str = "node",
                          **kw) -> List[str]:
        names = []
        for i in range(count):
            name = f"{prefix}-{i}"
            self.store.create(make_virtual_node(name, gpus=gpus, **kw))
            names.append(name)
        return names

    # ------------------------------------------------------------------ apply / wait
    def apply(self, manifest) -> List[Obj]:
        """Apply YAML text / dict / list of dicts. Create-or-update semantics."""
        if isinstance(manifest, str):
            docs = [d for d in _yaml.safe_load_all(manifest) if d]
        elif isinstance(manifest, dict):
            docs = [manifest]
        else:
            docs = list(manifest)
        out = []
        for doc in docs:
            kind = doc.get("kind")
            md = doc.get("metadata", {})
            cur = self.store.try_get(kind, md.get("namespace"), md.get("name", ""))
            if cur is None:
                out.append(self.store.create(doc))
            else:
                # create-or-update must tolerate racing controller writes (any
                # status write bumps the shared resourceVersion): re-read + retry
                from .kubecore.store import ApiError
                for attempt in range(50):
                    doc2 = dict(doc)
                    doc2.setdefault("metadata", {})["resourceVersion"] = \
                        cur["metadata"]["resourceVersion"]
                    try:
                        out.append(self.store.update(doc2))
                        break
                    except ApiError as e:
                        if e.reason != "Conflict" or attempt == 49:
                            raise
                        cur = self.store.get(kind, md.get("namespace"),
                                             md.get("name", ""))
        return out

    def delete_pcs(self, name: str, namespace: str = "default") -> None:
        self.store.delete(c.KIND_PCS, namespace, name)
        self.c_pcs.enqueue(namespace, name)

    def wait_for(self, predicate: Callable[[], bool], timeout: float = 30.0,
                 poll: float = 0.01, desc: str = "condition") -> None:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if predicate():
                return
            time.sleep(poll)
        raise TimeoutError(f"timed out waiting for {desc}")

    def wait_pcs_available(self, name: str, namespace: str = "default",
                           timeout: float = 30.0, min_available: Optional[int] = None)\
            -> Obj:
        def ok() -> bool:
            pcs = self.store.try_get(c.KIND_PCS, namespace, name)
            if pcs is None:
                return False
            want = min_available if min_available is not None \
                else int(pcs["spec"].get("replicas", 0))
            return int((pcs.get("status") or {}).get("availableReplicas", 0)) >= want
        self.wait_for(ok, timeout, desc=f"PCS {name} available")
        return self.store.get(c.KIND_PCS, namespace, name)

    def wait_pods_ready(self, selector: Dict[str, str], count: int,
                        namespace: str = "default", timeout: float = 30.0) -> None:
        def ok() -> bool:
            pods = self.store.list("Pod", namespace, selector)
            return sum(1 for p in pods if cond.pod_is_ready(p)) >= count
        self.wait_for(ok, timeout, desc=f"{count} ready pods for {selector}")

    def wait_deleted(self, kind: str, name: str, namespace: str = "default",
                     timeout: float = 30.0) -> None:
        self.wait_for(lambda: self.store.try_get(kind, namespace, name) is None,
                      timeout, desc=f"{kind} {name} deleted")
