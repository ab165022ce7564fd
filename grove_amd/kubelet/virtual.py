"""Virtual kubelet — KWOK-style node agents driving pod lifecycle.

The reference tests at scale with KWOK virtual nodes (hack/infra_manager/kwok.py,
100-node scale preset); this module is the in-process equivalent: it watches bound pods
and walks them Pending → Running → Ready with configurable latencies.

It also embeds the grove-initc contract (operator/initc/internal/wait.go:109): a pod
whose PodClique declares startsAfter dependencies does not start its main containers
until every parent PodClique has >= minAvailable Ready pods. The reference runs that
wait as an init container inside the pod; a virtual node runs no containers, so the
kubelet enforces the same predicate before the Running transition.
"""
from __future__ import annotations

import logging
import time
from typing import Dict, List, Optional

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError
from ..utils import conditions as cond
from ..controllers.manager import Result

log = logging.getLogger("grove.kubelet")


def startup_dependencies_met(store: Store, pod: Obj) -> bool:
    """grove-initc wait predicate: each parent PCLQ has >= minAvailable Ready pods."""
    ns = pod["metadata"].get("namespace", "default")
    pclq_name = pod["metadata"].get("labels", {}).get(c.LABEL_PODCLIQUE)
    if not pclq_name:
        return True
    pclq = store.try_get(c.KIND_PCLQ, ns, pclq_name, copy=False)
    if pclq is None:
        return True
    for dep_fqn in pclq["spec"].get("startsAfter") or []:
        dep = store.try_get(c.KIND_PCLQ, ns, dep_fqn, copy=False)
        if dep is None:
            return False
        min_avail = int(dep["spec"].get("minAvailable", 1))
        if int((dep.get("status") or {}).get("readyReplicas", 0)) < min_avail:
            return False
    return True


class VirtualKubelet:
    """Per-pod reconciler for pods bound to virtual nodes."""

    def __init__(self, store: Store, node_names: Optional[List[str]] = None,
                 startup_latency_s: float = 0.0, ready_latency_s: float = 0.0,
                 payload=None):
        self.store = store
        self.node_names = set(node_names) if node_names is not None else None
        self.startup_latency_s = startup_latency_s
        self.ready_latency_s = ready_latency_s
        self.payload = payload  # callable(pod)->None run at start (GPU nodes override)
        self._started_at: Dict[str, float] = {}

    EXTERNAL_KUBELET_ANNOTATION = "grove.io/external-kubelet"

    def handles(self, node_name: str) -> bool:
        if self.node_names is not None and node_name not in self.node_names:
            return False
        # nodes registered by a remote node agent run their own kubelet — the
        # in-process virtual kubelet must not race it for pod lifecycle
        node = self.store.try_get("Node", None, node_name, copy=False)
        if node is not None and (node["metadata"].get("annotations") or {}).get(
                self.EXTERNAL_KUBELET_ANNOTATION) == "true":
            return False
        return True

    def reconcile(self, namespace: str, name: str) -> Result:
        # zero-copy read: this loop only reads the pod; all mutations go via patch
        pod = self.store.try_get("Pod", namespace, name, copy=False)
        if pod is None:
            self._started_at.pop(f"{namespace}/{name}", None)
            return Result.DONE
        node = pod.get("spec", {}).get("nodeName")
        if not node or not self.handles(node) or pod["metadata"].get("deletionTimestamp"):
            return Result.DONE
        if cond.pod_is_ready(pod):
            return Result.DONE
        key = f"{namespace}/{name}"
        phase = (pod.get("status") or {}).get("phase", "Pending")

        if phase == "Pending":
            if not startup_dependencies_met(self.store, pod):
                return Result(requeue_after=0.02)
            t0 = self._started_at.setdefault(key, time.monotonic())
            remaining = self.startup_latency_s - (time.monotonic() - t0)
            if remaining > 0:
                return Result(requeue_after=remaining)
            if self.payload is not None:
                try:
                    self.payload(pod)
                except Exception:
                    log.exception("pod payload failed for %s", key)
                    self._fail(namespace, name, "PayloadFailed")
                    return Result.DONE
            if self.ready_latency_s <= 0:
                # zero-latency fast path: Running + Ready in ONE status patch
                # (halves the per-pod patch count on the serial latency chain)
                self._mark_ready(namespace, name, set_start=True)
                self._started_at.pop(key, None)
                return Result.DONE
            self._transition_running(namespace, name)
            self._started_at[key] = time.monotonic()
            return Result(requeue_after=self.ready_latency_s)

        if phase == "Running":
            t0 = self._started_at.get(key, time.monotonic())
            remaining = self.ready_latency_s - (time.monotonic() - t0)
            if remaining > 0:
                return Result(requeue_after=remaining)
            self._mark_ready(namespace, name)
            self._started_at.pop(key, None)
        return Result.DONE

    def _transition_running(self, ns: str, name: str) -> None:
        def upd(o: Obj) -> None:
            st = o.setdefault("status", {})
            st["phase"] = "Running"
            st["startTime"] = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
        try:
            self.store.patch("Pod", ns, name, upd, status=True,
                             return_copy=False)
        except ApiError:
            pass

    def _mark_ready(self, ns: str, name: str, set_start: bool = False) -> None:
        def upd(o: Obj) -> None:
            st = o.setdefault("status", {})
            st["phase"] = "Running"
            if set_start and not st.get("startTime"):
                st["startTime"] = time.strftime("%Y-%m-%dT%H:%M:%SZ",
                                                time.gmtime())
            conds = st.setdefault("conditions", [])
            for want in ("ContainersReady", "Ready"):
                for cd in conds:
                    if cd.get("type") == want:
                        cd["status"] = "True"
                        break
                else:
                    conds.append({"type": want, "status": "True",
                                  "reason": "KubeletReady",
                                  "lastTransitionTime": time.strftime(
                                      "%Y-%m-%dT%H:%M:%SZ", time.gmtime())})
        try:
            self.store.patch("Pod", ns, name, upd, status=True,
                             return_copy=False)
        except ApiError:
            pass

    def _fail(self, ns: str, name: str, reason: str) -> None:
        def upd(o: Obj) -> None:
            o.setdefault("status", {})["phase"] = "Failed"
            o["status"]["reason"] = reason
        try:
            self.store.patch("Pod", ns, name, upd, status=True,
                             return_copy=False)
        except ApiError:
            pass


def make_virtual_node(name: str, gpus: int = 0, cpu: str = "256",
                      memory: str = "2048Gi", pods: int = 512,
                      labels: Optional[Dict[str, str]] = None) -> Obj:
    """Node object for the in-process cluster. An 8-GPU node is one xGMI hive."""
    lbl = {
        "kubernetes.io/hostname": name,
        c.NODE_LABEL_GPU_COUNT: str(gpus),
    }
    if gpus:
        lbl[c.NODE_LABEL_XGMI_HIVE] = f"{name}-hive0"
        lbl[c.NODE_LABEL_GPU_PRODUCT] = "MI355X"
    lbl.update(labels or {})
    alloc = {"cpu": cpu, "memory": memory, "pods": str(pods)}
    if gpus:
        alloc[c.AMD_GPU_RESOURCE] = str(gpus)
    return {
        "apiVersion": "v1", "kind": "Node",
        "metadata": {"name": name, "labels": lbl},
        "spec": {},
        "status": {"allocatable": dict(alloc), "capacity": dict(alloc),
                   "conditions": [{"type": "Ready", "status": "True"}]},
    }
