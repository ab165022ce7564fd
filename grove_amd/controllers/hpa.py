"""HorizontalPodAutoscaler controller — multi-level autoscaling.

Parity role: the reference creates HPA objects (components/hpa/hpa.go:128) and relies on
the cluster's kube-controller-manager to drive them against metrics-server. This
in-process cluster has neither, so the HPA loop itself is implemented here: classic
utilization autoscaling (desired = ceil(current × avgUtilization / target)) against the
pod metric source, targeting PodClique /scale (spec.replicas) or PodCliqueScalingGroup
/scale — the reference's two scale targets (clique-level autoScalingConfig and
scaling-group scaleConfig).

Metric source: pod annotation `grove.io/cpu-usage` (millicores) — the synthetic
metrics-server stand-in used by tests and the soak/scale harness.
"""
from __future__ import annotations

import logging
import math
from typing import Any, Dict, List, Optional

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError
from ..utils.errors import report_api_error
from ..utils.quantity import cpu_millis
from .manager import Result

log = logging.getLogger("grove.hpa")

CPU_USAGE_ANNOTATION = "grove.io/cpu-usage"


class HPAReconciler:
    def __init__(self, store: Store, sync_period_s: float = 0.25,
                 tolerance: float = 0.1,
                 scale_down_stabilization_s: float = 10.0):
        self.store = store
        self.sync_period_s = sync_period_s
        self.tolerance = tolerance
        # kube HPA's downscale stabilization window (default 300s there; shortened
        # for the in-process loop): scale-in is applied only after the recommendation
        # has stayed below current for the whole window
        self.scale_down_stabilization_s = scale_down_stabilization_s
        self._below_since: Dict[str, float] = {}

    def reconcile(self, namespace: str, name: str) -> Result:
        hpa = self.store.try_get("HorizontalPodAutoscaler", namespace, name)
        if hpa is None or hpa["metadata"].get("deletionTimestamp"):
            return Result.DONE
        spec = hpa.get("spec") or {}
        ref = spec.get("scaleTargetRef") or {}
        target = self.store.try_get(ref.get("kind", ""), namespace, ref.get("name", ""))
        if target is None:
            return Result(requeue_after=self.sync_period_s)
        current = int(target["spec"].get("replicas", 1))

        desired = self._desired_replicas(namespace, hpa, ref, current)
        if desired is not None and desired < current:
            import time as _time
            key = f"{namespace}/{name}"
            since = self._below_since.setdefault(key, _time.monotonic())
            if _time.monotonic() - since < self.scale_down_stabilization_s:
                desired = current  # hold: inside the stabilization window
        elif desired is not None and desired >= current:
            self._below_since.pop(f"{namespace}/{name}", None)
        if desired is not None and desired != current:
            def scale(o: Obj) -> None:
                o["spec"]["replicas"] = desired
            try:
                self.store.patch(ref["kind"], namespace, ref["name"], scale)
                self.store.record_event(
                    hpa, "Normal", "SuccessfulRescale",
                    f"scaled {ref['kind']}/{ref['name']} {current} -> {desired}")
            except ApiError as e:
                report_api_error(self.store, ref["kind"], namespace, ref["name"],
                                 "HPA rescale", e)

        def upd(o: Obj) -> None:
            st = o.setdefault("status", {})
            st["currentReplicas"] = current
            st["desiredReplicas"] = desired if desired is not None else current
        try:
            self.store.patch("HorizontalPodAutoscaler", namespace, name, upd,
                             status=True)
        except ApiError as e:
            report_api_error(self.store, "HorizontalPodAutoscaler", namespace,
                             name, "write HPA status", e)
        return Result(requeue_after=self.sync_period_s)

    def _desired_replicas(self, ns: str, hpa: Obj, ref: Dict[str, Any],
                          current: int) -> Optional[int]:
        spec = hpa.get("spec") or {}
        min_r = int(spec.get("minReplicas", 1))
        max_r = int(spec.get("maxReplicas", current))
        util = self._target_utilization(spec)
        pods = self._target_pods(ns, ref)
        ratio: Optional[float] = None
        if util is not None and pods:
            usages, requests = [], []
            for p in pods:
                ann = p["metadata"].get("annotations") or {}
                if CPU_USAGE_ANNOTATION not in ann:
                    continue
                usages.append(cpu_millis(ann[CPU_USAGE_ANNOTATION]))
                req = 0
                for ctr in p["spec"].get("containers", []):
                    req += cpu_millis(((ctr.get("resources") or {})
                                       .get("requests") or {}).get("cpu", 0))
                requests.append(max(req, 1))
            if usages:
                avg_util = 100.0 * sum(usages) / sum(requests)
                ratio = avg_util / util
        if ratio is None:
            return max(min_r, min(max_r, current))
        if abs(ratio - 1.0) <= self.tolerance:
            desired = current
        else:
            desired = math.ceil(current * ratio)
        return max(min_r, min(max_r, desired))

    @staticmethod
    def _target_utilization(spec: Obj) -> Optional[float]:
        for m in spec.get("metrics") or []:
            res = m.get("resource") or {}
            if res.get("name") == "cpu":
                t = res.get("target") or {}
                if t.get("averageUtilization") is not None:
                    return float(t["averageUtilization"])
        return None

    def _target_pods(self, ns: str, ref: Dict[str, Any]) -> List[Obj]:
        if ref.get("kind") == c.KIND_PCLQ:
            return self.store.list("Pod", ns, {c.LABEL_PODCLIQUE: ref["name"]},
                                   copy_objects=False)
        if ref.get("kind") == c.KIND_PCSG:
            return self.store.list("Pod", ns, {c.LABEL_PCSG: ref["name"]},
                                   copy_objects=False)
        return []
