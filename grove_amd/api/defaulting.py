"""Defaulting admission for grove.io resources.

Behavior parity with the reference defaulting webhook + kubebuilder defaults:
/root/reference/operator/internal/webhook/admission/pcs/defaulting/podcliqueset.go:33-120,
/root/reference/operator/api/core/v1alpha1/{podcliqueset,podclique,scalinggroup}.go
(kubebuilder:default markers). Fresh implementation over unstructured dicts.
"""
from __future__ import annotations

from typing import Any, Dict, Optional

from . import constants as c

Obj = Dict[str, Any]


def default_pod_spec(pod_spec: Obj) -> None:
    if not pod_spec.get("restartPolicy"):
        pod_spec["restartPolicy"] = "Always"
    if pod_spec.get("terminationGracePeriodSeconds") is None:
        pod_spec["terminationGracePeriodSeconds"] = c.DEFAULT_TERMINATION_GRACE_SECONDS


def default_podcliqueset(pcs: Obj, _old: Optional[Obj] = None) -> None:
    spec = pcs.setdefault("spec", {})
    # kubebuilder:default=0 on spec.replicas (podcliqueset.go:64)
    spec.setdefault("replicas", 0)
    if spec.get("updateStrategy") is None:
        spec["updateStrategy"] = {"type": c.UPDATE_ROLLING_RECREATE}
    else:
        spec["updateStrategy"].setdefault("type", c.UPDATE_ROLLING_RECREATE)
    tmpl = spec.setdefault("template", {})
    tmpl.setdefault("cliqueStartupType", c.STARTUP_ANY_ORDER)
    if tmpl.get("terminationDelay") is None:
        tmpl["terminationDelay"] = f"{c.DEFAULT_TERMINATION_DELAY_SECONDS // 3600}h"
    if tmpl.get("headlessServiceConfig") is None:
        tmpl["headlessServiceConfig"] = {"publishNotReadyAddresses": True}
    for clique in tmpl.get("cliques") or []:
        cs = clique.setdefault("spec", {})
        if not cs.get("replicas"):
            cs["replicas"] = 1
        if cs.get("minAvailable") is None:
            cs["minAvailable"] = cs["replicas"]
        sc = cs.get("autoScalingConfig") or cs.get("scaleConfig")
        if sc is not None and sc.get("minReplicas") is None:
            sc["minReplicas"] = cs["replicas"]
        default_pod_spec(cs.setdefault("podSpec", {}))
    for sg in tmpl.get("podCliqueScalingGroups") or []:
        if sg.get("replicas") is None:
            sg["replicas"] = 1
        if sg.get("minAvailable") is None:
            sg["minAvailable"] = 1
        if sg.get("scaleConfig") is not None and sg["scaleConfig"].get("minReplicas") is None:
            sg["scaleConfig"]["minReplicas"] = sg["replicas"]


def default_podclique(pclq: Obj, _old: Optional[Obj] = None) -> None:
    spec = pclq.setdefault("spec", {})
    if not spec.get("replicas"):
        spec["replicas"] = 1
    if spec.get("minAvailable") is None:
        spec["minAvailable"] = spec["replicas"]
    default_pod_spec(spec.setdefault("podSpec", {}))


def default_pcsg(pcsg: Obj, _old: Optional[Obj] = None) -> None:
    spec = pcsg.setdefault("spec", {})
    if spec.get("replicas") is None:
        spec["replicas"] = 1
    if spec.get("minAvailable") is None:
        spec["minAvailable"] = 1


def parse_duration_seconds(d: Any) -> float:
    """Parse a metav1.Duration-style string ('4h', '30m', '90s', '1h30m') to seconds."""
    if d is None:
        return 0.0
    if isinstance(d, (int, float)):
        return float(d)
    s = str(d).strip()
    total, num = 0.0, ""
    units = {"h": 3600.0, "m": 60.0, "s": 1.0, "ms": 0.001, "us": 1e-6, "ns": 1e-9}
    i = 0
    while i < len(s):
        ch = s[i]
        if ch.isdigit() or ch in ".+-":
            num += ch
            i += 1
        else:
            u = ch
            if i + 1 < len(s) and s[i:i + 2] in units:
                u = s[i:i + 2]
                i += 1
            if u not in units or not num:
                raise ValueError(f"invalid duration {d!r}")
            total += float(num) * units[u]
            num = ""
            i += 1
    if num:
        raise ValueError(f"invalid duration {d!r}: missing unit")
    return total
