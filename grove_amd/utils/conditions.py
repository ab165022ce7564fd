"""metav1.Condition-style helpers over dict objects."""
from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

Obj = Dict[str, Any]


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


def get_condition(obj: Obj, ctype: str) -> Optional[Obj]:
    for cond in (obj.get("status") or {}).get("conditions") or []:
        if cond.get("type") == ctype:
            return cond
    return None


def condition_true(obj: Obj, ctype: str) -> bool:
    cond = get_condition(obj, ctype)
    return bool(cond and cond.get("status") == "True")


def set_condition(obj: Obj, ctype: str, status: bool, reason: str, message: str = "") -> bool:
    """Set/update a condition; returns True if it changed (status or reason)."""
    st = obj.setdefault("status", {})
    conds: List[Obj] = st.setdefault("conditions", [])
    sval = "True" if status else "False"
    for cond in conds:
        if cond.get("type") == ctype:
            changed = cond.get("status") != sval or cond.get("reason") != reason
            if cond.get("status") != sval:
                cond["lastTransitionTime"] = _now()
            cond["status"] = sval
            cond["reason"] = reason
            cond["message"] = message
            return changed
    conds.append({"type": ctype, "status": sval, "reason": reason, "message": message,
                  "lastTransitionTime": _now()})
    return True


def pod_condition_true(pod: Obj, ctype: str) -> bool:
    for cond in (pod.get("status") or {}).get("conditions") or []:
        if cond.get("type") == ctype:
            return cond.get("status") == "True"
    return False


def pod_is_scheduled(pod: Obj) -> bool:
    return bool(pod.get("spec", {}).get("nodeName")) or pod_condition_true(pod, "PodScheduled")


def pod_is_ready(pod: Obj) -> bool:
    return pod_condition_true(pod, "Ready")


def pod_is_gated(pod: Obj) -> bool:
    return bool(pod.get("spec", {}).get("schedulingGates"))
