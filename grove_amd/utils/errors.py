"""Coded error model + status.lastErrors recorder.

Parity source: operator/internal/errors/errors.go:101 (groveerr coded errors surfaced in
status.lastErrors) and controller/common/reconcileerrorrecorder.go.
"""
from __future__ import annotations

import time
from typing import Optional

from ..kubecore.store import Store, Obj, ApiError

# ERR_* code catalog (subset mirroring the reference's vocabulary)
ERR_SYNC_PODS = "ERR_SYNC_PODS"
ERR_SYNC_PODCLIQUE = "ERR_SYNC_PODCLIQUE"
ERR_SYNC_PODGANG = "ERR_SYNC_PODGANG"
ERR_SYNC_PCSG = "ERR_SYNC_PODCLIQUESCALINGGROUP"
ERR_SYNC_HPA = "ERR_SYNC_HPA"
ERR_GANG_TERMINATION = "ERR_GANG_TERMINATION"
ERR_ROLLING_UPDATE = "ERR_ROLLING_UPDATE"
ERR_RECONCILE = "ERR_RECONCILE"
ERR_DELETE = "ERR_DELETE"


class GroveError(Exception):
    def __init__(self, code: str, message: str, cause: Optional[Exception] = None):
        super().__init__(f"[{code}] {message}")
        self.code = code
        self.message = message
        self.cause = cause


def record_last_error(store: Store, kind: str, namespace: Optional[str], name: str,
                      code: str, message: str) -> None:
    def upd(o: Obj) -> None:
        st = o.setdefault("status", {})
        errs = st.setdefault("lastErrors", [])
        errs.append({"code": code, "description": message[:512],
                     "observedAt": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())})
        if len(errs) > 5:
            del errs[:-5]
    try:
        store.patch(kind, namespace, name, upd, status=True)
    except ApiError:
        pass


def clear_last_errors(store: Store, kind: str, namespace: Optional[str],
                      name: str) -> None:
    cur = store.try_get(kind, namespace, name)
    if cur is None or not (cur.get("status") or {}).get("lastErrors"):
        return  # nothing to clear — skip the patch round trip

    def upd(o: Obj) -> None:
        st = o.setdefault("status", {})
        if st.get("lastErrors"):
            st["lastErrors"] = []
    try:
        store.patch(kind, namespace, name, upd, status=True)
    except ApiError:
        pass
