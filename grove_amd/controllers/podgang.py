"""PodGang controller — the thin bridge from PodGangs to scheduler backends.

Parity source: operator/internal/controller/podgang/reconciler.go:49-80: on PodGang
spec change, resolve the backend from the grove.io/scheduler-name label (fallback
default) and call backend.sync_podgang.
"""
from __future__ import annotations

import logging

from ..api import constants as c
from ..kubecore.store import Store
from ..scheduler.backends import Registry
from .manager import Result

log = logging.getLogger("grove.podgang")


class PodGangReconciler:
    def __init__(self, store: Store, registry: Registry):
        self.store = store
        self.registry = registry

    def reconcile(self, namespace: str, name: str) -> Result:
        pg = self.store.try_get(c.KIND_PODGANG, namespace, name, copy=False)
        if pg is None or pg["metadata"].get("deletionTimestamp"):
            return Result.DONE
        backend = self.registry.resolve_for_podgang(pg)
        backend.sync_podgang(pg)
        return Result.DONE
