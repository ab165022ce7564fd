"""Store persistence — control-plane checkpoint/resume.

Parity role: the reference's control-plane state lives in etcd; CRD status + finalizers
+ expectations resync give it restart continuity. The in-process store is
memory-backed, so this module provides the durability half: atomic JSON snapshots of
every object table (+ the resourceVersion counter) and restore on startup. The operator
CLI snapshots periodically and on shutdown; controllers resync from the restored state
exactly like a controller-runtime cache re-list.
"""
from __future__ import annotations

import json
import os
import tempfile
import threading
from typing import Optional

from .store import Store


def save(store: Store, path: str) -> int:
    """Atomic snapshot; returns object count."""
    with store._lock:
        state = {
            "resourceVersion": str(next(store._rv)),
            "tables": {kind: list(tbl.objects.values())
                       for kind, tbl in store._tables.items()},
        }
    n = sum(len(v) for v in state["tables"].values())
    d = os.path.dirname(os.path.abspath(path))
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, prefix=".snapshot-")
    try:
        with os.fdopen(fd, "w") as f:
            json.dump(state, f, separators=(",", ":"))
        os.replace(tmp, path)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)
    return n


def load(store: Store, path: str) -> int:
    """Restore a snapshot into an empty store (objects re-indexed, uids preserved,
    rv counter resumed past the snapshot's high-water mark)."""
    import itertools
    with open(path) as f:
        state = json.load(f)
    n = 0
    with store._lock:
        for kind, objs in state["tables"].items():
            tbl = store._table(kind)
            for obj in objs:
                m = obj.get("metadata", {})
                key = (m.get("namespace", ""), m.get("name", ""))
                tbl.objects[key] = obj
                tbl.index_add(key, obj)
                if m.get("uid"):
                    store._uid_index[m["uid"]] = (kind, key[0], key[1])
                n += 1
        store._rv = itertools.count(int(state.get("resourceVersion", "1")) + 1)
    return n


class SnapshotLoop:
    """Background periodic snapshotter (operator CLI wiring)."""

    def __init__(self, store: Store, path: str, period_s: float = 10.0):
        self.store = store
        self.path = path
        self.period_s = period_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def start(self) -> "SnapshotLoop":
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="store-snapshot")
        self._thread.start()
        return self

    def _run(self) -> None:
        while not self._stop.wait(self.period_s):
            try:
                save(self.store, self.path)
            except Exception:
                pass

    def stop(self, final_snapshot: bool = True) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
        if final_snapshot:
            save(self.store, self.path)
