"""Timeline measurement harness.

Parity role: reference e2e/measurement/measurement.go:29-104 — milestone timeline
tracking (pods-created / pods-scheduled / pods-ready / gang-running) with per-gang
latencies and JSON export; this is the instrument behind the BASELINE.json metric
(PodGangs/sec + p50 time-to-all-Running).
"""
from __future__ import annotations

import threading
import time
from typing import Any, Dict, List, Optional

from ..api import constants as c
from ..kubecore.store import Store, Obj
from ..utils import conditions as cond


def percentile(values: List[float], p: float) -> Optional[float]:
    if not values:
        return None
    vs = sorted(values)
    idx = min(len(vs) - 1, max(0, int(round(p / 100.0 * (len(vs) - 1)))))
    return vs[idx]


class GangTimeline:
    __slots__ = ("name", "submitted", "pods_created", "scheduled", "running", "n_pods")

    def __init__(self, name: str, submitted: float):
        self.name = name
        self.submitted = submitted
        self.pods_created: Optional[float] = None
        self.scheduled: Optional[float] = None
        self.running: Optional[float] = None
        self.n_pods = 0


class Tracker:
    """Watches pods + PodGangs and records per-gang milestone times."""

    def __init__(self, store: Store):
        self.store = store
        self.gangs: Dict[str, GangTimeline] = {}
        self._lock = threading.Lock()
        self._pod_ready: Dict[str, Dict[str, float]] = {}  # gang -> pod -> ready ts
        self._pod_expected: Dict[str, int] = {}
        self._watches = []
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()

    def expect_gang(self, name: str, n_pods: int, submitted: Optional[float] = None):
        with self._lock:
            self.gangs[name] = GangTimeline(name, submitted or time.monotonic())
            self.gangs[name].n_pods = n_pods
            self._pod_ready[name] = {}
            self._pod_expected[name] = n_pods

    def start(self) -> "Tracker":
        for kind, handler in (("Pod", self._on_pod), (c.KIND_PODGANG, self._on_gang)):
            w = self.store.watch(kind, seed=True)
            self._watches.append(w)
            t = threading.Thread(target=self._pump, args=(w, handler), daemon=True)
            t.start()
            self._threads.append(t)
        return self

    def _pump(self, w, handler):
        import queue as _q
        while not self._stop.is_set():
            try:
                ev, obj = w.queue.get(timeout=0.2)
            except _q.Empty:
                continue
            try:
                handler(ev, obj)
            except Exception:
                pass

    def _on_pod(self, ev: str, pod: Obj) -> None:
        gang = pod["metadata"].get("labels", {}).get(c.LABEL_PODGANG)
        if not gang:
            return
        now = time.monotonic()
        with self._lock:
            tl = self.gangs.get(gang)
            if tl is None:
                return
            if tl.pods_created is None:
                # creation milestone = first pod seen; refined by expected count below
                pass
            if ev != "DELETED" and cond.pod_is_ready(pod):
                ready = self._pod_ready[gang]
                ready.setdefault(pod["metadata"]["name"], now)
                if tl.running is None and len(ready) >= self._pod_expected.get(gang, 1):
                    tl.running = now

    def _on_gang(self, ev: str, pg: Obj) -> None:
        name = pg["metadata"]["name"]
        now = time.monotonic()
        with self._lock:
            tl = self.gangs.get(name)
            if tl is None:
                return
            if tl.scheduled is None and cond.condition_true(pg, c.PODGANG_COND_SCHEDULED):
                tl.scheduled = now

    def stop(self) -> None:
        self._stop.set()
        for w in self._watches:
            w.stop()


    # ---- reporting ----
    def wait_all_running(self, timeout: float = 60.0) -> bool:
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            with self._lock:
                if all(t.running is not None for t in self.gangs.values()):
                    return True
            time.sleep(0.002)
        return False

    def summary(self) -> Dict[str, Any]:
        with self._lock:
            ttr = [(t.running - t.submitted) * 1000 for t in self.gangs.values()
                   if t.running is not None]
            tts = [(t.scheduled - t.submitted) * 1000 for t in self.gangs.values()
                   if t.scheduled is not None]
            n = len(self.gangs)
            done = sum(1 for t in self.gangs.values() if t.running is not None)
        return {
            "gangs_total": n,
            "gangs_running": done,
            "p50_time_to_running_ms": percentile(ttr, 50),
            "p95_time_to_running_ms": percentile(ttr, 95),
            "max_time_to_running_ms": max(ttr) if ttr else None,
            "p50_time_to_scheduled_ms": percentile(tts, 50),
        }


class RemoteTracker(Tracker):
    """Tracker over the wire: pumps the apiserver's ndjson watch streams through an
    HttpStoreClient instead of in-process store watches — the measurement instrument
    of the deployable (multi-process) shape, like the reference's client-go-based
    e2e/measurement/measurement.go:29-104 harness."""

    def __init__(self, client, pod_tap=None):
        super().__init__(store=None)  # type: ignore[arg-type]
        self.client = client
        # optional tap: every Pod event is ALSO forwarded here (lets a co-located
        # consumer — e.g. the bench node agent's dispatch collector — share one
        # wire stream instead of opening its own)
        self._pod_tap = pod_tap

    def start(self) -> "RemoteTracker":
        handlers = [("Pod", self._pod_with_tap),
                    (c.KIND_PODGANG, self._on_gang)]
        for kind, handler in handlers:
            t = threading.Thread(target=self._pump_remote, args=(kind, handler),
                                 daemon=True)
            t.start()
            self._threads.append(t)
        return self

    def _pod_with_tap(self, ev, obj):
        if self._pod_tap is not None:
            try:
                self._pod_tap(ev, obj)
            except Exception:
                pass
        self._on_pod(ev, obj)

    def _pump_remote(self, kind, handler):
        try:
            for ev, obj in self.client.watch_events(kind, None, seed=True):
                if self._stop.is_set():
                    return
                try:
                    handler(ev, obj)
                except Exception:
                    pass
        except Exception:
            pass  # stream torn down (server stop / tracker stop)

