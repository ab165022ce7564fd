"""Leader election — Lease-based, coordination.k8s.io semantics.

Parity source: the reference's controller-runtime leader election
(config/v1alpha1/types.go leaderElection, manager.go options). Campaigns on a Lease
object with optimistic concurrency: acquire when unheld or expired, renew every
renew_deadline, release on stop. Only the leader's callback runs; losing the lease
fires on_stopped_leading (the operator exits or demotes to standby).
"""
from __future__ import annotations

import threading
import time
from typing import Callable, Optional

from .store import Store, Obj, ApiError

LEASE_KIND = "Lease"


class LeaderElector:
    def __init__(self, store: Store, name: str, identity: str,
                 namespace: str = "grove-system",
                 lease_duration_s: float = 15.0,
                 renew_period_s: float = 5.0):
        self.store = store
        self.name = name
        self.identity = identity
        self.namespace = namespace
        self.lease_duration_s = lease_duration_s
        self.renew_period_s = renew_period_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.is_leader = threading.Event()

    # ------------------------------------------------------------------ campaign
    def _try_acquire_or_renew(self) -> bool:
        now = time.time()
        lease = self.store.try_get(LEASE_KIND, self.namespace, self.name)
        if lease is None:
            try:
                self.store.create({
                    "apiVersion": "coordination.k8s.io/v1", "kind": LEASE_KIND,
                    "metadata": {"name": self.name, "namespace": self.namespace},
                    "spec": {"holderIdentity": self.identity,
                             "leaseDurationSeconds": int(self.lease_duration_s),
                             "renewTime": now,
                             "acquireTime": now,
                             "leaseTransitions": 0},
                })
                return True
            except ApiError:
                return False
        spec = lease.get("spec") or {}
        holder = spec.get("holderIdentity")
        expired = now - float(spec.get("renewTime", 0)) > self.lease_duration_s
        if holder != self.identity and not expired:
            return False

        def take(o: Obj) -> None:
            s = o.setdefault("spec", {})
            if s.get("holderIdentity") != self.identity:
                s["leaseTransitions"] = int(s.get("leaseTransitions", 0)) + 1
                s["acquireTime"] = now
            s["holderIdentity"] = self.identity
            s["renewTime"] = now
        try:
            self.store.patch(LEASE_KIND, self.namespace, self.name, take, retries=1)
            return True
        except ApiError:
            return False

    def _run(self, on_started: Optional[Callable[[], None]],
             on_stopped: Optional[Callable[[], None]]) -> None:
        was_leader = False
        while not self._stop.is_set():
            ok = self._try_acquire_or_renew()
            if ok and not was_leader:
                was_leader = True
                self.is_leader.set()
                if on_started:
                    on_started()
            elif not ok and was_leader:
                was_leader = False
                self.is_leader.clear()
                if on_stopped:
                    on_stopped()
            self._stop.wait(self.renew_period_s if ok else self.renew_period_s / 2)
        if was_leader:
            self.release()
            self.is_leader.clear()
            if on_stopped:
                on_stopped()

    def start(self, on_started_leading: Optional[Callable[[], None]] = None,
              on_stopped_leading: Optional[Callable[[], None]] = None
              ) -> "LeaderElector":
        self._thread = threading.Thread(
            target=self._run, args=(on_started_leading, on_stopped_leading),
            daemon=True, name=f"leader-elector-{self.name}")
        self._thread.start()
        return self

    def release(self) -> None:
        def rel(o: Obj) -> None:
            s = o.setdefault("spec", {})
            if s.get("holderIdentity") == self.identity:
                s["holderIdentity"] = None
                s["renewTime"] = 0
        try:
            self.store.patch(LEASE_KIND, self.namespace, self.name, rel, retries=2)
        except ApiError:
            pass

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
