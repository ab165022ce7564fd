"""Unit tests for the cross-cutting utility packages (reference §2.8 parity:
workqueue semantics, index allocator, concurrency helpers, quantities, conditions,
hashes, placement score model)."""
import threading
import time

import pytest

from grove_amd.controllers.manager import Controller, Result, WorkQueue
from grove_amd.scheduler.placement import placement_score
from grove_amd.utils.concurrent import run_concurrently, run_concurrently_with_slow_start
from grove_amd.utils.conditions import set_condition, get_condition, condition_true
from grove_amd.utils.hashing import pcs_generation_hash, pod_template_hash
from grove_amd.utils.indexing import available_indices
from grove_amd.utils.quantity import cpu_millis, parse_quantity


class TestWorkQueue:
    def test_dedup_while_queued(self):
        q = WorkQueue()
        q.add(("ns", "a"))
        q.add(("ns", "a"))
        q.add(("ns", "b"))
        assert len(q) == 2

    def test_dirty_requeue_while_processing(self):
        q = WorkQueue()
        q.add(("ns", "a"))
        item = q.get()
        assert item == ("ns", "a")
        q.add(("ns", "a"))  # arrives while processing → marked dirty
        assert len(q) == 0
        q.done(item)
        assert q.get(timeout=0.5) == ("ns", "a")

    def test_delayed_add(self):
        q = WorkQueue()
        q.add_after(("ns", "a"), 0.05)
        assert q.get(timeout=0.01) is None
        assert q.get(timeout=0.5) == ("ns", "a")

    def test_rate_limit_backoff_grows(self):
        q = WorkQueue(base_delay=0.01)
        t0 = time.monotonic()
        q.add_rate_limited(("ns", "a"))
        q.get(timeout=0.5)
        q.done(("ns", "a"))
        q.add_rate_limited(("ns", "a"))
        assert q.get(timeout=0.005) is None  # second failure: 0.02s delay
        assert q.get(timeout=0.5) == ("ns", "a")
        q.forget(("ns", "a"))
        assert q._failures == {}

    def test_controller_requeue_after(self):
        seen = []

        def rec(ns, name):
            seen.append(time.monotonic())
            return Result(requeue_after=0.03) if len(seen) < 3 else Result.DONE
        ctrl = Controller("t", rec, workers=1)
        ctrl.start()
        ctrl.enqueue("ns", "x")
        deadline = time.monotonic() + 2
        while len(seen) < 3 and time.monotonic() < deadline:
            time.sleep(0.01)
        ctrl.stop()
        assert len(seen) == 3
        assert seen[1] - seen[0] >= 0.025

    def test_controller_backoff_on_exception(self):
        calls = []

        def rec(ns, name):
            calls.append(1)
            if len(calls) < 3:
                raise RuntimeError("boom")
            return Result.DONE
        ctrl = Controller("t", rec, workers=1)
        ctrl.start()
        ctrl.enqueue("ns", "x")
        deadline = time.monotonic() + 3
        while len(calls) < 3 and time.monotonic() < deadline:
            time.sleep(0.01)
        ctrl.stop()
        assert len(calls) == 3


class TestIndexing:
    def test_hole_filling(self):
        assert available_indices([0, 2, 5], 3) == [1, 3, 4]
        assert available_indices([], 2) == [0, 1]
        assert available_indices([1], 1) == [0]


class TestConcurrent:
    def test_run_concurrently_collects_errors(self):
        def ok():
            pass

        def bad():
            raise ValueError("x")
        errs = run_concurrently([("a", ok), ("b", bad), ("c", ok)])
        assert len(errs) == 1

    def test_slow_start_aborts_on_systematic_failure(self):
        attempts = []

        def bad():
            attempts.append(1)
            raise ValueError("always")
        tasks = [(f"t{i}", bad) for i in range(16)]
        errs = run_concurrently_with_slow_start(tasks, initial_batch=1)
        # batch 1 fails completely → stop: exactly 1 attempt, not 16
        assert len(attempts) == 1 and len(errs) == 1

    def test_slow_start_doubles(self):
        ran = []

        def ok(i):
            return lambda: ran.append(i)
        tasks = [(f"t{i}", ok(i)) for i in range(7)]
        errs = run_concurrently_with_slow_start(tasks, initial_batch=1)
        assert not errs and sorted(ran) == list(range(7))


class TestQuantity:
    def test_parse(self):
        assert parse_quantity("100m") == pytest.approx(0.1)
        assert parse_quantity("2") == 2
        assert parse_quantity("1Gi") == 2**30
        assert parse_quantity("1.5Ki") == 1536
        assert parse_quantity(None) == 0
        assert cpu_millis("250m") == 250
        assert cpu_millis("2") == 2000


class TestConditions:
    def test_set_and_transition(self):
        obj = {"status": {}}
        assert set_condition(obj, "Ready", True, "AllGood") is True
        t1 = get_condition(obj, "Ready")["lastTransitionTime"]
        # same status → no transition-time change, not "changed"
        assert set_condition(obj, "Ready", True, "AllGood") is False
        assert get_condition(obj, "Ready")["lastTransitionTime"] == t1
        assert condition_true(obj, "Ready")
        assert set_condition(obj, "Ready", False, "Broken") is True
        assert not condition_true(obj, "Ready")


class TestHashing:
    def test_generation_hash_stability(self):
        pcs = {"spec": {"template": {"cliques": [
            {"name": "a", "spec": {"podSpec": {"containers": [{"image": "x:1"}]}}}]}}}
        h1 = pcs_generation_hash(pcs)
        assert h1 == pcs_generation_hash(pcs)
        pcs["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
            "image"] = "x:2"
        assert pcs_generation_hash(pcs) != h1
        # replica-count changes do NOT change the hash (no rolling update)
        pcs["spec"]["replicas"] = 99
        h2 = pcs_generation_hash(pcs)
        pcs["spec"]["replicas"] = 1
        assert pcs_generation_hash(pcs) == h2

    def test_pod_template_hash_priority_class_sensitivity(self):
        a = pod_template_hash("w", {"containers": []}, "")
        b = pod_template_hash("w", {"containers": []}, "high")
        assert a != b


class TestPlacementScore:
    def test_score_model(self):
        # single GPU: fabric max (no collective bound)
        assert placement_score(1, 1) == pytest.approx(153.0 * 7)
        # one hive: per-link ring bandwidth
        assert placement_score(1, 8) == pytest.approx(153.0)
        # split across nodes: NIC-bound, monotonically worse
        s2 = placement_score(2, 8)
        s4 = placement_score(4, 8)
        assert s2 < 153.0 and s4 < s2
        assert s2 == pytest.approx(25.0)


class TestDeletionOrder:
    def test_victim_priority(self):
        """deletionsort.go parity: gated → pending → unscheduled → not-ready →
        younger first."""
        from grove_amd.controllers.podclique import PodCliqueReconciler

        def pod(name, ts, gated=False, phase="Running", scheduled=True, ready=True):
            conds = []
            if scheduled:
                conds.append({"type": "PodScheduled", "status": "True"})
            if ready:
                conds.append({"type": "Ready", "status": "True"})
            return {"metadata": {"name": name, "creationTimestamp": ts},
                    "spec": {"schedulingGates": [{"name": "g"}] if gated else [],
                             "nodeName": "n" if scheduled else None},
                    "status": {"phase": phase, "conditions": conds}}

        ready_old = pod("ready-old", "2026-01-01T00:00:00Z")
        ready_new = pod("ready-new", "2026-01-02T00:00:00Z")
        gated = pod("gated", "2026-01-01T12:00:00Z", gated=True, phase="Pending",
                    scheduled=False, ready=False)
        pending = pod("pending", "2026-01-01T12:00:00Z", phase="Pending",
                      scheduled=False, ready=False)
        unready = pod("unready", "2026-01-01T12:00:00Z", ready=False)
        order = [p["metadata"]["name"] for p in PodCliqueReconciler._deletion_order(
            [ready_old, ready_new, unready, pending, gated])]
        assert order[0] == "gated"
        assert order[1] == "pending"
        assert order[2] == "unready"
        # among ready pods, the younger one is deleted first
        assert order[3] == "ready-new" and order[4] == "ready-old"
