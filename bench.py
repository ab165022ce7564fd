#!/usr/bin/env python3
"""bench.py — flagship benchmark: gang scheduling throughput on MI355X.

Measures the BASELINE.json metric: PodGangs/sec scheduled + p50 time-to-all-Running for
gangs of N GPU pods on an N-GPU MI355X node (gang sizes 1/2/4/8 = the driver's scaling
sweep). One step = submit one PodCliqueSet with `--gangs-per-step` replicas (each
replica is one PodGang of N pods, 1 amd.com/gpu each) and drive the full control plane —
admission → PCS/PCLQ reconcile → gated pods → PodGang init → ungating → xGMI-aware gang
placement → dispatch to the per-GPU node-agent rank → MFMA GEMM payload on the assigned
GPU → Ready → gang Running.

Transports (the HEADLINE number is the deployable shape):
  http   — operator+scheduler run as a SEPARATE OS process serving the kube-style HTTP
           apiserver; this bench process acts as the node agent over the wire
           (HttpStoreClient watch/patch with bearer-token auth), exactly the
           multi-process deployment `python -m grove_amd operator/agent` ships. This is
           the reference-equivalent path (e2e/measurement/measurement.go crosses a real
           apiserver) and the default headline.
  inproc — everything in one process against the in-memory store (upper bound; also
           reported so the wire overhead is visible).
Default --transport both runs inproc first, then http, and reports http as the
headline with the inproc numbers nested under "inproc".

Distributed layout (torchrun, one rank per GPU): rank 0 runs the control plane + the
node agent for GPU 0; every rank serves a dispatch loop over a gloo control group and
joins one RCCL all-reduce heartbeat per dispatch cycle (backend "nccl" = RCCL over
xGMI). Pods run their payload on their scheduler-assigned GPU and then complete
(Succeeded), releasing the GPU for the next gang — the serving-job pattern.

Prints ONE JSON line from rank 0 per the driver contract.
"""
from __future__ import annotations

import argparse
import json
import os
import socket
import subprocess
import sys
import threading
import time
from typing import Any, Dict, List, Optional, Tuple

import torch

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

from grove_amd.api import constants as c  # noqa: E402
from grove_amd.cluster import Cluster  # noqa: E402
from grove_amd.controllers.manager import Controller, Result  # noqa: E402
from grove_amd.harness.measurement import (Tracker, RemoteTracker,  # noqa: E402
                                           percentile)
from grove_amd.kubelet.virtual import (make_virtual_node,  # noqa: E402
                                       startup_dependencies_met)
from grove_amd.kubelet import gpunode  # noqa: E402
from grove_amd.kubecore.store import ApiError, Obj  # noqa: E402
from grove_amd.kubecore.httpclient import HttpStoreClient  # noqa: E402

Task = Tuple[str, str, int, str, tuple]  # ns, name, gpu, payload kind, dims


# --------------------------------------------------------------------------- dispatch
class DispatchKubelet:
    """Collects bound, startup-unblocked pods for the cycle-based dispatcher."""

    def __init__(self, store):
        self.store = store
        self._lock = threading.Lock()
        self._pending: List[Task] = []
        self._seen: set = set()

    def reconcile(self, ns: str, name: str) -> Result:
        pod = self.store.try_get("Pod", ns, name)
        if pod is None:
            return Result.DONE
        key = f"{ns}/{name}"
        if key in self._seen:
            return Result.DONE
        if not pod.get("spec", {}).get("nodeName"):
            return Result.DONE
        if (pod.get("status") or {}).get("phase") != "Pending":
            return Result.DONE
        if not startup_dependencies_met(self.store, pod):
            return Result(requeue_after=0.005)
        gpu = gpunode.assigned_gpu(pod)
        kind, dims = gpunode.parse_payload(pod)
        with self._lock:
            if key in self._seen:
                return Result.DONE
            self._seen.add(key)
            self._pending.append((ns, name, gpu if gpu is not None else 0, kind, dims))
        return Result.DONE

    def drain(self) -> List[Task]:
        with self._lock:
            out, self._pending = self._pending, []
            return out

    def complete(self, ns: str, name: str) -> None:
        def upd(o: Obj) -> None:
            st = o.setdefault("status", {})
            st["phase"] = "Succeeded"
            conds = st.setdefault("conditions", [])
            for want in ("ContainersReady", "Ready"):
                for cd in conds:
                    if cd.get("type") == want:
                        cd["status"] = "True"
                        break
                else:
                    conds.append({"type": want, "status": "True",
                                  "reason": "PayloadComplete"})
        try:
            self.store.patch("Pod", ns, name, upd, status=True,
                             return_copy=False)
        except ApiError:
            pass


class Comm:
    """gloo control channel + RCCL heartbeat. Degenerates to local calls at world=1."""

    def __init__(self):
        self.world = int(os.environ.get("WORLD_SIZE", "1"))
        self.rank = int(os.environ.get("RANK", "0"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        self.dist = None
        self.nccl_group = None
        self._hb = None
        if self.world > 1:
            import torch.distributed as dist
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            dist.init_process_group("gloo")
            self.dist = dist
            if torch.cuda.is_available():
                n_dev = torch.cuda.device_count()
                self.local_rank = self.local_rank % max(1, n_dev)
                torch.cuda.set_device(self.local_rank)
                if n_dev >= self.world:
                    # one rank per GPU: RCCL heartbeat group over xGMI
                    self.nccl_group = dist.new_group(backend="nccl")
                    self._hb = torch.ones(1,
                                          device=f"cuda:{self.local_rank}")
                # else: oversubscribed protocol rehearsal (2 ranks / 1 GPU) —
                # RCCL cannot place two ranks on one device; gloo-only control

    def broadcast(self, obj=None):
        if self.dist is None:
            return obj
        box = [obj]
        self.dist.broadcast_object_list(box, src=0)
        return box[0]

    def gather(self, obj):
        if self.dist is None:
            return [obj]
        out = [None] * self.world if self.rank == 0 else None
        self.dist.gather_object(obj, out, dst=0)
        return out

    def heartbeat(self):
        """One RCCL all-reduce over xGMI per dispatch cycle (synchronized — an
        unsynchronized 4-byte all-reduce every cycle would pile thousands of kernels
        onto the stream over a long run)."""
        if self.nccl_group is not None:
            self.dist.all_reduce(self._hb, group=self.nccl_group)
            torch.cuda.synchronize()

    def barrier_sync(self):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        if self.dist is not None:
            self.dist.barrier()

    def max_over_ranks(self, value: float) -> float:
        if self.dist is None:
            return value
        t = torch.tensor([value], dtype=torch.float64)
        self.dist.all_reduce(t, op=self.dist.ReduceOp.MAX)
        return float(t.item())


def run_tasks(tasks, results_sink):
    for (ns, name, gpu, kind, dims) in tasks:
        try:
            gpunode.run_payload_descriptor(kind, dims, gpu)
            results_sink.append((ns, name, None))
        except Exception as e:  # pod payload failure → reported, not fatal
            results_sink.append((ns, name, str(e)))


def bench_pcs(name: str, gangs: int, gang_size: int, payload: str) -> Dict[str, Any]:
    return {
        "apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
        "metadata": {"name": name},
        "spec": {
            "replicas": gangs,
            "template": {
                "cliques": [{
                    "name": "inf",
                    "annotations": {gpunode.PAYLOAD_SHAPE_ANNOTATION: payload},
                    "spec": {
                        "roleName": "inference",
                        "replicas": gang_size,
                        "minAvailable": gang_size,
                        "podSpec": {"containers": [{
                            "name": "model", "image": "grove-bench-payload",
                            "resources": {"requests": {
                                "cpu": "1", c.AMD_GPU_RESOURCE: "1"}},
                        }]},
                    },
                }],
            },
        },
    }


# --------------------------------------------------------------------------- planes
class InprocPlane:
    """Control plane in this process against the in-memory store."""

    transport = "inproc"

    def __init__(self, args, n_gpus: int):
        cluster = Cluster(concurrent_syncs=args.workers)
        kubelet = DispatchKubelet(cluster.store)
        # replace the virtual kubelet with the dispatch kubelet (payload on ranks)
        cluster.c_kubelet.stop()
        dispatch_ctrl = cluster.manager.add_controller(
            Controller("dispatch-kubelet", kubelet.reconcile, workers=2))

        def on_pod(ev, obj, _old):
            md = obj["metadata"]
            if obj.get("spec", {}).get("nodeName"):
                dispatch_ctrl.enqueue(md.get("namespace", "default"), md["name"])
        cluster.manager.watch("Pod", on_pod)
        cluster.start()
        cluster.store.create(
            make_virtual_node("mi355x-0", gpus=n_gpus, cpu="10240", pods=65536))
        self.cluster = cluster
        self.kubelet = kubelet
        self.tracker = Tracker(cluster.store).start()

    def submit(self, pcs: Obj) -> None:
        self.cluster.store.create(pcs)

    def teardown_step(self, name: str) -> None:
        try:
            self.cluster.store.delete(c.KIND_PCS, "default", name)
        except Exception:
            pass

    def drain(self) -> List[Task]:
        return self.kubelet.drain()

    def complete(self, ns: str, name: str) -> None:
        self.kubelet.complete(ns, name)

    def stop(self) -> None:
        self.tracker.stop()
        self.cluster.stop()


class HttpPlane:
    """The deployable shape: operator+scheduler as a separate OS process behind the
    HTTP apiserver; this process is the node agent + measurement client, all traffic
    over the wire (reference measurement.go:29-104 crosses a real
    apiserver the same way)."""

    transport = "http"
    AGENT_TOKEN = "bench-agent-token"

    def __init__(self, args, n_gpus: int):
        port = _free_port()
        self.base = f"http://127.0.0.1:{port}"
        cfg_path = os.path.join(REPO, "gpurun_out", f".bench_op_{port}.yaml")
        os.makedirs(os.path.dirname(cfg_path), exist_ok=True)
        with open(cfg_path, "w") as f:
            f.write("servers:\n"
                    f"  api: {{enabled: true, host: 127.0.0.1, port: {port}}}\n"
                    "controllers:\n"
                    f"  podCliqueSet: {{concurrentSyncs: {args.workers}}}\n"
                    f"  podClique: {{concurrentSyncs: {args.workers}}}\n"
                    f"  podCliqueScalingGroup: {{concurrentSyncs: {args.workers}}}\n"
                    "logLevel: warn\n")
        env = {**os.environ, "GROVE_AGENT_TOKEN": self.AGENT_TOKEN}
        # the operator must not inherit torchrun's rank env (it is a plain process)
        for k in ("RANK", "LOCAL_RANK", "WORLD_SIZE", "MASTER_ADDR", "MASTER_PORT",
                  "GROUP_RANK", "LOCAL_WORLD_SIZE", "TORCHELASTIC_RUN_ID"):
            env.pop(k, None)
        self.operator = subprocess.Popen(
            [sys.executable, "-m", "grove_amd", "operator", "--config-file", cfg_path],
            cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
            env=env)
        self.client = HttpStoreClient(self.base, timeout=15.0,
                                      token=self.AGENT_TOKEN)
        deadline = time.monotonic() + 60
        import urllib.request
        while True:
            try:
                urllib.request.urlopen(f"{self.base}/healthz", timeout=0.5)
                break
            except Exception:
                if time.monotonic() > deadline or self.operator.poll() is not None:
                    out = self.operator.stdout.read()[:2000] \
                        if self.operator.poll() is not None else ""
                    raise RuntimeError(f"operator process failed to serve: {out}")
                time.sleep(0.05)
        node = make_virtual_node("mi355x-0", gpus=n_gpus, cpu="10240", pods=65536)
        node["metadata"].setdefault("annotations", {})[
            "grove.io/external-kubelet"] = "true"
        self.client.create(node)
        self._lock = threading.Lock()
        self._pending: List[Task] = []
        self._deferred: List[Obj] = []  # startup-dep-blocked pods, re-checked in drain
        self._seen: set = set()
        self._stop = threading.Event()
        # one wire stream serves both the dispatch collector (tap) and the
        # tracker's milestone handler
        self.tracker = RemoteTracker(self.client, pod_tap=self._on_pod).start()

    def _on_pod(self, ev: str, pod: Obj) -> None:
        if ev == "DELETED":
            return
        md = pod["metadata"]
        key = f"{md.get('namespace', 'default')}/{md['name']}"
        if key in self._seen:
            return
        if not pod.get("spec", {}).get("nodeName"):
            return
        if (pod.get("status") or {}).get("phase") != "Pending":
            return
        self._admit(key, pod)

    def _admit(self, key: str, pod: Obj) -> None:
        # pods without a grove-initc container have no startup deps (pod.go:315);
        # dep-blocked pods go to the deferred list, re-checked each drain()
        has_initc = any(ic.get("name") == "grove-initc"
                        for ic in pod["spec"].get("initContainers", []))
        if has_initc and not startup_dependencies_met(self.client, pod):
            with self._lock:
                if key not in self._seen:
                    self._deferred.append(pod)
            return
        gpu = gpunode.assigned_gpu(pod)
        kind, dims = gpunode.parse_payload(pod)
        md = pod["metadata"]
        with self._lock:
            if key in self._seen:
                return
            self._seen.add(key)
            self._pending.append((md.get("namespace", "default"), md["name"],
                                  gpu if gpu is not None else 0, kind, dims))

    def submit(self, pcs: Obj) -> None:
        self.client.create(pcs)

    def teardown_step(self, name: str) -> None:
        try:
            self.client.delete(c.KIND_PCS, "default", name)
        except Exception:
            pass

    def drain(self) -> List[Task]:
        with self._lock:
            deferred, self._deferred = self._deferred, []
            out, self._pending = self._pending, []
        for pod in deferred:
            md = pod["metadata"]
            self._admit(f"{md.get('namespace', 'default')}/{md['name']}", pod)
        if deferred:
            with self._lock:
                out.extend(self._pending)
                self._pending = []
        return out

    def complete(self, ns: str, name: str) -> None:
        try:
            self.client.merge_patch("Pod", ns, name, {"status": {
                "phase": "Succeeded",
                "conditions": [
                    {"type": "ContainersReady", "status": "True",
                     "reason": "PayloadComplete"},
                    {"type": "Ready", "status": "True",
                     "reason": "PayloadComplete"}]}}, status=True)
        except ApiError:
            pass

    def stop(self) -> None:
        self._stop.set()
        self.tracker.stop()
        self.operator.terminate()
        try:
            self.operator.wait(timeout=10)
        except subprocess.TimeoutExpired:
            self.operator.kill()


def _calibrate() -> Dict[str, Any]:
    """Per-box calibration (NOTES r1 item 5): host load + a fixed spin workload so
    readers can normalize gangs/s across differently-loaded boxes. spin_ms is the
    wall time of a fixed pure-Python loop (~100 ms on an idle modern core)."""
    t0 = time.perf_counter()
    x = 0
    for i in range(2_000_000):
        x += i * 3 // 7
    spin_ms = (time.perf_counter() - t0) * 1000
    try:
        load1, load5, _ = os.getloadavg()
    except OSError:
        load1 = load5 = None
    return {"spin_ms": round(spin_ms, 2), "loadavg_1m": load1,
            "loadavg_5m": load5, "cpu_count": os.cpu_count()}


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


# --------------------------------------------------------------------------- driver
def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=None)
    ap.add_argument("--steps", type=int, default=12)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--gangs-per-step", type=int, default=4)
    ap.add_argument("--gang-size", type=int, default=None)
    ap.add_argument("--payload", type=str, default="2048x2048x2048x1",
                    help="per-pod GEMM payload MxNxKxiters")
    ap.add_argument("--step-timeout", type=float, default=120.0)
    ap.add_argument("--workers", type=int, default=4,
                    help="per-controller worker threads (ConcurrentSyncs)")
    ap.add_argument("--inflight", type=int, default=2,
                    help="pipelined steps kept in flight (1 = fully serial)")
    ap.add_argument("--transport", choices=("both", "http", "inproc"),
                    default="both",
                    help="both = inproc + http (http is the headline)")
    args = ap.parse_args()

    comm = Comm()
    n_gpus = args.gpus or comm.world
    gang_size = args.gang_size or n_gpus
    if gang_size > n_gpus * max(1, comm.world):
        sys.exit(f"--gang-size {gang_size} cannot fit: the bench node advertises "
                 f"{n_gpus} amd.com/gpu per rank (use --gpus >= gang size)")

    if comm.rank != 0:
        _serve_agent(comm)
        return

    try:
        _run_rank0(comm, args, n_gpus, gang_size)
    except BaseException:
        # release agent ranks before dying — a hung collective would otherwise keep
        # every rank alive until the launcher's timeout
        if comm.dist is not None:
            try:
                comm.broadcast({"type": "stop"})
                comm.barrier_sync()
            except Exception:
                pass
        raise


def _measure_allreduce_busbw(comm: Comm, mb: int = 256, iters: int = 5) -> float:
    t = torch.ones(mb * 1024 * 1024 // 4, device=f"cuda:{comm.local_rank}")
    for _ in range(2):
        comm.dist.all_reduce(t, group=comm.nccl_group)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        comm.dist.all_reduce(t, group=comm.nccl_group)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    n = comm.world
    return 2 * (n - 1) / n * t.numel() * 4 / dt / 1e9


def _serve_agent(comm: Comm) -> None:
    """Node-agent ranks: execute dispatched payloads until stop.

    Protocol (all collective, so every rank stays matched with rank 0):
      run     — execute my assignment, RCCL heartbeat, gather results
      mark    — cuda.synchronize + barrier; record local timestamp
      elapsed — all-reduce MAX of my (last mark − previous mark)
      stop    — final barrier, exit
    """
    marks: List[float] = []
    while True:
        instr = comm.broadcast(None)
        kind = instr.get("type")
        if kind == "stop":
            comm.barrier_sync()
            return
        if kind == "mark":
            comm.barrier_sync()
            marks.append(time.perf_counter())
            continue
        if kind == "elapsed":
            local = marks[-1] - marks[-2] if len(marks) >= 2 else 0.0
            comm.max_over_ranks(local)
            continue
        if kind == "busbw":
            _measure_allreduce_busbw(comm)
            continue
        results: List[Tuple[str, str, Optional[str]]] = []
        if kind == "run":
            mine = instr.get("assign", {}).get(comm.rank, [])
            run_tasks(mine, results)
        comm.heartbeat()
        comm.gather(results)


def _drive(plane, comm: Comm, args, gang_size: int,
           collect_busbw: bool = False) -> Dict[str, Any]:
    """Prime + warm up + run the timed region on one control plane; return metrics."""
    tracker = plane.tracker
    world = comm.world

    def dispatch_cycle(idle_sleep: float = 0.0015) -> int:
        tasks = plane.drain()
        if not tasks and comm.dist is None:
            time.sleep(idle_sleep)
            return 0
        assign: Dict[int, list] = {}
        for t in tasks:
            gpu = t[2]
            rank = gpu % max(1, world)
            assign.setdefault(rank, []).append(t)
        if comm.dist is not None:
            comm.broadcast({"type": "run", "assign": assign})
        local_results: List[Tuple[str, str, Optional[str]]] = []
        run_tasks(assign.get(0, []), local_results)
        comm.heartbeat()
        gathered = comm.gather(local_results)
        n = 0
        for rank_results in gathered or []:
            for (ns, name, err) in rank_results or []:
                plane.complete(ns, name)
                n += 1
        if n == 0 and comm.dist is not None:
            time.sleep(idle_sleep)
        return n

    def submit_step(step_id: int, timed_epoch: str) -> str:
        name = f"bench-{timed_epoch}-{step_id}"
        pcs = bench_pcs(name, args.gangs_per_step, gang_size, args.payload)
        submit = time.monotonic()
        for g in range(args.gangs_per_step):
            tracker.expect_gang(f"{name}-{g}", gang_size, submit)
        plane.submit(pcs)
        return name

    def wait_step(name: str) -> None:
        deadline = time.monotonic() + args.step_timeout
        while time.monotonic() < deadline:
            dispatch_cycle()
            if all(tracker.gangs[f"{name}-{g}"].running is not None
                   for g in range(args.gangs_per_step)):
                # full lifecycle: tear the step's PCS down once its gangs ran —
                # without this a long run accumulates thousands of finished CR
                # trees and list/watch costs grow (1000-step soak measured 163
                # gangs/s with p95 121 ms before; bounded-store behavior after)
                plane.teardown_step(name)
                return
        raise TimeoutError(
            f"step {name} did not reach all-Running ({plane.transport})")

    def run_steps(n: int, timed_epoch: str) -> None:
        """Pipelined: keep up to --inflight steps' gangs in flight."""
        window: List[str] = []
        for s in range(n):
            window.append(submit_step(s, timed_epoch))
            if len(window) >= max(1, args.inflight):
                wait_step(window.pop(0))
        for name in window:
            wait_step(name)

    # priming (setup, before the W official warmup steps): a cold box pays
    # page-cache/HIP-module/thread-pool costs on the first gangs — measured 222 vs
    # 330 gangs/s first-run-vs-steady on one box. A fixed handful of priming gangs
    # reaches steady state regardless of the driver's chosen -W. The wire plane
    # needs a longer ramp (TCP/uvicorn/operator-process warmup: a 2000-step soak
    # sustains 194 gangs/s where 4-prime short runs report ~120).
    prime = int(os.environ.get("GROVE_BENCH_PRIME",
                               "16" if plane.transport == "http" else "4"))
    run_steps(prime, "prime")
    run_steps(args.warmup, "warm")

    # RCCL-over-xGMI evidence: bus bandwidth of a 256 MB all-reduce across all ranks
    # (ring algorithm: busbw = 2*(n-1)/n * bytes / t). Runs on the driver's 8-GPU
    # scale sweep; skipped when single-rank or no GPU.
    rccl_busbw = None
    if collect_busbw and comm.nccl_group is not None and comm.world > 1:
        comm.broadcast({"type": "busbw"})
        rccl_busbw = _measure_allreduce_busbw(comm)

    # timed region (barrier + synchronize on both sides; MAX over ranks)
    if comm.dist is not None:
        comm.broadcast({"type": "mark"})
    comm.barrier_sync()
    t0 = time.perf_counter()
    run_steps(args.steps, "timed")
    if comm.dist is not None:
        comm.broadcast({"type": "mark"})
    comm.barrier_sync()
    t1 = time.perf_counter()
    if comm.dist is not None:
        comm.broadcast({"type": "elapsed"})
    elapsed = comm.max_over_ranks(t1 - t0)

    timed_gangs = [tracker.gangs[f"bench-timed-{s}-{g}"]
                   for s in range(args.steps) for g in range(args.gangs_per_step)]
    ttr = [(t.running - t.submitted) * 1000 for t in timed_gangs if t.running]
    tts = [(t.scheduled - t.submitted) * 1000 for t in timed_gangs if t.scheduled]
    n_gangs = args.steps * args.gangs_per_step
    return {
        "transport": plane.transport,
        "value": round(n_gangs / elapsed, 3),
        "elapsed_s": round(elapsed, 4),
        "ms_per_step": round(elapsed / args.steps * 1000, 3),
        "p50_time_to_running_ms": round(percentile(ttr, 50), 3) if ttr else None,
        "p95_time_to_running_ms": round(percentile(ttr, 95), 3) if ttr else None,
        "p50_time_to_scheduled_ms": round(percentile(tts, 50), 3) if tts else None,
        "rccl_allreduce_busbw_gbps": round(rccl_busbw, 1) if rccl_busbw else None,
    }


def _run_rank0(comm: Comm, args, n_gpus: int, gang_size: int) -> None:
    # the control plane is a latency chain of ~6 thread handoffs per gang; the
    # default 5 ms GIL switch interval throttles those handoffs (same-box A/B:
    # +6-11% gangs/s at 0.2 ms). The http-mode operator process sets its own.
    sys.setswitchinterval(0.0002)
    # GC tuning for the latency tail: the default gen0 threshold (700) fires
    # collections mid-gang at high object churn; raising thresholds + freezing
    # startup objects moves collections off the critical path (A/B on tail
    # percentiles; the store's object graph is acyclic dicts, so cycles are rare)
    if os.environ.get("GROVE_GC_TUNE", "1") != "0":
        import gc
        gc.collect()
        gc.freeze()
        gc.set_threshold(100000, 50, 50)

    results: Dict[str, Dict[str, Any]] = {}
    order = {"both": ["inproc", "http"], "http": ["http"],
             "inproc": ["inproc"]}[args.transport]
    for i, transport in enumerate(order):
        plane = (InprocPlane if transport == "inproc" else HttpPlane)(args, n_gpus)
        try:
            results[transport] = _drive(plane, comm, args, gang_size,
                                        collect_busbw=(i == len(order) - 1))
        finally:
            plane.stop()

    if comm.dist is not None:
        comm.broadcast({"type": "stop"})
        comm.barrier_sync()

    headline = results.get("http") or results["inproc"]
    result = {
        "metric": "podgangs_per_sec",
        "value": headline["value"],
        "unit": "gangs/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": headline["ms_per_step"],
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "p50_time_to_running_ms": headline["p50_time_to_running_ms"],
        "p95_time_to_running_ms": headline["p95_time_to_running_ms"],
        "p50_time_to_scheduled_ms": headline["p50_time_to_scheduled_ms"],
        "calibration": _calibrate(),
        "config": {
            "model": "gang-scheduled inference PodCliqueSet (1 clique, "
                     f"{gang_size} pods x 1 amd.com/gpu, MFMA bf16 GEMM payload)",
            "global_batch": args.gangs_per_step,
            "seq_len": None,
            "parallelism": f"gang{gang_size}",
            "gang_size": gang_size,
            "gangs_per_step": args.gangs_per_step,
            "payload": args.payload,
            "inflight": args.inflight,
            "concurrent_syncs": args.workers,
            "transport": headline["transport"],
            "operator_version": __import__("grove_amd").__version__,
            "scheduler": "amd-gang-scheduler (native xGMI Filter/Score)",
            "rccl_allreduce_busbw_gbps": headline["rccl_allreduce_busbw_gbps"],
        },
    }
    if "inproc" in results and headline["transport"] != "inproc":
        result["inproc"] = {k: results["inproc"][k] for k in (
            "value", "ms_per_step", "p50_time_to_running_ms",
            "p95_time_to_running_ms", "p50_time_to_scheduled_ms")}
        result["inproc"]["unit"] = "gangs/s"
    print(json.dumps(result), flush=True)


if __name__ == "__main__":
    main()
