#!/usr/bin/env python3
"""Scale sweep: N-pod PodCliqueSet all-Ready latency (reference envelope:
e2e/tests/scale/scale_test.go:166 — 1000 pods in 600 s on 100 KWOK nodes).

Usage: python scripts/scale_sweep.py [pods ...] (default 1000 10000)
Writes profiles/scaleN.json-style records to stdout; optional --profile prints
a sampled flamegraph-ish top list.
"""
import collections
import json
import sys
import threading
import time
import traceback

sys.path.insert(0, __file__.rsplit("/", 2)[0])
sys.setswitchinterval(0.0002)

from grove_amd import Cluster  # noqa: E402
from grove_amd.api import constants as c  # noqa: E402


def scale_pcs(name, replicas):
    return {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
            "metadata": {"name": name},
            "spec": {"replicas": replicas, "template": {"cliques": [{
                "name": "srv",
                "spec": {"roleName": "srv", "replicas": 2, "minAvailable": 2,
                         "podSpec": {"containers": [{
                             "name": "m", "image": "img",
                             "resources": {"requests": {"cpu": "1"}}}]}}}]}}}


def run(pods: int, workers: int = 8, profile: bool = False):
    gangs = pods // 2
    nodes = max(100, pods // 10)
    cl = Cluster(concurrent_syncs=workers).start()
    cl.add_virtual_nodes(nodes, cpu="64", pods=256)
    counts = collections.Counter()
    stop = [False]
    if profile:
        def sampler():
            while not stop[0]:
                for tid, frame in sys._current_frames().items():
                    st = traceback.extract_stack(frame, limit=3)
                    if st:
                        counts[";".join(f"{f.filename.rsplit('/', 1)[-1]}:{f.name}"
                                        for f in st[-2:])] += 1
                time.sleep(0.01)
        threading.Thread(target=sampler, daemon=True).start()
    t0 = time.monotonic()
    cl.store.create(scale_pcs("sweep", gangs))
    created = ready = None
    while True:
        pcs = cl.store.try_get(c.KIND_PCS, "default", "sweep", copy=False)
        avail = int((pcs.get("status") or {}).get("availableReplicas", 0))
        if created is None:
            n = len(cl.store.list("Pod", "default",
                                  {c.LABEL_PART_OF: "sweep"}, copy_objects=False))
            if n >= pods:
                created = time.monotonic() - t0
        if avail >= gangs:
            ready = time.monotonic() - t0
            break
        time.sleep(0.25)
        if time.monotonic() - t0 > 900:
            break
    stop[0] = True
    import os
    t_spin = time.perf_counter()
    x = 0
    for i in range(2_000_000):
        x += i * 3 // 7
    spin_ms = (time.perf_counter() - t_spin) * 1000
    rec = {"pods": pods, "gangs": gangs, "nodes": nodes,
           "created_s": round(created, 1) if created else None,
           "all_ready_s": round(ready, 1) if ready else None,
           "workers": workers,
           "calibration": {"spin_ms": round(spin_ms, 1),
                           "loadavg_1m": round(os.getloadavg()[0], 1)}}
    print(json.dumps(rec), flush=True)
    if profile:
        for k, v in counts.most_common(15):
            print(f"{v:7d} {k}", flush=True)
    t0 = time.monotonic()
    cl.delete_pcs("sweep")
    cl.wait_deleted(c.KIND_PCS, "sweep", timeout=300)
    print(json.dumps({"delete_s": round(time.monotonic() - t0, 1)}), flush=True)
    cl.stop()
    return rec


if __name__ == "__main__":
    profile = "--profile" in sys.argv
    sizes = [int(a) for a in sys.argv[1:] if a.isdigit()] or [1000, 10000]
    for n in sizes:
        run(n, profile=profile)
