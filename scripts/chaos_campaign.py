#!/usr/bin/env python3
"""Multi-seed chaos campaign: random pod kills, node cordons/losses and replica flips
under load; every seed must converge to fully-Available. Usage:
    python scripts/chaos_campaign.py [n_seeds] [ops_per_seed]
"""
import random
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from grove_amd import Cluster
from grove_amd.api import constants as c
from grove_amd.utils import conditions as cc


def run_seed(seed: int, ops: int) -> bool:
    rng = random.Random(seed)
    cl = Cluster().start()
    try:
        cl.add_virtual_nodes(4, gpus=8, prefix="h")
        pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
               "metadata": {"name": "x"},
               "spec": {"replicas": 2, "template": {"terminationDelay": "300ms",
                   "cliques": [{"name": "w", "spec": {
                       "roleName": "w", "replicas": 3, "minAvailable": 3,
                       "podSpec": {"containers": [{"name": "m", "image": "i",
                           "resources": {"requests": {
                               "cpu": "1", c.AMD_GPU_RESOURCE: "1"}}}]}}}]}}}
        cl.store.create(pcs)
        cl.wait_pcs_available("x", timeout=30)
        for _ in range(ops):
            op = rng.randrange(4)
            if op == 0:
                pods = cl.store.list("Pod", "default", copy_objects=False)
                if pods:
                    try:
                        cl.store.delete("Pod", "default",
                                        rng.choice(pods)["metadata"]["name"])
                    except Exception:
                        pass
            elif op == 1:
                n = f"h-{rng.randrange(4)}"
                try:
                    cl.store.patch("Node", None, n, lambda o: o["spec"].update(
                        unschedulable=rng.random() < 0.5))
                except Exception:
                    pass
            elif op == 2:
                r = rng.choice([1, 2, 3])
                try:
                    cl.store.patch(c.KIND_PCS, "default", "x",
                                   lambda o: o["spec"].update(replicas=r))
                    cl.c_pcs.enqueue("default", "x")
                except Exception:
                    pass
            time.sleep(0.15)
        for i in range(4):
            try:
                cl.store.patch("Node", None, f"h-{i}",
                               lambda o: o["spec"].update(unschedulable=False))
            except Exception:
                pass
        deadline = time.monotonic() + 60
        while time.monotonic() < deadline:
            p = cl.store.get(c.KIND_PCS, "default", "x")
            want = int(p["spec"]["replicas"])
            pods = cl.store.list("Pod", "default", {c.LABEL_PART_OF: "x"},
                                 copy_objects=False)
            live = [q for q in pods if (q.get("status") or {}).get("phase")
                    not in ("Succeeded", "Failed")]
            if int((p.get("status") or {}).get("availableReplicas", 0)) >= want \
                    and len(live) == want * 3 \
                    and all(cc.pod_is_ready(q) for q in live):
                return True
            time.sleep(0.2)
        return False
    finally:
        cl.stop()


def main() -> int:
    n_seeds = int(sys.argv[1]) if len(sys.argv) > 1 else 10
    ops = int(sys.argv[2]) if len(sys.argv) > 2 else 12
    fails = []
    for seed in range(1, n_seeds + 1):
        ok = run_seed(seed, ops)
        print(f"seed {seed}: {'converged' if ok else 'FAILED'}")
        if not ok:
            fails.append(seed)
    print("failures:", fails or "none")
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
