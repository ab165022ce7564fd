"""Chaos suite: randomized failure injection under churn, asserting convergence.

The reference's GT tests inject single failures; this goes further (seeded, bounded):
random pod kills, node cordons/uncordons and a node loss while gangs churn — the
system must converge to fully-Available with exact object counts and no orphans.
"""
import random
import time

import pytest

from grove_amd.api import constants as c
from grove_amd.utils import conditions as cond


def _pcs(name, replicas, pods, gpus=0):
    res = {"cpu": "100m"}
    if gpus:
        res[c.AMD_GPU_RESOURCE] = str(gpus)
    return {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
            "metadata": {"name": name},
            "spec": {"replicas": replicas, "template": {
                "terminationDelay": "500ms",
                "cliques": [{"name": "w", "spec": {
                    "roleName": "w", "replicas": pods, "minAvailable": pods,
                    "podSpec": {"containers": [{
                        "name": "m", "image": "i",
                        "resources": {"requests": res}}]}}}]}}}


@pytest.mark.timeout(300)
def test_chaos_pod_kills_converge(cluster):
    rng = random.Random(1234)
    cluster.add_virtual_nodes(6, cpu="64", pods=256)
    for i in range(4):
        cluster.store.create(_pcs(f"ch{i}", 3, 3))
    for i in range(4):
        cluster.wait_pcs_available(f"ch{i}", timeout=30)

    # 30 random pod kills over ~6 seconds while everything keeps reconciling
    for _ in range(30):
        pods = cluster.store.list("Pod", "default", copy_objects=False)
        if pods:
            victim = rng.choice(pods)["metadata"]["name"]
            try:
                cluster.store.delete("Pod", "default", victim)
            except Exception:
                pass
        time.sleep(0.2)

    # convergence: every PCS fully available again, exact pod counts, no orphans
    for i in range(4):
        cluster.wait_pcs_available(f"ch{i}", timeout=60)
    deadline = time.monotonic() + 60
    while time.monotonic() < deadline:
        pods = cluster.store.list("Pod", "default", copy_objects=False)
        from grove_amd.utils import conditions as cc
        if len(pods) == 4 * 3 * 3 and all(cc.pod_is_ready(p) for p in pods):
            break
        time.sleep(0.2)
    pods = cluster.store.list("Pod", "default")
    assert len(pods) == 36
    # every pod belongs to a live PCLQ (no orphans) and carries a dense index
    pclqs = {q["metadata"]["name"] for q in cluster.store.list(c.KIND_PCLQ)}
    for p in pods:
        assert p["metadata"]["labels"][c.LABEL_PODCLIQUE] in pclqs
    for q in cluster.store.list(c.KIND_PCLQ):
        idx = sorted(int(p["metadata"]["labels"][c.LABEL_POD_INDEX])
                     for p in cluster.store.list(
                         "Pod", "default",
                         {c.LABEL_PODCLIQUE: q["metadata"]["name"]}))
        assert idx == list(range(len(idx))), f"index holes in {q['metadata']['name']}"


@pytest.mark.timeout(300)
def test_chaos_node_churn_with_gangs(cluster):
    """GPU gangs keep their all-or-nothing guarantee while nodes cordon/uncordon and
    one node disappears entirely."""
    rng = random.Random(99)
    names = cluster.add_virtual_nodes(3, gpus=8, prefix="hive")
    for i in range(2):
        cluster.store.create(_pcs(f"g{i}", 1, 4, gpus=1))
    for i in range(2):
        cluster.wait_pcs_available(f"g{i}", timeout=30)

    for step in range(10):
        n = rng.choice(names[1:])  # never touch hive-0 so capacity always exists
        flip = rng.random() < 0.5
        try:
            cluster.store.patch("Node", None, n,
                                lambda o: o["spec"].update(unschedulable=flip))
        except Exception:
            pass
        if step == 5:
            cluster.store.delete("Node", None, names[-1])
            names = names[:-1]
        time.sleep(0.3)
    for n in names:
        try:
            cluster.store.patch("Node", None, n,
                                lambda o: o["spec"].update(unschedulable=False))
        except Exception:
            pass

    for i in range(2):
        cluster.wait_pcs_available(f"g{i}", timeout=90)
    # gang invariant held: every gang's pods are co-resident with distinct GPU ids
    for i in range(2):
        pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: f"g{i}"},
                                  filter_fn=lambda p: (p.get("status") or {})
                                  .get("phase") != "Failed")
        live = [p for p in pods if p["spec"].get("nodeName")]
        assert len(live) == 4
        per_node = {}
        for p in live:
            ids = (p["metadata"].get("annotations") or {}).get(
                "scheduling.amd.com/gpu-ids", "")
            per_node.setdefault(p["spec"]["nodeName"], []).extend(
                ids.split(",") if ids else [])
        for node, ids in per_node.items():
            assert len(ids) == len(set(ids)), f"GPU double-assignment on {node}"


@pytest.mark.timeout(300)
def test_rolling_update_with_concurrent_failure(cluster):
    """Reference hard-part (d): rolling update coherence while a non-updating replica
    fails and gang-terminates mid-update — everything must converge to the new
    template, fully available."""
    cluster.add_virtual_nodes(3, cpu="64", pods=256)
    pcs = _pcs("rux", 3, 2)
    pcs["spec"]["template"]["terminationDelay"] = "400ms"
    pcs["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v1"
    cluster.store.create(pcs)
    cluster.wait_pcs_available("rux", timeout=30)

    def bump(o):
        o["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
            "image"] = "img:v2"
    cluster.store.patch(c.KIND_PCS, "default", "rux", bump)
    cluster.c_pcs.enqueue("default", "rux")

    # while the update runs, kill both pods of replica 2 and cordon nothing —
    # replacements reschedule; if the breach outlives terminationDelay the replica
    # gang-terminates and recreates straight onto the new template
    time.sleep(0.2)
    for p in cluster.store.list("Pod", "default",
                                {c.LABEL_PCS_REPLICA_INDEX: "2",
                                 c.LABEL_PART_OF: "rux"}):
        try:
            cluster.store.delete("Pod", "default", p["metadata"]["name"])
        except Exception:
            pass

    def done():
        p = cluster.store.get(c.KIND_PCS, "default", "rux")
        prog = (p.get("status") or {}).get("updateProgress") or {}
        pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "rux"},
                                  copy_objects=False)
        from grove_amd.utils import conditions as cc
        return (prog.get("updateEndedAt")
                and len(pods) == 6
                and all(p["spec"]["containers"][0]["image"] == "img:v2"
                        for p in pods)
                and all(cc.pod_is_ready(p) for p in pods))
    cluster.wait_for(done, timeout=120, desc="update + failure convergence")
    cluster.wait_pcs_available("rux", timeout=30)
