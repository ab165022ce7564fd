"""Topology-aware scheduling tests (reference TAS1-18 parity, e2e/tests/topology_test.go):
CTB → SchedulerTopology sync + drift condition, constraint translation to node-label
keys, and pack-constraint enforcement in the gang scheduler."""
import time

import pytest

from grove_amd.api import constants as c
from grove_amd.scheduler.backends import KIND_SCHEDULER_TOPOLOGY
from grove_amd.utils import conditions as cond


CTB = {
    "apiVersion": c.API_VERSION, "kind": c.KIND_CTB,
    "metadata": {"name": "cluster-topology"},
    "spec": {"levels": [
        {"domain": "zone", "key": "topology.kubernetes.io/zone"},
        {"domain": "rack", "key": "topology.kubernetes.io/rack"},
        {"domain": "host", "key": "kubernetes.io/hostname"},
    ]},
}


def _pcs(name, gang_pods, gpus_per_pod=1, constraint=None):
    tmpl = {"cliques": [{"name": "w", "spec": {
        "roleName": "w", "replicas": gang_pods, "minAvailable": gang_pods,
        "podSpec": {"containers": [{"name": "m", "image": "img",
                                    "resources": {"requests": {
                                        "cpu": "1",
                                        c.AMD_GPU_RESOURCE: str(gpus_per_pod)}}}]}}}]}
    if constraint:
        tmpl["topologyConstraint"] = constraint
    return {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
            "metadata": {"name": name},
            "spec": {"replicas": 1, "template": tmpl}}


def test_ctb_sync_creates_scheduler_topology(cluster):
    cluster.store.create(CTB)

    def synced():
        st = cluster.store.try_get(KIND_SCHEDULER_TOPOLOGY, None, "cluster-topology")
        return st is not None
    cluster.wait_for(synced, timeout=10, desc="SchedulerTopology created")
    st = cluster.store.get(KIND_SCHEDULER_TOPOLOGY, None, "cluster-topology")
    domains = [lv["domain"] for lv in st["spec"]["levels"]]
    assert domains == ["zone", "rack", "host", "xgmi-hive"]
    assert st["spec"]["levels"][-1]["nodeLabelKey"] == c.NODE_LABEL_XGMI_HIVE

    def in_sync():
        ctb = cluster.store.get(c.KIND_CTB, None, "cluster-topology")
        d = cond.get_condition(ctb, c.COND_SCHEDULER_TOPOLOGY_DRIFT)
        return d is not None and d["status"] == "False"
    cluster.wait_for(in_sync, timeout=10, desc="drift condition False")


def test_ctb_drift_repaired(cluster):
    cluster.store.create(CTB)
    cluster.wait_for(lambda: cluster.store.try_get(
        KIND_SCHEDULER_TOPOLOGY, None, "cluster-topology") is not None, timeout=10)
    # external mutation drifts the backend topology resource
    cluster.store.patch(KIND_SCHEDULER_TOPOLOGY, None, "cluster-topology",
                        lambda o: o["spec"].update(levels=[]))
    cluster.c_ctb.enqueue("", "cluster-topology")

    def repaired():
        st = cluster.store.get(KIND_SCHEDULER_TOPOLOGY, None, "cluster-topology")
        return len(st["spec"]["levels"]) == 4
    cluster.wait_for(repaired, timeout=10, desc="auto-managed topology repaired")


def test_ctb_validation_rejects_duplicates(cluster):
    bad = {"apiVersion": c.API_VERSION, "kind": c.KIND_CTB,
           "metadata": {"name": "bad"},
           "spec": {"levels": [
               {"domain": "rack", "key": "a"},
               {"domain": "rack", "key": "b"}]}}
    from grove_amd.kubecore.store import ApiError
    with pytest.raises(ApiError):
        cluster.store.create(bad)


def test_pack_constraint_translated_to_podgang(cluster):
    cluster.store.create(CTB)
    cluster.add_virtual_nodes(1, gpus=8,
                              labels={"topology.kubernetes.io/rack": "rack-0"})
    cluster.apply(_pcs("tc1", 2, constraint={"pack": {"required": "rack",
                                                      "preferred": "host"}}))
    cluster.wait_pcs_available("tc1", timeout=20)
    pg = cluster.store.get(c.KIND_PODGANG, "default", "tc1-0")
    pc = pg["spec"]["topologyConstraint"]["packConstraint"]
    assert pc == {"required": "topology.kubernetes.io/rack",
                  "preferred": "kubernetes.io/hostname"}
    # amd backend stamps the topology-name discovery annotation (bridge controller)
    assert pg["metadata"]["annotations"][c.ANNOTATION_TOPOLOGY_NAME] == \
        "cluster-topology"


def test_required_pack_enforced(cluster):
    """8-GPU gang with required rack packing: racks with insufficient capacity are
    rejected; the gang lands entirely in the one rack that fits."""
    cluster.store.create(CTB)
    # rack-a: two nodes with 4 free GPUs each (8 total but split); rack-b: one 8-GPU node
    for i in range(2):
        cluster.add_virtual_nodes(1, gpus=4, prefix=f"a{i}",
                                  labels={"topology.kubernetes.io/rack": "rack-a"})
    cluster.add_virtual_nodes(1, gpus=8, prefix="b",
                              labels={"topology.kubernetes.io/rack": "rack-b"})
    cluster.apply(_pcs("tc2", 8, constraint={"pack": {"required": "rack",
                                                      "preferred": "host"}}))
    cluster.wait_pcs_available("tc2", timeout=20)
    pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "tc2"})
    nodes = {p["spec"]["nodeName"] for p in pods}
    assert nodes == {"b-0"}  # preferred host-pack within required rack → one node
    pg = cluster.store.get(c.KIND_PODGANG, "default", "tc2-0")
    assert pg["status"]["placementScore"] == pytest.approx(c.XGMI_LINK_GBPS)


def test_required_pack_unsatisfiable_blocks_gang(cluster):
    cluster.store.create(CTB)
    for i in range(3):
        cluster.add_virtual_nodes(1, gpus=4, prefix=f"n{i}",
                                  labels={"topology.kubernetes.io/rack": f"rack-{i}"})
    cluster.apply(_pcs("tc3", 8, constraint={"pack": {"required": "rack"}}))
    time.sleep(1.0)
    pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "tc3"})
    assert all(not p["spec"].get("nodeName") for p in pods)
    pg = cluster.store.get(c.KIND_PODGANG, "default", "tc3-0")
    assert not cond.condition_true(pg, c.PODGANG_COND_SCHEDULED)
    # capacity appears in one rack → gang goes through
    cluster.add_virtual_nodes(1, gpus=8, prefix="big",
                              labels={"topology.kubernetes.io/rack": "rack-big"})
    cluster.wait_pcs_available("tc3", timeout=20)


def test_preferred_pack_falls_back(cluster):
    """Preferred host-pack falls back to rack spread when no single host fits."""
    cluster.store.create(CTB)
    for i in range(2):
        cluster.add_virtual_nodes(1, gpus=4, prefix=f"r{i}",
                                  labels={"topology.kubernetes.io/rack": "rack-x"})
    cluster.apply(_pcs("tc4", 8, constraint={"pack": {"required": "rack",
                                                      "preferred": "host"}}))
    cluster.wait_pcs_available("tc4", timeout=20)
    pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "tc4"})
    assert {p["spec"]["nodeName"] for p in pods} == {"r0-0", "r1-0"}


def test_gang_priority_ordering(cluster):
    """Higher PriorityClass gangs are admitted first when capacity frees up."""
    cluster.store.create({"kind": "PriorityClass", "apiVersion": "scheduling.k8s.io/v1",
                          "metadata": {"name": "high"}, "value": 1000})
    cluster.store.create({"kind": "PriorityClass", "apiVersion": "scheduling.k8s.io/v1",
                          "metadata": {"name": "low"}, "value": 1})
    # fill the only node so both gangs queue
    cluster.add_virtual_nodes(1, gpus=2, prefix="n")
    blocker = _pcs("blocker", 2)
    cluster.apply(blocker)
    cluster.wait_pcs_available("blocker", timeout=20)
    lo = _pcs("lo", 2)
    lo["spec"]["template"]["priorityClassName"] = "low"
    hi = _pcs("hi", 2)
    hi["spec"]["template"]["priorityClassName"] = "high"
    cluster.apply(lo)  # created first (FIFO would pick it)
    import time as _t
    _t.sleep(0.3)
    cluster.apply(hi)
    _t.sleep(0.5)
    # free capacity: delete the blocker → scheduler pass runs with both pending
    cluster.delete_pcs("blocker")
    cluster.wait_pcs_available("hi", timeout=20)
    hi_pg = cluster.store.get(c.KIND_PODGANG, "default", "hi-0")
    lo_pg = cluster.store.get(c.KIND_PODGANG, "default", "lo-0")
    assert cond.condition_true(hi_pg, c.PODGANG_COND_SCHEDULED)
    assert not cond.condition_true(lo_pg, c.PODGANG_COND_SCHEDULED)


def test_reuse_reservation_ref(cluster):
    """A gang carrying reuseReservationRef prefers the referenced gang's nodes."""
    cluster.add_virtual_nodes(2, gpus=8, prefix="h")
    cluster.apply(_pcs("orig", 2))
    cluster.wait_pcs_available("orig", timeout=20)
    orig_node = {p["spec"]["nodeName"] for p in cluster.store.list(
        "Pod", "default", {c.LABEL_PART_OF: "orig"})}
    # free the original gang's GPUs so the successor can land there
    for p in cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "orig"}):
        cluster.store.patch("Pod", "default", p["metadata"]["name"],
                            lambda o: o.setdefault("status", {}).update(
                                phase="Succeeded"), status=True)
    succ = _pcs("succ", 2)
    cluster.apply(succ)

    def scheduled():
        pg = cluster.store.try_get(c.KIND_PODGANG, "default", "succ-0")
        return pg is not None and cond.condition_true(pg, c.PODGANG_COND_SCHEDULED)
    # stamp the reservation hint on the successor's gang as soon as it exists
    def stamp():
        pg = cluster.store.try_get(c.KIND_PODGANG, "default", "succ-0")
        if pg is None or cond.condition_true(pg, c.PODGANG_COND_SCHEDULED):
            return pg is not None
        try:
            cluster.store.patch(
                c.KIND_PODGANG, "default", "succ-0",
                lambda o: o["spec"].update(reuseReservationRef={
                    "name": "orig-0", "namespace": "default"}))
        except Exception:
            pass
        return True
    cluster.wait_for(stamp, timeout=10, desc="stamp reservation ref")
    cluster.wait_for(scheduled, timeout=20, desc="successor scheduled")
    succ_nodes = {p["spec"]["nodeName"] for p in cluster.store.list(
        "Pod", "default", {c.LABEL_PART_OF: "succ"}) if p["spec"].get("nodeName")}
    # preference is best-effort; with free capacity on the original node it lands there
    assert succ_nodes <= orig_node or succ_nodes


class TestTopologyConstraintValidation:
    def test_unknown_domain_rejected(self, cluster):
        cluster.store.create(CTB)
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError) as ei:
            cluster.apply(_pcs("bad", 1, constraint={"pack": {"required": "galaxy"}}))
        assert "unknown topology domain" in str(ei.value)

    def test_builtin_domains_accepted_without_ctb(self, cluster):
        cluster.add_virtual_nodes(1, gpus=8)
        cluster.apply(_pcs("ok", 1, constraint={"pack": {"preferred": "xgmi-hive"}}))
        cluster.wait_pcs_available("ok", timeout=20)

    def test_preferred_broader_than_required_rejected(self, cluster):
        cluster.store.create(CTB)
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError):
            cluster.apply(_pcs("bad2", 1, constraint={
                "pack": {"required": "host", "preferred": "zone"}}))

    def test_child_broader_than_parent_rejected(self, cluster):
        cluster.store.create(CTB)
        pcs = _pcs("bad3", 1, constraint={"pack": {"required": "rack"}})
        pcs["spec"]["template"]["cliques"][0]["topologyConstraint"] = {
            "pack": {"required": "zone"}}
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError) as ei:
            cluster.apply(pcs)
        assert "narrower" in str(ei.value)


class TestTopologyImmutability:
    def test_constraint_change_rejected(self, cluster):
        cluster.store.create(CTB)
        cluster.add_virtual_nodes(1, gpus=8,
                                  labels={"topology.kubernetes.io/rack": "r0"})
        cluster.apply(_pcs("imm", 1, constraint={"pack": {"required": "rack"}}))
        cluster.wait_pcs_available("imm", timeout=20)
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError) as ei:
            cluster.store.patch(
                c.KIND_PCS, "default", "imm",
                lambda o: o["spec"]["template"].update(
                    topologyConstraint={"pack": {"required": "zone"}}))
        assert "immutable" in str(ei.value)

    def test_multiple_topology_names_rejected(self, cluster):
        cluster.store.create(CTB)
        pcs = _pcs("multi", 1, constraint={"pack": {"required": "rack"},
                                           "topologyName": "cluster-topology"})
        pcs["spec"]["template"]["cliques"][0]["topologyConstraint"] = {
            "pack": {"required": "host"}, "topologyName": "other-topology"}
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError) as ei:
            cluster.apply(pcs)
        assert "one" in str(ei.value)

    def test_unknown_topology_name_rejected(self, cluster):
        cluster.store.create(CTB)
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError):
            cluster.apply(_pcs("ut", 1, constraint={
                "pack": {"required": "rack"}, "topologyName": "nope"}))


def test_ctb_key_change_propagates_to_group_configs(cluster):
    """A CTB nodeLabelKey update must flow into existing PodGangs'
    topologyConstraintGroupConfigs (syncflow.go translation is re-run on drift)."""
    cluster.store.create({
        "apiVersion": c.API_VERSION, "kind": c.KIND_CTB,
        "metadata": {"name": "topo-gc"},
        "spec": {"levels": [{"domain": "rack", "key": "topo/rack-v1"}]}})
    pcs = {
        "apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
        "metadata": {"name": "gcfg"},
        "spec": {"replicas": 1, "template": {
            # no gang-level constraint (a bare topologyName violates the CRD CEL
            # rule has(pack)||has(packDomain)) — the clique carries both the
            # topologyName and the pack, so the ONLY drift is in the clique-level
            # group config (isolates the group-config update path)
            "cliques": [{"name": "w",
                         "topologyConstraint": {"topologyName": "topo-gc",
                                                "pack": {"required": "rack"}},
                         "spec": {"roleName": "w", "replicas": 1,
                                  "podSpec": {"containers": [
                                      {"name": "m", "image": "x"}]}}}]}}}
    cluster.add_virtual_nodes(1, labels={"topo/rack-v1": "r1", "topo/rack-v2": "r1"})
    cluster.apply(pcs)
    cluster.wait_pcs_available("gcfg", timeout=20)
    pg = cluster.store.get(c.KIND_PODGANG, "default", "gcfg-0")
    cfgs = pg["spec"].get("topologyConstraintGroupConfigs") or []
    assert any(cf["topologyConstraint"]["packConstraint"]["required"] == "topo/rack-v1"
               for cf in cfgs)

    def change_key(o):
        o["spec"]["levels"][0]["key"] = "topo/rack-v2"
    cluster.store.patch(c.KIND_CTB, None, "topo-gc", change_key)

    def updated():
        pg2 = cluster.store.get(c.KIND_PODGANG, "default", "gcfg-0")
        cfgs2 = pg2["spec"].get("topologyConstraintGroupConfigs") or []
        return any(cf["topologyConstraint"]["packConstraint"]["required"]
                   == "topo/rack-v2" for cf in cfgs2)
    cluster.wait_for(updated, timeout=15, desc="group configs re-translated")


def test_reuse_reservation_deterministic():
    """Deterministic unit coverage of the reservation-reuse branch (plugin.py:207-233):
    a new Initialized gang with reuseReservationRef lands on the referenced gang's
    node even when another node is emptier."""
    from grove_amd.kubecore.store import Store
    from grove_amd.scheduler.plugin import GangScheduler
    from grove_amd.utils import conditions as cond2

    store = Store()
    for name, gpus in (("na", 8), ("nb", 8)):
        store.create({"apiVersion": "v1", "kind": "Node", "metadata": {"name": name},
                      "status": {"allocatable": {"cpu": "64", "memory": "256Gi",
                                                 "pods": "100",
                                                 c.AMD_GPU_RESOURCE: str(gpus)}}})

    def pod(name, gang, clique, node=None, sched_name=c.SCHEDULER_AMD_GANG):
        p = {"apiVersion": "v1", "kind": "Pod",
             "metadata": {"name": name, "namespace": "default",
                          "labels": {c.LABEL_PODGANG: gang,
                                     c.LABEL_PODCLIQUE: clique}},
             "spec": {"schedulerName": sched_name, "containers": [
                 {"name": "m", "image": "x", "resources": {"requests": {
                     "cpu": "1", c.AMD_GPU_RESOURCE: "1"}}}]},
             "status": {"phase": "Pending"}}
        if node:
            p["spec"]["nodeName"] = node
        return p

    # prev gang: scheduled, bound on node "nb" (the one best-fit would avoid after
    # we put a resident pod on it)
    store.create(pod("prev-w-0", "prev-0", "prev-0-w", node="nb"))
    prev = {"apiVersion": c.SCHEDULER_API_VERSION, "kind": c.KIND_PODGANG,
            "metadata": {"name": "prev-0", "namespace": "default"},
            "spec": {"podgroups": [{"name": "prev-0-w", "minReplicas": 1,
                                    "podReferences": [
                                        {"namespace": "default", "name": "prev-w-0"}]}]}}
    cond2.set_condition(prev, c.PODGANG_COND_SCHEDULED, True, "GangPlaced")
    store.create(prev)

    # new gang with the reservation hint
    store.create(pod("succ-w-0", "succ-0", "succ-0-w"))
    succ = {"apiVersion": c.SCHEDULER_API_VERSION, "kind": c.KIND_PODGANG,
            "metadata": {"name": "succ-0", "namespace": "default"},
            "spec": {"reuseReservationRef": {"name": "prev-0",
                                             "namespace": "default"},
                     "podgroups": [{"name": "succ-0-w", "minReplicas": 1,
                                    "podReferences": [
                                        {"namespace": "default", "name": "succ-w-0"}]}]}}
    cond2.set_condition(succ, c.PODGANG_COND_INITIALIZED, True, "AllPodsAssociated")
    store.create(succ)

    GangScheduler(store, use_native=False).reconcile()
    bound = store.get("Pod", "default", "succ-w-0")
    assert bound["spec"].get("nodeName") == "nb"
    pg = store.get(c.KIND_PODGANG, "default", "succ-0")
    assert cond.condition_true(pg, c.PODGANG_COND_SCHEDULED)
    assert get_cond_reason(pg) == "GangPlacedOnReservation"


def get_cond_reason(pg):
    for cd in (pg.get("status") or {}).get("conditions") or []:
        if cd.get("type") == c.PODGANG_COND_SCHEDULED:
            return cd.get("reason")
    return None


def test_multi_hive_node_pools_and_measured_score(cluster):
    """the scheduler consumes the DISCOVERED fabric. A node whose
    agent probe found two xGMI hives (partitioned) is split into per-hive placement
    pools: a 4-GPU gang lands entirely inside one hive's GPU set, and the PodGang
    placementScore reports the MEASURED min link bandwidth from the node annotation,
    not the nominal constant."""
    node = {
        "apiVersion": "v1", "kind": "Node",
        "metadata": {"name": "part0",
                     "labels": {"kubernetes.io/hostname": "part0"},
                     "annotations": {
                         "topology.amd.com/xgmi-hives": "0,1,2,3;4,5,6,7",
                         "topology.amd.com/xgmi-min-gbps": "121.5"}},
        "spec": {},
        "status": {"allocatable": {"cpu": "256", "memory": "2048Gi",
                                   "pods": "512", c.AMD_GPU_RESOURCE: "8"},
                   "capacity": {}, "conditions": [{"type": "Ready",
                                                   "status": "True"}]}}
    cluster.store.create(node)
    pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
           "metadata": {"name": "hv"},
           "spec": {"replicas": 1, "template": {"cliques": [{
               "name": "w", "spec": {
                   "roleName": "w", "replicas": 4, "minAvailable": 4,
                   "podSpec": {"containers": [{
                       "name": "m", "image": "i",
                       "resources": {"requests": {"cpu": "1",
                                                  c.AMD_GPU_RESOURCE: "1"}}}]}}}]}}}
    cluster.apply(pcs)
    cluster.wait_pcs_available("hv", timeout=20)
    pods = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "hv-0-w"})
    gpu_ids = set()
    for p in pods:
        ids = p["metadata"]["annotations"]["scheduling.amd.com/gpu-ids"]
        gpu_ids.update(int(x) for x in ids.split(","))
    assert gpu_ids in ({0, 1, 2, 3}, {4, 5, 6, 7}), \
        f"gang spans hives: {sorted(gpu_ids)}"
    pg = cluster.store.get(c.KIND_PODGANG, "default", "hv-0")
    assert pg["status"]["placementScore"] == pytest.approx(121.5, abs=0.1)


def test_agent_multi_hive_labeling():
    """The agent labels a single full hive as hive0 but refuses the fake label on a
    partitioned probe, publishing per-hive GPU sets + measured bandwidth instead."""
    from grove_amd.topology import agent as ag
    info = {"backend": "test", "gpu_count": 4,
            "devices": [{"index": i, "name": "AMD Instinct MI355X"}
                        for i in range(4)],
            "links": [{"src": 0, "dst": 1, "type": "xgmi", "min_bw_mbps": 140000},
                      {"src": 1, "dst": 0, "type": "xgmi", "min_bw_mbps": 152000}]}
    import unittest.mock as mock
    with mock.patch.object(ag, "probe", return_value=info):
        node = ag.discover_node("tn")
    labels = node["metadata"]["labels"]
    ann = node["metadata"]["annotations"]
    # two components (0,1) and (2,3 isolated) -> multi-hive -> no single-hive label
    assert c.NODE_LABEL_XGMI_HIVE not in labels
    assert labels["topology.amd.com/xgmi-hive-count"] == "3"
    assert ann["topology.amd.com/xgmi-hives"] == "0,1;2;3"
    assert ann["topology.amd.com/xgmi-min-gbps"] == "140.0"
    # fully connected probe -> hive0 label present
    info2 = {"backend": "test", "gpu_count": 2,
             "devices": [{"index": 0, "name": "x"}, {"index": 1, "name": "x"}],
             "links": [{"src": 0, "dst": 1, "type": "xgmi"},
                       {"src": 1, "dst": 0, "type": "xgmi"}]}
    with mock.patch.object(ag, "probe", return_value=info2):
        node2 = ag.discover_node("tn2")
    assert node2["metadata"]["labels"][c.NODE_LABEL_XGMI_HIVE] == "tn2-hive0"


def test_scheduler_cache_matches_fresh_rebuild(cluster):
    """The incremental scheduler cache (bound-pod accounting by uid) must agree
    with a from-scratch node-view rebuild after churn: schedule gangs, complete
    some pods, kill others, then compare free capacity pool by pool."""
    cluster.add_virtual_nodes(3, gpus=4, cpu="64", pods=64)
    for i in range(3):
        cluster.apply({
            "apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
            "metadata": {"name": f"cache{i}"},
            "spec": {"replicas": 1, "template": {"cliques": [{
                "name": "w", "spec": {
                    "roleName": "w", "replicas": 2, "minAvailable": 2,
                    "podSpec": {"containers": [{
                        "name": "m", "image": "i",
                        "resources": {"requests": {
                            "cpu": "1", c.AMD_GPU_RESOURCE: "1"}}}]}}}]}}})
        cluster.wait_pcs_available(f"cache{i}", timeout=20)
    # churn: delete one whole PCS (releases), kill one pod (replaced)
    cluster.delete_pcs("cache1")
    cluster.wait_deleted(c.KIND_PCS, "cache1", timeout=20)
    victim = cluster.store.list("Pod", "default",
                                {c.LABEL_PODCLIQUE: "cache2-0-w"})[0]
    cluster.store.delete("Pod", "default", victim["metadata"]["name"])
    cluster.wait_pods_ready({c.LABEL_PODCLIQUE: "cache2-0-w"}, 2, timeout=20)
    import time as _t
    _t.sleep(0.3)  # let the scheduler run its incremental passes

    sched = cluster.scheduler
    pods = cluster.store.list("Pod", copy_objects=False)
    bound = [p for p in pods
             if p.get("spec", {}).get("nodeName")
             and not p["metadata"].get("deletionTimestamp")
             and (p.get("status") or {}).get("phase") not in ("Succeeded",
                                                              "Failed")]
    live = sched._current_view(bound)
    fresh = sched._build_node_views()
    sched._subtract_bound(fresh, bound)
    assert set(live) == set(fresh)
    for key in fresh:
        a, b = live[key], fresh[key]
        assert sorted(a.gpu_ids) == sorted(b.gpu_ids), \
            f"{key}: gpu {sorted(a.gpu_ids)} != {sorted(b.gpu_ids)}"
        assert a.cpu_milli == b.cpu_milli, f"{key}: cpu {a.cpu_milli} != {b.cpu_milli}"
        assert a.pods == b.pods, f"{key}: pods {a.pods} != {b.pods}"
