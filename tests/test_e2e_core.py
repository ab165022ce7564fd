"""End-to-end control-plane tests on the in-process cluster (CPU-only, virtual nodes).

Mirrors the reference e2e suites (gang scheduling GS*, startup ordering SO*,
gang termination GT* — operator/e2e/tests/) against the in-process cluster.
"""
import time

import pytest

from grove_amd.api import constants as c
from grove_amd.utils import conditions as cond


def _gpu_pcs(name="g1", replicas=1, cliques=(("ldr", 1, 1), ("wkr", 3, 3)),
             sg=None, gpus_per_pod=1, startup=None):
    cl = []
    for cname, reps, minav in cliques:
        cl.append({"name": cname, "spec": {
            "roleName": cname, "replicas": reps, "minAvailable": minav,
            "podSpec": {"containers": [{"name": "main", "image": "dummy",
                                        "resources": {"requests": {
                                            "cpu": "1",
                                            c.AMD_GPU_RESOURCE: str(gpus_per_pod)}}}]}}})
    tmpl = {"cliques": cl}
    if sg:
        tmpl["podCliqueScalingGroups"] = sg
    if startup:
        tmpl["cliqueStartupType"] = startup
    return {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
            "metadata": {"name": name},
            "spec": {"replicas": replicas, "template": tmpl}}


def test_simple1_reaches_available(cluster, simple1_yaml):
    cluster.add_virtual_nodes(4)
    cluster.apply(simple1_yaml)
    pcs = cluster.wait_pcs_available("simple1", timeout=20)
    st = pcs["status"]
    assert st["availableReplicas"] == 1
    assert st["podGangStatuses"] == [{"name": "simple1-0", "phase": "Running"}]
    # ownership chain objects exist
    s = cluster.store
    assert s.try_get("Service", "default", "simple1-0") is not None
    assert s.try_get("ServiceAccount", "default", "simple1") is not None
    assert s.try_get("Secret", "default", "simple1-ic-sat") is not None
    assert s.try_get(c.KIND_PCSG, "default", "simple1-0-sga") is not None
    # HPAs: one for pca (clique autoscaling), one for the scaling group
    hpas = {h["metadata"]["name"] for h in s.list("HorizontalPodAutoscaler")}
    assert hpas == {"simple1-0-pca", "simple1-0-sga"}
    # pods: 3 pca + 2 pcb + 2 pcc + 2 pcd
    pods = s.list("Pod", "default", {c.LABEL_PART_OF: "simple1"})
    assert len(pods) == 9
    assert all(p["spec"].get("nodeName") for p in pods)
    assert all(not p["spec"].get("schedulingGates") for p in pods)


def test_pod_contract(cluster, simple1_yaml):
    cluster.add_virtual_nodes(2)
    cluster.apply(simple1_yaml)
    cluster.wait_pcs_available("simple1", timeout=20)
    pods = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "simple1-0-pca"})
    assert len(pods) == 3
    idx = sorted(int(p["metadata"]["labels"][c.LABEL_POD_INDEX]) for p in pods)
    assert idx == [0, 1, 2]
    p = pods[0]
    i = p["metadata"]["labels"][c.LABEL_POD_INDEX]
    assert p["spec"]["hostname"] == f"simple1-0-pca-{i}"
    assert p["spec"]["subdomain"] == "simple1-0"
    assert p["metadata"]["labels"][c.LABEL_PODGANG] == "simple1-0"
    env = {e["name"]: e["value"] for e in p["spec"]["containers"][0]["env"]}
    assert env[c.ENV_PCS_NAME] == "simple1"
    assert env[c.ENV_PCS_INDEX] == "0"
    assert env[c.ENV_PCLQ_NAME] == "simple1-0-pca"
    assert env[c.ENV_HEADLESS_SERVICE] == "simple1-0.default.svc.cluster.local"
    assert env[c.ENV_POD_INDEX] == i
    # PCSG member pods carry the PCSG env vars
    pods_b = cluster.store.list("Pod", "default",
                                {c.LABEL_PODCLIQUE: "simple1-0-sga-0-pcb"})
    envb = {e["name"]: e["value"] for e in pods_b[0]["spec"]["containers"][0]["env"]}
    assert envb[c.ENV_PCSG_NAME] == "simple1-0-sga"
    assert envb[c.ENV_PCSG_TEMPLATE_NUM_PODS] == "4"  # pcb(2)+pcc(2)


def test_gang_all_or_nothing(cluster):
    """GS parity: a gang that cannot fully fit binds no pods at all."""
    cluster.add_virtual_nodes(1, gpus=8)
    cluster.apply(_gpu_pcs("big", cliques=(("w", 12, 12),)))  # 12 GPUs > 8
    time.sleep(1.0)
    pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "big"})
    assert len(pods) == 12
    assert all(not p["spec"].get("nodeName") for p in pods)
    pg = cluster.store.get(c.KIND_PODGANG, "default", "big-0")
    assert not cond.condition_true(pg, c.PODGANG_COND_SCHEDULED)


def test_gang_fits_after_capacity_added(cluster):
    cluster.apply(_gpu_pcs("g8", cliques=(("w", 8, 8),)))
    time.sleep(0.5)
    assert all(not p["spec"].get("nodeName")
               for p in cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "g8"}))
    cluster.add_virtual_nodes(1, gpus=8, prefix="gpu")
    pcs = cluster.wait_pcs_available("g8", timeout=20)
    assert pcs["status"]["availableReplicas"] == 1
    pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "g8"})
    # all 8 pods on the single 8-GPU node, distinct GPU ids
    assert {p["spec"]["nodeName"] for p in pods} == {"gpu-0"}
    ids = sorted(int((p["metadata"].get("annotations") or {})["scheduling.amd.com/gpu-ids"])
                 for p in pods)
    assert ids == list(range(8))


def test_xgmi_packing_prefers_single_hive(cluster):
    """Score parity with BASELINE north star: an 8-GPU gang lands in ONE hive even when
    split placements exist."""
    cluster.add_virtual_nodes(2, gpus=8, prefix="hive")
    # pre-load hive-0 with a 4-GPU single pod so only hive-1 can take a full 8-gang
    cluster.apply(_gpu_pcs("small", cliques=(("s", 1, 1),), gpus_per_pod=4))
    cluster.wait_pcs_available("small", timeout=10)
    cluster.apply(_gpu_pcs("inst", cliques=(("w", 8, 8),)))
    cluster.wait_pcs_available("inst", timeout=20)
    pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "inst"})
    nodes = {p["spec"]["nodeName"] for p in pods}
    assert len(nodes) == 1
    pg = cluster.store.get(c.KIND_PODGANG, "default", "inst-0")
    assert pg["status"]["placementScore"] == pytest.approx(c.XGMI_LINK_GBPS, abs=1.0)


def test_startup_ordering_explicit(cluster):
    """SO parity: pod of a dependent clique becomes Ready only after its parent clique
    has minAvailable Ready pods."""
    pcs = _gpu_pcs("so", cliques=(("a", 2, 2), ("b", 2, 2)), gpus_per_pod=0,
                   startup=c.STARTUP_EXPLICIT)
    pcs["spec"]["template"]["cliques"][1]["spec"]["startsAfter"] = ["a"]
    cluster.add_virtual_nodes(2)
    cluster.apply(pcs)
    cluster.wait_pcs_available("so", timeout=20)
    pclq_b = cluster.store.get(c.KIND_PCLQ, "default", "so-0-b")
    assert pclq_b["spec"]["startsAfter"] == ["so-0-a"]
    # dependent pods carry the visible grove-initc init container (pod.go:315-371)
    b_pods = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "so-0-b"})
    ic = b_pods[0]["spec"]["initContainers"][0]
    assert ic["name"] == "grove-initc"
    assert "--podcliques=so-0-a:2" in ic["args"]
    assert any(v.get("secret", {}).get("secretName") == "so-ic-sat"
               for v in b_pods[0]["spec"]["volumes"])
    assert any(e["name"] == c.ENV_PCLQ_NAME for e in ic["env"])
    # parent-clique pods have no init container injected
    a_pods = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "so-0-a"})
    assert not any(x.get("name") == "grove-initc"
                   for x in a_pods[0]["spec"].get("initContainers", []))


def test_in_order_startup_dependencies(cluster):
    pcs = _gpu_pcs("io", cliques=(("a", 1, 1), ("b", 1, 1)), gpus_per_pod=0,
                   startup=c.STARTUP_IN_ORDER)
    cluster.add_virtual_nodes(1)
    cluster.apply(pcs)
    cluster.wait_pcs_available("io", timeout=20)
    assert cluster.store.get(c.KIND_PCLQ, "default", "io-0-a")["spec"]["startsAfter"] == []
    assert cluster.store.get(c.KIND_PCLQ, "default", "io-0-b")["spec"]["startsAfter"] == \
        ["io-0-a"]


def test_scaled_podgangs_created_beyond_minavailable(cluster):
    sg = [{"name": "sg", "cliqueNames": ["w"], "replicas": 3, "minAvailable": 1}]
    pcs = _gpu_pcs("sc", cliques=(("w", 2, 2),), sg=sg, gpus_per_pod=0)
    cluster.add_virtual_nodes(2)
    cluster.apply(pcs)
    cluster.wait_pcs_available("sc", timeout=20)
    gangs = sorted(g["metadata"]["name"]
                   for g in cluster.store.list(c.KIND_PODGANG, "default",
                                               {c.LABEL_PART_OF: "sc"}))
    assert gangs == ["sc-0", "sc-0-sg-0", "sc-0-sg-1"]
    scaled = cluster.store.get(c.KIND_PODGANG, "default", "sc-0-sg-0")
    assert scaled["metadata"]["labels"][c.LABEL_BASE_PODGANG] == "sc-0"
    # member PCLQs of scaled replicas carry the base-podgang label
    q = cluster.store.get(c.KIND_PCLQ, "default", "sc-0-sg-1-w")
    assert q["metadata"]["labels"][c.LABEL_BASE_PODGANG] == "sc-0"
    # all pods eventually scheduled + ready
    cluster.wait_pods_ready({c.LABEL_PART_OF: "sc"}, 6, timeout=20)


def test_pcs_delete_cleans_up(cluster, simple1_yaml):
    cluster.add_virtual_nodes(2)
    cluster.apply(simple1_yaml)
    cluster.wait_pcs_available("simple1", timeout=20)
    cluster.delete_pcs("simple1")
    cluster.wait_deleted(c.KIND_PCS, "simple1", timeout=20)
    for kind in (c.KIND_PCLQ, c.KIND_PCSG, c.KIND_PODGANG, "Pod"):
        leftovers = cluster.store.list(kind, "default", {c.LABEL_PART_OF: "simple1"})
        assert leftovers == [], f"leftover {kind}"


def test_pod_replacement_after_failure(cluster):
    """A killed pod is recreated and rescheduled (same clique, hole-filled index)."""
    cluster.add_virtual_nodes(1, gpus=8, prefix="gpu")
    cluster.apply(_gpu_pcs("rep", cliques=(("w", 4, 2),)))
    cluster.wait_pcs_available("rep", timeout=20)
    victim = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "rep-0-w"})[0]
    cluster.store.delete("Pod", "default", victim["metadata"]["name"])
    cluster.c_pclq.enqueue("default", "rep-0-w")

    def healed():
        pods = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "rep-0-w"})
        from grove_amd.utils import conditions as cc
        return len(pods) == 4 and sum(1 for p in pods if cc.pod_is_ready(p)) == 4
    cluster.wait_for(healed, timeout=20, desc="pod replacement")
    idx = sorted(int(p["metadata"]["labels"][c.LABEL_POD_INDEX])
                 for p in cluster.store.list("Pod", "default",
                                             {c.LABEL_PODCLIQUE: "rep-0-w"}))
    assert idx == [0, 1, 2, 3]


def test_pcsg_scale_out_creates_scaled_gang(cluster):
    sg = [{"name": "sg", "cliqueNames": ["w"], "replicas": 1, "minAvailable": 1}]
    pcs = _gpu_pcs("hs", cliques=(("w", 1, 1),), sg=sg, gpus_per_pod=0)
    cluster.add_virtual_nodes(1)
    cluster.apply(pcs)
    cluster.wait_pcs_available("hs", timeout=20)
    # scale the PCSG (what the HPA would do via /scale)
    cluster.store.patch(c.KIND_PCSG, "default", "hs-0-sg",
                        lambda o: o["spec"].update(replicas=3))
    cluster.c_pcsg.enqueue("default", "hs-0-sg")

    def scaled():
        gangs = {g["metadata"]["name"]
                 for g in cluster.store.list(c.KIND_PODGANG, "default",
                                             {c.LABEL_PART_OF: "hs"})}
        return gangs == {"hs-0", "hs-0-sg-0", "hs-0-sg-1"}
    cluster.wait_for(scaled, timeout=20, desc="scaled podgangs")
    cluster.wait_pods_ready({c.LABEL_PART_OF: "hs"}, 3, timeout=20)
    # scale back in: scaled gangs + pods go away
    cluster.store.patch(c.KIND_PCSG, "default", "hs-0-sg",
                        lambda o: o["spec"].update(replicas=1))
    cluster.c_pcsg.enqueue("default", "hs-0-sg")

    def shrunk():
        gangs = {g["metadata"]["name"]
                 for g in cluster.store.list(c.KIND_PODGANG, "default",
                                             {c.LABEL_PART_OF: "hs"})}
        pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "hs"})
        return gangs == {"hs-0"} and len(pods) == 1
    cluster.wait_for(shrunk, timeout=20, desc="scale-in cleanup")


def test_non_default_namespace(cluster):
    """Workloads in a custom namespace: full chain works and stays isolated."""
    cluster.add_virtual_nodes(1)
    pcs = _gpu_pcs("nsx", cliques=(("w", 2, 2),), gpus_per_pod=0)
    pcs["metadata"]["namespace"] = "team-a"
    cluster.store.create(pcs)
    cluster.wait_for(
        lambda: int((cluster.store.try_get(c.KIND_PCS, "team-a", "nsx") or {})
                    .get("status", {}).get("availableReplicas", 0)) >= 1,
        timeout=20, desc="team-a PCS available")
    assert cluster.store.list("Pod", "team-a", {c.LABEL_PART_OF: "nsx"})
    assert not cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "nsx"})
    pg = cluster.store.get(c.KIND_PODGANG, "team-a", "nsx-0")
    assert pg["spec"]["podgroups"][0]["podReferences"][0]["namespace"] == "team-a"


def test_zero_replica_pcs(cluster):
    """kubebuilder default spec.replicas=0: valid, creates RBAC but no gangs/pods."""
    cluster.add_virtual_nodes(1)
    pcs = _gpu_pcs("zr", cliques=(("w", 1, 1),), gpus_per_pod=0)
    del pcs["spec"]["replicas"]
    cluster.store.create(pcs)
    import time as _t
    _t.sleep(0.5)
    out = cluster.store.get(c.KIND_PCS, "default", "zr")
    assert out["spec"]["replicas"] == 0
    assert cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "zr"}) == []
    assert cluster.store.list(c.KIND_PODGANG, "default", {c.LABEL_PART_OF: "zr"}) == []
    assert cluster.store.try_get("ServiceAccount", "default", "zr") is not None
    assert int(out.get("status", {}).get("availableReplicas", 0)) == 0


def test_multi_gpu_pods_in_gang(cluster):
    """Gang of 2 pods x 4 GPUs each fills one hive; GPU ids partition exactly."""
    cluster.add_virtual_nodes(1, gpus=8, prefix="hive")
    cluster.apply(_gpu_pcs("mg", cliques=(("w", 2, 2),), gpus_per_pod=4))
    cluster.wait_pcs_available("mg", timeout=20)
    pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "mg"})
    ids = []
    for p in pods:
        ids.extend(int(x) for x in
                   p["metadata"]["annotations"]["scheduling.amd.com/gpu-ids"].split(","))
        assert len(p["metadata"]["annotations"][
            "scheduling.amd.com/gpu-ids"].split(",")) == 4
    assert sorted(ids) == list(range(8))


def test_reapply_is_idempotent(cluster, simple1_yaml):
    cluster.add_virtual_nodes(2)
    cluster.apply(simple1_yaml)
    cluster.wait_pcs_available("simple1", timeout=20)
    pods_before = {p["metadata"]["name"]
                   for p in cluster.store.list("Pod", "default",
                                               {c.LABEL_PART_OF: "simple1"})}
    gen = cluster.store.get(c.KIND_PCS, "default", "simple1")["metadata"]["generation"]
    cluster.apply(simple1_yaml)  # identical re-apply
    import time as _t
    _t.sleep(0.8)
    after = cluster.store.get(c.KIND_PCS, "default", "simple1")
    assert after["metadata"]["generation"] == gen  # no spec change detected
    pods_after = {p["metadata"]["name"]
                  for p in cluster.store.list("Pod", "default",
                                              {c.LABEL_PART_OF: "simple1"})}
    assert pods_after == pods_before  # zero churn


def test_reconcile_trigger_annotation(cluster):
    """register.go:70-232 parity: stamping grove.io/reconcile-trigger on a PCS forces
    a reconcile even with no spec change (our watch enqueues all PCS events)."""
    pcs = _gpu_pcs("rt", cliques=(("a", 1, 1),), gpus_per_pod=0)
    cluster.add_virtual_nodes(1)
    cluster.apply(pcs)
    cluster.wait_pcs_available("rt", timeout=20)
    time.sleep(0.3)
    n0 = cluster.c_pcs.reconcile_count

    def bump(o):
        o["metadata"].setdefault("annotations", {})[
            c.ANNOTATION_RECONCILE_TRIGGER] = "manual-1"
    cluster.store.patch(c.KIND_PCS, "default", "rt", bump)
    cluster.wait_for(lambda: cluster.c_pcs.reconcile_count > n0, timeout=10,
                     desc="annotation-triggered reconcile")


def test_pcsg_startup_dependency_scoping(cluster):
    """ADVICE r1 parity fix (componentutils.GenerateDependencyNamesForBasePodGang):
    with Explicit startup and a scaling group, (a) a standalone clique depending on a
    PCSG member clique waits on ALL [0, minAvailable) PCSG replicas; (b) a scaled PCSG
    replica's member (j >= minAvailable) depends only on its OWN replica's siblings;
    (c) cross-gang deps from a scaled replica are dropped."""
    pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
           "metadata": {"name": "dep"},
           "spec": {"replicas": 1, "template": {
               "cliqueStartupType": c.STARTUP_EXPLICIT,
               "cliques": [
                   {"name": "ld", "spec": {"roleName": "ld", "replicas": 1,
                       "podSpec": {"containers": [{"name": "m", "image": "i"}]}}},
                   {"name": "wk", "spec": {"roleName": "wk", "replicas": 1,
                       "startsAfter": ["ld"],
                       "podSpec": {"containers": [{"name": "m", "image": "i"}]}}},
                   {"name": "tail", "spec": {"roleName": "t", "replicas": 1,
                       "startsAfter": ["wk"],
                       "podSpec": {"containers": [{"name": "m", "image": "i"}]}}},
               ],
               "podCliqueScalingGroups": [
                   {"name": "sg", "cliqueNames": ["ld", "wk"],
                    "replicas": 3, "minAvailable": 2}]}}}
    cluster.add_virtual_nodes(3)
    cluster.apply(pcs)
    cluster.wait_pcs_available("dep", timeout=30)
    # (a) standalone "tail" depends on wk of ALL base replicas [0, minAvailable)
    tail = cluster.store.get(c.KIND_PCLQ, "default", "dep-0-tail")
    assert sorted(tail["spec"]["startsAfter"]) == \
        ["dep-0-sg-0-wk", "dep-0-sg-1-wk"]
    # base member wk@j<minAvailable also waits on ld of all base replicas
    wk0 = cluster.store.get(c.KIND_PCLQ, "default", "dep-0-sg-0-wk")
    assert sorted(wk0["spec"]["startsAfter"]) == \
        ["dep-0-sg-0-ld", "dep-0-sg-1-ld"]
    # (b) scaled replica j=2 depends only on its own gang's ld
    wk2 = cluster.store.get(c.KIND_PCLQ, "default", "dep-0-sg-2-wk")
    assert wk2["spec"]["startsAfter"] == ["dep-0-sg-2-ld"]
    # (c) ld has no deps anywhere
    for j in range(3):
        ld = cluster.store.get(c.KIND_PCLQ, "default", f"dep-0-sg-{j}-ld")
        assert ld["spec"]["startsAfter"] == []


def test_child_drift_repaired_despite_sync_fingerprint(cluster):
    """The structural-sync fingerprint must never mask child drift: deleting a
    PodClique out from under the PCS (a structural change with no PCS spec
    change) gets repaired — the fingerprint includes child generations, so the
    vanished child forces a full resync."""
    cluster.add_virtual_nodes(1)
    pcs = _gpu_pcs("drift", cliques=(("a", 1, 1), ("b", 1, 1)), gpus_per_pod=0)
    cluster.apply(pcs)
    cluster.wait_pcs_available("drift", timeout=20)
    uid = cluster.store.get(c.KIND_PCLQ, "default", "drift-0-b")["metadata"]["uid"]
    cluster.store.delete(c.KIND_PCLQ, "default", "drift-0-b")

    def repaired():
        q = cluster.store.try_get(c.KIND_PCLQ, "default", "drift-0-b")
        return q is not None and q["metadata"]["uid"] != uid \
            and not q["metadata"].get("deletionTimestamp")
    cluster.wait_for(repaired, timeout=20, desc="deleted PodClique recreated")
    cluster.wait_pcs_available("drift", timeout=20)
