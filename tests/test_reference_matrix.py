"""Reference e2e matrix coverage: scenario tests mapped 1:1 to
the reference's e2e case IDs (GS*/SO*/GT*/OD*/RU*/TAS*; see docs/E2E_MATRIX.md for
the full test-ID → test-function mapping). Each docstring names the reference test
function it mirrors (behavior spec, fresh implementation on the in-process cluster
with virtual nodes standing in for KWOK)."""
import time

import pytest

from grove_amd.api import constants as c
from grove_amd.utils import conditions as cond


def _pcs(name, replicas=1, cliques=(("a", 1, 1),), sg=None, startup=None,
         termination_delay="1h", strategy=None, gpus=0):
    cl = []
    for (n, r, m) in cliques:
        spec = {"roleName": n, "replicas": r, "minAvailable": m,
                "podSpec": {"containers": [{
                    "name": "m", "image": "img:v1",
                    "resources": {"requests": ({"cpu": "1", c.AMD_GPU_RESOURCE:
                                                str(gpus)} if gpus else
                                               {"cpu": "1"})}}]}}
        cl.append({"name": n, "spec": spec})
    tmpl = {"cliques": cl, "terminationDelay": termination_delay}
    if sg:
        tmpl["podCliqueScalingGroups"] = sg
    if startup:
        tmpl["cliqueStartupType"] = startup
    spec = {"replicas": replicas, "template": tmpl}
    if strategy:
        spec["updateStrategy"] = {"type": strategy}
    return {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
            "metadata": {"name": name}, "spec": spec}


def _pods(cluster, selector, ns="default"):
    return cluster.store.list("Pod", ns, selector)


def _cordon_all(cluster, flag=True):
    for n in cluster.store.list("Node"):
        cluster.store.patch("Node", None, n["metadata"]["name"],
                            lambda o: o["spec"].update(unschedulable=flag))


# --------------------------------------------------------------------------- GS
def test_gs4_pcs_and_pcsg_scaling_full_replicas(cluster):
    """GS4 (gang_scheduling_test.go:217): PCS replicas scaled up AND the PCSG
    scaled up — every new replica tree gets its own gangs and reaches ready."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 1, "minAvailable": 1}]
    cluster.add_virtual_nodes(4, cpu="64", pods=64)
    cluster.apply(_pcs("gs4", cliques=(("a", 1, 1), ("b", 1, 1)), sg=sg))
    cluster.wait_pcs_available("gs4", timeout=20)
    # scale PCS 1 -> 2 and the replica-0 PCSG 1 -> 2
    cluster.store.patch(c.KIND_PCS, "default", "gs4",
                        lambda o: o["spec"].update(replicas=2))
    cluster.store.patch(c.KIND_PCSG, "default", "gs4-0-sg",
                        lambda o: o["spec"].update(replicas=2))
    cluster.wait_pcs_available("gs4", timeout=30)
    cluster.wait_for(
        lambda: len(_pods(cluster, {c.LABEL_PART_OF: "gs4"})) == 5,
        timeout=20, desc="2 replicas x (a + sg members) + scaled member")
    # scaled PCSG replica got its own scaled PodGang
    assert cluster.store.try_get(c.KIND_PODGANG, "default", "gs4-0-sg-0") \
        is not None
    assert cluster.store.try_get(c.KIND_PODGANG, "default", "gs4-1") is not None


def test_gs5_gang_schedules_at_min_replicas(cluster):
    """GS5 (gang_scheduling_test.go:283): a clique with minAvailable < replicas is
    gang-admitted once capacity fits the MINIMUM; the rest schedule as capacity
    arrives."""
    cluster.add_virtual_nodes(1, cpu="3", pods=64)  # fits exactly 3 x cpu:1 pods
    cluster.apply(_pcs("gs5", cliques=(("w", 10, 3),)))
    # min gang (3 pods) must schedule + become ready; 7 stay pending
    cluster.wait_pods_ready({c.LABEL_PODCLIQUE: "gs5-0-w"}, 3, timeout=20)
    pods = _pods(cluster, {c.LABEL_PODCLIQUE: "gs5-0-w"})
    assert sum(1 for p in pods if p["spec"].get("nodeName")) == 3
    assert len(pods) == 10
    # capacity arrives -> everyone lands
    cluster.add_virtual_nodes(7, cpu="1", pods=8, prefix="late")
    cluster.wait_pods_ready({c.LABEL_PODCLIQUE: "gs5-0-w"}, 10, timeout=20)


def test_gs6_pcsg_min_available_scaled_gangs(cluster):
    """GS6/GS7 (gang_scheduling_test.go:351,463): a PCSG with minAvailable=1,
    replicas=3 forms ONE base gang (replica 0) and TWO scaled gangs, each
    independently schedulable; scaled gangs wait for the base gang."""
    sg = [{"name": "sx", "cliqueNames": ["b", "c"], "replicas": 3,
           "minAvailable": 1}]
    cluster.add_virtual_nodes(8, cpu="8", pods=64)
    cluster.apply(_pcs("gs6", cliques=(("a", 1, 1), ("b", 1, 1), ("c", 1, 1)),
                       sg=sg))
    cluster.wait_pcs_available("gs6", timeout=30)
    gangs = {g["metadata"]["name"]
             for g in cluster.store.list(c.KIND_PODGANG, "default",
                                         {c.LABEL_PART_OF: "gs6"})}
    assert gangs == {"gs6-0", "gs6-0-sx-0", "gs6-0-sx-1"}
    for sgname in ("gs6-0-sx-0", "gs6-0-sx-1"):
        pg = cluster.store.get(c.KIND_PODGANG, "default", sgname)
        assert pg["metadata"]["labels"][c.LABEL_BASE_PODGANG] == "gs6-0"
        # PCS availability only needs PCSG minAvailable=1, so the scaled gangs
        # may still be mid-bind here — wait for their Scheduled stamp
        cluster.wait_for(
            lambda n=sgname: cond.condition_true(
                cluster.store.get(c.KIND_PODGANG, "default", n),
                c.PODGANG_COND_SCHEDULED),
            timeout=15, desc=f"{sgname} Scheduled")
    cluster.wait_pods_ready({c.LABEL_PART_OF: "gs6"}, 7, timeout=20)


def test_gs9_pcs_scaling_with_min_replicas(cluster):
    """GS9 (gang_scheduling_test.go:679): scaling the PCS adds whole replica trees
    whose gangs admit at clique minAvailable while capacity is scarce."""
    cluster.add_virtual_nodes(1, cpu="2", pods=64)
    cluster.apply(_pcs("gs9", replicas=1, cliques=(("w", 2, 2),)))
    cluster.wait_pcs_available("gs9", timeout=20)
    cluster.store.patch(c.KIND_PCS, "default", "gs9",
                        lambda o: o["spec"].update(replicas=2))
    # replica 1's gang cannot fit -> its pods stay unscheduled, replica 0 untouched
    cluster.wait_for(
        lambda: len(_pods(cluster, {c.LABEL_PART_OF: "gs9"})) == 4,
        timeout=20, desc="4 pods exist")
    time.sleep(0.3)
    r1 = _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "1"})
    assert all(not p["spec"].get("nodeName") for p in r1), \
        "partial gang must not schedule"
    cluster.add_virtual_nodes(1, cpu="2", pods=64, prefix="extra")
    cluster.wait_pcs_available("gs9", timeout=20, min_available=2)


# --------------------------------------------------------------------------- SO
def test_so2_inorder_startup_with_min_replicas(cluster):
    """SO2/SO4 (startup_ordering_test.go:108,218): with InOrder startup and
    minAvailable < replicas, the dependent clique starts once the parent reaches
    minAvailable ready pods (not full replicas)."""
    cluster.add_virtual_nodes(2, cpu="8", pods=64)
    pcs = _pcs("so2", cliques=(("a", 3, 2), ("b", 1, 1)),
               startup=c.STARTUP_IN_ORDER)
    cluster.apply(pcs)
    cluster.wait_pcs_available("so2", timeout=20)
    b = cluster.store.get(c.KIND_PCLQ, "default", "so2-0-b")
    assert b["spec"]["startsAfter"] == ["so2-0-a"]
    b_pods = _pods(cluster, {c.LABEL_PODCLIQUE: "so2-0-b"})
    ic = b_pods[0]["spec"]["initContainers"][0]
    # waits for the parent's minAvailable (2), not its full replicas (3)
    assert "--podcliques=so2-0-a:2" in ic["args"]


# --------------------------------------------------------------------------- GT
def test_gt2_pcsg_owned_breach_terminates_pcs_replica(cluster):
    """GT2 (gang_termination_test.go:57): a breach in a PCSG member clique (the
    PCSG's MinAvailableBreached, not a standalone clique's) gang-terminates the
    whole PCS replica after terminationDelay — standalone cliques of the same
    replica are recreated too."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 2, "minAvailable": 2}]
    cluster.add_virtual_nodes(3, cpu="8", pods=64)
    cluster.apply(_pcs("gt2", cliques=(("a", 1, 1), ("b", 1, 1)), sg=sg,
                       termination_delay="300ms"))
    cluster.wait_pcs_available("gt2", timeout=20)
    a_uid = cluster.store.get(c.KIND_PCLQ, "default", "gt2-0-a")["metadata"]["uid"]
    _cordon_all(cluster)
    # kill a PCSG member pod -> PCSG availableReplicas drops below minAvailable
    victim = _pods(cluster, {c.LABEL_PODCLIQUE: "gt2-0-sg-0-b"})[0]
    cluster.store.delete("Pod", "default", victim["metadata"]["name"])

    def pcsg_breached():
        g = cluster.store.get(c.KIND_PCSG, "default", "gt2-0-sg")
        return cond.condition_true(g, c.COND_MIN_AVAILABLE_BREACHED)
    cluster.wait_for(pcsg_breached, timeout=20, desc="PCSG MinAvailableBreached")

    # nodes stay cordoned so the breach cannot heal before terminationDelay —
    # the whole replica (including standalone clique a) must be recreated
    def replica_recreated():
        q = cluster.store.try_get(c.KIND_PCLQ, "default", "gt2-0-a")
        return q is not None and q["metadata"]["uid"] != a_uid
    cluster.wait_for(replica_recreated, timeout=30,
                     desc="standalone clique recreated by gang termination")
    _cordon_all(cluster, False)
    cluster.wait_pcs_available("gt2", timeout=30)


def test_gt6_scaled_pod_deletion_does_not_terminate_pcs_replica(cluster):
    """GT6 (gang_termination_test.go:307): deleting a pod of a SCALED PCSG replica
    (base gang minimum still met) must not fire PCS-scope gang termination — the
    standalone cliques keep their identity and only the scaled pod is replaced."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 2, "minAvailable": 1}]
    cluster.add_virtual_nodes(3, cpu="8", pods=64)
    cluster.apply(_pcs("gt6", cliques=(("a", 1, 1), ("b", 1, 1)), sg=sg,
                       termination_delay="300ms"))
    cluster.wait_pcs_available("gt6", timeout=20)
    a_uid = cluster.store.get(c.KIND_PCLQ, "default", "gt6-0-a")["metadata"]["uid"]
    # scaled replica j=1 (>= minAvailable) pod
    victim = _pods(cluster, {c.LABEL_PODCLIQUE: "gt6-0-sg-1-b"})[0]
    cluster.store.delete("Pod", "default", victim["metadata"]["name"])
    time.sleep(0.9)  # > terminationDelay
    # standalone clique survived (no PCS-scope termination)
    a = cluster.store.get(c.KIND_PCLQ, "default", "gt6-0-a")
    assert a["metadata"]["uid"] == a_uid
    # the scaled pod was replaced and the set is whole again (generous timeout:
    # the replacement traverses podReferences refresh + ungate + schedule, and
    # the suite may share a heavily loaded box)
    cluster.wait_pods_ready({c.LABEL_PODCLIQUE: "gt6-0-sg-1-b"}, 1, timeout=45)


# --------------------------------------------------------------------------- OD
def test_od2_manual_deletion_creates_updated_pod(cluster):
    """OD2 (update/ondelete_test.go:72): with OnDelete, a template change does not
    touch running pods; deleting one manually yields a replacement built from the
    NEW template (new pod-template-hash)."""
    cluster.add_virtual_nodes(2, cpu="8", pods=64)
    cluster.apply(_pcs("od2", cliques=(("w", 2, 1),), strategy=c.UPDATE_ON_DELETE))
    cluster.wait_pcs_available("od2", timeout=20)
    old_pods = _pods(cluster, {c.LABEL_PODCLIQUE: "od2-0-w"})
    old_hash = old_pods[0]["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH]
    cur = cluster.store.get(c.KIND_PCS, "default", "od2")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)

    def pclq_updated():
        q = cluster.store.get(c.KIND_PCLQ, "default", "od2-0-w")
        return q["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] != old_hash
    cluster.wait_for(pclq_updated, timeout=20, desc="PCLQ spec propagated")
    time.sleep(0.3)
    # no automatic pod replacement (OD1 semantics)
    pods = _pods(cluster, {c.LABEL_PODCLIQUE: "od2-0-w"})
    assert {p["metadata"]["uid"] for p in pods} == \
        {p["metadata"]["uid"] for p in old_pods}
    # manual delete -> replacement carries the new hash + image
    cluster.store.delete("Pod", "default", old_pods[0]["metadata"]["name"])

    def replaced():
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: "od2-0-w"})
        return len(ps) == 2 and any(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] != old_hash
            for p in ps)
    cluster.wait_for(replaced, timeout=20, desc="updated replacement pod")
    new_pod = [p for p in _pods(cluster, {c.LABEL_PODCLIQUE: "od2-0-w"})
               if p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] != old_hash][0]
    assert new_pod["spec"]["containers"][0]["image"] == "img:v2"


def test_od5_pcsg_manual_deletion_creates_updated_replica(cluster):
    """OD5 (update/ondelete_test.go:293): OnDelete at PCSG scope — member pods
    survive the template change until manually deleted, then rebuild updated."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 2, "minAvailable": 1}]
    cluster.add_virtual_nodes(2, cpu="8", pods=64)
    cluster.apply(_pcs("od5", cliques=(("b", 1, 1),), sg=sg,
                       strategy=c.UPDATE_ON_DELETE))
    cluster.wait_pcs_available("od5", timeout=20)
    old = _pods(cluster, {c.LABEL_PODCLIQUE: "od5-0-sg-1-b"})[0]
    old_hash = old["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH]
    cur = cluster.store.get(c.KIND_PCS, "default", "od5")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)

    def member_pclq_updated():
        q = cluster.store.get(c.KIND_PCLQ, "default", "od5-0-sg-1-b")
        return q["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] != old_hash
    cluster.wait_for(member_pclq_updated, timeout=20, desc="member PCLQ updated")
    time.sleep(0.2)
    assert _pods(cluster, {c.LABEL_PODCLIQUE: "od5-0-sg-1-b"})[0][
        "metadata"]["uid"] == old["metadata"]["uid"]
    cluster.store.delete("Pod", "default", old["metadata"]["name"])

    def replaced():
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: "od5-0-sg-1-b"})
        return len(ps) == 1 and ps[0]["metadata"]["labels"][
            c.LABEL_POD_TEMPLATE_HASH] != old_hash
    cluster.wait_for(replaced, timeout=20, desc="updated member replacement")


def test_od9_strategy_transition_resumes_update(cluster):
    """OD9 (update/ondelete_test.go:567): switching OnDelete -> RollingRecreate
    with a pending template change starts replacing pods automatically."""
    cluster.add_virtual_nodes(2, cpu="8", pods=64)
    cluster.apply(_pcs("od9", cliques=(("w", 2, 1),), strategy=c.UPDATE_ON_DELETE))
    cluster.wait_pcs_available("od9", timeout=20)
    old_hash = _pods(cluster, {c.LABEL_PODCLIQUE: "od9-0-w"})[0][
        "metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH]
    cur = cluster.store.get(c.KIND_PCS, "default", "od9")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    time.sleep(0.3)
    assert all(p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == old_hash
               for p in _pods(cluster, {c.LABEL_PODCLIQUE: "od9-0-w"}))
    cur = cluster.store.get(c.KIND_PCS, "default", "od9")
    cur["spec"]["updateStrategy"] = {"type": c.UPDATE_ROLLING_RECREATE}
    cluster.apply(cur)

    def all_updated():
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: "od9-0-w"})
        return len(ps) == 2 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] != old_hash
            and cond.pod_is_ready(p) for p in ps)
    cluster.wait_for(all_updated, timeout=30,
                     desc="rolling update resumed after strategy switch")


# --------------------------------------------------------------------------- RU
def test_ru11_scale_out_during_update_gets_new_template(cluster):
    """RU11 (update/rolling_recreate_test.go:293): a PCS replica added while a
    rolling update is in flight is created directly from the NEW template."""
    cluster.add_virtual_nodes(3, cpu="16", pods=64)
    cluster.apply(_pcs("ru11", replicas=2, cliques=(("w", 2, 1),)))
    cluster.wait_pcs_available("ru11", timeout=20)
    cur = cluster.store.get(c.KIND_PCS, "default", "ru11")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cur["spec"]["replicas"] = 3
    cluster.apply(cur)
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "w", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def replica2_new():
        ps = _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "2"})
        return len(ps) == 2 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == new_hash
            for p in ps)
    cluster.wait_for(replica2_new, timeout=25,
                     desc="scale-out replica born on the new template")
    cluster.wait_pcs_available("ru11", timeout=40)
    # eventually every pod carries the new template
    def all_new():
        ps = _pods(cluster, {c.LABEL_PART_OF: "ru11"})
        return len(ps) == 6 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == new_hash
            for p in ps)
    cluster.wait_for(all_new, timeout=40, desc="rolling update completed")


def test_ru12_scale_in_during_update_completes(cluster):
    """RU12/RU13 (rolling_recreate_test.go:345,412): scaling the PCS in while an
    update is running removes the excess replica trees and the update still ends
    (updateProgress.updateEndedAt set)."""
    cluster.add_virtual_nodes(3, cpu="16", pods=64)
    cluster.apply(_pcs("ru12", replicas=3, cliques=(("w", 2, 1),)))
    cluster.wait_pcs_available("ru12", timeout=20)
    cur = cluster.store.get(c.KIND_PCS, "default", "ru12")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cur["spec"]["replicas"] = 1
    cluster.apply(cur)

    def done():
        pcs = cluster.store.get(c.KIND_PCS, "default", "ru12")
        st = pcs.get("status") or {}
        prog = st.get("updateProgress") or {}
        return bool(prog.get("updateEndedAt")) \
            and int(st.get("availableReplicas", 0)) >= 1
    cluster.wait_for(done, timeout=40, desc="update finished after scale-in")
    assert len(_pods(cluster, {c.LABEL_PART_OF: "ru12"})) == 2
    assert cluster.store.try_get(c.KIND_PCLQ, "default", "ru12-2-w") is None


def test_ru14_pcsg_scale_out_during_update(cluster):
    """RU14/RU15 (rolling_recreate_test.go:465,522): PCSG replicas added around an
    update come up on the new template; existing members roll."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 1, "minAvailable": 1}]
    cluster.add_virtual_nodes(3, cpu="16", pods=64)
    cluster.apply(_pcs("ru14", cliques=(("b", 1, 1),), sg=sg))
    cluster.wait_pcs_available("ru14", timeout=20)
    cur = cluster.store.get(c.KIND_PCS, "default", "ru14")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    cluster.store.patch(c.KIND_PCSG, "default", "ru14-0-sg",
                        lambda o: o["spec"].update(replicas=2))
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "b", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def all_new():
        ps = _pods(cluster, {c.LABEL_PART_OF: "ru14"})
        return len(ps) == 2 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == new_hash
            and cond.pod_is_ready(p) for p in ps)
    cluster.wait_for(all_new, timeout=40,
                     desc="scaled member + rolled member both on new template")


# --------------------------------------------------------------------------- TAS
def _rack_nodes(cluster, racks, per_rack, gpus=8):
    from grove_amd.kubelet.virtual import make_virtual_node
    for r in range(racks):
        for i in range(per_rack):
            n = make_virtual_node(f"r{r}n{i}", gpus=gpus, cpu="64", pods=64)
            n["metadata"]["labels"]["topology.kubernetes.io/rack"] = f"rack{r}"
            cluster.store.create(n)


_CTB = {"apiVersion": c.API_VERSION, "kind": c.KIND_CTB,
        "metadata": {"name": "default-topology"},
        "spec": {"levels": [
            {"domain": "rack", "key": "topology.kubernetes.io/rack"},
            {"domain": "host", "key": "kubernetes.io/hostname"}]}}


def test_tas9_pcs_plus_clique_constraint(cluster):
    """TAS9 (topology_test.go:585): gang-level pack (rack) + clique-level pack
    (host) — the clique's pods pack onto one host inside the gang's rack."""
    cluster.store.create(_CTB)
    _rack_nodes(cluster, racks=2, per_rack=2, gpus=4)
    pcs = _pcs("tas9", cliques=(("w", 2, 2),), gpus=1)
    pcs["spec"]["template"]["topologyConstraint"] = {"pack": {"required": "rack"}}
    pcs["spec"]["template"]["cliques"][0]["topologyConstraint"] = {
        "pack": {"required": "host"}}
    cluster.apply(pcs)
    cluster.wait_pcs_available("tas9", timeout=20)
    pods = _pods(cluster, {c.LABEL_PODCLIQUE: "tas9-0-w"})
    hosts = {p["spec"]["nodeName"] for p in pods}
    assert len(hosts) == 1  # clique packed to one host (inside one rack a fortiori)


def test_tas10_pcsg_scaling_with_constraint(cluster):
    """TAS10 (topology_test.go:635): each PCSG replica packs within its own
    domain of the PCSG's pack key; scaled replicas get the same constraint."""
    cluster.store.create(_CTB)
    _rack_nodes(cluster, racks=3, per_rack=1, gpus=4)
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 2, "minAvailable": 1,
           "topologyConstraint": {"pack": {"required": "host"}}}]
    pcs = _pcs("tas10", cliques=(("b", 2, 2),), sg=sg, gpus=1)
    cluster.apply(pcs)
    cluster.wait_pcs_available("tas10", timeout=20)
    for j in (0, 1):
        # PCS availability only needs PCSG minAvailable=1 — the scaled replica
        # (j=1) may still be binding; wait for both pods to land first
        cluster.wait_for(
            lambda j=j: all(
                p["spec"].get("nodeName")
                for p in _pods(cluster, {c.LABEL_PODCLIQUE: f"tas10-0-sg-{j}-b"})
            ) and len(_pods(cluster,
                            {c.LABEL_PODCLIQUE: f"tas10-0-sg-{j}-b"})) == 2,
            timeout=15, desc=f"sg replica {j} bound")
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: f"tas10-0-sg-{j}-b"})
        assert len({p["spec"]["nodeName"] for p in ps}) == 1, \
            f"PCSG replica {j} spans hosts"
    # scaled gang carries the translated constraint
    pg = cluster.store.get(c.KIND_PODGANG, "default", "tas10-0-sg-0")
    assert pg["spec"]["topologyConstraint"]["packConstraint"]["required"] == \
        "kubernetes.io/hostname"


def test_tas14_multi_replica_rack_constraint(cluster):
    """TAS14/TAS16 (topology_test.go:927,1099): every PCS replica's gang packs
    into ONE rack; different replicas may use different racks."""
    cluster.store.create(_CTB)
    _rack_nodes(cluster, racks=2, per_rack=2, gpus=2)
    pcs = _pcs("tas14", replicas=2, cliques=(("w", 3, 3),), gpus=1)
    pcs["spec"]["template"]["topologyConstraint"] = {"pack": {"required": "rack"}}
    cluster.apply(pcs)
    cluster.wait_pcs_available("tas14", timeout=20)
    node_rack = {n["metadata"]["name"]:
                 n["metadata"]["labels"]["topology.kubernetes.io/rack"]
                 for n in cluster.store.list("Node")}
    for r in (0, 1):
        ps = _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: str(r)})
        racks = {node_rack[p["spec"]["nodeName"]] for p in ps}
        assert len(racks) == 1, f"replica {r} spans racks {racks}"


def test_tas17_heterogeneous_gpu_cluster(cluster):
    """TAS17 (topology_test.go:1200): nodes with different GPU counts — an 8-GPU
    gang must land on the node that actually has 8 free GPUs."""
    from grove_amd.kubelet.virtual import make_virtual_node
    cluster.store.create(make_virtual_node("small", gpus=2, cpu="64", pods=64))
    cluster.store.create(make_virtual_node("big", gpus=8, cpu="64", pods=64))
    pcs = _pcs("tas17", cliques=(("w", 8, 8),), gpus=1)
    cluster.apply(pcs)
    cluster.wait_pcs_available("tas17", timeout=20)
    pods = _pods(cluster, {c.LABEL_PODCLIQUE: "tas17-0-w"})
    assert all(p["spec"]["nodeName"] == "big" for p in pods)


def test_ru16_pcsg_scale_in_during_update(cluster):
    """RU16/RU17 (rolling_recreate_test.go:580,640): scaling the PCSG in while a
    rolling update runs removes the excess member trees and the update still
    completes on the survivors."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 3, "minAvailable": 1}]
    cluster.add_virtual_nodes(3, cpu="16", pods=64)
    cluster.apply(_pcs("ru16", cliques=(("b", 1, 1),), sg=sg))
    cluster.wait_pcs_available("ru16", timeout=20)
    cur = cluster.store.get(c.KIND_PCS, "default", "ru16")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    cluster.store.patch(c.KIND_PCSG, "default", "ru16-0-sg",
                        lambda o: o["spec"].update(replicas=1))
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "b", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def settled():
        if cluster.store.try_get(c.KIND_PCLQ, "default", "ru16-0-sg-2-b"):
            return False
        ps = _pods(cluster, {c.LABEL_PART_OF: "ru16"})
        return len(ps) == 1 and ps[0]["metadata"]["labels"][
            c.LABEL_POD_TEMPLATE_HASH] == new_hash and cond.pod_is_ready(ps[0])
    cluster.wait_for(settled, timeout=40,
                     desc="scale-in applied and survivor updated")


def test_ru18_clique_scale_out_during_update(cluster):
    """RU18/RU20 (rolling_recreate_test.go:704,832): scaling a PodClique's
    replicas (the HPA path — PCLQ spec patched directly) while the template
    update is in flight: new pods are born on the NEW template and the update
    finishes across the larger set."""
    cluster.add_virtual_nodes(2, cpu="16", pods=64)
    pcs = _pcs("ru18", cliques=(("w", 2, 1),))
    # autoscaled clique: replica preservation keeps the HPA's scale decisions
    # (podclique.go:284 parity) while the template rolls
    pcs["spec"]["template"]["cliques"][0]["spec"]["autoScalingConfig"] = {
        "minReplicas": 1, "maxReplicas": 5}
    cluster.apply(pcs)
    cluster.wait_pcs_available("ru18", timeout=20)
    cur = cluster.store.get(c.KIND_PCS, "default", "ru18")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    # HPA-style direct PCLQ scale mid-update
    cluster.store.patch(c.KIND_PCLQ, "default", "ru18-0-w",
                        lambda o: o["spec"].update(replicas=3))
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "w", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def all_updated():
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: "ru18-0-w"})
        return len(ps) == 3 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == new_hash
            and cond.pod_is_ready(p) for p in ps)
    cluster.wait_for(all_updated, timeout=40,
                     desc="3 pods, all on the new template")


def test_gs11_pcs_and_pcsg_scaling_with_min_replicas(cluster):
    """GS11 (gang_scheduling_test.go:886): PCS replicas AND PCSG replicas scaled
    while capacity only fits the gang minima — base gangs admit at minAvailable,
    scaled gangs wait, everything lands once capacity arrives."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 2, "minAvailable": 1}]
    cluster.add_virtual_nodes(1, cpu="2", pods=64)
    cluster.apply(_pcs("gs11", replicas=1, cliques=(("a", 1, 1), ("b", 1, 1)),
                       sg=sg))
    # base gang (a + sg replica 0) = 2 pods fits exactly; scaled replica 1 pends
    cluster.wait_pods_ready({c.LABEL_PCS_REPLICA_INDEX: "0"}, 2, timeout=20)
    sc = _pods(cluster, {c.LABEL_PODCLIQUE: "gs11-0-sg-1-b"})
    assert sc and all(not p["spec"].get("nodeName") for p in sc)
    # scale the PCS out too: replica 1's tree pends entirely
    cluster.store.patch(c.KIND_PCS, "default", "gs11",
                        lambda o: o["spec"].update(replicas=2))
    cluster.wait_for(
        lambda: len(_pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "1"})) == 3,
        timeout=20, desc="replica 1 pods created")
    time.sleep(0.3)
    assert all(not p["spec"].get("nodeName")
               for p in _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "1"}))
    cluster.add_virtual_nodes(3, cpu="4", pods=64, prefix="cap")
    cluster.wait_pcs_available("gs11", timeout=30, min_available=2)
    cluster.wait_pods_ready({c.LABEL_PART_OF: "gs11"}, 6, timeout=30)


def test_od7_on_delete_multiple_replicas(cluster):
    """OD7 (update/ondelete_test.go:447): OnDelete across several PCS replicas —
    each replica's pods wait for their own manual deletion; deleting one
    replica's pod must not touch the others."""
    cluster.add_virtual_nodes(2, cpu="16", pods=64)
    cluster.apply(_pcs("od7", replicas=3, cliques=(("w", 1, 1),),
                       strategy=c.UPDATE_ON_DELETE))
    cluster.wait_pcs_available("od7", timeout=20)
    pods0 = {p["metadata"]["uid"]: p
             for p in _pods(cluster, {c.LABEL_PART_OF: "od7"})}
    old_hash = next(iter(pods0.values()))["metadata"]["labels"][
        c.LABEL_POD_TEMPLATE_HASH]
    cur = cluster.store.get(c.KIND_PCS, "default", "od7")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    time.sleep(0.4)
    assert {p["metadata"]["uid"]
            for p in _pods(cluster, {c.LABEL_PART_OF: "od7"})} == set(pods0)
    # delete replica 1's pod only
    victim = _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "1"})[0]
    cluster.store.delete("Pod", "default", victim["metadata"]["name"])

    def replica1_updated():
        ps = _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "1"})
        return len(ps) == 1 and ps[0]["metadata"]["labels"][
            c.LABEL_POD_TEMPLATE_HASH] != old_hash and cond.pod_is_ready(ps[0])
    cluster.wait_for(replica1_updated, timeout=20, desc="replica 1 updated")
    # replicas 0 and 2 untouched (still old template, same uids)
    for r in ("0", "2"):
        ps = _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: r})
        assert ps[0]["metadata"]["uid"] in pods0
        assert ps[0]["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == old_hash


def test_gt3_min_replicas_pcs_owned_termination(cluster):
    """GT3 (gang_termination_test.go:119): a clique running at replicas=3,
    minAvailable=2 tolerates ONE pod loss (no termination), but a second loss
    breaches minAvailable and gang-terminates the replica after the delay."""
    cluster.add_virtual_nodes(3, cpu="8", pods=64)
    cluster.apply(_pcs("gt3", cliques=(("w", 3, 2),),
                       termination_delay="400ms"))
    cluster.wait_pcs_available("gt3", timeout=20)
    uid0 = cluster.store.get(c.KIND_PCLQ, "default", "gt3-0-w")["metadata"]["uid"]
    _cordon_all(cluster)
    # one loss: 2 >= minAvailable -> tolerated
    victims = _pods(cluster, {c.LABEL_PODCLIQUE: "gt3-0-w"})
    cluster.store.delete("Pod", "default", victims[0]["metadata"]["name"])
    time.sleep(0.9)  # > terminationDelay
    assert cluster.store.get(c.KIND_PCLQ, "default",
                             "gt3-0-w")["metadata"]["uid"] == uid0, \
        "one pod loss within minAvailable must not gang-terminate"
    # second loss: 1 < minAvailable -> breach -> termination fires
    left = [p for p in _pods(cluster, {c.LABEL_PODCLIQUE: "gt3-0-w"})
            if p["spec"].get("nodeName")]
    cluster.store.delete("Pod", "default", left[0]["metadata"]["name"])

    def recreated():
        q = cluster.store.try_get(c.KIND_PCLQ, "default", "gt3-0-w")
        return q is not None and q["metadata"]["uid"] != uid0
    cluster.wait_for(recreated, timeout=25, desc="gang termination after breach")
    _cordon_all(cluster, False)
    cluster.wait_pcs_available("gt3", timeout=30)


def test_gt4_min_replicas_pcsg_owned_termination(cluster):
    """GT4 (gang_termination_test.go:170): a PCSG at replicas=3, minAvailable=2
    tolerates losing ONE member replica; losing a second drops
    availableReplicas below minAvailable and gang-terminates the PCS replica."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 3, "minAvailable": 2}]
    cluster.add_virtual_nodes(4, cpu="8", pods=64)
    cluster.apply(_pcs("gt4", cliques=(("b", 1, 1),), sg=sg,
                       termination_delay="400ms"))
    cluster.wait_pcs_available("gt4", timeout=20)
    sg_uid = cluster.store.get(c.KIND_PCSG, "default", "gt4-0-sg")["metadata"]["uid"]
    _cordon_all(cluster)
    # lose replica 2 (scaled): 2 available >= minAvailable -> recycle only, no
    # PCS-scope termination
    v2 = _pods(cluster, {c.LABEL_PODCLIQUE: "gt4-0-sg-2-b"})[0]
    cluster.store.delete("Pod", "default", v2["metadata"]["name"])
    time.sleep(0.9)
    assert cluster.store.get(c.KIND_PCSG, "default",
                             "gt4-0-sg")["metadata"]["uid"] == sg_uid
    # lose a base replica too: 1 available < minAvailable -> breach -> the PCS
    # replica is gang-terminated (member PCLQs recreated with fresh uids)
    uid_b0 = cluster.store.get(c.KIND_PCLQ, "default",
                               "gt4-0-sg-0-b")["metadata"]["uid"]
    v0 = _pods(cluster, {c.LABEL_PODCLIQUE: "gt4-0-sg-0-b"})[0]
    cluster.store.delete("Pod", "default", v0["metadata"]["name"])

    def terminated():
        q = cluster.store.try_get(c.KIND_PCLQ, "default", "gt4-0-sg-0-b")
        return q is not None and q["metadata"]["uid"] != uid_b0
    cluster.wait_for(terminated, timeout=25,
                     desc="PCS-scope termination after PCSG minAvailable breach")
    _cordon_all(cluster, False)
    cluster.wait_pcs_available("gt4", timeout=30)


def test_tas12_large_scaling_ratio(cluster):
    """TAS12 (topology_test.go:774): a PCSG scaled to MANY replicas under a
    host-pack constraint — every one of the 6 replicas (each 2 pods) must pack
    onto one host; the scheduler fans them across hosts without violating any
    per-replica constraint."""
    cluster.store.create(_CTB)
    _rack_nodes(cluster, racks=3, per_rack=2, gpus=4)
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 6, "minAvailable": 2,
           "topologyConstraint": {"pack": {"required": "host"}}}]
    pcs = _pcs("tas12", cliques=(("b", 2, 2),), sg=sg, gpus=1)
    cluster.apply(pcs)
    cluster.wait_pcs_available("tas12", timeout=30)
    # PCS availability only needs minAvailable PCSG replicas — wait for ALL 12
    # pods (scaled replicas included) before asserting per-replica placement
    cluster.wait_pods_ready({c.LABEL_PART_OF: "tas12"}, 12, timeout=30)
    for j in range(6):
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: f"tas12-0-sg-{j}-b"})
        assert len(ps) == 2
        assert len({p["spec"]["nodeName"] for p in ps}) == 1, \
            f"PCSG replica {j} spans hosts"


def test_ru21_clique_scale_in_before_update(cluster):
    """RU20/RU21 (rolling_recreate_test.go:832,897): scaling a PodClique IN just
    before/while the template rolls — the update completes across the reduced
    set and no orphan pods survive."""
    cluster.add_virtual_nodes(2, cpu="16", pods=64)
    pcs = _pcs("ru21", cliques=(("w", 4, 1),))
    pcs["spec"]["template"]["cliques"][0]["spec"]["autoScalingConfig"] = {
        "minReplicas": 1, "maxReplicas": 6}
    cluster.apply(pcs)
    cluster.wait_pcs_available("ru21", timeout=20)
    # HPA-style scale-in first, then the template change
    cluster.store.patch(c.KIND_PCLQ, "default", "ru21-0-w",
                        lambda o: o["spec"].update(replicas=2))
    cur = cluster.store.get(c.KIND_PCS, "default", "ru21")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "w", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def settled():
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: "ru21-0-w"})
        return len(ps) == 2 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == new_hash
            and cond.pod_is_ready(p) for p in ps)
    cluster.wait_for(settled, timeout=40, desc="2 pods, all on new template")


def test_tas16_three_level_hierarchy(cluster):
    """TAS16 (topology_test.go:1099): zone > rack > host CTB; the gang requires
    zone packing and prefers host packing — with a host that fits, everything
    lands on one host; when no single host fits, it falls back within one zone
    but never crosses zones."""
    ctb = {"apiVersion": c.API_VERSION, "kind": c.KIND_CTB,
           "metadata": {"name": "default-topology"},
           "spec": {"levels": [
               {"domain": "zone", "key": "topology.kubernetes.io/zone"},
               {"domain": "rack", "key": "topology.kubernetes.io/rack"},
               {"domain": "host", "key": "kubernetes.io/hostname"}]}}
    cluster.store.create(ctb)
    from grove_amd.kubelet.virtual import make_virtual_node
    for z in range(2):
        for r in range(2):
            for h in range(2):
                n = make_virtual_node(f"z{z}r{r}h{h}", gpus=2, cpu="64", pods=64)
                n["metadata"]["labels"]["topology.kubernetes.io/zone"] = f"z{z}"
                n["metadata"]["labels"]["topology.kubernetes.io/rack"] = f"z{z}r{r}"
                cluster.store.create(n)
    # 2-GPU gang: fits one host -> host-preferred packing
    pcs = _pcs("t16a", cliques=(("w", 2, 2),), gpus=1)
    pcs["spec"]["template"]["topologyConstraint"] = {
        "pack": {"required": "zone", "preferred": "host"}}
    cluster.apply(pcs)
    cluster.wait_pcs_available("t16a", timeout=20)
    pods = _pods(cluster, {c.LABEL_PODCLIQUE: "t16a-0-w"})
    assert len({p["spec"]["nodeName"] for p in pods}) == 1
    # 6-GPU gang: no host fits (2 GPUs each) -> falls back inside ONE zone
    pcs = _pcs("t16b", cliques=(("w", 6, 6),), gpus=1)
    pcs["spec"]["template"]["topologyConstraint"] = {
        "pack": {"required": "zone", "preferred": "host"}}
    cluster.apply(pcs)
    cluster.wait_pcs_available("t16b", timeout=20)
    pods = _pods(cluster, {c.LABEL_PODCLIQUE: "t16b-0-w"})
    zones = {p["spec"]["nodeName"][:2] for p in pods}
    assert len(zones) == 1, f"gang crossed zones: {zones}"
    assert len({p["spec"]["nodeName"] for p in pods}) == 3  # 3 hosts x 2 GPUs


def test_so4_explicit_startup_with_min_replicas(cluster):
    """SO4 (startup_ordering_test.go:218): Explicit startsAfter DAG with
    minAvailable < replicas — the dependent waits on the parent's minAvailable,
    and the grove-initc flag carries exactly that threshold."""
    cluster.add_virtual_nodes(2, cpu="8", pods=64)
    pcs = _pcs("so4", cliques=(("a", 4, 2), ("b", 2, 1), ("z", 1, 1)),
               startup=c.STARTUP_EXPLICIT)
    pcs["spec"]["template"]["cliques"][1]["spec"]["startsAfter"] = ["a"]
    pcs["spec"]["template"]["cliques"][2]["spec"]["startsAfter"] = ["a", "b"]
    cluster.apply(pcs)
    cluster.wait_pcs_available("so4", timeout=20)
    z_pods = _pods(cluster, {c.LABEL_PODCLIQUE: "so4-0-z"})
    ic = z_pods[0]["spec"]["initContainers"][0]
    assert "--podcliques=so4-0-a:2" in ic["args"]  # parent a's minAvailable
    assert "--podcliques=so4-0-b:1" in ic["args"]  # parent b's minAvailable
    assert cluster.store.get(c.KIND_PCLQ, "default",
                             "so4-0-z")["spec"]["startsAfter"] == \
        ["so4-0-a", "so4-0-b"]


def test_gs7_scaled_gangs_schedule_independently(cluster):
    """GS7 (gang_scheduling_test.go:463): with PCSG minAvailable=1, replicas=3,
    each scaled gang admits independently — under scarcity one scaled gang binds
    while the other stays pending without blocking it; relief schedules the rest."""
    sg = [{"name": "sx", "cliqueNames": ["b"], "replicas": 3, "minAvailable": 1}]
    cluster.add_virtual_nodes(1, cpu="3", pods=64)
    cluster.apply(_pcs("gs7", cliques=(("a", 1, 1), ("b", 1, 1)), sg=sg))
    cluster.wait_pcs_available("gs7", timeout=20)  # base gang (a + sx j0) fits
    cluster.wait_for(lambda: len(_pods(cluster, {c.LABEL_PART_OF: "gs7"})) == 4,
                     timeout=15, desc="4 pods exist")
    # capacity 3 cpu, 4 one-cpu pods: exactly one scaled gang binds
    cluster.wait_for(
        lambda: sum(1 for p in _pods(cluster, {c.LABEL_PART_OF: "gs7"})
                    if p["spec"].get("nodeName")) == 3,
        timeout=15, desc="3 of 4 pods bound")
    time.sleep(0.3)  # settle: the fourth must stay pending, not flap
    unbound = [p for p in _pods(cluster, {c.LABEL_PART_OF: "gs7"})
               if not p["spec"].get("nodeName")]
    assert len(unbound) == 1, "exactly one scaled gang must remain pending"
    cluster.add_virtual_nodes(1, cpu="2", pods=64, prefix="extra")
    cluster.wait_pods_ready({c.LABEL_PART_OF: "gs7"}, 4, timeout=20)


def test_tas11_clique_constraint_without_parent(cluster):
    """TAS11 (topology_test.go:697): a member clique carries a host pack with NO
    PCSG- or PCS-level constraint — each PCSG replica's clique packs one host on
    its own, and the scaled gang carries the clique constraint as a group config
    rather than gang-level."""
    cluster.store.create(_CTB)
    _rack_nodes(cluster, racks=2, per_rack=2, gpus=4)
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 2, "minAvailable": 1}]
    pcs = _pcs("tas11", cliques=(("b", 2, 2),), sg=sg, gpus=1)
    pcs["spec"]["template"]["cliques"][0]["topologyConstraint"] = {
        "pack": {"required": "host"}}
    cluster.apply(pcs)
    cluster.wait_pcs_available("tas11", timeout=20)
    for j in (0, 1):
        sel = {c.LABEL_PODCLIQUE: f"tas11-0-sg-{j}-b"}
        cluster.wait_for(
            lambda s=sel: len(_pods(cluster, s)) == 2
            and all(p["spec"].get("nodeName") for p in _pods(cluster, s)),
            timeout=15, desc=f"sg replica {j} bound")
        ps = _pods(cluster, sel)
        assert len({p["spec"]["nodeName"] for p in ps}) == 1, \
            f"sg replica {j} clique spans hosts"
    pg = cluster.store.get(c.KIND_PODGANG, "default", "tas11-0-sg-0")
    cfgs = pg["spec"].get("topologyConstraintGroupConfigs") or []
    assert any(g["topologyConstraint"]["packConstraint"]["required"]
               == "kubernetes.io/hostname" for g in cfgs), cfgs
    assert not (pg["spec"].get("topologyConstraint") or {}), \
        "no gang-level constraint expected"


def test_tas20_unavailable_topology_level_surfaces(cluster):
    """TAS20 (topology_test.go unavailable-level scenario): a required pack level
    that labels NO node is surfaced as a Warning Event on the PodGang instead of a
    silently pending gang; once nodes carry the level, the gang schedules."""
    cluster.store.create(_CTB)
    cluster.add_virtual_nodes(2, cpu="8", pods=64)  # nodes lack the rack label
    pcs = _pcs("tas20", cliques=(("w", 2, 2),))
    pcs["spec"]["template"]["topologyConstraint"] = {"pack": {"required": "rack"}}
    cluster.apply(pcs)

    def surfaced():
        return any(e["reason"] == "UnsatisfiableTopologyConstraint"
                   and e["involvedObject"]["name"] == "tas20-0"
                   for e in cluster.store.events)
    cluster.wait_for(surfaced, timeout=15, desc="unsatisfiable-level event")
    for n in cluster.store.list("Node"):
        cluster.store.patch(
            "Node", None, n["metadata"]["name"],
            lambda o: o["metadata"].setdefault("labels", {}).update(
                {"topology.kubernetes.io/rack": "rack0"}))
    cluster.wait_pcs_available("tas20", timeout=20)


def test_tas22_topology_cel_validation(cluster):
    """TAS22 (CRD CEL rules, podcliqueset.go x-kubernetes-validations): the
    apiserver-side equivalents of the TopologyConstraint CEL rules — empty
    constraint, empty pack, pack+packDomain mutual exclusion — reject on create;
    constraints are immutable on update."""
    cluster.store.create(_CTB)

    def pcs(name, tc):
        p = _pcs(name, cliques=(("w", 1, 1),))
        p["spec"]["template"]["topologyConstraint"] = tc
        return p

    for tc in ({}, {"pack": {}},
               {"pack": {"required": "rack"}, "packDomain": "rack"}):
        with pytest.raises(Exception):
            cluster.store.create(pcs("tas22-bad", tc))
    cluster.store.create(pcs("tas22", {"pack": {"required": "rack"}}))
    with pytest.raises(Exception):
        cluster.store.patch(
            c.KIND_PCS, "default", "tas22",
            lambda o: o["spec"]["template"].update(
                topologyConstraint={"pack": {"preferred": "rack"}}))


def test_gs10_pcs_scaling_min_replicas_advanced(cluster):
    """GS10 (gang_scheduling_test.go:754): scaling the PCS under scarcity — the
    NEW replica's gang admits at the clique minAvailable (2 of 3), the remainder
    lands when capacity arrives; the original replica is never disturbed."""
    cluster.add_virtual_nodes(1, cpu="5", pods=64)
    cluster.apply(_pcs("gs10", replicas=1, cliques=(("w", 3, 2),)))
    cluster.wait_pcs_available("gs10", timeout=20)
    r0_uids = {p["metadata"]["uid"]
               for p in _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "0"})}
    cluster.store.patch(c.KIND_PCS, "default", "gs10",
                        lambda o: o["spec"].update(replicas=2))
    # 2 cpu left: replica 1 admits at minAvailable=2, third pod stays pending
    cluster.wait_for(
        lambda: sum(1 for p in _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "1"})
                    if p["spec"].get("nodeName")) == 2,
        timeout=20, desc="replica 1 admitted at min")
    time.sleep(0.3)
    r1 = _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "1"})
    assert len(r1) == 3
    assert sum(1 for p in r1 if not p["spec"].get("nodeName")) == 1
    assert {p["metadata"]["uid"]
            for p in _pods(cluster, {c.LABEL_PCS_REPLICA_INDEX: "0"})} == r0_uids
    cluster.add_virtual_nodes(1, cpu="1", pods=8, prefix="late")
    cluster.wait_pods_ready({c.LABEL_PART_OF: "gs10"}, 6, timeout=20)


def test_tas6_standalone_clique_with_pcs_zone_constraint(cluster):
    """TAS6 (topology_test.go:417): a PCS-level zone pack applies to a standalone
    clique — all its pods land in ONE zone even though several zones have room."""
    cluster.store.create({
        "apiVersion": c.API_VERSION, "kind": c.KIND_CTB,
        "metadata": {"name": "zoned"},
        "spec": {"levels": [
            {"domain": "zone", "key": "topology.kubernetes.io/zone"},
            {"domain": "host", "key": "kubernetes.io/hostname"}]}})
    from grove_amd.kubelet.virtual import make_virtual_node
    for z in range(2):
        for i in range(2):
            n = make_virtual_node(f"z{z}n{i}", gpus=0, cpu="2", pods=16)
            n["metadata"]["labels"]["topology.kubernetes.io/zone"] = f"zone{z}"
            cluster.store.create(n)
    pcs = _pcs("tas6", cliques=(("w", 4, 4),))
    pcs["spec"]["template"]["topologyConstraint"] = {"pack": {"required": "zone"}}
    cluster.apply(pcs)
    cluster.wait_pcs_available("tas6", timeout=20)
    zones = {cluster.store.get("Node", None, p["spec"]["nodeName"])
             ["metadata"]["labels"]["topology.kubernetes.io/zone"]
             for p in _pods(cluster, {c.LABEL_PODCLIQUE: "tas6-0-w"})}
    assert len(zones) == 1, f"clique spans zones {zones}"


def test_tas8_full_hierarchy_cascading_constraints(cluster):
    """TAS8 (topology_test.go:529): three-level cascade — PCS packs the gang into
    one rack, the PCSG packs each replica onto one host inside it, and every
    level is satisfied simultaneously."""
    cluster.store.create(_CTB)
    _rack_nodes(cluster, racks=2, per_rack=2, gpus=4)
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 2, "minAvailable": 2,
           "topologyConstraint": {"pack": {"required": "host"}}}]
    pcs = _pcs("tas8", cliques=(("a", 1, 1), ("b", 2, 2)), sg=sg, gpus=1)
    pcs["spec"]["template"]["topologyConstraint"] = {"pack": {"required": "rack"}}
    cluster.apply(pcs)
    cluster.wait_pcs_available("tas8", timeout=20)
    node_rack = {n["metadata"]["name"]:
                 n["metadata"]["labels"]["topology.kubernetes.io/rack"]
                 for n in cluster.store.list("Node")}
    pods = _pods(cluster, {c.LABEL_PART_OF: "tas8"})
    assert len({node_rack[p["spec"]["nodeName"]] for p in pods}) == 1, \
        "gang spans racks"
    for j in (0, 1):
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: f"tas8-0-sg-{j}-b"})
        assert len({p["spec"]["nodeName"] for p in ps}) == 1, \
            f"sg replica {j} spans hosts"


def test_od6_on_delete_mixed_cliques_and_pcsg(cluster):
    """OD6 (update/ondelete_test.go:361): OnDelete with a standalone clique AND a
    PCSG — a spec change replaces nothing; deleting a PCSG member pod recreates
    it on the new template while the standalone clique keeps its old pod."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 1, "minAvailable": 1}]
    cluster.add_virtual_nodes(2, cpu="16", pods=64)
    cluster.apply(_pcs("od6", cliques=(("a", 1, 1), ("b", 1, 1)), sg=sg,
                       strategy=c.UPDATE_ON_DELETE))
    cluster.wait_pcs_available("od6", timeout=20)
    pods0 = {p["metadata"]["name"]: p["metadata"]["labels"][
        c.LABEL_POD_TEMPLATE_HASH]
        for p in _pods(cluster, {c.LABEL_PART_OF: "od6"})}
    cur = cluster.store.get(c.KIND_PCS, "default", "od6")
    for cl in cur["spec"]["template"]["cliques"]:
        cl["spec"]["podSpec"]["containers"][0]["image"] = "img:v2"
    cluster.apply(cur)
    time.sleep(0.4)
    assert {p["metadata"]["name"]
            for p in _pods(cluster, {c.LABEL_PART_OF: "od6"})} == set(pods0)
    victim = _pods(cluster, {c.LABEL_PODCLIQUE: "od6-0-sg-0-b"})[0]
    old_hash_b = victim["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH]
    cluster.store.delete("Pod", "default", victim["metadata"]["name"])

    def sg_pod_updated():
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: "od6-0-sg-0-b"})
        return len(ps) == 1 and ps[0]["metadata"]["labels"][
            c.LABEL_POD_TEMPLATE_HASH] != old_hash_b and cond.pod_is_ready(ps[0])
    cluster.wait_for(sg_pod_updated, timeout=20, desc="sg pod on new template")
    a_pods = _pods(cluster, {c.LABEL_PODCLIQUE: "od6-0-a"})
    assert len(a_pods) == 1
    assert a_pods[0]["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == \
        pods0[a_pods[0]["metadata"]["name"]], "standalone clique must keep its pod"


def test_ru15_pcsg_scale_out_before_update(cluster):
    """RU15 (rolling_recreate_test.go:522): the PCSG is scaled OUT and settles
    BEFORE the template update starts — the update then rolls every member,
    including the previously scaled-out one."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 1, "minAvailable": 1}]
    cluster.add_virtual_nodes(3, cpu="16", pods=64)
    cluster.apply(_pcs("ru15", cliques=(("b", 1, 1),), sg=sg))
    cluster.wait_pcs_available("ru15", timeout=20)
    cluster.store.patch(c.KIND_PCSG, "default", "ru15-0-sg",
                        lambda o: o["spec"].update(replicas=2))
    cluster.wait_pods_ready({c.LABEL_PART_OF: "ru15"}, 2, timeout=20)
    cur = cluster.store.get(c.KIND_PCS, "default", "ru15")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "b", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def all_new():
        ps = _pods(cluster, {c.LABEL_PART_OF: "ru15"})
        return len(ps) == 2 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == new_hash
            and cond.pod_is_ready(p) for p in ps)
    cluster.wait_for(all_new, timeout=60, desc="both PCSG members rolled")


def test_ru17_pcsg_scale_in_before_update(cluster):
    """RU17 (rolling_recreate_test.go:640): the PCSG is scaled IN and settles
    BEFORE the update — the update rolls only the survivors."""
    sg = [{"name": "sg", "cliqueNames": ["b"], "replicas": 3, "minAvailable": 1}]
    cluster.add_virtual_nodes(3, cpu="16", pods=64)
    cluster.apply(_pcs("ru17", cliques=(("b", 1, 1),), sg=sg))
    cluster.wait_pcs_available("ru17", timeout=20)
    cluster.store.patch(c.KIND_PCSG, "default", "ru17-0-sg",
                        lambda o: o["spec"].update(replicas=1))
    cluster.wait_for(
        lambda: cluster.store.try_get(c.KIND_PCLQ, "default", "ru17-0-sg-2-b")
        is None and len(_pods(cluster, {c.LABEL_PART_OF: "ru17"})) == 1,
        timeout=20, desc="scale-in settled")
    cur = cluster.store.get(c.KIND_PCS, "default", "ru17")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "b", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def survivor_new():
        ps = _pods(cluster, {c.LABEL_PART_OF: "ru17"})
        return len(ps) == 1 and ps[0]["metadata"]["labels"][
            c.LABEL_POD_TEMPLATE_HASH] == new_hash and cond.pod_is_ready(ps[0])
    cluster.wait_for(survivor_new, timeout=60, desc="survivor rolled")


def test_ru19_clique_scale_out_before_update(cluster):
    """RU19 (rolling_recreate_test.go:704): an autoscaled clique is scaled out
    (HPA path, direct PCLQ patch) and settles BEFORE the update — the update then
    rolls the full scaled set, preserving the HPA's replica decision."""
    cluster.add_virtual_nodes(2, cpu="16", pods=64)
    pcs = _pcs("ru19", cliques=(("w", 2, 1),))
    pcs["spec"]["template"]["cliques"][0]["spec"]["autoScalingConfig"] = {
        "minReplicas": 1, "maxReplicas": 5}
    cluster.apply(pcs)
    cluster.wait_pcs_available("ru19", timeout=20)
    cluster.store.patch(c.KIND_PCLQ, "default", "ru19-0-w",
                        lambda o: o["spec"].update(replicas=3))
    cluster.wait_pods_ready({c.LABEL_PODCLIQUE: "ru19-0-w"}, 3, timeout=20)
    cur = cluster.store.get(c.KIND_PCS, "default", "ru19")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "w", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def all_updated():
        ps = _pods(cluster, {c.LABEL_PODCLIQUE: "ru19-0-w"})
        return len(ps) == 3 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == new_hash
            and cond.pod_is_ready(p) for p in ps)
    cluster.wait_for(all_updated, timeout=60,
                     desc="3 pods rolled, HPA replicas preserved")


def test_ru13_pcs_scale_in_of_final_updating_replica(cluster):
    """RU13 (rolling_recreate_test.go:400): while the FINAL replica is the one
    still updating, the PCS is scaled in past it — the update must end cleanly
    with every surviving pod on the new template."""
    cluster.add_virtual_nodes(3, cpu="16", pods=64)
    cluster.apply(_pcs("ru13", replicas=2, cliques=(("w", 1, 1),)))
    cluster.wait_pcs_available("ru13", timeout=20, min_available=2)
    cur = cluster.store.get(c.KIND_PCS, "default", "ru13")
    cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"]["containers"][0][
        "image"] = "img:v2"
    cluster.apply(cur)

    # wait until exactly one replica has finished updating (the other is the
    # final in-flight ordinal), then scale that final replica away
    def one_done():
        st = (cluster.store.get(c.KIND_PCS, "default", "ru13").get("status")
              or {})
        prog = st.get("updateProgress") or {}
        return int(st.get("updatedReplicas") or 0) >= 1 and \
            not prog.get("updateEndedAt")
    cluster.wait_for(one_done, timeout=40, desc="first replica updated")
    cluster.store.patch(c.KIND_PCS, "default", "ru13",
                        lambda o: o["spec"].update(replicas=1))
    from grove_amd.utils.hashing import pod_template_hash
    new_hash = pod_template_hash(
        "w", cur["spec"]["template"]["cliques"][0]["spec"]["podSpec"])

    def ended_clean():
        pcs_obj = cluster.store.get(c.KIND_PCS, "default", "ru13")
        prog = (pcs_obj.get("status") or {}).get("updateProgress") or {}
        ps = _pods(cluster, {c.LABEL_PART_OF: "ru13"})
        return prog.get("updateEndedAt") and len(ps) == 1 and all(
            p["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] == new_hash
            and cond.pod_is_ready(p) for p in ps)
    cluster.wait_for(ended_clean, timeout=60, desc="update ended after scale-in")


def test_gs12_complex_multi_pcsg_scaling(cluster):
    """GS12 (gang_scheduling_test.go:1009): two PCSGs with different minAvailable
    scaled out and back in — base gang covers [0,minAvailable) of each group,
    every scaled replica gets its own gang, scale-in GCs the scaled gangs and
    their pods, and the set stays available throughout."""
    sg = [{"name": "pf", "cliqueNames": ["p"], "replicas": 2, "minAvailable": 1},
          {"name": "dc", "cliqueNames": ["d"], "replicas": 2, "minAvailable": 2}]
    cluster.add_virtual_nodes(4, cpu="16", pods=64)
    cluster.apply(_pcs("gs12", cliques=(("a", 1, 1), ("p", 1, 1), ("d", 1, 1)),
                       sg=sg))
    cluster.wait_pcs_available("gs12", timeout=20)
    gangs = {g["metadata"]["name"]
             for g in cluster.store.list(c.KIND_PODGANG, "default",
                                         {c.LABEL_PART_OF: "gs12"})}
    # base covers a + pf j0 + dc j0,j1; pf j1 is the one scaled gang
    assert gangs == {"gs12-0", "gs12-0-pf-0"}, gangs
    # scale pf 2->4 and dc 2->3: two more pf scaled gangs, one dc scaled gang
    cluster.store.patch(c.KIND_PCSG, "default", "gs12-0-pf",
                        lambda o: o["spec"].update(replicas=4))
    cluster.store.patch(c.KIND_PCSG, "default", "gs12-0-dc",
                        lambda o: o["spec"].update(replicas=3))
    cluster.wait_for(
        lambda: {g["metadata"]["name"]
                 for g in cluster.store.list(c.KIND_PODGANG, "default",
                                             {c.LABEL_PART_OF: "gs12"})}
        == {"gs12-0", "gs12-0-pf-0", "gs12-0-pf-1", "gs12-0-pf-2",
            "gs12-0-dc-0"},
        timeout=20, desc="scaled gangs created")
    cluster.wait_pods_ready({c.LABEL_PART_OF: "gs12"}, 8, timeout=30)
    # scale pf back to 2: scaled gangs pf-1/pf-2 and their pods are GC'd
    cluster.store.patch(c.KIND_PCSG, "default", "gs12-0-pf",
                        lambda o: o["spec"].update(replicas=2))
    cluster.wait_for(
        lambda: {g["metadata"]["name"]
                 for g in cluster.store.list(c.KIND_PODGANG, "default",
                                             {c.LABEL_PART_OF: "gs12"})}
        == {"gs12-0", "gs12-0-pf-0", "gs12-0-dc-0"},
        timeout=20, desc="scaled gangs GCd")
    cluster.wait_for(
        lambda: len(_pods(cluster, {c.LABEL_PART_OF: "gs12"})) == 6,
        timeout=20, desc="scaled pods removed")
    cluster.wait_pcs_available("gs12", timeout=20)
