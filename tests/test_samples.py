"""Every committed sample manifest must deploy and reach Available (BASELINE.json's
five configs are covered by simple1/disaggregated/multinode/agentic + the bench)."""
import os

import pytest

from grove_amd.api import constants as c

SAMPLES = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                       "samples")


def _apply_sample(cluster, fname):
    with open(os.path.join(SAMPLES, fname)) as f:
        return cluster.apply(f.read())


def test_sample_simple1(cluster):
    cluster.add_virtual_nodes(2)
    _apply_sample(cluster, "simple1.yaml")
    cluster.wait_pcs_available("simple1", timeout=30)


def test_sample_disaggregated_prefill_decode(cluster):
    cluster.store.create({"apiVersion": c.API_VERSION, "kind": c.KIND_CTB,
                          "metadata": {"name": "t"},
                          "spec": {"levels": [
                              {"domain": "host", "key": "kubernetes.io/hostname"}]}})
    cluster.add_virtual_nodes(1, gpus=8, prefix="mi355x")
    _apply_sample(cluster, "disaggregated.yaml")
    pcs = cluster.wait_pcs_available("disagg", timeout=30)
    assert pcs["status"]["availableReplicas"] == 1
    # whole instance packed on one 8-GPU hive, score = xGMI per-link bw
    pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "disagg"})
    assert len(pods) == 8 and {p["spec"]["nodeName"] for p in pods} == {"mi355x-0"}
    pg = cluster.store.get(c.KIND_PODGANG, "default", "disagg-0")
    assert pg["status"]["placementScore"] == pytest.approx(c.XGMI_LINK_GBPS)


def test_sample_multinode_leader_worker(cluster):
    cluster.store.create({"apiVersion": c.API_VERSION, "kind": c.KIND_CTB,
                          "metadata": {"name": "t"},
                          "spec": {"levels": [
                              {"domain": "rack", "key": "topology.kubernetes.io/rack"},
                              {"domain": "host", "key": "kubernetes.io/hostname"}]}})
    cluster.add_virtual_nodes(1, gpus=8, prefix="mi355x",
                              labels={"topology.kubernetes.io/rack": "r1"})
    _apply_sample(cluster, "multinode-leader-worker.yaml")
    cluster.wait_pcs_available("dsr1", timeout=30)
    # workers wait for the leader (explicit startup ordering)
    wkr = cluster.store.get(c.KIND_PCLQ, "default", "dsr1-0-inst-0-wkr")
    assert wkr["spec"]["startsAfter"] == ["dsr1-0-inst-0-ldr"]


def test_sample_agentic_pipeline(cluster):
    cluster.add_virtual_nodes(2, gpus=8)
    _apply_sample(cluster, "agentic-pipeline.yaml")
    for name in ("router", "model-a", "model-b"):
        cluster.wait_pcs_available(name, timeout=30)
    # HPAs exist for every autoscaled clique
    hpas = {h["metadata"]["name"]
            for h in cluster.store.list("HorizontalPodAutoscaler")}
    assert {"router-0-rt", "router-1-rt", "model-a-0-inf", "model-b-0-inf"} <= hpas


def test_node_loss_recovery(cluster):
    """Node dies → pods evicted → breach → gang terminated → rescheduled elsewhere."""
    import time
    cluster.add_virtual_nodes(2, gpus=8, prefix="hive")
    pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
           "metadata": {"name": "nl"},
           "spec": {"replicas": 1, "template": {
               "terminationDelay": "300ms",
               "cliques": [{"name": "w", "spec": {
                   "roleName": "w", "replicas": 4, "minAvailable": 4,
                   "podSpec": {"containers": [{
                       "name": "m", "image": "i",
                       "resources": {"requests": {c.AMD_GPU_RESOURCE: "1"}}}]}}}]}}}
    cluster.store.create(pcs)
    cluster.wait_pcs_available("nl", timeout=30)
    node = cluster.store.list("Pod", "default",
                              {c.LABEL_PART_OF: "nl"})[0]["spec"]["nodeName"]
    cluster.store.delete("Node", None, node)

    def recovered():
        pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "nl"})
        from grove_amd.utils import conditions as cc
        return (len(pods) == 4
                and all(cc.pod_is_ready(p) for p in pods)
                and all(p["spec"].get("nodeName") != node for p in pods))
    cluster.wait_for(recovered, timeout=60, desc="gang recovered on surviving node")
