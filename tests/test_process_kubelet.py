"""Process-backed pod runtime: pods run as real OS subprocesses with the injected
GROVE_* env, startup ordering enforced by the real grove-initc waiter over the HTTP
apiserver, and phases following the process lifecycle."""
import time

import pytest

from grove_amd.api import constants as c
from grove_amd.controllers.manager import Controller
from grove_amd.kubecore.apiserver import ApiServer
from grove_amd.kubelet.process import ProcessKubelet


@pytest.fixture()
def process_cluster(cluster):
    api = ApiServer(cluster.store, port=18177).start()
    kubelet = ProcessKubelet(cluster.store, api_url=api.url)
    cluster.c_kubelet.stop()  # replace the virtual kubelet
    ctrl = cluster.manager.add_controller(
        Controller("process-kubelet", kubelet.reconcile, workers=4))
    ctrl.start()
    orig_watch = cluster.store.watch("Pod")

    import threading

    def pump():
        import queue as q
        while True:
            try:
                ev, obj = orig_watch.queue.get(timeout=0.5)
            except q.Empty:
                continue
            except Exception:
                return
            if obj.get("spec", {}).get("nodeName"):
                ctrl.enqueue(obj["metadata"].get("namespace", "default"),
                             obj["metadata"]["name"])
    t = threading.Thread(target=pump, daemon=True)
    t.start()
    yield cluster, kubelet
    kubelet.shutdown()
    orig_watch.stop()
    ctrl.stop()
    api.stop()


@pytest.mark.timeout(180)
def test_pods_run_as_processes_with_startup_ordering(process_cluster):
    cluster, kubelet = process_cluster
    cluster.add_virtual_nodes(1)
    pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
           "metadata": {"name": "proc"},
           "spec": {"replicas": 1, "template": {
               "cliqueStartupType": c.STARTUP_EXPLICIT,
               "cliques": [
                   {"name": "a",
                    "annotations": {"grove.io/payload": "none"},
                    "spec": {"roleName": "a", "replicas": 1,
                             "podSpec": {"containers": [{"name": "m", "image": "i"}]}}},
                   {"name": "b",
                    "annotations": {"grove.io/payload": "none"},
                    "spec": {"roleName": "b", "replicas": 1, "startsAfter": ["a"],
                             "podSpec": {"containers": [{"name": "m", "image": "i"}]}}},
               ]}}}
    cluster.store.create(pcs)

    def all_done():
        pods = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "proc"})
        return (len(pods) == 2
                and all((p.get("status") or {}).get("phase") == "Succeeded"
                        for p in pods))
    cluster.wait_for(all_done, timeout=120, desc="both pod processes exited 0")
    # ordering: b's process could only finish after a was Ready (initc over HTTP)
    pods = {p["metadata"]["labels"][c.LABEL_PODCLIQUE]: p
            for p in cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "proc"})}
    assert pods["proc-0-a"]["status"]["phase"] == "Succeeded"
    assert pods["proc-0-b"]["status"]["phase"] == "Succeeded"
    # status rollup is eventually consistent — wait, don't read-after-write
    cluster.wait_pcs_available("proc", timeout=10)
