"""HTTP apiserver + grove-initc waiter + operator config tests."""
import json
import threading
import time
import urllib.request

import pytest

from grove_amd.api import constants as c


@pytest.fixture()
def served_cluster(cluster):
    from grove_amd.kubecore.apiserver import ApiServer
    api = ApiServer(cluster.store, port=18133)
    api.start()
    yield cluster, api
    api.stop()


def _get(url):
    with urllib.request.urlopen(url, timeout=5) as r:
        return json.loads(r.read())


def test_apiserver_crud_and_selectors(served_cluster, simple1_yaml):
    cluster, api = served_cluster
    cluster.add_virtual_nodes(2)
    # create a PCS over HTTP
    import yaml as _yaml
    pcs = list(_yaml.safe_load_all(simple1_yaml))[0]
    req = urllib.request.Request(
        f"{api.url}/apis/grove.io/v1alpha1/namespaces/default/podcliquesets",
        data=json.dumps(pcs).encode(), method="POST",
        headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=5) as r:
        assert r.status == 201
    cluster.wait_pcs_available("simple1", timeout=20)
    # list pods with label selector over HTTP
    out = _get(f"{api.url}/api/v1/namespaces/default/pods"
               f"?labelSelector={c.LABEL_PODCLIQUE}=simple1-0-pca")
    assert len(out["items"]) == 3
    # get a single PCLQ
    q = _get(f"{api.url}/apis/grove.io/v1alpha1/namespaces/default/"
             f"podcliques/simple1-0-pca")
    assert q["status"]["readyReplicas"] == 3
    # cluster-scoped list
    nodes = _get(f"{api.url}/api/v1/nodes")
    assert len(nodes["items"]) == 2
    # health + metrics
    assert _get(f"{api.url}/healthz")["status"] == "ok"
    with urllib.request.urlopen(f"{api.url}/metrics", timeout=5) as r:
        assert b"grove_store_objects" in r.read()
    # 404 surface
    try:
        _get(f"{api.url}/apis/grove.io/v1alpha1/namespaces/default/podcliques/nope")
        assert False
    except urllib.error.HTTPError as e:
        assert e.code == 404


def test_initc_waits_for_parent_clique(served_cluster):
    cluster, api = served_cluster
    from grove_amd.initc import wait_for_parents
    cluster.add_virtual_nodes(1)
    pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
           "metadata": {"name": "ic"},
           "spec": {"replicas": 1, "template": {
               "cliqueStartupType": c.STARTUP_EXPLICIT,
               "cliques": [
                   {"name": "a", "spec": {"roleName": "a", "replicas": 2,
                                          "podSpec": {"containers": [{"name": "m",
                                                                     "image": "i"}]}}},
                   {"name": "b", "spec": {"roleName": "b", "startsAfter": ["a"],
                                          "podSpec": {"containers": [{"name": "m",
                                                                     "image": "i"}]}}},
               ]}}}
    results = {}

    def waiter():
        results["ok"] = wait_for_parents(
            "default", "ic-0", [("ic-0-a", 2)], server=api.url, timeout=20,
            poll=0.05)
    t = threading.Thread(target=waiter)
    t.start()
    time.sleep(0.2)
    assert "ok" not in results  # blocked before workload exists
    cluster.store.create(pcs)
    t.join(timeout=25)
    assert results.get("ok") is True


def test_initc_cli_parse():
    from grove_amd.initc import parse_podcliques
    assert parse_podcliques(["x-0-a:2", "x-0-b:1"]) == [("x-0-a", 2), ("x-0-b", 1)]
    with pytest.raises(ValueError):
        parse_podcliques(["nope"])


def test_operator_config_load(tmp_path):
    from grove_amd.config import load_configuration
    from grove_amd.kubecore.store import ApiError
    p = tmp_path / "cfg.yaml"
    p.write_text("""
client: {qps: 200, burst: 300}
controllers:
  podCliqueSet: {concurrentSyncs: 8}
servers:
  api: {enabled: true, port: 9999}
authorizer: {enabled: false}
network: {autoXGMIDomainEnabled: true}
scheduler: {default: amd-gang-scheduler}
logLevel: debug
""")
    cfg = load_configuration(str(p))
    assert cfg.client_qps == 200 and cfg.concurrent_syncs("podCliqueSet") == 8
    assert cfg.api_server.enabled and cfg.api_server.port == 9999
    assert not cfg.authorizer_enabled and cfg.auto_xgmi_domain_enabled
    # invalid scheduler rejected
    p.write_text("scheduler: {default: volcano}")
    with pytest.raises(ApiError):
        load_configuration(str(p))


def test_default_config():
    from grove_amd.config import default_configuration
    cfg = default_configuration()
    assert cfg.default_scheduler == c.SCHEDULER_AMD_GANG
    assert cfg.authorizer_enabled and cfg.topology_aware_scheduling_enabled


def test_watch_stream_ndjson(served_cluster):
    cluster, api = served_cluster
    import threading
    import urllib.request
    events = []

    def consume():
        req = urllib.request.urlopen(
            f"{api.url}/api/v1/namespaces/default/pods?watch=true&seed=false",
            timeout=10)
        for raw in req:
            line = raw.decode().strip()
            if line:
                events.append(json.loads(line))
            if len(events) >= 2:
                return
    t = threading.Thread(target=consume, daemon=True)
    t.start()
    time.sleep(0.3)
    cluster.store.create({"kind": "Pod", "metadata": {"name": "w1"},
                          "spec": {"containers": []}})
    cluster.store.patch("Pod", "default", "w1",
                        lambda o: o["metadata"].setdefault("labels", {}).update(x="1"))
    t.join(timeout=10)
    assert len(events) >= 2
    assert events[0]["type"] == "ADDED"
    assert events[0]["object"]["metadata"]["name"] == "w1"
    assert events[1]["type"] == "MODIFIED"


def test_tls_apiserver(cluster, tmp_path):
    import ssl
    import urllib.request
    from grove_amd.kubecore.apiserver import ApiServer
    from grove_amd.kubecore.certs import (ensure_cert_secret, write_cert_files)
    data = ensure_cert_secret(cluster.store)
    crt, key = write_cert_files(data, str(tmp_path / "tls"))
    api = ApiServer(cluster.store, port=18233, ssl_certfile=crt, ssl_keyfile=key)
    api.start()
    try:
        assert api.url == "https://127.0.0.1:18233"  # TLS-aware url property
        ctx = ssl.create_default_context(cafile=crt)
        with urllib.request.urlopen("https://localhost:18233/healthz",
                                    context=ctx, timeout=5) as r:
            assert json.loads(r.read())["status"] == "ok"
    finally:
        api.stop()


def test_default_scheduler_backend_no_gang(simple1_yaml):
    """kube backend parity: pods carry default-scheduler, no gang admission — pods are
    scheduled individually as they ungate; gates still serialize startup."""
    from grove_amd import Cluster
    from grove_amd.api import constants as c
    cl = Cluster(scheduler_name=c.SCHEDULER_DEFAULT, use_native_scheduler=False).start()
    try:
        cl.add_virtual_nodes(2)
        cl.apply(simple1_yaml)
        cl.wait_pcs_available("simple1", timeout=30)
        pods = cl.store.list("Pod", "default", {c.LABEL_PART_OF: "simple1"})
        assert all(p["spec"]["schedulerName"] == c.SCHEDULER_DEFAULT for p in pods)
        # PodGangs exist (contract objects) but no gang placement score was set
        pg = cl.store.get(c.KIND_PODGANG, "default", "simple1-0")
        assert pg["metadata"]["labels"][c.LABEL_SCHEDULER_NAME] == c.SCHEDULER_DEFAULT
    finally:
        cl.stop()


def test_debug_profile_endpoint(served_cluster, simple1_yaml):
    cluster, api = served_cluster
    cluster.add_virtual_nodes(1)
    cluster.apply(simple1_yaml)
    with urllib.request.urlopen(
            f"{api.url}/debug/profile?seconds=0.5&interval_ms=5", timeout=10) as r:
        body = r.read().decode()
    # collapsed-stack lines: "frame;frame;... count"
    assert body.strip() and all(
        line.rsplit(" ", 1)[-1].isdigit() for line in body.strip().splitlines())


def test_httpclient_roundtrip_and_watch(served_cluster):
    """HttpStoreClient (the client-go analog): CRUD + patch + ndjson watch."""
    cluster, api = served_cluster
    from grove_amd.kubecore.httpclient import HttpStoreClient
    from grove_amd.kubecore.store import ApiError
    cl = HttpStoreClient(api.url)
    node = cl.create({"kind": "Node", "metadata": {"name": "hc-n0"},
                      "spec": {}, "status": {"allocatable": {"cpu": "4"}}})
    assert node["metadata"]["uid"]
    assert cl.get("Node", None, "hc-n0")["metadata"]["name"] == "hc-n0"
    got = cl.list("Node", None)
    assert any(n["metadata"]["name"] == "hc-n0" for n in got)
    cl.patch("Node", None, "hc-n0",
             lambda o: o["metadata"].setdefault("labels", {}).update(zone="z1"))
    assert cl.get("Node", None, "hc-n0")["metadata"]["labels"]["zone"] == "z1"
    # watch stream sees a subsequent event
    events = []
    import threading

    def consume():
        for ev, obj in cl.watch_events("Pod", "default", seed=False):
            events.append((ev, obj["metadata"]["name"]))
            return
    t = threading.Thread(target=consume, daemon=True)
    t.start()
    time.sleep(0.3)
    cl.create({"kind": "Pod", "metadata": {"name": "hc-p0", "namespace": "default"},
               "spec": {"containers": []}})
    t.join(timeout=10)
    assert events == [("ADDED", "hc-p0")]
    cl.delete("Pod", "default", "hc-p0")
    with pytest.raises(ApiError):
        cl.get("Pod", "default", "hc-p0")


def test_cli_apply_get_roundtrip(served_cluster, simple1_yaml, tmp_path, capsys):
    """The kubectl-analog CLI surface: `grove-amd apply -f` then `get` (table/yaml/
    json) against the running apiserver."""
    from grove_amd.__main__ import main

    cluster, api = served_cluster
    f = tmp_path / "pcs.yaml"
    f.write_text(simple1_yaml)
    server = "http://127.0.0.1:18133"
    assert main(["apply", "-f", str(f), "--server", server]) == 0
    out = capsys.readouterr().out
    assert "created PodCliqueSet/simple1" in out
    # re-apply = kubectl-apply create-or-patch: second pass PATCHes and reports
    # "configured" instead of erroring on 409
    assert main(["apply", "-f", str(f), "--server", server]) == 0
    assert "configured PodCliqueSet/simple1" in capsys.readouterr().out

    deadline = time.time() + 30
    while time.time() < deadline:
        if cluster.store.list(c.KIND_PCLQ, namespace="default"):
            break
        time.sleep(0.05)

    assert main(["get", "podcliquesets", "--server", server]) == 0
    out = capsys.readouterr().out
    assert "simple1" in out and "NAME" in out
    assert main(["get", "podcliques", "--server", server, "-o", "json"]) == 0
    items = json.loads(capsys.readouterr().out)
    assert any(o["metadata"]["name"].startswith("simple1-0-") for o in items)
    assert main(["get", "podcliquesets", "simple1", "--server", server,
                 "-o", "yaml"]) == 0
    assert "kind: PodCliqueSet" in capsys.readouterr().out
    # unknown resource → error exit
    assert main(["get", "bogus", "--server", server]) == 1
    capsys.readouterr()


def test_cli_version_and_crd_render(capsys):
    from grove_amd.__main__ import main
    assert main(["version"]) == 0
    assert "grove-amd" in capsys.readouterr().out
    assert main(["install-crds"]) == 0
    out = capsys.readouterr().out
    assert "podcliquesets.grove.io" in out and "podgangs.scheduler.grove.io" in out


def test_api_discovery(served_cluster):
    """client-go discovery analog: /apis lists both grove groups; the group-version
    endpoint returns an APIResourceList with namespaced flags and verbs."""
    groups = _get("http://127.0.0.1:18133/apis")
    names = {g["name"] for g in groups["groups"]}
    assert {"grove.io", "scheduler.grove.io"} <= names
    rl = _get("http://127.0.0.1:18133/apis/grove.io/v1alpha1")
    assert rl["kind"] == "APIResourceList"
    by_name = {r["name"]: r for r in rl["resources"]}
    assert by_name["podcliquesets"]["namespaced"] is True
    assert by_name["clustertopologybindings"]["namespaced"] is False
    assert "watch" in by_name["podcliques"]["verbs"]
    assert "podcliquesets/status" in by_name
    rl2 = _get("http://127.0.0.1:18133/apis/scheduler.grove.io/v1alpha1")
    assert any(r["name"] == "podgangs" for r in rl2["resources"])


def test_http_merge_patch(served_cluster):
    """kubectl patch --type=merge analog: recursive merge, null deletes."""
    import urllib.request
    cluster, api = served_cluster
    cluster.store.create({"apiVersion": "grove.io/v1alpha1", "kind": c.KIND_PCLQ,
                          "metadata": {"name": "mp", "namespace": "default",
                                       "labels": {"keep": "a", "drop": "b"}},
                          "spec": {"roleName": "r", "replicas": 1,
                                   "podSpec": {"containers": []}}})
    patch = {"metadata": {"labels": {"drop": None, "new": "c"}},
             "spec": {"replicas": 3}}
    req = urllib.request.Request(
        "http://127.0.0.1:18133/apis/grove.io/v1alpha1/namespaces/default/"
        "podcliques/mp", data=json.dumps(patch).encode(), method="PATCH",
        headers={"Content-Type": "application/merge-patch+json"})
    with urllib.request.urlopen(req, timeout=5) as r:
        out = json.loads(r.read())
    assert out["spec"]["replicas"] == 3
    assert out["metadata"]["labels"] == {"keep": "a", "new": "c"}
    assert out["spec"]["roleName"] == "r"  # untouched fields survive
    # client-side single-round-trip analog
    from grove_amd.kubecore.httpclient import HttpStoreClient
    cli = HttpStoreClient("http://127.0.0.1:18133")
    out2 = cli.merge_patch(c.KIND_PCLQ, "default", "mp", {"spec": {"replicas": 5}})
    assert out2["spec"]["replicas"] == 5


def test_cli_get_watch(served_cluster, tmp_path):
    """`grove-amd get -w` streams watch events from the apiserver."""
    import subprocess
    import sys
    cluster, api = served_cluster
    proc = subprocess.Popen(
        [sys.executable, "-m", "grove_amd", "get", "podcliques",
         "--server", "http://127.0.0.1:18133", "-w"],
        stdout=subprocess.PIPE, text=True)
    try:
        time.sleep(0.3)
        cluster.store.create({"apiVersion": "grove.io/v1alpha1",
                              "kind": c.KIND_PCLQ,
                              "metadata": {"name": "wq", "namespace": "default"},
                              "spec": {"roleName": "r", "replicas": 1,
                                       "podSpec": {"containers": []}}})
        deadline = time.time() + 10
        line = ""
        while time.time() < deadline:
            line = proc.stdout.readline()
            if "wq" in line:
                break
        assert "ADDED" in line and "wq" in line
    finally:
        proc.kill()
        proc.wait()


def test_metrics_include_controller_counters(served_cluster):
    """/metrics serves store gauges plus the Cluster's controller counters when the
    operator wires metrics_fn (manager.go metrics-endpoint parity)."""
    from grove_amd.kubecore.apiserver import ApiServer
    cluster, _ = served_cluster
    api2 = ApiServer(cluster.store, port=18134, metrics_fn=cluster.metrics_lines)
    api2.start()
    try:
        import urllib.request
        with urllib.request.urlopen("http://127.0.0.1:18134/metrics", timeout=5) as r:
            body = r.read().decode()
        assert "grove_store_objects" in body
        assert 'grove_reconcile_total{controller="podcliqueset"}' in body
    finally:
        api2.stop()


def test_initc_in_process_transport(cluster):
    """initc without --server: the attach() in-process Store transport."""
    from grove_amd import initc

    initc.attach(cluster.store)
    try:
        cluster.store.create({
            "apiVersion": "v1", "kind": "Pod",
            "metadata": {"name": "ip-0", "namespace": "default",
                         "labels": {c.LABEL_PODGANG: "ipg-0",
                                    c.LABEL_PODCLIQUE: "ipg-0-a"}},
            "spec": {"containers": []},
            "status": {"phase": "Running",
                       "conditions": [{"type": "Ready", "status": "True"}]}})
        assert initc.wait_for_parents(
            "default", "ipg-0", [("ipg-0-a", 1)], server=None, timeout=5)
    finally:
        initc.attach(None)


def test_http_auth_identity(served_cluster, simple1_yaml):
    """Wire auth: unauthenticated HTTP callers run as system:anonymous and the
    Authorizer rejects their mutations of grove-managed resources; a bearer token
    matching an SA-token Secret maps to that service account; a static token maps
    to its configured user (the wire surface must not default to the
    operator identity)."""
    import urllib.error
    import yaml as _yaml
    cluster, api = served_cluster
    cluster.add_virtual_nodes(2)
    cluster.apply(simple1_yaml)
    cluster.wait_pcs_available("simple1", timeout=20)
    q = cluster.store.get(c.KIND_PCLQ, "default", "simple1-0-pca")
    # anonymous PUT of a managed PodClique -> 403
    q2 = json.loads(json.dumps(q))
    q2["metadata"]["labels"]["tamper"] = "yes"
    req = urllib.request.Request(
        f"{api.url}/apis/grove.io/v1alpha1/namespaces/default/podcliques/"
        f"simple1-0-pca", data=json.dumps(q2).encode(), method="PUT",
        headers={"Content-Type": "application/json"})
    try:
        urllib.request.urlopen(req, timeout=5)
        assert False, "anonymous mutation of a managed resource must be rejected"
    except urllib.error.HTTPError as e:
        assert e.code == 403
    # anonymous DELETE of a managed pod -> 403
    pod = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "simple1-0-pca"})[0]
    req = urllib.request.Request(
        f"{api.url}/api/v1/namespaces/default/pods/{pod['metadata']['name']}",
        method="DELETE")
    try:
        urllib.request.urlopen(req, timeout=5)
        assert False
    except urllib.error.HTTPError as e:
        assert e.code == 403
    # SA-token secret auth: the <pcs>-ic-sat secret's token authenticates as the SA
    from grove_amd.api import namegen
    sec_name = namegen.initc_sa_token_secret_name("simple1")
    cluster.store.patch("Secret", "default", sec_name,
                        lambda o: o.setdefault("stringData", {}).update(
                            {"token": "sa-tok-123"}))
    req = urllib.request.Request(
        f"{api.url}/apis/grove.io/v1alpha1/namespaces/default/podcliques/"
        f"simple1-0-pca", data=json.dumps(q2).encode(), method="PUT",
        headers={"Content-Type": "application/json",
                 "Authorization": "Bearer sa-tok-123"})
    try:
        urllib.request.urlopen(req, timeout=5)
        assert False, "the pod SA is not exempt either"
    except urllib.error.HTTPError as e:
        assert e.code in (403, 409)  # authenticated but not exempt (or RV conflict)
    # static-token auth as the node agent (exempt) succeeds
    from grove_amd.kubecore.apiserver import ApiServer
    from grove_amd.kubecore.identity import NODE_AGENT_USER
    api2 = ApiServer(cluster.store, port=18135,
                     auth_tokens={"agent-tok": NODE_AGENT_USER})
    api2.start()
    try:
        cur = cluster.store.get(c.KIND_PCLQ, "default", "simple1-0-pca")
        cur["metadata"]["labels"]["tamper"] = "agent"
        req = urllib.request.Request(
            f"{api2.url}/apis/grove.io/v1alpha1/namespaces/default/podcliques/"
            f"simple1-0-pca", data=json.dumps(cur).encode(), method="PUT",
            headers={"Content-Type": "application/json",
                     "Authorization": "Bearer agent-tok"})
        with urllib.request.urlopen(req, timeout=5) as r:
            assert r.status == 200
    finally:
        api2.stop()


def test_lpx_backend_rejects_topology_constraints():
    """lpx backend parity (scheduler/lpx/backend.go:68-86): schedulerName-only
    dispatch; workloads with topology constraints are rejected at validation."""
    from grove_amd.kubecore.store import Store, ApiError
    from grove_amd.scheduler.backends import Registry
    st = Store()
    reg = Registry(st, default="lpx-scheduler")
    lpx = reg.get("lpx-scheduler")
    pod = {"spec": {}}
    lpx.prepare_pod(pod)
    assert pod["spec"]["schedulerName"] == "lpx-scheduler"
    ok_pcs = {"spec": {"template": {"cliques": [{"name": "a", "spec": {}}]}}}
    lpx.validate_podcliqueset(ok_pcs)  # no constraints -> fine
    bad = {"spec": {"template": {
        "cliques": [{"name": "a", "spec": {},
                     "topologyConstraint": {"pack": {"required": "rack"}}}]}}}
    with pytest.raises(ApiError, match="not supported by the lpx-scheduler"):
        lpx.validate_podcliqueset(bad)
    # config accepts the lpx profile
    from grove_amd.config import load_configuration
    import tempfile
    with tempfile.NamedTemporaryFile("w", suffix=".yaml", delete=False) as f:
        f.write("scheduler: {default: lpx-scheduler, "
                "profiles: [lpx-scheduler, amd-gang-scheduler]}\n")
        path = f.name
    cfg = load_configuration(path)
    assert cfg.default_scheduler == "lpx-scheduler"
