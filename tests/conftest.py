import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REFERENCE = "/root/reference"


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires a real MI355X GPU")
    config.addinivalue_line("markers", "reference: requires the read-only reference tree")


def pytest_collection_modifyitems(config, items):
    import torch
    has_gpu = torch.cuda.is_available()
    has_ref = os.path.isdir(REFERENCE)
    for item in items:
        if "gpu" in item.keywords and not has_gpu:
            item.add_marker(pytest.mark.skip(reason="no GPU in this environment"))
        if "reference" in item.keywords and not has_ref:
            item.add_marker(pytest.mark.skip(reason="reference tree not mounted"))


@pytest.hookimpl(hookwrapper=True)
def pytest_runtest_makereport(item, call):
    out = yield
    rep = out.get_result()
    setattr(item, f"rep_{rep.when}", rep)


@pytest.fixture()
def cluster(request):
    from grove_amd import Cluster
    cl = Cluster(use_native_scheduler=None if _native_built() else False).start()
    yield cl
    # diagnostics collector (reference e2e/diagnostics/collector.go parity):
    # on failure, dump resource counts + unhealthy objects before teardown
    rep = getattr(request.node, "rep_call", None)
    if rep is not None and rep.failed:
        import sys
        print("\n=== cluster diagnostics (on failure) ===", file=sys.stderr)
        print("store:", cl.store.stats(), file=sys.stderr)
        for pg in cl.store.list("PodGang"):
            refs = {g.get("name"): [r.get("name") for r in
                                    g.get("podReferences") or []]
                    for g in (pg.get("spec") or {}).get("podgroups") or []}
            print(f"  PodGang {pg['metadata']['name']} podReferences: {refs}",
                  file=sys.stderr)
        for kind in ("PodCliqueSet", "PodClique", "PodCliqueScalingGroup", "PodGang"):
            for o in cl.store.list(kind):
                st = o.get("status") or {}
                print(f"  {kind} {o['metadata']['name']}: "
                      f"{ {k: v for k, v in st.items() if k != 'conditions'} } "
                      f"conds={[(cd.get('type'), cd.get('status')) for cd in st.get('conditions', [])]}",
                      file=sys.stderr)
        pods = cl.store.list("Pod")
        gated = sum(1 for p in pods if p["spec"].get("schedulingGates"))
        bound = sum(1 for p in pods if p["spec"].get("nodeName"))
        print(f"  pods={len(pods)} gated={gated} bound={bound}", file=sys.stderr)
        for ev in cl.store.events[-10:]:
            print("  event:", ev.get("reason"), ev.get("message"), file=sys.stderr)
        # PCSG write history: rv-ordered spec.replicas transitions (who stomped?)
        try:
            tbl = cl.store._table("PodCliqueScalingGroup")
            for (rv, ev2, o) in tbl.history[-40:]:
                print(f"  pcsg-history rv={rv} {ev2} {o['metadata']['name']} "
                      f"replicas={(o.get('spec') or {}).get('replicas')} "
                      f"gen={o['metadata'].get('generation')}",
                      file=sys.stderr)
        except Exception as e:
            print("  pcsg-history unavailable:", e, file=sys.stderr)
    cl.stop()


def _native_built() -> bool:
    try:
        from grove_amd.scheduler import _sched  # noqa: F401
        return True
    except Exception:
        return False


@pytest.fixture()
def simple1_yaml():
    return """
apiVersion: grove.io/v1alpha1
kind: PodCliqueSet
metadata:
  name: simple1
spec:
  replicas: 1
  template:
    cliques:
      - name: pca
        spec:
          roleName: rolea
          replicas: 3
          podSpec:
            containers:
              - name: pca
                image: nginx:latest
                resources: {requests: {cpu: 10m}}
          autoScalingConfig:
            maxReplicas: 5
      - name: pcb
        spec:
          roleName: roleb
          replicas: 2
          podSpec:
            containers:
              - name: pcb
                image: nginx:latest
                resources: {requests: {cpu: 10m}}
      - name: pcc
        spec:
          roleName: rolec
          replicas: 2
          podSpec:
            containers:
              - name: pcc
                image: nginx:latest
                resources: {requests: {cpu: 10m}}
      - name: pcd
        spec:
          roleName: roled
          replicas: 2
          podSpec:
            containers:
              - name: pcd
                image: nginx:latest
                resources: {requests: {cpu: 10m}}
    podCliqueScalingGroups:
      - name: sga
        cliqueNames: [pcb, pcc]
        scaleConfig:
          maxReplicas: 6
"""
