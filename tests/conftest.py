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


@pytest.fixture()
def cluster():
    from grove_amd import Cluster
    cl = Cluster(use_native_scheduler=None if _native_built() else False).start()
    yield cl
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
