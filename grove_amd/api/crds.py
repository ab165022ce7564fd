"""CRD rendering + installer.

Parity source: operator/cmd/install-crds + internal/crdinstaller/installer.go:18 (CRDs
embedded in the binary, applied via server-side apply) and the generated CRD YAML under
api/core/v1alpha1/crds/. Here the CRDs are rendered from the declared schema
(api/schema.py) — structure-complete openAPIV3 schemas with printcolumns — and the
installer applies them to a kube-style apiserver (ours or a real one) over HTTP, or
writes them to disk for `kubectl apply`.
"""
from __future__ import annotations

import json
from typing import Any, Dict, List

import yaml

from . import constants as c
from .schema import SCHEMAS, PASSTHROUGH

FIELD_MANAGER = "grove-crd-installer"

_META = {
    "podcliquesets.grove.io": dict(group="grove.io", kind="PodCliqueSet",
                                   plural="podcliquesets", singular="podcliqueset",
                                   shortNames=["pcs"], scope="Namespaced"),
    "podcliques.grove.io": dict(group="grove.io", kind="PodClique",
                                plural="podcliques", singular="podclique",
                                shortNames=["pclq"], scope="Namespaced"),
    "podcliquescalinggroups.grove.io": dict(group="grove.io",
                                            kind="PodCliqueScalingGroup",
                                            plural="podcliquescalinggroups",
                                            singular="podcliquescalinggroup",
                                            shortNames=["pcsg"], scope="Namespaced"),
    "clustertopologybindings.grove.io": dict(group="grove.io",
                                             kind="ClusterTopologyBinding",
                                             plural="clustertopologybindings",
                                             singular="clustertopologybinding",
                                             shortNames=["ctb"], scope="Cluster"),
    "podgangs.scheduler.grove.io": dict(group="scheduler.grove.io", kind="PodGang",
                                        plural="podgangs", singular="podgang",
                                        shortNames=[], scope="Namespaced"),
}

_PRINTCOLUMNS = {
    "podcliquesets.grove.io": [
        {"name": "Replicas", "type": "integer", "jsonPath": ".spec.replicas"},
        {"name": "Available", "type": "integer",
         "jsonPath": ".status.availableReplicas"},
        {"name": "Age", "type": "date", "jsonPath": ".metadata.creationTimestamp"},
    ],
    "podcliques.grove.io": [
        {"name": "Replicas", "type": "integer", "jsonPath": ".spec.replicas"},
        {"name": "Ready", "type": "integer", "jsonPath": ".status.readyReplicas"},
        {"name": "Age", "type": "date", "jsonPath": ".metadata.creationTimestamp"},
    ],
    "podcliquescalinggroups.grove.io": [
        {"name": "Replicas", "type": "integer", "jsonPath": ".spec.replicas"},
        {"name": "Available", "type": "integer",
         "jsonPath": ".status.availableReplicas"},
    ],
    "clustertopologybindings.grove.io": [],
    "podgangs.scheduler.grove.io": [
        {"name": "Phase", "type": "string", "jsonPath": ".status.phase"},
        {"name": "Score", "type": "number", "jsonPath": ".status.placementScore"},
    ],
}


def _to_openapi(node) -> Dict[str, Any]:
    if node == PASSTHROUGH:
        return {"type": "object", "x-kubernetes-preserve-unknown-fields": True}
    if not node:
        return {"x-kubernetes-preserve-unknown-fields": True}
    props = {k: _to_openapi(v) for k, v in node.items()}
    return {"type": "object", "properties": props}


def render_crd(crd_name: str) -> Dict[str, Any]:
    meta = _META[crd_name]
    tree = dict(SCHEMAS[crd_name])
    schema = _to_openapi(tree)
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": crd_name},
        "spec": {
            "group": meta["group"],
            "names": {"kind": meta["kind"], "plural": meta["plural"],
                      "singular": meta["singular"],
                      "shortNames": meta["shortNames"]},
            "scope": meta["scope"],
            "versions": [{
                "name": "v1alpha1",
                "served": True,
                "storage": True,
                "schema": {"openAPIV3Schema": schema},
                "subresources": {"status": {}},
                "additionalPrinterColumns": _PRINTCOLUMNS[crd_name],
            }],
        },
    }


def render_all() -> List[Dict[str, Any]]:
    return [render_crd(n) for n in sorted(SCHEMAS)]


def write_crds(directory: str) -> List[str]:
    import os
    os.makedirs(directory, exist_ok=True)
    paths = []
    for crd in render_all():
        path = f"{directory}/{crd['metadata']['name']}.yaml"
        with open(path, "w") as f:
            yaml.safe_dump(crd, f, sort_keys=False)
        paths.append(path)
    return paths


def install_crds(server_url: str) -> int:
    """Apply rendered CRDs to an apiextensions-speaking apiserver over HTTP."""
    import urllib.request
    n = 0
    for crd in render_all():
        req = urllib.request.Request(
            f"{server_url}/apis/apiextensions.k8s.io/v1/customresourcedefinitions"
            f"?fieldManager={FIELD_MANAGER}",
            data=json.dumps(crd).encode(), method="POST",
            headers={"Content-Type": "application/json"})
        try:
            urllib.request.urlopen(req, timeout=10)
            n += 1
        except Exception:
            pass  # already exists / server applies its own merge
    return n
