"""CRD rendering + installer.

Parity source: operator/cmd/install-crds + internal/crdinstaller/installer.go:18 (CRDs
embedded in the binary, applied via server-side apply) and the generated CRD YAML under
api/core/v1alpha1/crds/. The CRDs are rendered from the full structural schemas in
api/openapi.py (byte-compatible with the reference YAML modulo descriptions —
tests/test_crd_parity.py holds the diff empty) with the reference's printer columns,
scale/status subresources and short names; the installer applies them to a kube-style
apiserver (ours or a real one) over HTTP, or writes them to disk for `kubectl apply`.
"""
from __future__ import annotations

import json
from typing import Any, Dict, List

import yaml

from .openapi import schemas

FIELD_MANAGER = "grove-crd-installer"

# kind -> CRD identity (names contract: operator/api/core/v1alpha1/crds/*.yaml)
CRD_META: Dict[str, Dict[str, Any]] = {
    "PodCliqueSet": dict(
        group="grove.io", plural="podcliquesets", singular="podcliqueset",
        listKind="PodCliqueSetList", shortNames=["pcs"], scope="Namespaced"),
    "PodClique": dict(
        group="grove.io", plural="podcliques", singular="podclique",
        listKind="PodCliqueList", shortNames=["pclq"], scope="Namespaced"),
    "PodCliqueScalingGroup": dict(
        group="grove.io", plural="podcliquescalinggroups",
        singular="podcliquescalinggroup", listKind="PodCliqueScalingGroupList",
        shortNames=["pcsg"], scope="Namespaced"),
    "ClusterTopologyBinding": dict(
        group="grove.io", plural="clustertopologybindings",
        singular="clustertopologybinding", listKind="ClusterTopologyBindingList",
        shortNames=["ct"], scope="Cluster"),
    "PodGang": dict(
        group="scheduler.grove.io", plural="podgangs", singular="podgang",
        listKind="PodGangList", shortNames=["pg"], scope="Namespaced"),
}


def _col(name: str, path: str, type_: str = "integer", priority: int = 0) -> Dict:
    col: Dict[str, Any] = {"jsonPath": path, "name": name, "type": type_}
    if priority:
        col["priority"] = priority
    return col


_AGE = _col("Age", ".metadata.creationTimestamp", "date")

PRINTER_COLUMNS: Dict[str, List[Dict[str, Any]]] = {
    "PodCliqueSet": [
        _col("Replicas", ".spec.replicas"),
        _col("Available", ".status.availableReplicas"),
        _col("Updated", ".status.updatedReplicas"),
        _col("PCLQs-Updated", ".status.updateProgress.updatedPodCliquesCount"),
        _col("PCLQs-Total", ".status.updateProgress.totalPodCliquesCount"),
        _col("PCSGs-Updated",
             ".status.updateProgress.updatedPodCliqueScalingGroupsCount"),
        _col("PCSGs-Total",
             ".status.updateProgress.totalPodCliqueScalingGroupsCount"),
        _AGE,
    ],
    "PodClique": [
        _col("MinAvail", ".spec.minAvailable"),
        _col("Replicas", ".spec.replicas"),
        _col("Ready", ".status.readyReplicas"),
        _col("Scheduled", ".status.scheduledReplicas"),
        _col("Updated", ".status.updatedReplicas"),
        _col("Gated", ".status.scheduleGatedReplicas", priority=1),
        _col("MinBreached",
             '.status.conditions[?(@.type=="MinAvailableBreached")].status',
             "string", priority=1),
        _AGE,
    ],
    "PodCliqueScalingGroup": [
        _col("MinAvail", ".spec.minAvailable"),
        _col("Replicas", ".spec.replicas"),
        _col("Available", ".status.availableReplicas"),
        _col("Scheduled", ".status.scheduledReplicas"),
        _col("Updated", ".status.updatedReplicas"),
        _col("PCLQs-Updated", ".status.updateProgress.updatedPodCliquesCount"),
        _col("PCLQs-Total", ".status.updateProgress.totalPodCliquesCount"),
        _col("MinBreached",
             '.status.conditions[?(@.type=="MinAvailableBreached")].status',
             "string", priority=1),
        _AGE,
    ],
    "ClusterTopologyBinding": [
        _col("Domains", ".spec.levels[*].domain", "string"),
        _AGE,
    ],
    "PodGang": [
        _col("Phase", ".status.phase", "string"),
        _AGE,
    ],
}

# HPA /scale subresource wiring (reference subresources blocks)
SCALE_SUBRESOURCE: Dict[str, Dict[str, str]] = {
    "PodCliqueSet": {"labelSelectorPath": ".status.hpaPodSelector",
                     "specReplicasPath": ".spec.replicas",
                     "statusReplicasPath": ".status.replicas"},
    "PodClique": {"labelSelectorPath": ".status.hpaPodSelector",
                  "specReplicasPath": ".spec.replicas",
                  "statusReplicasPath": ".status.replicas"},
    "PodCliqueScalingGroup": {"labelSelectorPath": ".status.selector",
                              "specReplicasPath": ".spec.replicas",
                              "statusReplicasPath": ".status.replicas"},
}


def render_crd_for_kind(kind: str) -> Dict[str, Any]:
    meta = CRD_META[kind]
    subresources: Dict[str, Any] = {}
    if kind in SCALE_SUBRESOURCE:
        subresources["scale"] = dict(SCALE_SUBRESOURCE[kind])
    subresources["status"] = {}
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"{meta['plural']}.{meta['group']}"},
        "spec": {
            "group": meta["group"],
            "names": {"kind": kind, "listKind": meta["listKind"],
                      "plural": meta["plural"], "shortNames": meta["shortNames"],
                      "singular": meta["singular"]},
            "scope": meta["scope"],
            "versions": [{
                "additionalPrinterColumns": PRINTER_COLUMNS[kind],
                "name": "v1alpha1",
                "schema": {"openAPIV3Schema": schemas()[kind]},
                "served": True,
                "storage": True,
                "subresources": subresources,
            }],
        },
    }


def render_crd(crd_name: str) -> Dict[str, Any]:
    for kind, meta in CRD_META.items():
        if f"{meta['plural']}.{meta['group']}" == crd_name:
            return render_crd_for_kind(kind)
    raise KeyError(crd_name)


def render_all() -> List[Dict[str, Any]]:
    names = sorted(f"{m['plural']}.{m['group']}" for m in CRD_META.values())
    return [render_crd(n) for n in names]


class _NoAliasDumper(yaml.SafeDumper):
    def ignore_aliases(self, data):  # shared schema blocks must inline, not anchor
        return True


def write_crds(directory: str) -> List[str]:
    import os
    os.makedirs(directory, exist_ok=True)
    paths = []
    for crd in render_all():
        path = f"{directory}/{crd['metadata']['name']}.yaml"
        with open(path, "w") as f:
            f.write("# GENERATED FILE — do not edit. Rendered by grove_amd.api.crds\n"
                    "# (python -m grove_amd install-crds --output-dir crds) from the\n"
                    "# structural schemas in grove_amd/api/openapi.py; the embedded\n"
                    "# standard-Kubernetes subtrees are vendored controller-gen\n"
                    "# output (scripts/vendor_k8s_schemas.py). Byte-compatibility\n"
                    "# with the reference CRD schema is the contract\n"
                    "# (tests/test_crd_parity.py holds the semantic diff empty).\n"
                    "---\n")
            yaml.dump(crd, f, Dumper=_NoAliasDumper, sort_keys=True,
                      default_flow_style=False)
        paths.append(path)
    return paths


_SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


def install_crds(server_url: str) -> int:
    """Apply rendered CRDs to an apiextensions-speaking apiserver over HTTP.

    Running in-cluster (the install-crds init container, crdinstaller/installer.go
    parity) it authenticates with the pod's projected ServiceAccount token and
    trusts the cluster CA; AlreadyExists falls back to server-side-apply-style
    PUT so upgrades replace the schema (GREP-436)."""
    import os
    import ssl
    import urllib.error
    import urllib.request
    headers = {"Content-Type": "application/json"}
    ctx = None
    token_path = os.path.join(_SA_DIR, "token")
    if os.path.exists(token_path):
        with open(token_path) as f:
            headers["Authorization"] = f"Bearer {f.read().strip()}"
    ca_path = os.path.join(_SA_DIR, "ca.crt")
    if os.path.exists(ca_path):
        ctx = ssl.create_default_context(cafile=ca_path)

    def call(method: str, url: str, body) -> bool:
        req = urllib.request.Request(url, data=json.dumps(body).encode(),
                                     method=method, headers=headers)
        try:
            urllib.request.urlopen(req, timeout=15, context=ctx)
            return True
        except urllib.error.HTTPError as e:
            if method == "POST" and e.code == 409:
                return False  # exists — caller retries as update
            return False
        except Exception:
            return False

    base = f"{server_url}/apis/apiextensions.k8s.io/v1/customresourcedefinitions"
    n = 0
    for crd in render_all():
        name = crd["metadata"]["name"]
        if call("POST", f"{base}?fieldManager={FIELD_MANAGER}", crd):
            n += 1
            continue
        # update path: fetch current resourceVersion, then PUT the new schema
        try:
            with urllib.request.urlopen(
                    urllib.request.Request(f"{base}/{name}", headers=headers),
                    timeout=15, context=ctx) as r:
                cur = json.loads(r.read())
            if cur.get("spec") == crd["spec"]:
                n += 1  # unchanged — idempotent re-run, no write
                continue
            crd = dict(crd)
            crd["metadata"] = dict(crd["metadata"],
                                   resourceVersion=cur["metadata"]
                                   ["resourceVersion"])
            if call("PUT", f"{base}/{name}?fieldManager={FIELD_MANAGER}", crd):
                n += 1
        except Exception:
            pass
    return n
