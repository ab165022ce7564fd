"""Opaque content hashes for rolling-update change detection.

Parity role: reference computeGenerationHash (podcliqueset/reconcilespec.go:118) hashes pod
templates + priorityClassName; pod-template-hash (component/utils/podclique.go:200) hashes a
single clique's pod template. Values are opaque correlation tokens — only equality matters —
so we use sha256 over canonical JSON, truncated to 10 hex chars.
"""
from __future__ import annotations

import hashlib
import json
from typing import Any, Dict, List


def _digest(payload: Any) -> str:
    raw = json.dumps(payload, sort_keys=True, separators=(",", ":"), default=str)
    return hashlib.sha256(raw.encode()).hexdigest()[:10]


def pcs_generation_hash(pcs: Dict[str, Any]) -> str:
    # Labels/annotations of the clique template are part of the digest (reference
    # reconcilespec.go:118-132): a label-only template change must trigger a
    # rolling update.
    tmpl = (pcs.get("spec") or {}).get("template") or {}
    payload: List[Any] = [tmpl.get("priorityClassName", "")]
    for cl in tmpl.get("cliques") or []:
        payload.append({"name": cl.get("name"),
                        "labels": cl.get("labels") or {},
                        "annotations": cl.get("annotations") or {},
                        "podSpec": (cl.get("spec") or {}).get("podSpec")})
    return _digest(payload)


def pod_template_hash(clique_name: str, pod_spec: Dict[str, Any], priority_class: str = "",
                      labels: Dict[str, str] = None, annotations: Dict[str, str] = None) -> str:
    return _digest({"name": clique_name, "podSpec": pod_spec,
                    "priorityClassName": priority_class,
                    "labels": labels or {}, "annotations": annotations or {}})
