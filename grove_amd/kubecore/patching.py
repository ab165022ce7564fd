"""Patch strategies: RFC 7386 JSON merge patch + strategic merge patch.

Parity role: kubectl/client-go Patch(types.MergePatchType) and
Patch(types.StrategicMergePatchType). Strategic merge differs from plain merge on
LISTS: fields with a patchMergeKey (containers by name, env by name, volumeMounts by
mountPath, ...) merge per-element instead of being replaced wholesale, and `$patch:
delete` / `$patch: replace` directives control element/collection behavior — the
semantics kubectl relies on to edit one container of a pod without clobbering the
rest.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional

# field name -> merge key (the corev1 patchMergeKey tags that matter for pod specs)
MERGE_KEYS: Dict[str, str] = {
    "containers": "name",
    "initContainers": "name",
    "ephemeralContainers": "name",
    "volumes": "name",
    "env": "name",
    "envFrom": "prefix",
    "volumeMounts": "mountPath",
    "volumeDevices": "devicePath",
    "ports": "containerPort",
    "hostAliases": "ip",
    "imagePullSecrets": "name",
    "schedulingGates": "name",
    "readinessGates": "conditionType",
    "resourceClaims": "name",
    "ownerReferences": "uid",
    "finalizers": None,  # set-style: union  # type: ignore[dict-item]
    "conditions": "type",
    "podgroups": "name",
    "cliques": "name",
}


def json_merge_patch(target: Any, patch: Any) -> Any:
    """RFC 7386: null deletes a key; objects merge recursively; else replace."""
    if not isinstance(patch, dict) or not isinstance(target, dict):
        return patch
    for k, v in patch.items():
        if v is None:
            target.pop(k, None)
        else:
            target[k] = json_merge_patch(target.get(k), v)
    return target


def strategic_merge_patch(target: Any, patch: Any,
                          field_name: Optional[str] = None) -> Any:
    if isinstance(patch, dict):
        directive = patch.get("$patch")
        if directive == "replace":
            out = {k: v for k, v in patch.items() if k != "$patch"}
            return out
        if directive == "delete":
            return None  # caller removes
        if not isinstance(target, dict):
            target = {}
        for k, v in patch.items():
            if k == "$patch":
                continue
            if v is None:
                target.pop(k, None)
                continue
            merged = strategic_merge_patch(target.get(k), v, field_name=k)
            if merged is None:
                target.pop(k, None)
            else:
                target[k] = merged
        return target
    if isinstance(patch, list):
        key = MERGE_KEYS.get(field_name or "")
        if key is None and field_name == "finalizers":
            base = list(target) if isinstance(target, list) else []
            for item in patch:
                if item not in base:
                    base.append(item)
            return base
        if key is None or not all(isinstance(i, dict) for i in patch):
            return patch  # atomic list: replace
        # merge-by-key list
        base: List[Dict[str, Any]] = [dict(i) for i in target] \
            if isinstance(target, list) else []
        replace_all = any(i.get("$patch") == "replace" for i in patch
                          if isinstance(i, dict) and len(i) == 1)
        if replace_all:
            return [i for i in patch
                    if not (isinstance(i, dict) and i.get("$patch") == "replace"
                            and len(i) == 1)]
        for item in patch:
            if item.get("$patch") == "delete":
                base = [b for b in base if b.get(key) != item.get(key)]
                continue
            for i, b in enumerate(base):
                if b.get(key) == item.get(key) and item.get(key) is not None:
                    base[i] = strategic_merge_patch(b, item, field_name=None)
                    break
            else:
                base.append(item)
        return base
    return patch
