"""Structural OpenAPI v3 schemas for the five grove CRDs — the byte-compatible
kubectl/CRD surface (BASELINE.json north star).

Hand-declared from the reference API contracts (behavior specs:
operator/api/core/v1alpha1/{podcliqueset,podclique,scalinggroup,
clustertopologybinding}.go and scheduler/api/core/v1alpha1/podgang.go, as rendered by
controller-gen into operator/api/core/v1alpha1/crds/*.yaml). The embedded
standard-Kubernetes type schemas (corev1.PodSpec, autoscaling/v2 MetricSpec,
metav1.Condition, resource.k8s.io claim templates) are VENDORED mechanical codegen
output — see scripts/vendor_k8s_schemas.py — loaded from _k8s_openapi.json.

tests/test_crd_parity.py diffs the rendered CRDs against the reference YAML
normalized (descriptions ignored); grove_amd/api/schemavalidate.py enforces these
schemas server-side in store admission.
"""
from __future__ import annotations

import copy
import json
import os
from typing import Any, Dict, List, Optional

Schema = Dict[str, Any]

_VENDOR_PATH = os.path.join(os.path.dirname(__file__), "_k8s_openapi.json")
with open(_VENDOR_PATH) as _f:
    _VENDORED: Dict[str, Schema] = json.load(_f)


def vendor(name: str) -> Schema:
    return copy.deepcopy(_VENDORED[name])


# ---------------------------------------------------------------- schema helpers
STR: Schema = {"type": "string"}


def int32(default: Optional[int] = None) -> Schema:
    s: Schema = {"format": "int32", "type": "integer"}
    if default is not None:
        s["default"] = default
    return s


def int64() -> Schema:
    return {"format": "int64", "type": "integer"}


def date_time() -> Schema:
    return {"format": "date-time", "type": "string"}


def enum(*values: str, default: Optional[str] = None) -> Schema:
    s: Schema = {"enum": list(values), "type": "string"}
    if default is not None:
        s["default"] = default
    return s


def obj(props: Dict[str, Schema], required: Optional[List[str]] = None,
        **extra: Any) -> Schema:
    s: Schema = {"properties": props, "type": "object"}
    if required:
        s["required"] = sorted(required)
    s.update(extra)
    return s


def arr(items: Schema, **extra: Any) -> Schema:
    s: Schema = {"items": items, "type": "array"}
    s.update(extra)
    return s


def str_arr() -> Schema:
    return arr({"type": "string"})


def map_str() -> Schema:
    return {"additionalProperties": {"type": "string"}, "type": "object"}


def conditions() -> Schema:
    return arr(vendor("metav1_condition"))


def last_errors() -> Schema:
    # groveerr coded error record surfaced in every status (errors.go contract)
    return arr(obj({
        "code": dict(STR),
        "description": dict(STR),
        "observedAt": date_time(),
    }, required=["code", "description", "observedAt"]))


# ---------------------------------------------------------------- shared grove blocks
_DOMAIN_NAME: Schema = {"maxLength": 63, "minLength": 1,
                        "pattern": "^[a-z][a-z0-9-]*$", "type": "string"}


def topology_constraint() -> Schema:
    """Operator-side TopologyConstraint (podcliqueset.go:269-338): pack {required|
    preferred} with CEL requiredness, deprecated packDomain, topologyName."""
    pack = obj({
        "preferred": dict(_DOMAIN_NAME),
        "required": dict(_DOMAIN_NAME),
    })
    pack["x-kubernetes-validations"] = [{
        "message": "pack must specify at least one of required or preferred",
        "reason": "FieldValueRequired",
        "rule": "has(self.required) || has(self.preferred)",
    }]
    tc = obj({
        "pack": pack,
        "packDomain": dict(_DOMAIN_NAME),
        "topologyName": dict(STR),
    })
    tc["x-kubernetes-validations"] = [
        {"fieldPath": ".pack",
         "message": "topologyConstraint must specify pack or deprecated packDomain",
         "reason": "FieldValueRequired",
         "rule": "has(self.pack) || has(self.packDomain)"},
        {"fieldPath": ".pack.required",
         "message": "must not set both pack.required and deprecated packDomain",
         "rule": "!(has(self.packDomain) && has(self.pack) && "
                 "has(self.pack.required))"},
    ]
    return tc


def scale_config() -> Schema:
    """AutoScalingConfig / ScaleConfig (podcliqueset.go HPA config): bounds +
    autoscaling/v2 metrics (vendored)."""
    return obj({
        "maxReplicas": int32(),
        "metrics": vendor("hpa_metrics"),
        "minReplicas": int32(),
    }, required=["maxReplicas"])


def resource_sharing(child_filters: List[str]) -> Schema:
    """ResourceSharing entry (resourcesharing.go): DRA claim shared at PCS/PCSG/
    clique scope; filter child keys vary by scope."""
    props: Dict[str, Schema] = {
        "name": dict(STR),
        "namespace": dict(STR),
        "scope": enum("AllReplicas", "PerReplica"),
    }
    if child_filters:
        props["filter"] = obj({k: str_arr() for k in child_filters})
    # field order in properties is canonicalized at render; declare filter with rest
    return arr(obj(props, required=["name", "scope"]))


def podclique_spec() -> Schema:
    """PodCliqueSpec (podclique.go:60-110) — shared verbatim by the PCS template's
    cliques[].spec and the PodClique CRD's spec."""
    return obj({
        "autoScalingConfig": scale_config(),
        "minAvailable": int32(),
        "podSpec": vendor("podspec"),
        "replicas": int32(),
        "roleName": dict(STR),
        "startsAfter": str_arr(),
    }, required=["podSpec", "replicas", "roleName"])


# ---------------------------------------------------------------- PodCliqueSet
def podcliqueset_schema() -> Schema:
    clique_template = obj({
        "annotations": map_str(),
        "labels": map_str(),
        "name": dict(STR),
        "resourceSharing": resource_sharing([]),
        "spec": podclique_spec(),
        "topologyConstraint": topology_constraint(),
    }, required=["name", "spec"])

    scaling_group_config = obj({
        "annotations": map_str(),
        "cliqueNames": str_arr(),
        "minAvailable": int32(default=1),
        "name": dict(STR),
        "replicas": int32(default=1),
        "resourceSharing": resource_sharing(["childCliqueNames"]),
        "scaleConfig": scale_config(),
        "topologyConstraint": topology_constraint(),
    }, required=["cliqueNames", "name"])

    template = obj({
        "cliqueStartupType": enum(
            "CliqueStartupTypeAnyOrder", "CliqueStartupTypeInOrder",
            "CliqueStartupTypeExplicit", default="CliqueStartupTypeAnyOrder"),
        "cliques": arr(clique_template, **{
            "x-kubernetes-list-map-keys": ["name"],
            "x-kubernetes-list-type": "map"}),
        "headlessServiceConfig": obj(
            {"publishNotReadyAddresses": {"default": True, "type": "boolean"}},
            required=["publishNotReadyAddresses"]),
        "podCliqueScalingGroups": arr(scaling_group_config),
        "priorityClassName": dict(STR),
        "resourceClaimTemplates": arr(vendor("resourceclaim_template")),
        "resourceSharing": resource_sharing(
            ["childCliqueNames", "childScalingGroupNames"]),
        "terminationDelay": dict(STR),
        "topologyConstraint": topology_constraint(),
    }, required=["cliques"])

    spec = obj({
        "replicas": int32(default=0),
        "template": template,
    }, required=["template"])

    update_progress = obj({
        "currentlyUpdating": arr(obj({
            "replicaIndex": int32(),
            "updateEndedAt": date_time(),
            "updateStartedAt": date_time(),
        }, required=["replicaIndex"])),
        "totalPodCliqueScalingGroupsCount": int32(default=0),
        "totalPodCliquesCount": int32(default=0),
        "updateEndedAt": date_time(),
        "updateStartedAt": date_time(),
        "updatedPodCliqueScalingGroupsCount": int32(default=0),
        "updatedPodCliquesCount": int32(default=0),
    })

    status = obj({
        "availableReplicas": int32(default=0),
        "conditions": conditions(),
        "currentGenerationHash": dict(STR),
        "hpaPodSelector": dict(STR),
        "lastErrors": last_errors(),
        "observedGeneration": int64(),
        "podGangStatuses": arr(obj({
            "conditions": conditions(),
            "name": dict(STR),
            "phase": enum("Pending", "Starting", "Running", "Failed", "Succeeded"),
        }, required=["name", "phase"])),
        "replicas": int32(),
        "updateProgress": update_progress,
        "updatedReplicas": int32(default=0),
    }, required=["availableReplicas", "updatedReplicas"])

    root = _root(spec, status)
    # packDomain deprecation ratchet (podcliqueset.go:36-38): existing objects may
    # keep packDomain, new objects must use pack.required
    dep = ("packDomain is deprecated and cannot be used on new workloads; "
           "use pack.required")
    root["x-kubernetes-validations"] = [
        {"fieldPath": ".spec.template.topologyConstraint.packDomain",
         "message": dep, "optionalOldSelf": True, "reason": "FieldValueForbidden",
         "rule": "oldSelf.hasValue() || !has(self.spec.template.topologyConstraint)"
                 " || !has(self.spec.template.topologyConstraint.packDomain)"},
        {"fieldPath": ".spec.template.cliques",
         "message": dep, "optionalOldSelf": True, "reason": "FieldValueForbidden",
         "rule": "oldSelf.hasValue() || !has(self.spec.template.cliques) || "
                 "self.spec.template.cliques.all(c, !has(c.topologyConstraint) || "
                 "!has(c.topologyConstraint.packDomain))"},
        {"fieldPath": ".spec.template.podCliqueScalingGroups",
         "message": dep, "optionalOldSelf": True, "reason": "FieldValueForbidden",
         "rule": "oldSelf.hasValue() || "
                 "!has(self.spec.template.podCliqueScalingGroups) || "
                 "self.spec.template.podCliqueScalingGroups.all(g, "
                 "!has(g.topologyConstraint) || "
                 "!has(g.topologyConstraint.packDomain))"},
    ]
    root["properties"]["spec"]["properties"]["updateStrategy"] = obj({
        "type": enum("RollingRecreate", "OnDelete", default="RollingRecreate"),
    })
    return root


# ---------------------------------------------------------------- PodClique
def podclique_schema() -> Schema:
    status = obj({
        "conditions": conditions(),
        "currentPodCliqueSetGenerationHash": dict(STR),
        "currentPodTemplateHash": dict(STR),
        "hpaPodSelector": dict(STR),
        "lastErrors": last_errors(),
        "observedGeneration": int64(),
        "readyReplicas": int32(default=0),
        "replicas": int32(),
        "scheduleGatedReplicas": int32(default=0),
        "scheduledReplicas": int32(default=0),
        "updateProgress": obj({
            "podCliqueSetGenerationHash": dict(STR),
            "podTemplateHash": dict(STR),
            "readyPodsSelectedToUpdate": obj({
                "completed": str_arr(),
                "current": dict(STR),
            }, required=["current"]),
            "updateEndedAt": date_time(),
            "updateStartedAt": date_time(),
        }, required=["podCliqueSetGenerationHash", "podTemplateHash"]),
        "updatedReplicas": int32(default=0),
    }, required=["readyReplicas", "scheduleGatedReplicas", "scheduledReplicas",
                 "updatedReplicas"])
    return _root(podclique_spec(), status)


# ---------------------------------------------------------------- PCSG
def podcliquescalinggroup_schema() -> Schema:
    spec = obj({
        "cliqueNames": str_arr(),
        "minAvailable": int32(default=1),
        "replicas": int32(default=1),
    }, required=["cliqueNames", "replicas"])
    status = obj({
        "availableReplicas": int32(default=0),
        "conditions": conditions(),
        "currentPodCliqueSetGenerationHash": dict(STR),
        "lastErrors": last_errors(),
        "observedGeneration": int64(),
        "replicas": int32(),
        "scheduledReplicas": int32(default=0),
        "selector": dict(STR),
        "updateProgress": obj({
            "podCliqueSetGenerationHash": dict(STR),
            "readyReplicaIndicesSelectedToUpdate": obj({
                "completed": arr(int32()),
                "current": int32(),
            }, required=["current"]),
            "totalPodCliquesCount": int32(default=0),
            "updateEndedAt": date_time(),
            "updateStartedAt": date_time(),
            "updatedPodCliquesCount": int32(default=0),
        }, required=["podCliqueSetGenerationHash", "updateStartedAt"]),
        "updatedReplicas": int32(default=0),
    }, required=["availableReplicas", "scheduledReplicas", "updatedReplicas"])
    return _root(spec, status)


# ---------------------------------------------------------------- CTB
def clustertopologybinding_schema() -> Schema:
    label_key: Schema = {
        "maxLength": 63, "minLength": 1,
        "pattern": "^(([A-Za-z0-9][-A-Za-z0-9_.]*)?[A-Za-z0-9]/)?"
                   "([A-Za-z0-9][-A-Za-z0-9_.]*)?[A-Za-z0-9]$",
        "type": "string"}
    spec = obj({
        "levels": arr(obj({
            "domain": dict(_DOMAIN_NAME),
            "key": label_key,
        }, required=["domain", "key"]), minItems=1),
        "schedulerTopologyBindings": arr(obj({
            "schedulerName": dict(STR),
            "topologyReference": dict(STR),
        }, required=["schedulerName", "topologyReference"])),
    }, required=["levels"])
    status = obj({
        "conditions": conditions(),
        "observedGeneration": int64(),
        "schedulerTopologyStatuses": arr(obj({
            "inSync": {"type": "boolean"},
            "message": dict(STR),
            "schedulerBackendTopologyObservedGeneration": int64(),
            "schedulerName": dict(STR),
            "topologyReference": dict(STR),
        }, required=["inSync", "schedulerName", "topologyReference"])),
    })
    return _root(spec, status)


# ---------------------------------------------------------------- PodGang
def podgang_schema() -> Schema:
    def pg_topology_constraint() -> Schema:
        # scheduler-side constraint: node-label keys already translated from CTB
        # levels (podgang.go:101-128) — free-form strings, no domain pattern
        return obj({
            "packConstraint": obj({
                "preferred": dict(STR),
                "required": dict(STR),
            }),
        })

    namespaced_name = obj({
        "name": dict(STR),
        "namespace": dict(STR),
    }, required=["name", "namespace"])

    spec = obj({
        "podgroups": arr(obj({
            "minReplicas": int32(),
            "name": dict(STR),
            "podReferences": arr(namespaced_name),
            "topologyConstraint": pg_topology_constraint(),
        }, required=["minReplicas", "name", "podReferences"])),
        "priorityClassName": dict(STR),
        "reuseReservationRef": namespaced_name,
        "topologyConstraint": pg_topology_constraint(),
        "topologyConstraintGroupConfigs": arr(obj({
            "name": dict(STR),
            "podGroupNames": str_arr(),
            "topologyConstraint": pg_topology_constraint(),
        }, required=["name", "podGroupNames"])),
    }, required=["podgroups"])
    status = obj({
        "conditions": conditions(),
        "phase": dict(STR),
        "placementScore": {"type": "number"},
    }, required=["phase"])
    return _root(spec, status)


def _root(spec: Schema, status: Schema) -> Schema:
    return obj({
        "apiVersion": dict(STR),
        "kind": dict(STR),
        "metadata": {"type": "object"},
        "spec": spec,
        "status": status,
    }, required=["spec"])


# ---------------------------------------------------------------- registry
def schemas() -> Dict[str, Schema]:
    """kind -> full openAPIV3Schema."""
    return {
        "PodCliqueSet": podcliqueset_schema(),
        "PodClique": podclique_schema(),
        "PodCliqueScalingGroup": podcliquescalinggroup_schema(),
        "ClusterTopologyBinding": clustertopologybinding_schema(),
        "PodGang": podgang_schema(),
    }
