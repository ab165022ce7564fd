"""Declared API surface — the field tree this stack reads, writes, or passes through.

Used by tests/test_crd_parity.py to diff the kubectl-visible surface against the
reference CRD schemas (field NAMES, recursively): every reference field must be either
implemented here or listed in KNOWN_GAPS with a reason. Subtrees marked PASSTHROUGH are
embedded upstream Kubernetes types (corev1.PodSpec, autoscaling metrics, DRA specs)
that the stack stores and forwards opaquely — their inner fields are accepted by
construction.
"""
from __future__ import annotations

from typing import Dict, List, Set

PASSTHROUGH = "__passthrough__"

_CONDITIONS = {
    "lastTransitionTime": {}, "message": {}, "observedGeneration": {},
    "reason": {}, "status": {}, "type": {},
}
_LAST_ERRORS = {"code": {}, "description": {}, "observedAt": {}}
_TOPO_CONSTRAINT = {
    "pack": {"preferred": {}, "required": {}},
    "packDomain": {},       # deprecated; accepted + translated
    "topologyName": {},
}
_RESOURCE_SHARING = {
    "name": {}, "namespace": {}, "scope": {},
    "filter": {"childCliqueNames": {}, "childScalingGroupNames": {}},
}
_SCALE_CONFIG = {
    "maxReplicas": {}, "minReplicas": {},
    "metrics": PASSTHROUGH,  # autoscaling/v2 MetricSpec embedded type
}
_CLIQUE_SPEC = {
    "roleName": {}, "replicas": {}, "minAvailable": {}, "startsAfter": {},
    "podSpec": PASSTHROUGH,  # corev1.PodSpec embedded type
    "autoScalingConfig": _SCALE_CONFIG,
}

SCHEMAS: Dict[str, dict] = {
    "podcliquesets.grove.io": {
        "apiVersion": {}, "kind": {}, "metadata": PASSTHROUGH,
        "spec": {
            "replicas": {},
            "updateStrategy": {"type": {}},
            "template": {
                "cliqueStartupType": {},
                "priorityClassName": {},
                "terminationDelay": {},
                "headlessServiceConfig": {"publishNotReadyAddresses": {}},
                "topologyConstraint": _TOPO_CONSTRAINT,
                "resourceClaimTemplates": {
                    "name": {}, "templateSpec": PASSTHROUGH},
                "resourceSharing": _RESOURCE_SHARING,
                "cliques": {
                    "name": {}, "labels": {}, "annotations": {},
                    "spec": _CLIQUE_SPEC,
                    "topologyConstraint": _TOPO_CONSTRAINT,
                    "resourceSharing": _RESOURCE_SHARING,
                },
                "podCliqueScalingGroups": {
                    "name": {}, "annotations": {}, "cliqueNames": {},
                    "replicas": {}, "minAvailable": {},
                    "scaleConfig": _SCALE_CONFIG,
                    "resourceSharing": _RESOURCE_SHARING,
                    "topologyConstraint": _TOPO_CONSTRAINT,
                },
            },
        },
        "status": {
            "replicas": {}, "availableReplicas": {}, "updatedReplicas": {},
            "observedGeneration": {}, "currentGenerationHash": {},
            "hpaPodSelector": {}, "conditions": _CONDITIONS,
            "lastErrors": _LAST_ERRORS,
            "podGangStatuses": {"name": {}, "phase": {},
                                "conditions": _CONDITIONS},
            "updateProgress": {
                "updateStartedAt": {}, "updateEndedAt": {},
                "currentlyUpdating": {"replicaIndex": {}, "updateStartedAt": {},
                                      "updateEndedAt": {}},
                "updatedReplicas": {},
                "totalPodCliquesCount": {}, "updatedPodCliquesCount": {},
                "totalPodCliqueScalingGroupsCount": {},
                "updatedPodCliqueScalingGroupsCount": {},
            },
        },
    },
    "podcliques.grove.io": {
        "apiVersion": {}, "kind": {}, "metadata": PASSTHROUGH,
        "spec": _CLIQUE_SPEC,
        "status": {
            "replicas": {}, "readyReplicas": {}, "scheduledReplicas": {},
            "scheduleGatedReplicas": {}, "updatedReplicas": {},
            "observedGeneration": {}, "conditions": _CONDITIONS,
            "lastErrors": _LAST_ERRORS, "hpaPodSelector": {},
            "currentPodTemplateHash": {}, "currentPodCliqueSetGenerationHash": {},
            "updateProgress": {
                "podCliqueSetGenerationHash": {}, "podTemplateHash": {},
                "updateStartedAt": {}, "updateEndedAt": {},
                "readyPodsSelectedToUpdate": {"completed": {}, "current": {}},
            },
        },
    },
    "podcliquescalinggroups.grove.io": {
        "apiVersion": {}, "kind": {}, "metadata": PASSTHROUGH,
        "spec": {"replicas": {}, "minAvailable": {}, "cliqueNames": {}},
        "status": {
            "replicas": {}, "scheduledReplicas": {}, "availableReplicas": {},
            "updatedReplicas": {}, "observedGeneration": {}, "selector": {},
            "conditions": _CONDITIONS, "lastErrors": _LAST_ERRORS,
            "currentPodCliqueSetGenerationHash": {},
            "updateProgress": {
                "podCliqueSetGenerationHash": {},
                "updateStartedAt": {}, "updateEndedAt": {},
                "readyReplicaIndicesSelectedToUpdate": {"completed": {},
                                                        "current": {}},
                "totalPodCliquesCount": {}, "updatedPodCliquesCount": {},
            },
        },
    },
    "clustertopologybindings.grove.io": {
        "apiVersion": {}, "kind": {}, "metadata": PASSTHROUGH,
        "spec": {
            "levels": {"domain": {}, "key": {}},
            "schedulerTopologyBindings": {"schedulerName": {},
                                          "topologyReference": {}},
        },
        "status": {
            "observedGeneration": {}, "conditions": _CONDITIONS,
            "schedulerTopologyStatuses": {
                "schedulerName": {}, "topologyReference": {}, "inSync": {},
                "message": {},
                "schedulerBackendTopologyObservedGeneration": {},
            },
        },
    },
    "podgangs.scheduler.grove.io": {
        "apiVersion": {}, "kind": {}, "metadata": PASSTHROUGH,
        "spec": {
            "priorityClassName": {},
            "reuseReservationRef": {"name": {}, "namespace": {}},
            "podgroups": {
                "name": {}, "minReplicas": {},
                "podReferences": {"name": {}, "namespace": {}},
                "topologyConstraint": {"packConstraint": {"required": {},
                                                          "preferred": {}}},
            },
            "topologyConstraint": {"packConstraint": {"required": {},
                                                      "preferred": {}}},
            "topologyConstraintGroupConfigs": {
                "name": {}, "podGroupNames": {},
                "topologyConstraint": {"packConstraint": {"required": {},
                                                          "preferred": {}}},
            },
        },
        "status": {"phase": {}, "placementScore": {}, "conditions": _CONDITIONS},
    },
}

# Reference fields deliberately not implemented yet, with reasons.
KNOWN_GAPS: Dict[str, List[str]] = {
    "podgangs.scheduler.grove.io": [],
    "podcliquesets.grove.io": [],
    "podcliques.grove.io": [],
    "podcliquescalinggroups.grove.io": [],
    "clustertopologybindings.grove.io": [],
}


def declared_paths(crd_name: str) -> Set[str]:
    """Flatten SCHEMAS[crd_name] into dotted paths; PASSTHROUGH marks subtree roots."""
    out: Set[str] = set()

    def walk(node, prefix: str):
        if node == PASSTHROUGH:
            out.add(prefix + ".*")
            return
        for k, v in node.items():
            p = f"{prefix}.{k}" if prefix else k
            out.add(p)
            walk(v, p)

    walk(SCHEMAS[crd_name], "")
    return out
