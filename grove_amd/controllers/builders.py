"""Builders for every child resource the operator manages.

Parity sources: pod build — podclique/components/pod/pod.go:69-371; PCLQ build —
podcliqueset/components/podclique/podclique.go:284-410; PCSG build —
components/podcliquescalinggroup/; Service — components/service/service.go; HPA —
components/hpa/hpa.go:128; RBAC — components/{serviceaccount,role,rolebinding}/;
PodGang skeleton — components/podgang/podgang.go:129. Fresh dict-based implementation.
"""
from __future__ import annotations

import copy
import os
from typing import Any, Dict, List, Optional

from ..api import constants as c
from ..api import namegen
from ..kubecore.store import Obj, owner_reference
from ..utils.hashing import pod_template_hash


def default_labels(pcs_name: str) -> Dict[str, str]:
    return {c.LABEL_MANAGED_BY: c.LABEL_MANAGED_BY_VALUE, c.LABEL_PART_OF: pcs_name}


def _startup_dependencies(pcs: Obj, clique_name: str, pcs_replica: int,
                          pcsg_cfg_name: Optional[str] = None,
                          pcsg_replica: Optional[int] = None) -> List[str]:
    """Resolve StartsAfter FQNs per startup type (podclique.go:341-455,
    componentutils.GenerateDependencyNamesForBasePodGang parity).

    Base-gang cliques (standalone, or PCSG member replica j < minAvailable) depend on
    ALL replicas in [0, minAvailable) of a dependency clique that belongs to a PCSG.
    Scaled-gang cliques (PCSG member, j >= minAvailable) depend only on siblings inside
    their OWN PCSG replica j; dependencies outside the PCSG are dropped (a scaled gang
    must not couple to another gang's readiness).
    """
    tmpl = pcs["spec"]["template"]
    startup = tmpl.get("cliqueStartupType", c.STARTUP_ANY_ORDER)
    cliques = tmpl.get("cliques") or []
    names = [cl["name"] for cl in cliques]
    deps: List[str] = []
    if startup == c.STARTUP_IN_ORDER:
        i = names.index(clique_name)
        if i > 0:
            deps = [names[i - 1]]
    elif startup == c.STARTUP_EXPLICIT:
        for cl in cliques:
            if cl["name"] == clique_name:
                deps = list(cl.get("spec", {}).get("startsAfter") or [])
    if not deps:
        return []
    pcs_name = pcs["metadata"]["name"]
    sgs = tmpl.get("podCliqueScalingGroups") or []

    def sg_of(name: str) -> Optional[Obj]:
        for sg in sgs:
            if name in (sg.get("cliqueNames") or []):
                return sg
        return None

    my_sg = next((s for s in sgs if s["name"] == pcsg_cfg_name), None) \
        if pcsg_cfg_name else None
    scaled = (my_sg is not None and pcsg_replica is not None
              and pcsg_replica >= int(my_sg.get("minAvailable", 1)))
    out: List[str] = []
    for d in deps:
        d_sg = sg_of(d)
        if scaled:
            if d_sg is not None and d_sg["name"] == pcsg_cfg_name:
                sg_fqn = namegen.pcsg_name(pcs_name, pcs_replica, pcsg_cfg_name)
                out.append(namegen.podclique_name(sg_fqn, pcsg_replica, d))
            # cross-gang dependency from a scaled gang: dropped
        elif d_sg is not None:
            sg_fqn = namegen.pcsg_name(pcs_name, pcs_replica, d_sg["name"])
            out.extend(namegen.podclique_name(sg_fqn, j, d)
                       for j in range(int(d_sg.get("minAvailable", 1))))
        else:
            out.append(namegen.podclique_name(pcs_name, pcs_replica, d))
    return out


def build_podclique(pcs: Obj, pcs_replica: int, clique_tmpl: Obj,
                    owner: Obj, owner_name: Optional[str] = None,
                    owner_replica: Optional[int] = None,
                    pcsg_name: Optional[str] = None,
                    pcsg_replica: Optional[int] = None,
                    podgang_name: Optional[str] = None,
                    base_podgang_name: Optional[str] = None) -> Obj:
    """Build a PodClique CR. owner is the PCS (standalone) or the PCSG (member)."""
    pcs_name = pcs["metadata"]["name"]
    namespace = pcs["metadata"].get("namespace", "default")
    cl_name = clique_tmpl["name"]
    if pcsg_name is not None:
        fqn = namegen.podclique_name(pcsg_name, pcsg_replica, cl_name)
        component = c.COMPONENT_PCSG_PODCLIQUE
    else:
        fqn = namegen.podclique_name(pcs_name, pcs_replica, cl_name)
        component = c.COMPONENT_PCS_PODCLIQUE
    if podgang_name is None:
        podgang_name = namegen.base_podgang_name(pcs_name, pcs_replica)
    spec = copy.deepcopy(clique_tmpl["spec"])
    tmpl = pcs["spec"]["template"]
    hash_ = pod_template_hash(cl_name, spec.get("podSpec", {}),
                              tmpl.get("priorityClassName", ""),
                              clique_tmpl.get("labels"),
                              clique_tmpl.get("annotations"))
    labels = {
        **default_labels(pcs_name),
        **(clique_tmpl.get("labels") or {}),
        c.LABEL_COMPONENT: component,
        c.LABEL_APP_NAME: fqn,
        c.LABEL_PCS_REPLICA_INDEX: str(pcs_replica),
        c.LABEL_PODGANG: podgang_name,
        c.LABEL_POD_TEMPLATE_HASH: hash_,
    }
    if pcsg_name is not None:
        labels[c.LABEL_PCSG] = pcsg_name
        labels[c.LABEL_PCSG_REPLICA_INDEX] = str(pcsg_replica)
    if base_podgang_name:
        labels[c.LABEL_BASE_PODGANG] = base_podgang_name
    # Recover the scaling-group CONFIG name from the PCSG FQN
    # ("<pcs>-<replica>-<sg>", namegen.go:85) for dependency scoping.
    pcsg_cfg_name = None
    if pcsg_name is not None:
        prefix = f"{pcs_name}-{pcs_replica}-"
        pcsg_cfg_name = pcsg_name[len(prefix):] if pcsg_name.startswith(prefix) \
            else pcsg_name
    spec["startsAfter"] = _startup_dependencies(
        pcs, cl_name, pcs_replica, pcsg_cfg_name, pcsg_replica)
    return {
        "apiVersion": c.API_VERSION,
        "kind": c.KIND_PCLQ,
        "metadata": {
            "name": fqn,
            "namespace": namespace,
            "labels": labels,
            "annotations": dict(clique_tmpl.get("annotations") or {}),
            "finalizers": [c.FINALIZER_PCLQ],
            "ownerReferences": [owner_reference(owner)],
        },
        "spec": spec,
    }


def match_by_fqn_suffix(fqn: str, named: List[Obj]) -> Optional[Obj]:
    """Resolve which template a generated FQN ("<owner>-<replica>-<name>") came from.
    Clique/scaling-group names may themselves contain dashes (DNS-1123), so pick the
    LONGEST "-<name>" suffix match — "x-0-model-a" must resolve to clique "model-a",
    not a sibling clique "a"."""
    best = None
    for item in named:
        nm = item.get("name", "")
        if nm and fqn.endswith("-" + nm) and (
                best is None or len(nm) > len(best.get("name", ""))):
            best = item
    return best


def build_pcsg(pcs: Obj, pcs_replica: int, sg_cfg: Obj) -> Obj:
    pcs_name = pcs["metadata"]["name"]
    fqn = namegen.pcsg_name(pcs_name, pcs_replica, sg_cfg["name"])
    labels = {
        **default_labels(pcs_name),
        c.LABEL_COMPONENT: c.COMPONENT_PCSG,
        c.LABEL_APP_NAME: fqn,
        c.LABEL_PCS_REPLICA_INDEX: str(pcs_replica),
    }
    return {
        "apiVersion": c.API_VERSION,
        "kind": c.KIND_PCSG,
        "metadata": {
            "name": fqn,
            "namespace": pcs["metadata"].get("namespace", "default"),
            "labels": labels,
            "annotations": dict(sg_cfg.get("annotations") or {}),
            "finalizers": [c.FINALIZER_PCSG],
            "ownerReferences": [owner_reference(pcs)],
        },
        "spec": {
            "replicas": sg_cfg.get("replicas", 1),
            "minAvailable": sg_cfg.get("minAvailable", 1),
            "cliqueNames": list(sg_cfg.get("cliqueNames") or []),
        },
    }


def build_pod(pcs: Obj, pclq: Obj, pod_index: int, scheduler_name: str,
              pcsg_template_num_pods: Optional[int] = None) -> Obj:
    """Gated pod with the full grove label/env/hostname contract (pod.go:138-371)."""
    pclq_name = pclq["metadata"]["name"]
    namespace = pclq["metadata"].get("namespace", "default")
    pcs_name = pclq["metadata"]["labels"].get(c.LABEL_PART_OF, "")
    pcs_replica = pclq["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX, "0")
    podgang_name = pclq["metadata"]["labels"].get(c.LABEL_PODGANG, "")
    pod_spec = copy.deepcopy(pclq["spec"].get("podSpec", {}))
    svc_name = namegen.headless_service_name(pcs_name, int(pcs_replica))

    env = [
        {"name": c.ENV_PCS_NAME, "value": pcs_name},
        {"name": c.ENV_PCS_INDEX, "value": str(pcs_replica)},
        {"name": c.ENV_PCLQ_NAME, "value": pclq_name},
        {"name": c.ENV_HEADLESS_SERVICE,
         "value": namegen.headless_service_address(pcs_name, int(pcs_replica), namespace)},
        {"name": c.ENV_POD_INDEX, "value": str(pod_index)},
    ]
    pcsg = pclq["metadata"]["labels"].get(c.LABEL_PCSG)
    if pcsg:
        env.append({"name": c.ENV_PCSG_NAME, "value": pcsg})
        env.append({"name": c.ENV_PCSG_INDEX,
                    "value": pclq["metadata"]["labels"].get(c.LABEL_PCSG_REPLICA_INDEX, "0")})
        if pcsg_template_num_pods is not None:
            env.append({"name": c.ENV_PCSG_TEMPLATE_NUM_PODS,
                        "value": str(pcsg_template_num_pods)})
    # startup ordering: inject the grove-initc init container (pod.go:315-371 parity) —
    # flags are --podcliques=<parentFQN>:<minAvailable>; the SA token secret
    # <pcs>-ic-sat is mounted so initc can watch pods. The image honors the
    # GROVE_INIT_CONTAINER_IMAGE env contract (pod/initcontainer.go:37).
    starts_after = pclq["spec"].get("startsAfter") or []
    if starts_after:
        def _parent_min_available(fqn: str) -> int:
            cl = match_by_fqn_suffix(
                fqn, ((pcs.get("spec") or {}).get("template") or {})
                .get("cliques") or [])
            if cl is not None:
                sp = cl.get("spec") or {}
                return int(sp.get("minAvailable") or sp.get("replicas", 1))
            return 1
        initc = {
            "name": "grove-initc",
            "image": os.environ.get("GROVE_INIT_CONTAINER_IMAGE",
                                    "grove-initc:latest"),
            "command": ["python", "-m", "grove_amd.initc"],
            "args": (["--namespace", namespace, "--podgang", podgang_name]
                     + [f"--podcliques={fqn}:{_parent_min_available(fqn)}"
                        for fqn in starts_after]),
            "volumeMounts": [{"name": "grove-ic-sat",
                              "mountPath": "/var/grove/sa-token",
                              "readOnly": True}],
        }
        ics = pod_spec.setdefault("initContainers", [])
        if not any(ic.get("name") == "grove-initc" for ic in ics):
            ics.insert(0, initc)
        vols = pod_spec.setdefault("volumes", [])
        if not any(v.get("name") == "grove-ic-sat" for v in vols):
            vols.append({"name": "grove-ic-sat", "secret": {
                "secretName": namegen.initc_sa_token_secret_name(pcs_name)}})

    for ctr in pod_spec.get("containers", []) + pod_spec.get("initContainers", []):
        ctr.setdefault("env", [])
        ctr["env"] = env + ctr["env"]

    claim_refs = pclq["spec"].get("resourceClaims") or []
    if claim_refs:
        pod_spec["resourceClaims"] = [
            {"name": r["name"], "resourceClaimName": r["resourceClaimName"]}
            for r in claim_refs]
        for ctr in pod_spec.get("containers", []):
            ctr.setdefault("resources", {}).setdefault("claims", [])
            ctr["resources"]["claims"].extend({"name": r["name"]} for r in claim_refs)

    pod_spec["schedulingGates"] = [{"name": c.POD_GANG_SCHEDULING_GATE}]
    pod_spec["schedulerName"] = scheduler_name
    pod_spec["hostname"] = namegen.pod_hostname(pclq_name, pod_index)
    pod_spec["subdomain"] = svc_name
    pod_spec["serviceAccountName"] = namegen.pod_service_account_name(pcs_name)
    pcs_tmpl = None
    labels = {
        **default_labels(pcs_name),
        c.LABEL_APP_NAME: pclq_name,
        c.LABEL_PODCLIQUE: pclq_name,
        c.LABEL_PODGANG: podgang_name,
        c.LABEL_PCS_REPLICA_INDEX: str(pcs_replica),
        c.LABEL_POD_INDEX: str(pod_index),
        c.LABEL_POD_TEMPLATE_HASH: pclq["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH, ""),
    }
    if pcsg:
        labels[c.LABEL_PCSG] = pcsg
        labels[c.LABEL_PCSG_REPLICA_INDEX] = pclq["metadata"]["labels"].get(
            c.LABEL_PCSG_REPLICA_INDEX, "0")
    base_pg = pclq["metadata"]["labels"].get(c.LABEL_BASE_PODGANG)
    if base_pg:
        labels[c.LABEL_BASE_PODGANG] = base_pg
    return {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {
            "generateName": f"{pclq_name}-",
            "namespace": namespace,
            "labels": labels,
            # clique-template annotations ride on the PCLQ and flow to its pods
            "annotations": dict(pclq["metadata"].get("annotations") or {}),
            "ownerReferences": [owner_reference(pclq)],
        },
        "spec": pod_spec,
        "status": {"phase": "Pending"},
    }


def build_headless_service(pcs: Obj, pcs_replica: int) -> Obj:
    pcs_name = pcs["metadata"]["name"]
    cfg = pcs["spec"]["template"].get("headlessServiceConfig") or {}
    name = namegen.headless_service_name(pcs_name, pcs_replica)
    return {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {
            "name": name,
            "namespace": pcs["metadata"].get("namespace", "default"),
            "labels": {
                **default_labels(pcs_name),
                c.LABEL_COMPONENT: c.COMPONENT_HEADLESS_SERVICE,
                c.LABEL_APP_NAME: name,
                c.LABEL_PCS_REPLICA_INDEX: str(pcs_replica),
            },
            "ownerReferences": [owner_reference(pcs)],
        },
        "spec": {
            "clusterIP": "None",
            "publishNotReadyAddresses": cfg.get("publishNotReadyAddresses", True),
            "selector": {
                c.LABEL_PART_OF: pcs_name,
                c.LABEL_PCS_REPLICA_INDEX: str(pcs_replica),
            },
        },
    }


def build_service_account(pcs: Obj) -> Obj:
    pcs_name = pcs["metadata"]["name"]
    return {
        "apiVersion": "v1", "kind": "ServiceAccount",
        "metadata": {
            "name": namegen.pod_service_account_name(pcs_name),
            "namespace": pcs["metadata"].get("namespace", "default"),
            "labels": {**default_labels(pcs_name),
                       c.LABEL_COMPONENT: c.COMPONENT_POD_SERVICE_ACCOUNT},
            "ownerReferences": [owner_reference(pcs)],
        },
    }


def build_role(pcs: Obj) -> Obj:
    pcs_name = pcs["metadata"]["name"]
    return {
        "apiVersion": "rbac.authorization.k8s.io/v1", "kind": "Role",
        "metadata": {
            "name": namegen.pod_role_name(pcs_name),
            "namespace": pcs["metadata"].get("namespace", "default"),
            "labels": {**default_labels(pcs_name), c.LABEL_COMPONENT: c.COMPONENT_POD_ROLE},
            "ownerReferences": [owner_reference(pcs)],
        },
        "rules": [{"apiGroups": [""], "resources": ["pods"],
                   "verbs": ["get", "list", "watch"]}],
    }


def build_role_binding(pcs: Obj) -> Obj:
    pcs_name = pcs["metadata"]["name"]
    ns = pcs["metadata"].get("namespace", "default")
    return {
        "apiVersion": "rbac.authorization.k8s.io/v1", "kind": "RoleBinding",
        "metadata": {
            "name": namegen.pod_role_binding_name(pcs_name),
            "namespace": ns,
            "labels": {**default_labels(pcs_name),
                       c.LABEL_COMPONENT: c.COMPONENT_POD_ROLE_BINDING},
            "ownerReferences": [owner_reference(pcs)],
        },
        "roleRef": {"apiGroup": "rbac.authorization.k8s.io", "kind": "Role",
                    "name": namegen.pod_role_name(pcs_name)},
        "subjects": [{"kind": "ServiceAccount",
                      "name": namegen.pod_service_account_name(pcs_name), "namespace": ns}],
    }


def build_sa_token_secret(pcs: Obj) -> Obj:
    pcs_name = pcs["metadata"]["name"]
    return {
        "apiVersion": "v1", "kind": "Secret",
        "metadata": {
            "name": namegen.initc_sa_token_secret_name(pcs_name),
            "namespace": pcs["metadata"].get("namespace", "default"),
            "labels": {**default_labels(pcs_name),
                       c.LABEL_COMPONENT: c.COMPONENT_SA_TOKEN_SECRET},
            "annotations": {"kubernetes.io/service-account.name":
                            namegen.pod_service_account_name(pcs_name)},
            "ownerReferences": [owner_reference(pcs)],
        },
        "type": "kubernetes.io/service-account-token",
    }


def build_hpa(pcs: Obj, target_kind: str, target_name: str, scale_cfg: Obj) -> Obj:
    pcs_name = pcs["metadata"]["name"]
    metrics = scale_cfg.get("metrics") or [{
        "type": "Resource",
        "resource": {"name": "cpu",
                     "target": {"type": "Utilization", "averageUtilization": 80}},
    }]
    return {
        "apiVersion": "autoscaling/v2", "kind": "HorizontalPodAutoscaler",
        "metadata": {
            "name": target_name,
            "namespace": pcs["metadata"].get("namespace", "default"),
            "labels": {**default_labels(pcs_name), c.LABEL_COMPONENT: c.COMPONENT_HPA},
            "ownerReferences": [owner_reference(pcs)],
        },
        "spec": {
            "scaleTargetRef": {"apiVersion": c.API_VERSION, "kind": target_kind,
                               "name": target_name},
            "minReplicas": scale_cfg.get("minReplicas", 1),
            "maxReplicas": scale_cfg["maxReplicas"],
            "metrics": metrics,
        },
    }


def build_podgang(pcs: Obj, name: str, scheduler_name: str,
                  base_podgang: Optional[str] = None,
                  priority_class: str = "") -> Obj:
    pcs_name = pcs["metadata"]["name"]
    labels = {
        **default_labels(pcs_name),
        c.LABEL_COMPONENT: c.COMPONENT_PODGANG,
        c.LABEL_APP_NAME: name,
        c.LABEL_SCHEDULER_NAME: scheduler_name,
    }
    if base_podgang:
        labels[c.LABEL_BASE_PODGANG] = base_podgang
    spec: Dict[str, Any] = {"podgroups": []}
    if priority_class:
        spec["priorityClassName"] = priority_class
    return {
        "apiVersion": c.SCHEDULER_API_VERSION,
        "kind": c.KIND_PODGANG,
        "metadata": {
            "name": name,
            "namespace": pcs["metadata"].get("namespace", "default"),
            "labels": labels,
            "ownerReferences": [owner_reference(pcs)],
        },
        "spec": spec,
        "status": {"conditions": [], "phase": "Pending"},
    }
