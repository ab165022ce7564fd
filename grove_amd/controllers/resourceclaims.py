"""Resource sharing — DRA ResourceClaims, MI355X-native.

Parity source: operator/internal/resourceclaim/{reconcile,resolve,naming}.go +
components/resourceclaim/ + pod injection (podclique/components/pod/pod.go:206-269):
named ResourceClaimTemplates declared on the PCS template are materialized as
ResourceClaim objects at PCS scope (AllReplicas = one claim for the whole set,
PerReplica = one claim per PCS replica) or PCSG scope, filtered to child cliques, and
referenced from pod spec.resourceClaims.

MI355X replacement for the reference's MNNVL/ComputeDomain path (mnnvl/):
`auto_xgmi_domain` creates one xGMI-domain claim per PCS replica (deviceClass
xgmi.amd.com) injected into every GPU-requesting clique, expressing "this gang shares
one xGMI hive" to DRA-aware schedulers — single-node 8×MI355X needs no cross-node
fabric CRD.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError, owner_reference

XGMI_DEVICE_CLASS = "xgmi.amd.com"
XGMI_TEMPLATE_NAME = "xgmi-domain"

ClaimRef = Dict[str, str]  # {"name": template, "resourceClaimName": claim}


def build_resource_claim(pcs: Obj, name: str, spec: Obj) -> Obj:
    return {
        "apiVersion": "resource.k8s.io/v1beta1", "kind": "ResourceClaim",
        "metadata": {
            "name": name,
            "namespace": pcs["metadata"].get("namespace", "default"),
            "labels": {
                c.LABEL_MANAGED_BY: c.LABEL_MANAGED_BY_VALUE,
                c.LABEL_PART_OF: pcs["metadata"]["name"],
                c.LABEL_COMPONENT: c.COMPONENT_RESOURCE_CLAIM,
            },
            "ownerReferences": [owner_reference(pcs)],
        },
        "spec": spec,
    }


def resolve_templates(pcs: Obj) -> Dict[str, Obj]:
    out: Dict[str, Obj] = {}
    for t in pcs["spec"]["template"].get("resourceClaimTemplates") or []:
        if t.get("name"):
            # templateSpec is a resourcev1.ResourceClaimTemplateSpec
            # (podcliqueset.go:417-422); the created ResourceClaim's spec is its
            # .spec subtree
            out[t["name"]] = (t.get("templateSpec") or {}).get("spec") or {}
    return out


def _filter_allows(entry: Obj, clique_name: str) -> bool:
    filt = entry.get("filter") or {}
    allowed = filt.get("childCliqueNames")
    return allowed is None or clique_name in allowed


def pcs_claims_for_replica(store: Store, pcs: Obj, r: int,
                           auto_xgmi_domain: bool) -> List[Tuple[Obj, Obj]]:
    """Returns [(claim_object, sharing_entry)] to ensure for PCS replica r."""
    pcs_name = pcs["metadata"]["name"]
    templates = resolve_templates(pcs)
    out: List[Tuple[Obj, Obj]] = []
    for entry in pcs["spec"]["template"].get("resourceSharing") or []:
        tname = entry.get("name")
        spec = templates.get(tname)
        if spec is None:
            continue
        scope = entry.get("scope", "AllReplicas")
        claim_name = f"{pcs_name}-{tname}" if scope == "AllReplicas" \
            else f"{pcs_name}-{r}-{tname}"
        out.append((build_resource_claim(pcs, claim_name, spec), entry))
    # xGMI-domain groups: one claim per (replica, group), shared by the group's
    # cliques (the MNNVL ComputeDomain-per-replica analog)
    groups = effective_xgmi_groups(pcs, auto_xgmi_domain)
    by_group: Dict[str, List[str]] = {}
    for clique, g in groups.items():
        by_group.setdefault(g, []).append(clique)
    for g, members in sorted(by_group.items()):
        entry = {"name": f"{XGMI_TEMPLATE_NAME}-{g}", "scope": "PerReplica",
                 "filter": {"childCliqueNames": sorted(members)}}
        claim = build_resource_claim(
            pcs, f"{pcs_name}-{r}-xgmi-{g}",
            {"devices": {"requests": [{"name": "xgmi-hive",
                                       "deviceClassName": XGMI_DEVICE_CLASS}]}})
        out.append((claim, entry))
    return out


def effective_xgmi_groups(pcs: Obj, auto_default: bool) -> Dict[str, str]:
    """clique name -> xGMI group, following the annotation hierarchy (PCS -> PCSG ->
    PCLQ, lower overrides; "none" opts out; non-GPU cliques silently skip inherited
    groups — auto-mnnvl.md:58-77 parity). With auto_default and no annotations, every
    GPU clique joins the implicit per-replica group "default"."""
    tmpl = pcs["spec"]["template"]
    pcs_group = (pcs["metadata"].get("annotations") or {}).get(
        c.ANNOTATION_XGMI_GROUP)
    sg_of = {}
    sg_group = {}
    for sg in tmpl.get("podCliqueScalingGroups") or []:
        g = (sg.get("annotations") or {}).get(c.ANNOTATION_XGMI_GROUP)
        for mn in sg.get("cliqueNames") or []:
            sg_of[mn] = sg["name"]
            if g is not None:
                sg_group[mn] = g
    gpu = set(_gpu_cliques(pcs))
    out: Dict[str, str] = {}
    for cl in tmpl.get("cliques") or []:
        name = cl["name"]
        own = (cl.get("annotations") or {}).get(c.ANNOTATION_XGMI_GROUP)
        eff = own if own is not None else sg_group.get(name, pcs_group)
        if eff is None and auto_default and name in gpu:
            eff = "default"
        if eff in (None, "none"):
            continue
        if name in gpu:
            out[name] = eff
        # non-GPU cliques silently skip INHERITED groups (explicit ones are rejected
        # at admission by validate_xgmi_groups)
    return out


def _gpu_cliques(pcs: Obj) -> List[str]:
    names = []
    for cl in pcs["spec"]["template"].get("cliques") or []:
        for ctr in (cl.get("spec", {}).get("podSpec", {}).get("containers") or []):
            req = ((ctr.get("resources") or {}).get("requests") or {})
            lim = ((ctr.get("resources") or {}).get("limits") or {})
            if c.AMD_GPU_RESOURCE in req or c.AMD_GPU_RESOURCE in lim:
                names.append(cl["name"])
                break
    return names


def _pcs_requests_gpus(pcs: Obj) -> bool:
    return bool(_gpu_cliques(pcs))


def ensure_claims(store: Store, claims: List[Tuple[Obj, Obj]]) -> None:
    for claim, _entry in claims:
        try:
            store.create(claim)
        except ApiError as e:
            if e.reason != "AlreadyExists":
                raise


def claim_refs_for_clique(claims: List[Tuple[Obj, Obj]],
                          clique_name: str) -> List[ClaimRef]:
    refs: List[ClaimRef] = []
    for claim, entry in claims:
        if _filter_allows(entry, clique_name):
            tname = entry.get("name")
            refs.append({"name": tname,
                         "resourceClaimName": claim["metadata"]["name"]})
    return refs


def clique_level_claims(store: Store, pcs: Obj, r: int,
                        clique_tmpl: Obj, pclq_fqn: str) -> List[Tuple[Obj, Obj]]:
    """Clique-level sharing (podclique/components/resourceclaim parity): AllReplicas =
    one claim shared by this clique across all PCS replicas; PerReplica = one claim per
    clique instance."""
    templates = resolve_templates(pcs)
    out: List[Tuple[Obj, Obj]] = []
    for entry in clique_tmpl.get("resourceSharing") or []:
        tname = entry.get("name")
        spec = templates.get(tname)
        if spec is None:
            continue
        if entry.get("scope", "AllReplicas") == "AllReplicas":
            cname = f"{pcs['metadata']['name']}-{clique_tmpl['name']}-{tname}"
        else:
            cname = f"{pclq_fqn}-{tname}"
        out.append((build_resource_claim(pcs, cname, spec), entry))
    return out


def pcsg_claims(pcs: Obj, sg_cfg: Obj, sg_fqn: str, replicas: int
                ) -> List[Tuple[Obj, Obj, Optional[int]]]:
    """PCSG-level sharing: [(claim, entry, replica_index_or_None_for_all)]."""
    templates = resolve_templates(pcs)
    out: List[Tuple[Obj, Obj, Optional[int]]] = []
    for entry in sg_cfg.get("resourceSharing") or []:
        tname = entry.get("name")
        spec = templates.get(tname)
        if spec is None:
            continue
        if entry.get("scope", "AllReplicas") == "AllReplicas":
            out.append((build_resource_claim(pcs, f"{sg_fqn}-{tname}", spec),
                        entry, None))
        else:
            for j in range(replicas):
                out.append((build_resource_claim(pcs, f"{sg_fqn}-{j}-{tname}", spec),
                            entry, j))
    return out
