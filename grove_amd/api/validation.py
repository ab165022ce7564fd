"""Validating admission for grove.io resources.

Behavior parity with /root/reference/operator/internal/webhook/admission/pcs/validation/
podcliqueset.go (structure checks, 45-char combined-name budget :44,1024-1040, startup DAG
acyclicity :464-489, PCSG membership rules :337-404,499-519, minAvailable/scaleConfig
constraints :573-589, immutable-field update rules :643-702). Fresh implementation.
"""
from __future__ import annotations

import re
from typing import Any, Dict, List, Optional

from . import constants as c
from .defaulting import parse_duration_seconds
from ..kubecore.store import invalid

Obj = Dict[str, Any]

MAX_COMBINED_RESOURCE_NAME_LENGTH = 45


def _err(path: str, msg: str):
    return invalid(f"{path}: {msg}")


def _check_name_budget(pcs_name: str, pcsg_name: str, pclq_name: str) -> None:
    total = len(pcs_name) + len(pcsg_name) + len(pclq_name)
    if total > MAX_COMBINED_RESOURCE_NAME_LENGTH:
        if pcsg_name:
            raise invalid(
                f"combined resource name length {total} exceeds 45-character limit required "
                f"for pod naming. Consider shortening: PodCliqueSet '{pcs_name}', "
                f"PodCliqueScalingGroup '{pcsg_name}', or PodClique '{pclq_name}'")
        raise invalid(
            f"combined resource name length {total} exceeds 45-character limit required "
            f"for pod naming. Consider shortening: PodCliqueSet '{pcs_name}' or "
            f"PodClique '{pclq_name}'")


def _check_startup_dag(cliques: List[Obj], startup_type: str) -> None:
    names = [cl.get("name") for cl in cliques]
    name_set = set(names)
    if startup_type != c.STARTUP_EXPLICIT:
        for cl in cliques:
            if cl.get("spec", {}).get("startsAfter"):
                raise _err("spec.template.cliques",
                           "startsAfter may only be set with CliqueStartupTypeExplicit")
        return
    deps = {}
    for cl in cliques:
        sa = cl.get("spec", {}).get("startsAfter") or []
        for d in sa:
            if d not in name_set:
                raise _err("spec.template.cliques",
                           f"clique {cl.get('name')!r} startsAfter references unknown clique {d!r}")
            if d == cl.get("name"):
                raise _err("spec.template.cliques",
                           f"clique {cl.get('name')!r} cannot start after itself")
        deps[cl.get("name")] = list(sa)
    # cycle check (iterative DFS, colors)
    WHITE, GRAY, BLACK = 0, 1, 2
    color = {n: WHITE for n in names}

    def visit(start: str) -> None:
        stack = [(start, iter(deps.get(start, ())))]
        color[start] = GRAY
        while stack:
            node, it = stack[-1]
            adv = next(it, None)
            if adv is None:
                color[node] = BLACK
                stack.pop()
            elif color[adv] == GRAY:
                raise _err("spec.template.cliques", "startsAfter dependencies form a cycle")
            elif color[adv] == WHITE:
                color[adv] = GRAY
                stack.append((adv, iter(deps.get(adv, ()))))

    for n in names:
        if color[n] == WHITE:
            visit(n)




_DNS1123_LABEL = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")
_DNS_SUBDOMAIN = re.compile(
    r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?(\.[a-z0-9]([-a-z0-9]*[a-z0-9])?)*$")
_ENV_VAR_NAME = re.compile(r"^[-._a-zA-Z][-._a-zA-Z0-9]*$")


def _validate_pod_spec(podspec: Obj, path: str, is_create: bool) -> None:
    """PodSpec sanity (validation/podcliqueset.go:591-642): operator-managed
    fields must not be user-set; env var names must be valid and unique."""
    if is_create:
        if podspec.get("topologySpreadConstraints"):
            raise _err(f"{path}.topologySpreadConstraints",
                       "must not be set (the gang scheduler owns placement)")
        if podspec.get("nodeName"):
            raise _err(f"{path}.nodeName",
                       "must not be set (the gang scheduler owns placement)")
    for kind_key in ("containers", "initContainers"):
        for ctr in podspec.get(kind_key) or []:
            seen_env = set()
            for ev in ctr.get("env") or []:
                nm = str(ev.get("name", ""))
                if not _ENV_VAR_NAME.match(nm):
                    raise _err(f"{path}.{kind_key}[{ctr.get('name')}].env",
                               f"invalid environment variable name {nm!r}")
                if nm in seen_env:
                    raise _err(f"{path}.{kind_key}[{ctr.get('name')}].env",
                               f"duplicate environment variable {nm!r}")
                seen_env.add(nm)


def _validate_resource_sharing(refs: List[Obj], template_names: set, path: str,
                               clique_names: Optional[set] = None,
                               group_names: Optional[set] = None,
                               allow_group_filter: bool = False) -> None:
    """ResourceSharing entries (validation/podcliqueset.go:139-233): name
    required+unique, scope enum, namespace only for external templates, filter
    children must exist within the declaring scope."""
    seen = set()
    for ref in refs:
        nm = ref.get("name", "")
        if not nm:
            raise _err(f"{path}.name", "reference name is required")
        if nm in seen:
            raise _err(f"{path}.name", f"duplicate reference {nm!r}")
        seen.add(nm)
        if ref.get("namespace") and nm in template_names:
            raise _err(f"{path}[{nm}].namespace",
                       "namespace must be empty when name matches an internal "
                       "resourceClaimTemplate")
        if ref.get("scope") not in ("AllReplicas", "PerReplica"):
            raise _err(f"{path}[{nm}].scope",
                       f"unsupported scope {ref.get('scope')!r}; "
                       "supported: AllReplicas, PerReplica")
        filt = ref.get("filter")
        if filt is None:
            continue
        child_cl = filt.get("childCliqueNames") or []
        child_sg = filt.get("childScalingGroupNames") or []
        if not child_cl and not (child_sg if allow_group_filter else ()):
            raise _err(f"{path}[{nm}].filter",
                       "filter must specify at least one child entry")
        if clique_names is not None:
            for cn in child_cl:
                if cn not in clique_names:
                    raise _err(f"{path}[{nm}].filter.childCliqueNames",
                               f"unknown clique {cn!r}")
        if not allow_group_filter and child_sg:
            raise _err(f"{path}[{nm}].filter.childScalingGroupNames",
                       "not allowed at this scope")
        if allow_group_filter and group_names is not None:
            for gn in child_sg:
                if gn not in group_names:
                    raise _err(f"{path}[{nm}].filter.childScalingGroupNames",
                               f"unknown scaling group {gn!r}")


def validate_podcliqueset(pcs: Obj, old: Optional[Obj] = None) -> None:
    name = pcs.get("metadata", {}).get("name", "")
    spec = pcs.get("spec") or {}
    if spec.get("replicas", 0) < 0:
        raise _err("spec.replicas", "must be >= 0")
    ustrat = (spec.get("updateStrategy") or {}).get("type")
    if ustrat not in (None, c.UPDATE_ROLLING_RECREATE, c.UPDATE_ON_DELETE):
        raise _err("spec.updateStrategy.type", f"unsupported strategy {ustrat!r}")
    tmpl = spec.get("template") or {}
    cliques = tmpl.get("cliques") or []
    if not cliques:
        raise _err("spec.template.cliques", "at least one PodClique is required")

    startup = tmpl.get("cliqueStartupType", c.STARTUP_ANY_ORDER)
    if startup not in (c.STARTUP_ANY_ORDER, c.STARTUP_IN_ORDER, c.STARTUP_EXPLICIT):
        raise _err("spec.template.cliqueStartupType", f"unsupported startup type {startup!r}")

    seen = set()
    for cl in cliques:
        cn = cl.get("name")
        if not cn:
            raise _err("spec.template.cliques", "clique name is required")
        if cn in seen:
            raise _err("spec.template.cliques", f"duplicate clique name {cn!r}")
        seen.add(cn)
        cs = cl.get("spec") or {}
        if not cs.get("roleName"):
            raise _err(f"spec.template.cliques[{cn}].spec.roleName", "roleName is required")
        reps = cs.get("replicas", 1)
        if reps < 0:
            raise _err(f"spec.template.cliques[{cn}].spec.replicas", "must be >= 0")
        ma = cs.get("minAvailable")
        if ma is not None:
            if ma < 1:
                raise _err(f"spec.template.cliques[{cn}].spec.minAvailable", "must be >= 1")
            if ma > reps:
                raise _err(f"spec.template.cliques[{cn}].spec.minAvailable",
                           "must not be greater than replicas")
        if not _DNS1123_LABEL.match(str(cn)):
            raise _err(f"spec.template.cliques[{cn}].name",
                       "must be a valid DNS-1123 label")
        podspec = cs.get("podSpec") or {}
        if not podspec.get("containers"):
            raise _err(f"spec.template.cliques[{cn}].spec.podSpec.containers",
                       "at least one container is required")
        for ctr in podspec.get("containers", []):
            for ev in ctr.get("env") or []:
                if str(ev.get("name", "")).startswith("GROVE_"):
                    raise _err(f"spec.template.cliques[{cn}].spec.podSpec",
                               f"env var {ev.get('name')!r} uses the reserved GROVE_ prefix")
        _validate_pod_spec(podspec, f"spec.template.cliques[{cn}].spec.podSpec",
                           is_create=(old is None))
        asc = cs.get("autoScalingConfig")
        if asc is not None:
            _validate_scale_config(asc, cs.get("minAvailable", reps),
                                   f"spec.template.cliques[{cn}].spec.autoScalingConfig")

    sg_member_cliques: set = set()
    sg_names: set = set()
    for sg in tmpl.get("podCliqueScalingGroups") or []:
        sgn = sg.get("name")
        if not sgn:
            raise _err("spec.template.podCliqueScalingGroups", "scaling group name is required")
        if sgn in sg_names:
            raise _err("spec.template.podCliqueScalingGroups", f"duplicate scaling group {sgn!r}")
        if not _DNS_SUBDOMAIN.match(str(sgn)):
            raise _err(f"spec.template.podCliqueScalingGroups[{sgn}].name",
                       "must be a valid DNS subdomain")
        sg_names.add(sgn)
        members = sg.get("cliqueNames") or []
        if not members:
            raise _err(f"spec.template.podCliqueScalingGroups[{sgn}].cliqueNames",
                       "at least one clique name is required")
        for mn in members:
            if mn not in seen:
                raise _err(f"spec.template.podCliqueScalingGroups[{sgn}].cliqueNames",
                           f"unknown clique {mn!r}")
            if mn in sg_member_cliques:
                raise _err(f"spec.template.podCliqueScalingGroups[{sgn}].cliqueNames",
                           f"clique {mn!r} belongs to more than one scaling group")
            sg_member_cliques.add(mn)
            _check_name_budget(name, sgn, mn)
        reps = sg.get("replicas", 1)
        ma = sg.get("minAvailable", 1)
        if ma < 1:
            raise _err(f"spec.template.podCliqueScalingGroups[{sgn}].minAvailable", "must be >= 1")
        if ma > reps:
            raise _err(f"spec.template.podCliqueScalingGroups[{sgn}].minAvailable",
                       "must not be greater than replicas")
        if sg.get("scaleConfig") is not None:
            _validate_scale_config(sg["scaleConfig"], ma,
                                   f"spec.template.podCliqueScalingGroups[{sgn}].scaleConfig")
        # scaling-group members must not have individual autoscaling
        for cl in cliques:
            if cl.get("name") in members and (cl.get("spec") or {}).get("autoScalingConfig"):
                raise _err(f"spec.template.cliques[{cl['name']}].spec.autoScalingConfig",
                           "cliques in a scaling group cannot define individual autoscaling")

    for cl in cliques:
        if cl.get("name") not in sg_member_cliques:
            _check_name_budget(name, "", cl.get("name", ""))

    # resource claim templates: required name (DNS subdomain), unique, device
    # requests present (validation/podcliqueset.go:121-137)
    template_names: set = set()
    for rct in tmpl.get("resourceClaimTemplates") or []:
        rn = rct.get("name", "")
        if not rn:
            raise _err("spec.template.resourceClaimTemplates.name",
                       "template name is required")
        if not _DNS_SUBDOMAIN.match(rn):
            raise _err(f"spec.template.resourceClaimTemplates[{rn}].name",
                       "must be a valid DNS subdomain")
        if rn in template_names:
            raise _err("spec.template.resourceClaimTemplates.name",
                       f"duplicate template {rn!r}")
        template_names.add(rn)
        reqs = (((rct.get("templateSpec") or {}).get("spec") or {})
                .get("devices") or {}).get("requests") or []
        if not reqs:
            raise _err(f"spec.template.resourceClaimTemplates[{rn}]"
                       ".templateSpec.spec.devices.requests",
                       "at least one device request is required")

    clique_names = set(seen)
    _validate_resource_sharing(tmpl.get("resourceSharing") or [], template_names,
                               "spec.template.resourceSharing",
                               clique_names=clique_names, group_names=sg_names,
                               allow_group_filter=True)
    for cl in cliques:
        _validate_resource_sharing(
            cl.get("resourceSharing") or [], template_names,
            f"spec.template.cliques[{cl.get('name')}].resourceSharing")
    for sg in tmpl.get("podCliqueScalingGroups") or []:
        _validate_resource_sharing(
            sg.get("resourceSharing") or [], template_names,
            f"spec.template.podCliqueScalingGroups[{sg.get('name')}]"
            ".resourceSharing",
            clique_names=set(sg.get("cliqueNames") or []))

    _check_startup_dag(cliques, startup)

    td = tmpl.get("terminationDelay")
    if td is not None and parse_duration_seconds(td) <= 0:
        raise _err("spec.template.terminationDelay", "must be greater than 0")

    if old is not None:
        _validate_pcs_update(pcs, old)


def _validate_scale_config(sc: Obj, min_available: int, path: str) -> None:
    maxr = sc.get("maxReplicas")
    if maxr is None:
        raise _err(f"{path}.maxReplicas", "maxReplicas is required")
    minr = sc.get("minReplicas")
    if minr is not None:
        if minr < 1:
            raise _err(f"{path}.minReplicas", "must be >= 1")
        if maxr < minr:
            raise _err(f"{path}.maxReplicas", "must be >= minReplicas")
        if minr < min_available:
            raise _err(f"{path}.minReplicas", "must not be less than minAvailable")


def _validate_pcs_update(new: Obj, old: Obj) -> None:
    """Immutable-field rules (validation/podcliqueset.go:643-702,967-1010)."""
    nt = (new.get("spec") or {}).get("template") or {}
    ot = (old.get("spec") or {}).get("template") or {}
    if nt.get("cliqueStartupType") != ot.get("cliqueStartupType"):
        raise _err("spec.template.cliqueStartupType", "field is immutable")
    new_names = [cl.get("name") for cl in nt.get("cliques") or []]
    old_names = [cl.get("name") for cl in ot.get("cliques") or []]
    if new_names != old_names:
        raise _err("spec.template.cliques", "clique names cannot be added, removed or reordered")
    new_sgs = {sg.get("name"): sg.get("cliqueNames") for sg in nt.get("podCliqueScalingGroups") or []}
    old_sgs = {sg.get("name"): sg.get("cliqueNames") for sg in ot.get("podCliqueScalingGroups") or []}
    if new_sgs.keys() != old_sgs.keys():
        raise _err("spec.template.podCliqueScalingGroups", "scaling groups cannot be added or removed")
    for k in new_sgs:
        if new_sgs[k] != old_sgs[k]:
            raise _err(f"spec.template.podCliqueScalingGroups[{k}].cliqueNames", "field is immutable")
    # topology constraints are immutable after creation at every level
    # (docs/user-guide/topology-aware-scheduling.md: add/modify/remove rejected)
    if nt.get("topologyConstraint") != ot.get("topologyConstraint"):
        raise _err("spec.template.topologyConstraint",
                   "topology constraints are immutable; recreate the PodCliqueSet")
    for new_cl, old_cl in zip(nt.get("cliques") or [], ot.get("cliques") or []):
        if new_cl.get("topologyConstraint") != old_cl.get("topologyConstraint"):
            raise _err(f"spec.template.cliques[{new_cl.get('name')}].topologyConstraint",
                       "topology constraints are immutable; recreate the PodCliqueSet")
    for new_sg, old_sg in zip(nt.get("podCliqueScalingGroups") or [],
                              ot.get("podCliqueScalingGroups") or []):
        if new_sg.get("topologyConstraint") != old_sg.get("topologyConstraint"):
            raise _err(
                f"spec.template.podCliqueScalingGroups[{new_sg.get('name')}]"
                f".topologyConstraint",
                "topology constraints are immutable; recreate the PodCliqueSet")


def validate_clustertopologybinding(ctb: Obj, old: Optional[Obj] = None) -> None:
    levels = (ctb.get("spec") or {}).get("levels") or []
    if not levels:
        raise _err("spec.levels", "at least one topology level is required")
    seen_d, seen_k = set(), set()
    for lv in levels:
        d, k = lv.get("domain"), lv.get("key")
        if not d or not k:
            raise _err("spec.levels", "each level requires domain and key")
        if d in seen_d:
            raise _err("spec.levels", f"duplicate domain {d!r}")
        if k in seen_k:
            raise _err("spec.levels", f"duplicate node label key {k!r}")
        seen_d.add(d)
        seen_k.add(k)


def validate_pcsg(pcsg: Obj, old: Optional[Obj] = None) -> None:
    spec = pcsg.get("spec") or {}
    reps = int(spec.get("replicas", 1))
    if reps < 0:
        raise _err("spec.replicas", "must be >= 0")
    ma = spec.get("minAvailable")
    if ma is not None and int(ma) < 1:
        raise _err("spec.minAvailable", "must be >= 1")
    if old is not None:
        if spec.get("cliqueNames") != (old.get("spec") or {}).get("cliqueNames"):
            raise _err("spec.cliqueNames", "field is immutable")


def validate_podclique(pclq: Obj, old: Optional[Obj] = None) -> None:
    spec = pclq.get("spec") or {}
    if int(spec.get("replicas", 1)) < 0:
        raise _err("spec.replicas", "must be >= 0")
    ma = spec.get("minAvailable")
    if ma is not None and int(ma) < 1:
        raise _err("spec.minAvailable", "must be >= 1")
    if old is not None and spec.get("roleName") != (old.get("spec") or {}).get("roleName"):
        raise _err("spec.roleName", "field is immutable")


class TopologyConstraintValidator:
    """Store-bound PCS validator for topology constraints (webhook/admission/pcs/
    validation/topologyconstraints.go:173-310 parity): pack domains must resolve
    against the ClusterTopologyBinding levels (plus the native host/xgmi-hive
    built-ins), and child constraints must be equal-or-narrower than their parent's
    (hierarchy rule: a PCSG/clique may not pack at a BROADER level than the PCS)."""

    BUILTIN_DOMAINS = ("host", "xgmi-hive")

    def __init__(self, store):
        self.store = store

    def _known_domains(self):
        # ordered broadest -> narrowest
        ordered = []
        for ctb in self.store.list(c.KIND_CTB):
            for lv in (ctb.get("spec") or {}).get("levels") or []:
                d = lv.get("domain")
                if d and d not in ordered:
                    ordered.append(d)
            break
        for d in self.BUILTIN_DOMAINS:
            if d not in ordered:
                ordered.append(d)
        return ordered

    @staticmethod
    def _domains_of(tc):
        if not tc:
            return (None, None)
        pack = tc.get("pack") or {}
        return (pack.get("required") or tc.get("packDomain"), pack.get("preferred"))

    def __call__(self, pcs, old=None):
        tmpl = (pcs.get("spec") or {}).get("template") or {}
        parent_tc = tmpl.get("topologyConstraint")
        # a single PCS cannot use multiple topology names (children inherit the
        # parent name when they omit it — topology-aware-scheduling.md:32,69)
        names = set()
        for tc in [parent_tc] + [cl.get("topologyConstraint")
                                 for cl in tmpl.get("cliques") or []] + \
                  [sg.get("topologyConstraint")
                   for sg in tmpl.get("podCliqueScalingGroups") or []]:
            if tc is not None and not tc.get("pack") and not tc.get("packDomain"):
                # CEL parity: has(self.pack) || has(self.packDomain)
                raise _err("spec.template.topologyConstraint",
                           "topologyConstraint must set pack or packDomain")
            if tc and tc.get("topologyName"):
                names.add(tc["topologyName"])
        if len(names) > 1:
            raise _err("spec.template", f"a PodCliqueSet may reference only one "
                                        f"topologyName, found {sorted(names)}")
        if names:
            known_ctbs = {ctb["metadata"]["name"]
                          for ctb in self.store.list(c.KIND_CTB)}
            missing = names - known_ctbs
            if missing and known_ctbs:
                raise _err("spec.template.topologyConstraint.topologyName",
                           f"unknown ClusterTopologyBinding {sorted(missing)}")
        has_any = bool(parent_tc) or any(
            cl.get("topologyConstraint") for cl in tmpl.get("cliques") or []) or any(
            sg.get("topologyConstraint")
            for sg in tmpl.get("podCliqueScalingGroups") or [])
        if not has_any:
            return
        known = self._known_domains()

        def check(tc, path):
            # CRD CEL-rule equivalents (podcliqueset.go TopologyConstraint
            # x-kubernetes-validations): a constraint must express SOMETHING,
            # pack{} may not be empty, and the deprecated packDomain is mutually
            # exclusive with pack.
            if "pack" in tc and not (tc.get("pack") or {}).keys() & \
                    {"required", "preferred"}:
                raise _err(f"{path}.pack",
                           "pack must set required and/or preferred")
            if not tc.get("pack") and not tc.get("packDomain"):
                raise _err(path, "topologyConstraint must set pack or packDomain")
            if tc.get("packDomain") and (tc.get("pack") or {}).keys() & \
                    {"required", "preferred"}:
                raise _err(f"{path}.packDomain",
                           "packDomain (deprecated) and pack are mutually exclusive")
            req, pref = self._domains_of(tc)
            for d, f in ((req, "required"), (pref, "preferred")):
                if d is not None and d not in known:
                    raise _err(f"{path}.pack.{f}",
                               f"unknown topology domain {d!r}; levels defined by the "
                               f"ClusterTopologyBinding: {known}")
            if req is not None and pref is not None:
                if known.index(pref) < known.index(req):
                    raise _err(f"{path}.pack.preferred",
                               "preferred level must be equal or narrower than required")
            return req

        parent_req = check(parent_tc, "spec.template.topologyConstraint")             if parent_tc else None
        for sg in tmpl.get("podCliqueScalingGroups") or []:
            tc = sg.get("topologyConstraint")
            if tc:
                child_req = check(
                    tc, f"spec.template.podCliqueScalingGroups[{sg.get('name')}]"
                        f".topologyConstraint")
                if parent_req and child_req and \
                        known.index(child_req) < known.index(parent_req):
                    raise _err(
                        f"spec.template.podCliqueScalingGroups[{sg.get('name')}]"
                        f".topologyConstraint.pack.required",
                        "must be equal or narrower than the PodCliqueSet constraint")
        for cl in tmpl.get("cliques") or []:
            tc = cl.get("topologyConstraint")
            if tc:
                child_req = check(
                    tc, f"spec.template.cliques[{cl.get('name')}].topologyConstraint")
                if parent_req and child_req and \
                        known.index(child_req) < known.index(parent_req):
                    raise _err(
                        f"spec.template.cliques[{cl.get('name')}]"
                        f".topologyConstraint.pack.required",
                        "must be equal or narrower than the PodCliqueSet constraint")


_DNS1123 = None


def validate_xgmi_groups(pcs: Obj, old: Optional[Obj] = None) -> None:
    """grove.io/xgmi-group admission (auto-mnnvl.md parity): value must be "none" or
    a DNS-1123 label; an explicit group on a clique with no amd.com/gpu request is a
    user error; the annotation is immutable after creation at every level."""
    import re
    global _DNS1123
    if _DNS1123 is None:
        _DNS1123 = re.compile(r"^[a-z0-9]([a-z0-9-]{0,61}[a-z0-9])?$")
    tmpl = (pcs.get("spec") or {}).get("template") or {}

    def check_value(v, path):
        if v is not None and v != "none" and not _DNS1123.match(v):
            raise _err(path, f"xgmi-group {v!r} must be 'none' or a DNS-1123 label")

    def gpu_clique(cl):
        for ctr in (cl.get("spec", {}).get("podSpec", {}).get("containers") or []):
            res = (ctr.get("resources") or {})
            if "amd.com/gpu" in (res.get("requests") or {}) \
                    or "amd.com/gpu" in (res.get("limits") or {}):
                return True
        return False

    check_value((pcs.get("metadata", {}).get("annotations") or {}).get(
        c.ANNOTATION_XGMI_GROUP), "metadata.annotations")
    for sg in tmpl.get("podCliqueScalingGroups") or []:
        check_value((sg.get("annotations") or {}).get(c.ANNOTATION_XGMI_GROUP),
                    f"spec.template.podCliqueScalingGroups[{sg.get('name')}]")
    for cl in tmpl.get("cliques") or []:
        v = (cl.get("annotations") or {}).get(c.ANNOTATION_XGMI_GROUP)
        check_value(v, f"spec.template.cliques[{cl.get('name')}]")
        if v not in (None, "none") and not gpu_clique(cl):
            raise _err(f"spec.template.cliques[{cl.get('name')}]",
                       f"explicit xgmi-group on a PodClique with no amd.com/gpu "
                       f"request")
    if old is not None:
        def ann_of(obj_or_tmpl_item, meta=False):
            src = obj_or_tmpl_item.get("metadata", {}) if meta else obj_or_tmpl_item
            return (src.get("annotations") or {}).get(c.ANNOTATION_XGMI_GROUP)
        if ann_of(pcs, meta=True) != ann_of(old, meta=True):
            raise _err("metadata.annotations",
                       "grove.io/xgmi-group is immutable after creation")
        ot = (old.get("spec") or {}).get("template") or {}
        for new_cl, old_cl in zip(tmpl.get("cliques") or [], ot.get("cliques") or []):
            if ann_of(new_cl) != ann_of(old_cl):
                raise _err(f"spec.template.cliques[{new_cl.get('name')}]",
                           "grove.io/xgmi-group is immutable after creation")
        for new_sg, old_sg in zip(tmpl.get("podCliqueScalingGroups") or [],
                                  ot.get("podCliqueScalingGroups") or []):
            if ann_of(new_sg) != ann_of(old_sg):
                raise _err(
                    f"spec.template.podCliqueScalingGroups[{new_sg.get('name')}]",
                    "grove.io/xgmi-group is immutable after creation")
