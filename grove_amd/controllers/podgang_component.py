"""PodGang computation + sync — the gang topology core.

Behavior parity: operator/internal/controller/podcliqueset/components/podgang/
{podgang.go:98-278, syncflow.go:147-817}: expected base PodGangs (standalone cliques +
PCSG replicas [0,minAvailable)) and scaled PodGangs (PCSG replicas >= minAvailable, one
gang each, labeled with their base gang); created Initialized=False, podReferences filled
from pods carrying the grove.io/podgang label, Initialized flipped True only when every
expected pod exists and is associated. Fresh implementation.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple

from ..api import constants as c
from ..api import namegen
from ..kubecore.store import Store, Obj, ApiError
from ..utils import conditions as cond
from ..utils import errors as groveerr
from . import builders

Obj = Dict[str, Any]


class ExpectedGang:
    __slots__ = ("name", "base", "groups", "constraint", "group_configs")

    def __init__(self, name: str, base: Optional[str], groups: List[Dict[str, Any]],
                 constraint: Optional[Dict[str, Any]] = None,
                 group_configs: Optional[List[Dict[str, Any]]] = None):
        self.name = name
        self.base = base      # base podgang name for scaled gangs, else None
        self.groups = groups  # [{"name": pclq_fqn, "minReplicas": int, "replicas": int}]
        self.constraint = constraint          # translated gang-level TopologyConstraint
        self.group_configs = group_configs or []  # TopologyConstraintGroupConfig[]


def domain_key_map(store: Store, topology_name: Optional[str] = None
                   ) -> Dict[str, str]:
    """CTB levels → {domain: nodeLabelKey}, with native defaults for host/xgmi-hive
    (components/podgang/syncflow.go:349-380 translation parity). topology_name picks
    the CTB the constraint references; otherwise the single CTB is the source."""
    m = {"xgmi-hive": c.NODE_LABEL_XGMI_HIVE, "host": "kubernetes.io/hostname"}
    for ctb in store.list(c.KIND_CTB):
        if topology_name and ctb["metadata"]["name"] != topology_name:
            continue
        for lv in (ctb.get("spec") or {}).get("levels") or []:
            key = lv.get("key")
            if lv.get("domain") and key:
                m[lv["domain"]] = key
        break  # one CTB is the source of truth
    return m


def translate_constraint(tc: Optional[Dict[str, Any]],
                         dmap: Dict[str, str]) -> Optional[Dict[str, Any]]:
    """PCS-style TopologyConstraint (pack.required/preferred domains, deprecated
    packDomain) → PodGang-style packConstraint holding node label KEYS."""
    if not tc:
        return None
    pack = tc.get("pack") or {}
    required = pack.get("required") or tc.get("packDomain")
    preferred = pack.get("preferred")
    out: Dict[str, str] = {}
    if required and required in dmap:
        out["required"] = dmap[required]
    if preferred and preferred in dmap:
        out["preferred"] = dmap[preferred]
    return {"packConstraint": out} if out else None


def compute_expected_podgangs(store: Store, pcs: Obj,
                              pcsg_by_name: Optional[Dict[str, Obj]] = None
                              ) -> List[ExpectedGang]:
    """syncflow.go:147-335 equivalent."""
    pcs_name = pcs["metadata"]["name"]
    ns = pcs["metadata"].get("namespace", "default")
    tmpl = pcs["spec"]["template"]
    cliques = tmpl.get("cliques") or []
    sg_cfgs = tmpl.get("podCliqueScalingGroups") or []
    sg_members = {m for sg in sg_cfgs for m in (sg.get("cliqueNames") or [])}
    clique_by_name = {cl["name"]: cl for cl in cliques}
    topo_name = None
    for tc in [tmpl.get("topologyConstraint")] + \
              [cl.get("topologyConstraint") for cl in cliques] + \
              [sg.get("topologyConstraint") for sg in sg_cfgs]:
        if tc and tc.get("topologyName"):
            topo_name = tc["topologyName"]
            break
    dmap = domain_key_map(store, topo_name)
    gang_tc = translate_constraint(tmpl.get("topologyConstraint"), dmap)
    out: List[ExpectedGang] = []

    def clique_tc(mn: str) -> Optional[Dict[str, Any]]:
        return translate_constraint(
            clique_by_name[mn].get("topologyConstraint")
            or clique_by_name[mn].get("spec", {}).get("topologyConstraint"), dmap)

    for r in range(int(pcs["spec"].get("replicas", 0))):
        base_name = namegen.base_podgang_name(pcs_name, r)
        groups: List[Dict[str, Any]] = []
        base_group_cfgs: List[Dict[str, Any]] = []
        for cl in cliques:
            if cl["name"] in sg_members:
                continue
            spec = cl.get("spec", {})
            fqn = namegen.podclique_name(pcs_name, r, cl["name"])
            groups.append({
                "name": fqn,
                "minReplicas": int(spec.get("minAvailable", spec.get("replicas", 1))),
                "replicas": int(spec.get("replicas", 1)),
            })
            tc = clique_tc(cl["name"])
            if tc:
                base_group_cfgs.append({"name": fqn, "podGroupNames": [fqn],
                                        "topologyConstraint": tc})
        for sg in sg_cfgs:
            sg_fqn = namegen.pcsg_name(pcs_name, r, sg["name"])
            sg_tc = translate_constraint(sg.get("topologyConstraint"), dmap)
            # live PCSG replica count (HPA may have scaled it)
            pcsg = pcsg_by_name.get(sg_fqn) if pcsg_by_name is not None \
                else store.try_get(c.KIND_PCSG, ns, sg_fqn)
            replicas = int((pcsg or {}).get("spec", {}).get("replicas", sg.get("replicas", 1)))
            min_avail = int((pcsg or {}).get("spec", {}).get(
                "minAvailable", sg.get("minAvailable", 1)))
            for j in range(min(min_avail, replicas)):
                sub_names = []
                for mn in sg.get("cliqueNames") or []:
                    spec = clique_by_name[mn].get("spec", {})
                    fqn = namegen.podclique_name(sg_fqn, j, mn)
                    sub_names.append(fqn)
                    groups.append({
                        "name": fqn,
                        "minReplicas": int(spec.get("minAvailable", spec.get("replicas", 1))),
                        "replicas": int(spec.get("replicas", 1)),
                    })
                if sg_tc:
                    # each PCSG replica packs within its own domain (parent subgroup,
                    # kai/backend.go:260 hierarchical-subgroup parity)
                    base_group_cfgs.append({"name": f"{sg_fqn}-{j}",
                                            "podGroupNames": sub_names,
                                            "topologyConstraint": sg_tc})
            for j in range(min_avail, replicas):
                sg_groups = []
                sub_cfgs: List[Dict[str, Any]] = []
                for mn in sg.get("cliqueNames") or []:
                    spec = clique_by_name[mn].get("spec", {})
                    fqn = namegen.podclique_name(sg_fqn, j, mn)
                    sg_groups.append({
                        "name": fqn,
                        "minReplicas": int(spec.get("minAvailable", spec.get("replicas", 1))),
                        "replicas": int(spec.get("replicas", 1)),
                    })
                    tc = clique_tc(mn)
                    if tc:
                        sub_cfgs.append({"name": fqn, "podGroupNames": [fqn],
                                         "topologyConstraint": tc})
                out.append(ExpectedGang(
                    namegen.scaled_podgang_name(sg_fqn, j - min_avail), base_name,
                    sg_groups, constraint=sg_tc or gang_tc, group_configs=sub_cfgs))
        out.append(ExpectedGang(base_name, None, groups, constraint=gang_tc,
                                group_configs=base_group_cfgs))
    return out


def sync_podgangs(store: Store, pcs: Obj, scheduler_name: str,
                  rec: "groveerr.StepRecorder" = None) -> None:
    """Create/update PodGangs to match expectations; GC stale ones; flip Initialized.
    Non-benign step failures are recorded to the PCS's status.lastErrors via rec."""
    ns = pcs["metadata"].get("namespace", "default")
    pcs_name = pcs["metadata"]["name"]
    rec = rec or groveerr.StepRecorder(store, c.KIND_PCS, ns, pcs_name)
    pcsg_by_name = {g["metadata"]["name"]: g for g in store.list(
        c.KIND_PCSG, ns, {c.LABEL_PART_OF: pcs_name}, copy_objects=False)}
    expected = compute_expected_podgangs(store, pcs, pcsg_by_name)
    expected_names = {g.name for g in expected}

    existing = store.list(c.KIND_PODGANG, ns, {c.LABEL_PART_OF: pcs_name}, copy_objects=False)
    existing_by_name = {pg["metadata"]["name"]: pg for pg in existing}
    for pg in existing:
        if pg["metadata"]["name"] not in expected_names:
            with rec.step(groveerr.ERR_SYNC_PODGANG,
                          benign=groveerr.BENIGN_DELETE,
                          detail=f"GC PodGang {pg['metadata']['name']}"):
                store.delete(c.KIND_PODGANG, ns, pg["metadata"]["name"])

    # index pods once per sync pass: podgang label -> podclique label -> [pod names]
    pods = store.list("Pod", ns, {c.LABEL_PART_OF: pcs_name}, copy_objects=False)
    by_gang_clique: Dict[Tuple[str, str], List[str]] = {}
    for p in pods:
        lbl = p["metadata"].get("labels", {})
        key = (lbl.get(c.LABEL_PODGANG, ""), lbl.get(c.LABEL_PODCLIQUE, ""))
        by_gang_clique.setdefault(key, []).append(p["metadata"]["name"])

    tmpl = pcs["spec"]["template"]
    priority_class = tmpl.get("priorityClassName", "")
    for gang in expected:
        cur = existing_by_name.get(gang.name)
        groups_spec = []
        all_created = True
        for g in gang.groups:
            refs = sorted(by_gang_clique.get((gang.name, g["name"]), []))
            if len(refs) < g["replicas"]:
                all_created = False
            groups_spec.append({
                "name": g["name"],
                "minReplicas": g["minReplicas"],
                "podReferences": [{"namespace": ns, "name": r} for r in refs],
            })
        if cur is None:
            obj = builders.build_podgang(pcs, gang.name, scheduler_name,
                                         base_podgang=gang.base,
                                         priority_class=priority_class)
            obj["spec"]["podgroups"] = groups_spec
            if gang.constraint:
                obj["spec"]["topologyConstraint"] = gang.constraint
            if gang.group_configs:
                obj["spec"]["topologyConstraintGroupConfigs"] = gang.group_configs
            cond.set_condition(obj, c.PODGANG_COND_INITIALIZED, False, "PendingPodCreation")
            try:
                cur = store.create(obj)
            except ApiError as e:
                if e.reason != "AlreadyExists":
                    rec.record(groveerr.ERR_SYNC_PODGANG,
                               f"create PodGang {gang.name}: {e.reason}: "
                               f"{e.message}")
                cur = store.try_get(c.KIND_PODGANG, ns, gang.name)
                if cur is None:
                    rec.retry_needed = True
                    continue

        def upd(o: Obj) -> None:
            o["spec"]["podgroups"] = groups_spec
            if gang.constraint:
                o["spec"]["topologyConstraint"] = gang.constraint
            if gang.group_configs:
                o["spec"]["topologyConstraintGroupConfigs"] = gang.group_configs
        if cur["spec"].get("podgroups") != groups_spec \
                or cur["spec"].get("topologyConstraint") != gang.constraint \
                or (gang.group_configs and cur["spec"].get(
                    "topologyConstraintGroupConfigs") != gang.group_configs):
            try:
                cur = store.patch(c.KIND_PODGANG, ns, gang.name, upd)
            except ApiError as e:
                if e.reason == "Conflict":
                    rec.retry_needed = True  # refs not written; do not latch
                elif e.reason not in groveerr.BENIGN_UPDATE:
                    rec.record(groveerr.ERR_SYNC_PODGANG,
                               f"patch PodGang {gang.name}: {e.reason}: "
                               f"{e.message}")
                continue

        initialized = cond.condition_true(cur, c.PODGANG_COND_INITIALIZED)
        if all_created and gang.groups and not initialized:
            def flip(o: Obj) -> None:
                cond.set_condition(o, c.PODGANG_COND_INITIALIZED, True, "AllPodsAssociated")
            with rec.step(groveerr.ERR_SYNC_PODGANG,
                          benign=groveerr.BENIGN_UPDATE,
                          detail=f"flip Initialized on {gang.name}"):
                store.patch(c.KIND_PODGANG, ns, gang.name, flip, status=True)
        elif not all_created and initialized:
            # pods lost after init (e.g. gang termination in flight) — keep Initialized
            pass


def try_complete_podgang(store: Store, ns: str, gang_name: str,
                         rec: "groveerr.StepRecorder" = None,
                         force: bool = False) -> None:
    """Latency fast-path: called inline from the PCLQ pass right
    after pod creation, so a gang whose pods all exist gets its podReferences filled
    and Initialized flipped in the SAME reconcile instead of waiting for the next
    PCS-scope sync_podgangs pass (saves two watch→queue→worker hops per gang on the
    serial time-to-running chain). The PCS pass remains the reconciling authority;
    this only performs the monotonic completion step.

    force=True refreshes podReferences even on an already-Initialized gang — the
    repair path for a replaced pod whose gang still references its dead predecessor
    (the gate-removal wedge: gated pod not in refs + Initialized=True)."""
    pg = store.try_get(c.KIND_PODGANG, ns, gang_name, copy=False)
    if pg is None or (not force
                      and cond.condition_true(pg, c.PODGANG_COND_INITIALIZED)):
        return
    groups = (pg.get("spec") or {}).get("podgroups") or []
    if not groups:
        return
    pods = store.list("Pod", ns, {c.LABEL_PODGANG: gang_name}, copy_objects=False)
    by_clique: Dict[str, List[str]] = {}
    for p in pods:
        lbl = p["metadata"].get("labels", {})
        by_clique.setdefault(lbl.get(c.LABEL_PODCLIQUE, ""), []).append(
            p["metadata"]["name"])
    groups_spec = []
    for g in groups:
        refs = sorted(by_clique.get(g["name"], []))
        # spec.replicas of the member PCLQ is the completion bar; fall back to
        # minReplicas when the PCLQ is not readable (it always is in-process)
        pclq = store.try_get(c.KIND_PCLQ, ns, g["name"], copy=False)
        want = int((pclq or {}).get("spec", {}).get(
            "replicas", g.get("minReplicas", 1)))
        if len(refs) < want:
            return  # not complete yet; the PCS pass will finish the job
        groups_spec.append({"name": g["name"],
                            "minReplicas": g.get("minReplicas", 1),
                            "podReferences": [{"namespace": ns, "name": r}
                                              for r in refs]})
    def fill(o: Obj) -> None:
        cur_groups = {g["name"]: g for g in o["spec"].get("podgroups") or []}
        for gs in groups_spec:
            if gs["name"] in cur_groups:
                cur_groups[gs["name"]]["podReferences"] = gs["podReferences"]
    cur_refs = {g["name"]: [r.get("name") for r in g.get("podReferences") or []]
                for g in groups}
    if cur_refs != {gs["name"]: [r["name"] for r in gs["podReferences"]]
                    for gs in groups_spec}:
        try:
            store.patch(c.KIND_PODGANG, ns, gang_name, fill)
        except ApiError:
            if rec is not None:
                rec.retry_needed = True
            return

    if not cond.condition_true(pg, c.PODGANG_COND_INITIALIZED):
        def flip(o: Obj) -> None:
            cond.set_condition(o, c.PODGANG_COND_INITIALIZED, True,
                               "AllPodsAssociated")
        try:
            store.patch(c.KIND_PODGANG, ns, gang_name, flip, status=True)
        except ApiError:
            pass
