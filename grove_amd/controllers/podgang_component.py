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
from . import builders

Obj = Dict[str, Any]


class ExpectedGang:
    __slots__ = ("name", "base", "groups")

    def __init__(self, name: str, base: Optional[str], groups: List[Dict[str, Any]]):
        self.name = name
        self.base = base      # base podgang name for scaled gangs, else None
        self.groups = groups  # [{"name": pclq_fqn, "minReplicas": int, "replicas": int}]


def compute_expected_podgangs(store: Store, pcs: Obj) -> List[ExpectedGang]:
    """syncflow.go:147-335 equivalent."""
    pcs_name = pcs["metadata"]["name"]
    ns = pcs["metadata"].get("namespace", "default")
    tmpl = pcs["spec"]["template"]
    cliques = tmpl.get("cliques") or []
    sg_cfgs = tmpl.get("podCliqueScalingGroups") or []
    sg_members = {m for sg in sg_cfgs for m in (sg.get("cliqueNames") or [])}
    clique_by_name = {cl["name"]: cl for cl in cliques}
    out: List[ExpectedGang] = []

    for r in range(int(pcs["spec"].get("replicas", 0))):
        base_name = namegen.base_podgang_name(pcs_name, r)
        groups: List[Dict[str, Any]] = []
        for cl in cliques:
            if cl["name"] in sg_members:
                continue
            spec = cl.get("spec", {})
            groups.append({
                "name": namegen.podclique_name(pcs_name, r, cl["name"]),
                "minReplicas": int(spec.get("minAvailable", spec.get("replicas", 1))),
                "replicas": int(spec.get("replicas", 1)),
            })
        for sg in sg_cfgs:
            sg_fqn = namegen.pcsg_name(pcs_name, r, sg["name"])
            # live PCSG replica count (HPA may have scaled it)
            pcsg = store.try_get(c.KIND_PCSG, ns, sg_fqn)
            replicas = int((pcsg or {}).get("spec", {}).get("replicas", sg.get("replicas", 1)))
            min_avail = int((pcsg or {}).get("spec", {}).get(
                "minAvailable", sg.get("minAvailable", 1)))
            for j in range(min(min_avail, replicas)):
                for mn in sg.get("cliqueNames") or []:
                    spec = clique_by_name[mn].get("spec", {})
                    groups.append({
                        "name": namegen.podclique_name(sg_fqn, j, mn),
                        "minReplicas": int(spec.get("minAvailable", spec.get("replicas", 1))),
                        "replicas": int(spec.get("replicas", 1)),
                    })
            for j in range(min_avail, replicas):
                sg_groups = []
                for mn in sg.get("cliqueNames") or []:
                    spec = clique_by_name[mn].get("spec", {})
                    sg_groups.append({
                        "name": namegen.podclique_name(sg_fqn, j, mn),
                        "minReplicas": int(spec.get("minAvailable", spec.get("replicas", 1))),
                        "replicas": int(spec.get("replicas", 1)),
                    })
                out.append(ExpectedGang(
                    namegen.scaled_podgang_name(sg_fqn, j - min_avail), base_name, sg_groups))
        out.append(ExpectedGang(base_name, None, groups))
    return out


def sync_podgangs(store: Store, pcs: Obj, scheduler_name: str) -> None:
    """Create/update PodGangs to match expectations; GC stale ones; flip Initialized."""
    ns = pcs["metadata"].get("namespace", "default")
    pcs_name = pcs["metadata"]["name"]
    expected = compute_expected_podgangs(store, pcs)
    expected_names = {g.name for g in expected}

    existing = store.list(c.KIND_PODGANG, ns, {c.LABEL_PART_OF: pcs_name})
    for pg in existing:
        if pg["metadata"]["name"] not in expected_names:
            try:
                store.delete(c.KIND_PODGANG, ns, pg["metadata"]["name"])
            except ApiError:
                pass

    # index pods once per sync pass: podgang label -> podclique label -> [pod names]
    pods = store.list("Pod", ns, {c.LABEL_PART_OF: pcs_name})
    by_gang_clique: Dict[Tuple[str, str], List[str]] = {}
    for p in pods:
        lbl = p["metadata"].get("labels", {})
        key = (lbl.get(c.LABEL_PODGANG, ""), lbl.get(c.LABEL_PODCLIQUE, ""))
        by_gang_clique.setdefault(key, []).append(p["metadata"]["name"])

    tmpl = pcs["spec"]["template"]
    priority_class = tmpl.get("priorityClassName", "")
    for gang in expected:
        cur = store.try_get(c.KIND_PODGANG, ns, gang.name)
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
            obj["spec"]["podGroups"] = groups_spec
            cond.set_condition(obj, c.PODGANG_COND_INITIALIZED, False, "PendingPodCreation")
            try:
                cur = store.create(obj)
            except ApiError:
                cur = store.try_get(c.KIND_PODGANG, ns, gang.name)
                if cur is None:
                    continue

        def upd(o: Obj) -> None:
            o["spec"]["podGroups"] = groups_spec
        if cur["spec"].get("podGroups") != groups_spec:
            try:
                cur = store.patch(c.KIND_PODGANG, ns, gang.name, upd)
            except ApiError:
                continue

        initialized = cond.condition_true(cur, c.PODGANG_COND_INITIALIZED)
        if all_created and gang.groups and not initialized:
            def flip(o: Obj) -> None:
                cond.set_condition(o, c.PODGANG_COND_INITIALIZED, True, "AllPodsAssociated")
            try:
                store.patch(c.KIND_PODGANG, ns, gang.name, flip, status=True)
            except ApiError:
                pass
        elif not all_created and initialized:
            # pods lost after init (e.g. gang termination in flight) — keep Initialized
            pass
