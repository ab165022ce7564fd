"""The gang scheduler — MI355X-native PodGang scheduling.

The reference delegates gang placement to external schedulers (KAI/Volcano) through a
backend layer (operator/internal/scheduler/types.go:35, registry.go:97) and ships no
scheduler plugin of its own. Per BASELINE.json this build implements the real thing:

- PreFilter/Permit semantics: a PodGang is admitted only when Initialized=True and every
  podGroup has >= minReplicas ungated, unbound pods; placement is computed for the whole
  gang and bound all-or-nothing (no partial gangs, no deadlock between gangs).
- Filter: node free capacity (cpu / memory / amd.com/gpu / pod count).
- Score: xGMI topology packing (placement.py) — the whole gang in one 8×MI355X hive
  saturates 7×≈153 GB/s per-GPU fabric; split placements are NIC-bound and score low.
- PlacementScore is reported on PodGang.status (scheduler/api podgang.go:189 parity).

Runs as a single-key controller: any pod/podgang/node event enqueues one scheduling pass,
which handles every pending gang FIFO and then individually schedules gangless pods
(default-scheduler parity — gates still serialize startup).
"""
from __future__ import annotations

import logging
import time
from typing import Any, Dict, List, Optional, Set, Tuple

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError
from ..utils import conditions as cond
from ..utils.errors import report_api_error
from ..utils.quantity import cpu_millis, parse_quantity
from .placement import Assignment, NodeFree, PodRequest, place_gang

log = logging.getLogger("grove.scheduler")

GPU_IDS_ANNOTATION = "scheduling.amd.com/gpu-ids"

try:
    from . import _sched as _native  # C++ core (pybind11)
except Exception:  # pragma: no cover - exercised on boxes without the built extension
    _native = None


def native_available() -> bool:
    return _native is not None


class GangScheduler:
    def __init__(self, store: Store, scheduler_name: str = c.SCHEDULER_AMD_GANG,
                 use_native: Optional[bool] = None):
        self.store = store
        self.scheduler_name = scheduler_name
        if use_native is None:
            use_native = _native is not None
        if use_native and _native is None:
            raise RuntimeError(
                "grove_amd.scheduler._sched native extension not built; "
                "run python -m grove_amd.ops.build (or __graft_entry__.build())")
        self.use_native = use_native
        self.passes = 0
        self.gangs_scheduled = 0
        # pod resource-request parse cache (requests are immutable for a pod's
        # lifetime; parsing quantities for every bound pod on every pass was the
        # dominant scheduler cost at 10k-pod scale)
        self._req_cache: Dict[str, PodRequest] = {}
        # scheduler cache (kube scheduler-cache analog): the live free-capacity
        # view survives across passes; bound pods are subtracted INCREMENTALLY by
        # uid instead of rebuilding + re-subtracting O(pods) node views every pass
        # (that rebuild made total scheduling work O(pods^2) at 10k-pod scale).
        # Invalidation: any Node add/remove/update (rv signature) or every
        # _REBUILD_EVERY passes forces a full rebuild.
        self._view: Optional[Dict[str, NodeFree]] = None
        self._node_sig: Optional[tuple] = None
        # pod uid -> (pool_key, PodRequest, gpu_ids) for release on termination
        self._consumed: Dict[str, tuple] = {}
        self._passes_since_rebuild = 0
        # event classification (fed by the cluster watch): a pass that was
        # triggered ONLY by pods becoming Ready does a per-gang rollup instead of
        # the O(all pods) classification scan — ready-churn dominates pass volume
        # at 10k-pod scale
        self._dirty_placement = True
        self._dirty_ready_gangs: set = set()
        self._rollup_forced = 0
        self._pool_of_gpu: Dict[tuple, str] = {}
        # (ns, gang, key) already surfaced as UnsatisfiableTopologyConstraint
        self._unsat_surfaced: set = set()

    _REBUILD_EVERY = 100

    # ------------------------------------------------------------------ events
    def note_pod_event(self, ev: str, obj, old) -> None:
        spec = obj.get("spec") or {}
        st = obj.get("status") or {}
        labels = obj["metadata"].get("labels") or {}
        if ev == "DELETED" or st.get("phase") in ("Succeeded", "Failed"):
            if spec.get("nodeName"):
                self._dirty_placement = True  # capacity released
            return
        if not spec.get("nodeName"):
            self._dirty_placement = True  # new/ungated pod may be placeable
            return
        was_ready = old is not None and cond.pod_is_ready(old)
        if cond.pod_is_ready(obj) and not was_ready:
            gang = labels.get(c.LABEL_PODGANG)
            if gang:
                self._dirty_ready_gangs.add(
                    (obj["metadata"].get("namespace", "default"), gang))

    def note_placement_event(self) -> None:
        """PodGang/Node change: placement inputs moved."""
        self._dirty_placement = True

    # ------------------------------------------------------------------ pass
    def reconcile(self, _ns: str = "", _name: str = "") -> None:
        """One scheduling pass over the cluster."""
        self.passes += 1
        self._rollup_forced += 1
        if not self._dirty_placement and self._rollup_forced < 50:
            # ready-churn only: per-gang rollup, no cluster-wide scan
            gangs = self._dirty_ready_gangs
            self._dirty_ready_gangs = set()
            if gangs:
                self._rollup_gangs(gangs)
            return
        self._dirty_placement = False
        self._dirty_ready_gangs = set()
        self._rollup_forced = 0
        pods = self.store.list("Pod", copy_objects=False)
        bound: List[Obj] = []
        pending_by_gang: Dict[Tuple[str, str], List[Obj]] = {}
        pending_single: List[Obj] = []
        for p in pods:
            if p["metadata"].get("deletionTimestamp"):
                continue
            if (p.get("status") or {}).get("phase") in ("Succeeded", "Failed"):
                continue  # terminal pods release their resources
            if p.get("spec", {}).get("nodeName"):
                bound.append(p)
                continue
            if cond.pod_is_gated(p):
                continue
            gang = p["metadata"].get("labels", {}).get(c.LABEL_PODGANG)
            if gang and p["spec"].get("schedulerName") == self.scheduler_name:
                key = (p["metadata"].get("namespace", "default"), gang)
                pending_by_gang.setdefault(key, []).append(p)
            else:
                pending_single.append(p)
        pods_by_name = {(p["metadata"].get("namespace", "default"),
                         p["metadata"]["name"]): p for p in pods}
        if not pending_by_gang and not pending_single:
            # nothing to place: skip the expensive node-view build/subtract
            # entirely (the common pass during ready-wait churn at scale) —
            # only the gang Ready rollup needs to run
            self._rollup_ready(pods_by_name)
            return
        nodes = self._current_view(bound)
        if not nodes:
            return
        node_list = list(nodes.values())

        # ---- gang scheduling, FIFO by PodGang creation
        gangs: List[Obj] = []
        for (ns, gname) in pending_by_gang:
            pg = self.store.try_get(c.KIND_PODGANG, ns, gname, copy=False)
            if pg is not None:
                gangs.append(pg)
        prio_values: Dict[str, int] = {
            p["metadata"]["name"]: int((p.get("value") or 0))
            for p in self.store.list("PriorityClass", copy_objects=False)}
        gangs.sort(key=lambda g: (
            -prio_values.get((g.get("spec") or {}).get("priorityClassName", ""), 0),
            g["metadata"].get("creationTimestamp", ""),
            int(g["metadata"].get("resourceVersion", "0"))))
        for pg in gangs:
            ns = pg["metadata"].get("namespace", "default")
            gname = pg["metadata"]["name"]
            gang_pods = pending_by_gang[(ns, gname)]
            if cond.condition_true(pg, c.PODGANG_COND_SCHEDULED):
                # gang already placed — late pods (replacements, scale-ups) go
                # individually, preferring the gang's existing nodes
                self._schedule_singles(node_list, gang_pods,
                                       prefer=self._gang_nodes(pg, pods_by_name))
                continue
            if not cond.condition_true(pg, c.PODGANG_COND_INITIALIZED):
                continue
            self._schedule_gang(node_list, pg, gang_pods, pods_by_name)

        # ---- individual (gangless / default-scheduler parity) pods
        self._schedule_singles(node_list, pending_single)

        # ---- PodGang Ready rollup
        self._rollup_ready(pods_by_name)

    # ------------------------------------------------------------------ node views
    def _current_view(self, bound: List[Obj]) -> Dict[str, NodeFree]:
        """Return the live free-capacity view, incrementally maintained."""
        node_objs = self.store.list("Node", copy_objects=False)
        sig = tuple(sorted((n["metadata"]["name"],
                            n["metadata"].get("resourceVersion", ""))
                           for n in node_objs))
        self._passes_since_rebuild += 1
        if (self._view is None or sig != self._node_sig
                or self._passes_since_rebuild >= self._REBUILD_EVERY):
            self._view = self._build_node_views()
            self._node_sig = sig
            self._consumed = {}
            self._passes_since_rebuild = 0
            # authoritative (node, gpu) -> pool map BEFORE subtraction: release
            # must return a pod's GPUs to the hive pool they came from, which the
            # post-subtraction free sets can no longer tell us
            self._pool_of_gpu = {
                (nf.node_name, g): key
                for key, nf in self._view.items() for g in nf.gpu_ids}
            self._subtract_bound(self._view, bound)
            for p in bound:
                self._note_consumed(p, None)
            return self._view
        pools_by_node: Optional[Dict[str, List[NodeFree]]] = None
        active_uids = set()
        for p in bound:
            uid = p["metadata"].get("uid", "")
            active_uids.add(uid)
            if uid in self._consumed:
                continue
            if pools_by_node is None:
                pools_by_node = {}
                for nf in self._view.values():
                    pools_by_node.setdefault(nf.node_name, []).append(nf)
            self._subtract_one(pools_by_node, p)
            self._note_consumed(p, None)
        # release terminal/deleted pods' resources back to their pool
        for uid in [u for u in self._consumed if u not in active_uids]:
            pool_key, req, gpu_ids = self._consumed.pop(uid)
            nf = self._view.get(pool_key)
            if nf is None:
                continue
            nf.cpu_milli += req.cpu_milli
            nf.mem_bytes += req.mem_bytes
            nf.pods += 1
            nf.gpu_ids = sorted(set(nf.gpu_ids) | set(gpu_ids))
        return self._view

    def _note_consumed(self, pod: Obj, pool_key: Optional[str]) -> None:
        uid = pod["metadata"].get("uid", "")
        if not uid:
            return
        req = self._pod_request(pod)
        ids_str = (pod["metadata"].get("annotations") or {}).get(
            GPU_IDS_ANNOTATION, "")
        gpu_ids = [int(x) for x in ids_str.split(",") if x != ""]
        if pool_key is None:
            node = pod.get("spec", {}).get("nodeName", "")
            pool_key = node
            if gpu_ids:
                pool_key = getattr(self, "_pool_of_gpu", {}).get(
                    (node, gpu_ids[0]), node)
        self._consumed[uid] = (pool_key, req, gpu_ids)

    def _subtract_one(self, pools_by_node: Dict[str, List[NodeFree]],
                      p: Obj) -> None:
        pools = pools_by_node.get(p["spec"].get("nodeName", ""))
        if not pools:
            return
        req = self._pod_request(p)
        ids = (p["metadata"].get("annotations") or {}).get(GPU_IDS_ANNOTATION, "")
        taken = {int(x) for x in ids.split(",") if x != ""} if ids else set()
        node = pools[0]
        if taken:
            for nf in pools:
                if taken & set(nf.gpu_ids):
                    node = nf
                    break
        node.cpu_milli -= req.cpu_milli
        node.mem_bytes -= req.mem_bytes
        node.pods -= 1
        if taken:
            for nf in pools:
                nf.gpu_ids = [g for g in nf.gpu_ids if g not in taken]
        elif req.gpus:
            del node.gpu_ids[: req.gpus]

    def _build_node_views(self) -> Dict[str, NodeFree]:
        """One NodeFree POOL per xGMI hive (schedule from the
        DISCOVERED fabric). The topology agent publishes the hive partition in the
        topology.amd.com/xgmi-hives annotation and the measured min per-link
        bandwidth in topology.amd.com/xgmi-min-gbps; a partitioned node becomes
        several pools (keys "<node>#h<k>") so a gang never silently spans hives,
        and the placement score reports the measured bandwidth."""
        out: Dict[str, NodeFree] = {}
        for n in self.store.list("Node", copy_objects=False):
            if (n.get("spec") or {}).get("unschedulable"):
                continue
            name = n["metadata"]["name"]
            alloc = (n.get("status") or {}).get("allocatable") or {}
            labels = n["metadata"].get("labels") or {}
            ann = n["metadata"].get("annotations") or {}
            gpus = int(parse_quantity(alloc.get(c.AMD_GPU_RESOURCE, 0)))
            cpu = cpu_millis(alloc.get("cpu", 0))
            mem = parse_quantity(alloc.get("memory", 0))
            pods = int(parse_quantity(alloc.get("pods", 250)))
            link = None
            try:
                link = float(ann.get("topology.amd.com/xgmi-min-gbps", ""))
            except ValueError:
                pass
            hives: List[List[int]] = []
            raw = ann.get("topology.amd.com/xgmi-hives", "")
            if raw:
                try:
                    hives = [[int(x) for x in part.split(",") if x != ""]
                             for part in raw.split(";") if part]
                except ValueError:
                    hives = []
            covered = {g for h in hives for g in h}
            if gpus and (not hives or covered != set(range(gpus))):
                hives = [list(range(gpus))]
            if not gpus:
                hives = [[]]
            n_pools = max(1, len(hives))
            for k, hive in enumerate(hives):
                key = name if k == 0 else f"{name}#h{k}"
                frac = (len(hive) / gpus) if gpus else 1.0 / n_pools
                out[key] = NodeFree(
                    key, int(cpu * frac), mem * frac, sorted(hive),
                    max(1, int(pods * frac)), labels,
                    node_name=name, link_gbps=link)
        return out

    def _subtract_bound(self, nodes: Dict[str, NodeFree], bound: List[Obj]) -> None:
        pools_by_node: Dict[str, List[NodeFree]] = {}
        for nf in nodes.values():
            pools_by_node.setdefault(nf.node_name, []).append(nf)
        for p in bound:
            pools = pools_by_node.get(p["spec"]["nodeName"])
            if not pools:
                continue
            req = self._pod_request(p)
            ids = (p["metadata"].get("annotations") or {}).get(GPU_IDS_ANNOTATION, "")
            taken = {int(x) for x in ids.split(",") if x != ""} if ids else set()
            # charge the pool holding the pod's GPUs (fall back to the first pool)
            node = pools[0]
            if taken:
                for nf in pools:
                    if taken & set(nf.gpu_ids):
                        node = nf
                        break
            node.cpu_milli -= req.cpu_milli
            node.mem_bytes -= req.mem_bytes
            node.pods -= 1
            if taken:
                for nf in pools:
                    nf.gpu_ids = [g for g in nf.gpu_ids if g not in taken]
            elif req.gpus:
                del node.gpu_ids[: req.gpus]

    def _pod_request(self, pod: Obj) -> PodRequest:
        uid = pod["metadata"].get("uid", "")
        cached = self._req_cache.get(uid)
        if cached is not None:
            return cached
        cpu = 0
        mem = 0.0
        gpus = 0
        for ctr in (pod.get("spec", {}).get("containers") or []):
            r = ((ctr.get("resources") or {}).get("requests")
                 or (ctr.get("resources") or {}).get("limits") or {})
            cpu += cpu_millis(r.get("cpu", 0))
            mem += parse_quantity(r.get("memory", 0))
            gpus += int(parse_quantity(r.get(c.AMD_GPU_RESOURCE, 0)))
        req = PodRequest(pod["metadata"]["name"], cpu, mem, gpus)
        if uid:
            if len(self._req_cache) > 50000:
                self._req_cache.clear()  # bound memory across churny lifetimes
            self._req_cache[uid] = req
        return req

    def _gang_nodes(self, pg: Obj, pods_by_name: Dict) -> Set[str]:
        ns = pg["metadata"].get("namespace", "default")
        names: Set[str] = set()
        for group in (pg.get("spec") or {}).get("podgroups") or []:
            for ref in group.get("podReferences") or []:
                p = pods_by_name.get((ns, ref.get("name", "")))
                if p and p.get("spec", {}).get("nodeName"):
                    names.add(p["spec"]["nodeName"])
        return names

    # ------------------------------------------------------------------ gang place
    def _schedule_gang(self, nodes: List[NodeFree], pg: Obj, gang_pods: List[Obj],
                       pods_by_name: Dict) -> None:
        ns = pg["metadata"].get("namespace", "default")
        groups = (pg.get("spec") or {}).get("podgroups") or []
        by_clique: Dict[str, List[Obj]] = {}
        for p in gang_pods:
            by_clique.setdefault(
                p["metadata"]["labels"].get(c.LABEL_PODCLIQUE, ""), []).append(p)

        # admission: every group needs >= minReplicas ungated unbound pods available
        chosen: List[Obj] = []
        for g in groups:
            want = int(g.get("minReplicas", 0))
            have = by_clique.get(g["name"], [])
            already = self._count_bound(ns, g, pods_by_name)
            need = max(0, want - already)
            if len(have) < need:
                return  # not admittable yet
            chosen.extend(have)  # place everything available, all-or-nothing on the mins

        # reuseReservationRef (podgang.go:120): a rolling update's replacement gang
        # prefers the placement of the gang it replaces — try its nodes first so the
        # new pods land on warm hives.
        ref = (pg.get("spec") or {}).get("reuseReservationRef") or {}
        if ref.get("name"):
            prev = self.store.try_get(c.KIND_PODGANG,
                                      ref.get("namespace") or ns, ref["name"])
            if prev is not None:
                prev_nodes = self._gang_nodes(prev, pods_by_name)
                pool = [n for n in nodes if n.node_name in prev_nodes]
                if pool:
                    res = self._place_gang_pods(pool, pg, chosen)
                    if res is not None:
                        assignments, score = res
                        by_name0 = {p["metadata"]["name"]: p for p in chosen}
                        for a in assignments:
                            self._bind(by_name0[a.pod], a)
                        self.gangs_scheduled += 1

                        def mark0(o: Obj) -> None:
                            cond.set_condition(o, c.PODGANG_COND_SCHEDULED, True,
                                               "GangPlacedOnReservation")
                            o["status"]["placementScore"] = round(score, 3)
                            o["status"]["phase"] = "Starting"
                        try:
                            self.store.patch(c.KIND_PODGANG, ns,
                                             pg["metadata"]["name"], mark0, status=True, return_copy=False)
                        except ApiError as e:
                            report_api_error(self.store, c.KIND_PODGANG, ns,
                                             pg["metadata"]["name"],
                                             "mark scheduled (reservation)", e)
                        return

        result = self._place_gang_pods(nodes, pg, chosen)
        if result is None and len(chosen) > sum(int(g.get("minReplicas", 0))
                                                for g in groups):
            # fall back to the gang minimum only
            chosen2: List[Obj] = []
            for g in groups:
                have = by_clique.get(g["name"], [])
                chosen2.extend(have[: int(g.get("minReplicas", 0))])
            chosen = chosen2
            result = self._place_gang_pods(nodes, pg, chosen)
        if result is None:
            self._surface_unsatisfiable(pg, ns)
            return  # Permit rollback: nothing bound
        assignments, score = result
        by_name = {p["metadata"]["name"]: p for p in chosen}
        for a in assignments:
            self._bind(by_name[a.pod], a)
        self.gangs_scheduled += 1

        def mark(o: Obj) -> None:
            cond.set_condition(o, c.PODGANG_COND_SCHEDULED, True, "GangPlaced")
            o["status"]["placementScore"] = round(score, 3)
            o["status"]["phase"] = "Starting"
        try:
            self.store.patch(c.KIND_PODGANG, ns, pg["metadata"]["name"], mark, status=True, return_copy=False)
        except ApiError as e:
            report_api_error(self.store, c.KIND_PODGANG, ns,
                             pg["metadata"]["name"], "mark gang scheduled", e)

    def _surface_unsatisfiable(self, pg: Obj, ns: str) -> None:
        """TAS20 parity (topology_test.go unavailable-level scenario): when a gang
        cannot place because its REQUIRED pack key labels no node at all, say so —
        a Warning Event on the PodGang (deduped per gang+key). Transient capacity
        shortages are not surfaced here; only a structurally absent topology level."""
        req = ((pg.get("spec") or {}).get("topologyConstraint") or {}) \
            .get("packConstraint", {}).get("required")
        if not req:
            return
        key = (ns, pg["metadata"]["name"], req)
        if key in self._unsat_surfaced:
            return
        for n in self.store.list("Node", copy_objects=False):
            if req in (n["metadata"].get("labels") or {}):
                return  # level exists somewhere — normal capacity wait
        self._unsat_surfaced.add(key)
        self.store.record_event(
            pg, "Warning", "UnsatisfiableTopologyConstraint",
            f"required pack key {req!r} labels no node in the cluster")

    def _count_bound(self, ns: str, group: Obj, pods_by_name: Dict) -> int:
        n = 0
        for ref in group.get("podReferences") or []:
            p = pods_by_name.get((ns, ref.get("name", "")))
            if p and p.get("spec", {}).get("nodeName"):
                n += 1
        return n

    def _place_gang_pods(self, nodes: List[NodeFree], pg: Obj, chosen: List[Obj]):
        """Constraint-aware all-or-nothing placement of a gang's pods.

        Applies the gang-level packConstraint plus topologyConstraintGroupConfigs
        (each config's pods must pack within one domain of ITS key) with full rollback
        when any part cannot place."""
        from .placement import place_gang_constrained, snapshot, restore

        spec = pg.get("spec") or {}
        tc = (spec.get("topologyConstraint") or {}).get("packConstraint") or {}
        req_key, pref_key = tc.get("required"), tc.get("preferred")
        cfgs = spec.get("topologyConstraintGroupConfigs") or []
        if not cfgs:
            return place_gang_constrained(nodes, [self._pod_request(p) for p in chosen],
                                          req_key, pref_key, place_fn=self._place)
        by_clique: Dict[str, List[Obj]] = {}
        for p in chosen:
            by_clique.setdefault(
                p["metadata"]["labels"].get(c.LABEL_PODCLIQUE, ""), []).append(p)

        if not req_key:
            return self._place_cfg_groups(nodes, cfgs, by_clique, chosen,
                                          req_key, pref_key)
        # gang-level REQUIRED key + subgroup configs: the parent constraint binds
        # ALL subgroups into ONE domain of req_key (KAI hierarchical-subgroup
        # semantics) — try each domain pool, fullest-fitting first
        domains: Dict[str, List[NodeFree]] = {}
        for n in nodes:
            v = n.labels.get(req_key)
            if v is not None:
                domains.setdefault(v, []).append(n)
        order = sorted(domains, key=lambda v: (
            -sum(len(n.gpu_ids) for n in domains[v]),
            -sum(n.cpu_milli for n in domains[v]), v))
        for v in order:
            res = self._place_cfg_groups(domains[v], cfgs, by_clique, chosen,
                                         req_key, pref_key)
            if res is not None:
                return res
        return None

    def _place_cfg_groups(self, nodes: List[NodeFree], cfgs: List[Obj],
                          by_clique: Dict[str, List[Obj]], chosen: List[Obj],
                          req_key, pref_key):
        """Place each topologyConstraintGroupConfig subgroup (then the remainder)
        within the given node pool, all-or-nothing with rollback."""
        from .placement import place_gang_constrained, snapshot, restore
        snap = snapshot(nodes)
        claimed: set = set()
        all_assignments: List[Assignment] = []
        worst_score = float("inf")
        for cfg in cfgs:
            sub_pods: List[Obj] = []
            for gname in cfg.get("podGroupNames") or []:
                sub_pods.extend(by_clique.get(gname, []))
            if not sub_pods:
                continue
            sub_tc = ((cfg.get("topologyConstraint") or {})
                      .get("packConstraint") or {})
            res = place_gang_constrained(
                nodes, [self._pod_request(p) for p in sub_pods],
                sub_tc.get("required") or req_key,
                sub_tc.get("preferred") or pref_key, place_fn=self._place)
            if res is None:
                restore(snap)
                return None
            assignments, score = res
            all_assignments.extend(assignments)
            worst_score = min(worst_score, score)
            claimed.update(p["metadata"]["name"] for p in sub_pods)
        rest = [p for p in chosen if p["metadata"]["name"] not in claimed]
        if rest:
            res = place_gang_constrained(
                nodes, [self._pod_request(p) for p in rest], req_key, pref_key,
                place_fn=self._place)
            if res is None:
                restore(snap)
                return None
            assignments, score = res
            all_assignments.extend(assignments)
            worst_score = min(worst_score, score)
        return all_assignments, (worst_score if worst_score != float("inf") else 0.0)

    def _place(self, nodes: List[NodeFree], reqs: List[PodRequest]):
        if self.use_native and _native is not None:
            flat_nodes = [(n.name, n.cpu_milli, float(n.mem_bytes), list(n.gpu_ids),
                           n.pods, float(n.link_gbps)) for n in nodes]
            flat_pods = [(p.name, p.cpu_milli, float(p.mem_bytes), p.gpus) for p in reqs]
            res = _native.place_gang(flat_nodes, flat_pods)
            if res is None:
                return None
            assignments, score, consumed = res
            # apply consumption back onto the python node views
            node_by_name = {n.name: n for n in nodes}
            for (nname, cpu, mem, gpu_ids, pods_, _link) in consumed:
                n = node_by_name[nname]
                n.cpu_milli, n.mem_bytes, n.gpu_ids, n.pods = cpu, mem, list(gpu_ids), pods_
            return ([Assignment(p, nd, list(g)) for (p, nd, g) in assignments], score)
        return place_gang(nodes, reqs)

    # ------------------------------------------------------------------ singles
    def _schedule_singles(self, nodes: List[NodeFree], pods: List[Obj],
                          prefer: Optional[Set[str]] = None) -> None:
        for p in pods:
            req = self._pod_request(p)
            order = nodes
            if prefer:
                order = sorted(nodes, key=lambda n: n.node_name not in prefer)
            else:
                # pack GPU pods (best-fit), spread cpu pods (worst-fit) lightly
                order = sorted(nodes, key=lambda n: len(n.gpu_ids)) if req.gpus \
                    else sorted(nodes, key=lambda n: -n.cpu_milli)
            for n in order:
                if n.cpu_milli >= req.cpu_milli and n.mem_bytes >= req.mem_bytes \
                        and len(n.gpu_ids) >= req.gpus and n.pods >= 1:
                    n.cpu_milli -= req.cpu_milli
                    n.mem_bytes -= req.mem_bytes
                    n.pods -= 1
                    gpu_ids = n.gpu_ids[: req.gpus]
                    del n.gpu_ids[: req.gpus]
                    self._bind(p, Assignment(p["metadata"]["name"], n.node_name,
                                             gpu_ids))
                    uid = p["metadata"].get("uid", "")
                    if uid:
                        # record against the POOL (n.name), not the node — a
                        # multi-hive node's release must return GPUs to the
                        # right hive pool
                        self._consumed[uid] = (n.name, req, list(gpu_ids))
                    break

    # ------------------------------------------------------------------ bind
    def _bind(self, pod: Obj, a: Assignment) -> None:
        ns = pod["metadata"].get("namespace", "default")
        # consumption was already applied to the live view by the placement —
        # register it under the pod uid so the incremental pass never
        # double-subtracts this pod when it shows up bound
        uid = pod["metadata"].get("uid", "")
        if uid:
            self._consumed[uid] = (a.node, self._pod_request(pod),
                                   list(a.gpu_ids))

        def apply(o: Obj) -> None:
            if o["metadata"].get("deletionTimestamp"):
                raise ApiError(409, "Conflict", "pod deleting")
            o["spec"]["nodeName"] = a.node.split("#", 1)[0]
            if a.gpu_ids:
                o["metadata"].setdefault("annotations", {})[GPU_IDS_ANNOTATION] = \
                    ",".join(str(g) for g in a.gpu_ids)
            st = o.setdefault("status", {})
            conds = st.setdefault("conditions", [])
            for cd in conds:
                if cd.get("type") == "PodScheduled":
                    cd["status"] = "True"
                    break
            else:
                conds.append({"type": "PodScheduled", "status": "True",
                              "reason": "Scheduled",
                              "lastTransitionTime": time.strftime(
                                  "%Y-%m-%dT%H:%M:%SZ", time.gmtime())})
        try:
            self.store.patch("Pod", ns, pod["metadata"]["name"], apply,
                             return_copy=False)
        except ApiError as e:
            report_api_error(self.store, "Pod", ns, pod["metadata"]["name"],
                             "bind pod", e)

    # ------------------------------------------------------------------ ready rollup
    def _rollup_gangs(self, gangs: set) -> None:
        """Targeted Ready rollup for gangs whose pods just became Ready."""
        for (ns, gname) in gangs:
            pg = self.store.try_get(c.KIND_PODGANG, ns, gname, copy=False)
            if pg is None or not cond.condition_true(pg, c.PODGANG_COND_SCHEDULED) \
                    or cond.condition_true(pg, c.PODGANG_COND_READY):
                continue
            pods = {p["metadata"]["name"]: p for p in self.store.list(
                "Pod", ns, {c.LABEL_PODGANG: gname}, copy_objects=False)}
            ready = True
            for group in (pg.get("spec") or {}).get("podgroups") or []:
                n = sum(1 for ref in group.get("podReferences") or []
                        if (p := pods.get(ref.get("name", ""))) is not None
                        and cond.pod_is_ready(p))
                if n < int(group.get("minReplicas", 0)):
                    ready = False
                    break
            if ready:
                def mark(o: Obj) -> None:
                    cond.set_condition(o, c.PODGANG_COND_READY, True,
                                       "AllPodGroupsReady")
                    o["status"]["phase"] = "Running"
                try:
                    self.store.patch(c.KIND_PODGANG, ns, gname, mark,
                                     status=True, return_copy=False)
                except ApiError as e:
                    report_api_error(self.store, c.KIND_PODGANG, ns, gname,
                                     "mark gang ready", e)

    def _rollup_ready(self, pods_by_name: Dict) -> None:
        for pg in self.store.list(c.KIND_PODGANG, copy_objects=False):
            if not cond.condition_true(pg, c.PODGANG_COND_SCHEDULED):
                continue
            if cond.condition_true(pg, c.PODGANG_COND_READY):
                continue
            ns = pg["metadata"].get("namespace", "default")
            ready = True
            for group in (pg.get("spec") or {}).get("podgroups") or []:
                n = 0
                for ref in group.get("podReferences") or []:
                    p = pods_by_name.get((ns, ref.get("name", "")))
                    if p and cond.pod_is_ready(p):
                        n += 1
                if n < int(group.get("minReplicas", 0)):
                    ready = False
                    break
            if ready:
                def mark(o: Obj) -> None:
                    cond.set_condition(o, c.PODGANG_COND_READY, True, "AllPodGroupsReady")
                    o["status"]["phase"] = "Running"
                try:
                    self.store.patch(c.KIND_PODGANG, ns, pg["metadata"]["name"], mark,
                                     status=True)
                except ApiError as e:
                    report_api_error(self.store, c.KIND_PODGANG, ns,
                                     pg["metadata"]["name"], "mark gang ready", e)
