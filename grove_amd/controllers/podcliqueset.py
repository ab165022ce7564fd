"""PodCliqueSet controller — the orchestrator.

Behavior parity: operator/internal/controller/podcliqueset/ — finalizer + generation-hash
change detection (reconcilespec.go:41-158), dependency-grouped component sync
(reconcilespec.go:162-282: G1 RBAC/Service/HPA, G2 standalone PodCliques, G3 PCSG+PodGang),
gang termination at PCS-replica scope (components/podcliquesetreplica/gangterminate.go),
rolling-update orchestration one replica at a time (rollingupdate.go:37-296), and status
aggregation (reconcilestatus.go). Fresh MI355X-native implementation.
"""
from __future__ import annotations

import calendar
import logging
import time
from typing import Any, Dict, List, Optional, Set, Tuple

from ..api import constants as c
from ..api import namegen
from ..api.defaulting import parse_duration_seconds
from ..kubecore.store import Store, Obj, ApiError
from ..utils import conditions as cond
from ..utils.hashing import pcs_generation_hash, pod_template_hash
from ..utils import errors as groveerr
from . import builders
from .manager import Result
from .podgang_component import sync_podgangs
from . import resourceclaims

log = logging.getLogger("grove.podcliqueset")


def _iso_to_epoch(ts: str) -> float:
    """UTC condition timestamp → epoch seconds. calendar.timegm is DST-proof
    (time.mktime - time.timezone is off by an hour during DST). A malformed
    timestamp must NOT reset the breach clock (that would postpone gang
    termination forever), so parse failure returns 0.0 == 'breached long ago'."""
    try:
        return float(calendar.timegm(time.strptime(ts, "%Y-%m-%dT%H:%M:%SZ")))
    except Exception:
        return 0.0


class PodCliqueSetReconciler:
    def __init__(self, store: Store, scheduler_name: str = c.SCHEDULER_AMD_GANG,
                 auto_xgmi_domain: bool = False):
        self.store = store
        self.scheduler_name = scheduler_name
        self.auto_xgmi_domain = auto_xgmi_domain

    # ------------------------------------------------------------------ entry
    def reconcile(self, namespace: str, name: str) -> Result:
        """Top-level reconcile; errors are recorded to status.lastErrors with their
        ERR_* code (reconcileerrorrecorder.go parity) and re-raised for backoff."""
        try:
            res = self._reconcile(namespace, name)
            groveerr.clear_last_errors(self.store, c.KIND_PCS, namespace, name)
            return res
        except groveerr.GroveError as e:
            groveerr.record_last_error(self.store, c.KIND_PCS, namespace, name,
                                       e.code, e.message)
            raise
        except Exception as e:
            groveerr.record_last_error(self.store, c.KIND_PCS, namespace, name,
                                       groveerr.ERR_RECONCILE, str(e))
            raise

    def _reconcile(self, namespace: str, name: str) -> Result:
        pcs = self.store.try_get(c.KIND_PCS, namespace, name)
        if pcs is None:
            return Result.DONE
        if pcs["metadata"].get("deletionTimestamp"):
            return self._reconcile_delete(pcs)
        if c.FINALIZER_PCS not in (pcs["metadata"].get("finalizers") or []):
            def add_fin(o: Obj) -> None:
                o["metadata"].setdefault("finalizers", [])
                if c.FINALIZER_PCS not in o["metadata"]["finalizers"]:
                    o["metadata"]["finalizers"].append(c.FINALIZER_PCS)
            pcs = self.store.patch(c.KIND_PCS, namespace, name, add_fin)

        self._process_generation_hash(pcs)
        pcs = self.store.get(c.KIND_PCS, namespace, name)
        res = self._sync_resources(pcs)
        term = self._gang_termination(pcs)
        self._orchestrate_rolling_update(pcs)
        self._reconcile_status(namespace, name)
        if term is not None:
            return Result(requeue_after=term)
        return res

    # ------------------------------------------------------------------ delete
    def _reconcile_delete(self, pcs: Obj) -> Result:
        ns, name = pcs["metadata"].get("namespace"), pcs["metadata"]["name"]
        sel = {c.LABEL_PART_OF: name}
        remaining = 0
        for kind in (c.KIND_PCLQ, c.KIND_PCSG):
            for obj in self.store.list(kind, ns, sel):
                remaining += 1
                try:
                    self.store.delete(kind, ns, obj["metadata"]["name"])
                except ApiError:
                    pass
        if remaining:
            return Result(requeue_after=0.02)
        for pg in self.store.list(c.KIND_PODGANG, ns, sel):
            try:
                self.store.delete(c.KIND_PODGANG, ns, pg["metadata"]["name"])
            except ApiError:
                pass

        def rm(o: Obj) -> None:
            o["metadata"]["finalizers"] = [
                f for f in o["metadata"].get("finalizers", []) if f != c.FINALIZER_PCS]
        try:
            self.store.patch(c.KIND_PCS, ns, name, rm)
        except ApiError:
            pass
        return Result.DONE

    # ------------------------------------------------------------------ generation hash
    def _process_generation_hash(self, pcs: Obj) -> None:
        """reconcilespec.go:72-158: persist template hash; on change, start update."""
        ns, name = pcs["metadata"].get("namespace"), pcs["metadata"]["name"]
        new_hash = pcs_generation_hash(pcs)
        st = pcs.get("status") or {}
        cur_hash = st.get("currentGenerationHash")
        if cur_hash == new_hash:
            return

        def upd(o: Obj) -> None:
            s = o.setdefault("status", {})
            if s.get("currentGenerationHash") and s.get("currentGenerationHash") != new_hash:
                s["updateProgress"] = {
                    "updateStartedAt": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
                    "currentlyUpdating": None,
                    "updatedReplicas": [],
                }
            s["currentGenerationHash"] = new_hash
        try:
            self.store.patch(c.KIND_PCS, ns, name, upd, status=True)
        except ApiError:
            pass

    # ------------------------------------------------------------------ spec sync
    def _sync_resources(self, pcs: Obj) -> Result:
        ns = pcs["metadata"].get("namespace", "default")
        name = pcs["metadata"]["name"]
        replicas = int(pcs["spec"].get("replicas", 0))
        tmpl = pcs["spec"]["template"]
        sg_cfgs = tmpl.get("podCliqueScalingGroups") or []
        sg_members = {m for sg in sg_cfgs for m in (sg.get("cliqueNames") or [])}

        # ---- G1: RBAC + token secret + per-replica headless Service + HPAs
        self._ensure(builders.build_service_account(pcs))
        self._ensure(builders.build_role(pcs))
        self._ensure(builders.build_role_binding(pcs))
        self._ensure(builders.build_sa_token_secret(pcs))
        existing_svcs = {s["metadata"]["name"] for s in self.store.list(
            "Service", ns, {c.LABEL_PART_OF: name,
                            c.LABEL_COMPONENT: c.COMPONENT_HEADLESS_SERVICE})}
        for r in range(replicas):
            svc_name = namegen.headless_service_name(name, r)
            if svc_name not in existing_svcs:
                self._ensure(builders.build_headless_service(pcs, r))
        for svc_name in existing_svcs:
            idx = svc_name.rsplit("-", 1)[-1]
            if idx.isdigit() and int(idx) >= replicas:
                try:
                    self.store.delete("Service", ns, svc_name)
                except ApiError:
                    pass
        self._sync_hpas(pcs)

        # ---- G2: standalone PodCliques per replica (+ shared ResourceClaims)
        pclq_by_name = {q["metadata"]["name"]: q for q in self.store.list(
            c.KIND_PCLQ, ns, {c.LABEL_PART_OF: name,
                              c.LABEL_COMPONENT: c.COMPONENT_PCS_PODCLIQUE},
            copy_objects=False)}
        expected_pclqs: Set[str] = set()
        for r in range(replicas):
            claims = resourceclaims.pcs_claims_for_replica(
                self.store, pcs, r, self.auto_xgmi_domain)
            resourceclaims.ensure_claims(self.store, claims)
            for cl in tmpl.get("cliques") or []:
                if cl["name"] in sg_members:
                    continue
                fqn = namegen.podclique_name(name, r, cl["name"])
                expected_pclqs.add(fqn)
                refs = resourceclaims.claim_refs_for_clique(claims, cl["name"])
                cl_claims = resourceclaims.clique_level_claims(
                    self.store, pcs, r, cl, fqn)
                resourceclaims.ensure_claims(self.store, cl_claims)
                refs = refs + resourceclaims.claim_refs_for_clique(
                    cl_claims, cl["name"])
                self._sync_pclq(pcs, r, cl, fqn, owner=pcs, claim_refs=refs,
                                cur=pclq_by_name.get(fqn))
        # GC excess standalone PCLQs (scale-in / replica removal)
        for pclq in pclq_by_name.values():
            if pclq["metadata"]["name"] not in expected_pclqs:
                try:
                    self.store.delete(c.KIND_PCLQ, ns, pclq["metadata"]["name"])
                except ApiError:
                    pass

        # ---- G3a: PCSGs per config per replica
        pcsg_names = {g["metadata"]["name"] for g in self.store.list(
            c.KIND_PCSG, ns, {c.LABEL_PART_OF: name}, copy_objects=False)}
        expected_pcsgs: Set[str] = set()
        for r in range(replicas):
            for sg in sg_cfgs:
                fqn = namegen.pcsg_name(name, r, sg["name"])
                expected_pcsgs.add(fqn)
                if fqn not in pcsg_names:
                    try:
                        self.store.create(builders.build_pcsg(pcs, r, sg))
                    except ApiError:
                        pass
        for pcsg in self.store.list(c.KIND_PCSG, ns, {
                c.LABEL_PART_OF: name, c.LABEL_COMPONENT: c.COMPONENT_PCSG}):
            if pcsg["metadata"]["name"] not in expected_pcsgs:
                try:
                    self.store.delete(c.KIND_PCSG, ns, pcsg["metadata"]["name"])
                except ApiError:
                    pass

        # ---- G3b: PodGangs
        sync_podgangs(self.store, pcs, self.scheduler_name)
        return Result.DONE

    def _sync_pclq(self, pcs: Obj, r: int, clique_tmpl: Obj, fqn: str, owner: Obj,
                   claim_refs=None, cur="__lookup__") -> None:
        ns = pcs["metadata"].get("namespace", "default")
        if cur == "__lookup__":
            cur = self.store.try_get(c.KIND_PCLQ, ns, fqn)
        if cur is None:
            obj = builders.build_podclique(pcs, r, clique_tmpl, owner)
            obj["spec"]["updateStrategy"] = (pcs["spec"].get("updateStrategy") or {}).get(
                "type", c.UPDATE_ROLLING_RECREATE)
            if claim_refs:
                obj["spec"]["resourceClaims"] = claim_refs
            try:
                self.store.create(obj)
            except ApiError:
                pass
            return
        if cur["metadata"].get("deletionTimestamp"):
            return
        # propagate template changes only to replicas selected for update (GREP-393)
        new_hash = pod_template_hash(clique_tmpl["name"],
                                     clique_tmpl["spec"].get("podSpec", {}),
                                     pcs["spec"]["template"].get("priorityClassName", ""),
                                     clique_tmpl.get("labels"),
                                     clique_tmpl.get("annotations"))
        if cur["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH) != new_hash \
                and self._replica_selected_for_update(pcs, r):
            def upd(o: Obj) -> None:
                o["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] = new_hash
                o["spec"]["podSpec"] = clique_tmpl["spec"].get("podSpec", {})
                # HPA-aware replica preservation (podclique.go:284): never stomp the
                # replicas of an autoscaled clique
                fields = ("minAvailable",) if clique_tmpl["spec"].get(
                    "autoScalingConfig") else ("replicas", "minAvailable")
                for f in fields:
                    if f in clique_tmpl["spec"]:
                        o["spec"][f] = clique_tmpl["spec"][f]
            try:
                self.store.patch(c.KIND_PCLQ, ns, fqn, upd)
            except ApiError:
                pass
        # HPA-aware replica preservation: never stomp replicas on scaled cliques —
        # only non-scaled fields drift-corrected here.

    def _replica_selected_for_update(self, pcs: Obj, r: int) -> bool:
        if (pcs["spec"].get("updateStrategy") or {}).get("type") == c.UPDATE_ON_DELETE:
            return True  # OnDelete: spec propagates immediately, pods wait for the user
        prog = (pcs.get("status") or {}).get("updateProgress")
        if prog is None:
            return True  # no update in flight → initial create path
        cu = prog.get("currentlyUpdating")
        return cu is not None and int(cu.get("replicaIndex", -1)) == r

    def _sync_hpas(self, pcs: Obj) -> None:
        ns = pcs["metadata"].get("namespace", "default")
        name = pcs["metadata"]["name"]
        replicas = int(pcs["spec"].get("replicas", 0))
        tmpl = pcs["spec"]["template"]
        sg_members = {m for sg in tmpl.get("podCliqueScalingGroups") or []
                      for m in (sg.get("cliqueNames") or [])}
        expected: Dict[str, Tuple[str, str, Obj]] = {}
        for r in range(replicas):
            for cl in tmpl.get("cliques") or []:
                asc = (cl.get("spec") or {}).get("autoScalingConfig")
                if asc and cl["name"] not in sg_members:
                    t = namegen.podclique_name(name, r, cl["name"])
                    expected[t] = (c.KIND_PCLQ, t, asc)
            for sg in tmpl.get("podCliqueScalingGroups") or []:
                if sg.get("scaleConfig"):
                    t = namegen.pcsg_name(name, r, sg["name"])
                    expected[t] = (c.KIND_PCSG, t, sg["scaleConfig"])
        existing = {h["metadata"]["name"]: h for h in self.store.list(
            "HorizontalPodAutoscaler", ns,
            {c.LABEL_PART_OF: name, c.LABEL_COMPONENT: c.COMPONENT_HPA})}
        for hname, (kind, target, cfg) in expected.items():
            if hname not in existing:
                self._ensure(builders.build_hpa(pcs, kind, target, cfg))
        for hname in existing:
            if hname not in expected:
                try:
                    self.store.delete("HorizontalPodAutoscaler", ns, hname)
                except ApiError:
                    pass

    def _ensure(self, obj: Obj) -> None:
        try:
            self.store.create(obj)
        except ApiError as e:
            if e.reason != "AlreadyExists":
                raise

    # ------------------------------------------------------------------ gang termination
    def _gang_termination(self, pcs: Obj) -> Optional[float]:
        """PCS-replica-scope gang termination (gangterminate.go:69-332). Returns seconds
        until the next pending termination (for requeue), or None."""
        ns = pcs["metadata"].get("namespace", "default")
        name = pcs["metadata"]["name"]
        delay = parse_duration_seconds(
            pcs["spec"]["template"].get("terminationDelay", "4h"))
        next_wait: Optional[float] = None
        now = time.time()

        pclqs = self.store.list(c.KIND_PCLQ, ns, {c.LABEL_PART_OF: name}, copy_objects=False)
        pcsgs = {g["metadata"]["name"]: g for g in
                 self.store.list(c.KIND_PCSG, ns, {c.LABEL_PART_OF: name},
                                 copy_objects=False)}
        by_replica: Dict[int, List[Obj]] = {}
        for q in pclqs:
            # PCS-scope constituents are STANDALONE cliques + the PCSGs themselves;
            # PCSG member cliques are judged through their PCSG's MinAvailableBreached
            # (getPCSReplicaDeletionWork parity) and recycled at PCSG scope.
            if q["metadata"]["labels"].get(c.LABEL_COMPONENT) != c.COMPONENT_PCS_PODCLIQUE:
                continue
            ridx = q["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX)
            if ridx is not None and ridx.isdigit():
                by_replica.setdefault(int(ridx), []).append(q)
        for g in pcsgs.values():
            ridx = g["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX)
            if ridx is not None and ridx.isdigit():
                by_replica.setdefault(int(ridx), [])

        cu = ((pcs.get("status") or {}).get("updateProgress") or {}) \
            .get("currentlyUpdating") or {}
        updating_replica = int(cu.get("replicaIndex", -1))
        for ridx, constituents in by_replica.items():
            # suppression: a replica mid-rolling-update dips below MinAvailable by
            # design; never gang-terminate it while it is the one being updated
            if ridx == updating_replica:
                continue
            # suppression: in-flight termination for this replica's PCSGs
            replica_pcsgs = [g for g in pcsgs.values()
                             if g["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX)
                             == str(ridx)]
            if any(cond.condition_true(g, c.COND_GANG_TERMINATION_IN_PROGRESS)
                   for g in replica_pcsgs):
                continue
            breach_since: Optional[float] = None
            for q in constituents:
                bc = cond.get_condition(q, c.COND_MIN_AVAILABLE_BREACHED)
                if bc and bc.get("status") == "True":
                    ts = _iso_to_epoch(bc.get("lastTransitionTime", ""))
                    breach_since = ts if breach_since is None else min(breach_since, ts)
            for g in replica_pcsgs:
                bc = cond.get_condition(g, c.COND_MIN_AVAILABLE_BREACHED)
                if bc and bc.get("status") == "True":
                    ts = _iso_to_epoch(bc.get("lastTransitionTime", ""))
                    breach_since = ts if breach_since is None else min(breach_since, ts)
            if breach_since is None:
                continue
            remaining = delay - (now - breach_since)
            if remaining > 0:
                next_wait = remaining if next_wait is None else min(next_wait, remaining)
                continue
            # fire: mark PCSGs in-progress first (action-first crash ordering), then
            # delete all PodCliques of the replica
            log.info("gang termination fired for %s/%s replica %d", ns, name, ridx)
            for g in replica_pcsgs:
                def mark(o: Obj) -> None:
                    cond.set_condition(o, c.COND_GANG_TERMINATION_IN_PROGRESS, True,
                                       c.REASON_GANG_TERMINATION_ACTIVE)
                try:
                    self.store.patch(c.KIND_PCSG, ns, g["metadata"]["name"], mark,
                                     status=True)
                except ApiError:
                    pass
            self.store.delete_collection(c.KIND_PCLQ, ns, {
                c.LABEL_PART_OF: name, c.LABEL_PCS_REPLICA_INDEX: str(ridx)})
            self.store.record_event(pcs, "Warning", "GangTerminated",
                                    f"gang-terminated replica {ridx}")
        return next_wait

    # ------------------------------------------------------------------ rolling update
    def _orchestrate_rolling_update(self, pcs: Obj) -> None:
        """rollingupdate.go:37-296: one replica at a time, ordered no-scheduled-pods →
        breached → ordinal."""
        ns = pcs["metadata"].get("namespace", "default")
        name = pcs["metadata"]["name"]
        st = pcs.get("status") or {}
        prog = st.get("updateProgress")
        if not prog or prog.get("updateEndedAt"):
            return
        if (pcs["spec"].get("updateStrategy") or {}).get("type") == c.UPDATE_ON_DELETE:
            # OnDelete only records progress; spec propagation happens lazily for all
            # replicas and pods are replaced by the user.
            return
        new_hash = st.get("currentGenerationHash", "")
        replicas = int(pcs["spec"].get("replicas", 0))
        tmpl = pcs["spec"]["template"]

        def replica_pclqs(r: int) -> List[Obj]:
            return self.store.list(c.KIND_PCLQ, ns, {
                c.LABEL_PART_OF: name, c.LABEL_PCS_REPLICA_INDEX: str(r)},
                copy_objects=False)

        def pclq_expected_hash(q: Obj) -> str:
            # longest-suffix match: clique names may contain dashes (DNS-1123)
            cl = builders.match_by_fqn_suffix(q["metadata"]["name"],
                                              tmpl.get("cliques") or [])
            if cl is not None:
                return pod_template_hash(cl["name"], cl["spec"].get("podSpec", {}),
                                         tmpl.get("priorityClassName", ""),
                                         cl.get("labels"), cl.get("annotations"))
            return ""

        def replica_updated(r: int) -> bool:
            # Judged from the pods themselves, not PCLQ status: a freshly patched PCLQ
            # still carries pre-patch status counts (informer-staleness class the
            # reference handles with its expectations store, expect/expectations.go).
            qs = replica_pclqs(r)
            if not qs:
                return False
            from ..utils import conditions as _cond
            for q in qs:
                if q["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH) \
                        != pclq_expected_hash(q):
                    return False
                want = int(q["spec"].get("replicas", 1))
                min_avail = int(q["spec"].get("minAvailable", 1))
                pods = self.store.list("Pod", ns, {
                    c.LABEL_PODCLIQUE: q["metadata"]["name"]}, copy_objects=False)
                expected_hash = q["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH)
                if len(pods) != want:
                    return False
                n_ready = 0
                for p in pods:
                    if p["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH) \
                            != expected_hash:
                        return False  # outdated pod still draining
                    if _cond.pod_is_ready(p):
                        n_ready += 1
                if n_ready < min_avail:
                    return False
            return True

        cu = prog.get("currentlyUpdating")
        if cu is not None:
            r = int(cu.get("replicaIndex", -1))
            if not replica_updated(r):
                return  # still updating this replica
            def done(o: Obj) -> None:
                p = o["status"].get("updateProgress") or {}
                ur = p.setdefault("updatedReplicas", [])
                if r not in ur:
                    ur.append(r)
                p["currentlyUpdating"] = None
                o["status"]["updateProgress"] = p
            try:
                self.store.patch(c.KIND_PCS, ns, name, done, status=True)
            except ApiError:
                return
            prog = dict(prog, currentlyUpdating=None,
                        updatedReplicas=list(prog.get("updatedReplicas", [])) + [r])

        done_set = set(prog.get("updatedReplicas") or [])
        pending = [r for r in range(replicas) if r not in done_set
                   and not replica_updated(r)]
        if not pending:
            def finish(o: Obj) -> None:
                p = o["status"].get("updateProgress") or {}
                p["updateEndedAt"] = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
                p["currentlyUpdating"] = None
                o["status"]["updateProgress"] = p
            try:
                self.store.patch(c.KIND_PCS, ns, name, finish, status=True)
            except ApiError:
                pass
            return

        # ordering: no-scheduled-pods first, then breached, then ordinal (:182-209)
        def order(r: int):
            qs = replica_pclqs(r)
            sched = sum(int((q.get("status") or {}).get("scheduledReplicas", 0)) for q in qs)
            breached = any(cond.condition_true(q, c.COND_MIN_AVAILABLE_BREACHED) for q in qs)
            return (0 if sched == 0 else 1, 0 if breached else 1, r)
        target = sorted(pending, key=order)[0]

        def select(o: Obj) -> None:
            p = o["status"].get("updateProgress") or {}
            p["currentlyUpdating"] = {
                "replicaIndex": target,
                "startedAt": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            }
            o["status"]["updateProgress"] = p
        try:
            self.store.patch(c.KIND_PCS, ns, name, select, status=True)
        except ApiError:
            pass

    # ------------------------------------------------------------------ status
    def _reconcile_status(self, namespace: str, name: str) -> None:
        pcs = self.store.try_get(c.KIND_PCS, namespace, name)
        if pcs is None or pcs["metadata"].get("deletionTimestamp"):
            return
        ns = pcs["metadata"].get("namespace", "default")
        replicas = int(pcs["spec"].get("replicas", 0))
        tmpl = pcs["spec"]["template"]
        sg_members = {m for sg in tmpl.get("podCliqueScalingGroups") or []
                      for m in (sg.get("cliqueNames") or [])}

        pclqs = self.store.list(c.KIND_PCLQ, ns, {c.LABEL_PART_OF: name}, copy_objects=False)
        pcsgs = self.store.list(c.KIND_PCSG, ns, {c.LABEL_PART_OF: name}, copy_objects=False)
        pclq_by_replica: Dict[int, List[Obj]] = {}
        for q in pclqs:
            if q["metadata"]["labels"].get(c.LABEL_COMPONENT) != c.COMPONENT_PCS_PODCLIQUE:
                continue
            ridx = q["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX, "")
            if ridx.isdigit():
                pclq_by_replica.setdefault(int(ridx), []).append(q)
        pcsg_by_replica: Dict[int, List[Obj]] = {}
        for g in pcsgs:
            ridx = g["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX, "")
            if ridx.isdigit():
                pcsg_by_replica.setdefault(int(ridx), []).append(g)

        available = 0
        for r in range(replicas):
            qs = pclq_by_replica.get(r, [])
            gs = pcsg_by_replica.get(r, [])
            expected_standalone = sum(1 for cl in tmpl.get("cliques") or []
                                      if cl["name"] not in sg_members)
            expected_sgs = len(tmpl.get("podCliqueScalingGroups") or [])
            if len(qs) < expected_standalone or len(gs) < expected_sgs:
                continue
            ok = all(int((q.get("status") or {}).get("readyReplicas", 0))
                     >= int(q["spec"].get("minAvailable", 1)) for q in qs)
            ok = ok and all(int((g.get("status") or {}).get("availableReplicas", 0))
                            >= int(g["spec"].get("minAvailable", 1)) for g in gs)
            if ok:
                available += 1

        # rolling-update bookkeeping for updatedReplicas count
        st = pcs.get("status") or {}
        prog = st.get("updateProgress")
        updated = len((prog or {}).get("updatedReplicas") or []) if prog else replicas

        # per-gang phase rollup (podcliqueset.go PodGangStatus)
        gang_statuses: List[Dict[str, Any]] = []
        pclq_of = {q["metadata"]["name"]: q for q in pclqs}
        for pg in self.store.list(c.KIND_PODGANG, ns, {c.LABEL_PART_OF: name},
                                  copy_objects=False):
            phase = "Pending"
            groups = (pg.get("spec") or {}).get("podgroups") or []
            if groups:
                sched = all(
                    int((pclq_of.get(g["name"], {}).get("status") or {})
                        .get("scheduledReplicas", 0)) >= int(g.get("minReplicas", 0))
                    for g in groups)
                ready = all(
                    int((pclq_of.get(g["name"], {}).get("status") or {})
                        .get("readyReplicas", 0)) >= int(g.get("minReplicas", 0))
                    for g in groups)
                if ready:
                    phase = "Running"
                elif sched:
                    phase = "Starting"
            gang_statuses.append({"name": pg["metadata"]["name"], "phase": phase})
            # Unhealthy condition (scheduler/api podgang.go:152-171): a scheduled gang
            # whose member clique breached MinAvailable; cleared on recovery
            unhealthy = any(
                cond.condition_true(pclq_of.get(g["name"], {}),
                                    c.COND_MIN_AVAILABLE_BREACHED)
                for g in groups)
            was = cond.condition_true(pg, c.PODGANG_COND_UNHEALTHY)
            if unhealthy != was and cond.condition_true(pg, c.PODGANG_COND_SCHEDULED):
                def flip_unhealthy(o: Obj, v=unhealthy) -> None:
                    cond.set_condition(o, c.PODGANG_COND_UNHEALTHY, v,
                                       "MinAvailableBreached" if v else "Recovered")
                try:
                    self.store.patch(c.KIND_PODGANG, ns, pg["metadata"]["name"],
                                     flip_unhealthy, status=True)
                except ApiError:
                    pass

        def upd(o: Obj) -> None:
            s = o.setdefault("status", {})
            s["replicas"] = replicas
            s["availableReplicas"] = available
            s["updatedReplicas"] = updated
            s["observedGeneration"] = o["metadata"].get("generation")
            s["podGangStatuses"] = sorted(gang_statuses, key=lambda x: x["name"])
            s["hpaPodSelector"] = f"{c.LABEL_PART_OF}={name}"
        try:
            self.store.patch(c.KIND_PCS, namespace, name, upd, status=True)
        except ApiError:
            pass
