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


def _currently_updating_indices(prog: Obj) -> Set[int]:
    out: Set[int] = set()
    for e in prog.get("currentlyUpdating") or []:
        try:
            out.add(int(e.get("replicaIndex", -1)))
        except (TypeError, ValueError):
            pass
    return out


class PodCliqueSetReconciler:
    def __init__(self, store: Store, scheduler_name: str = c.SCHEDULER_AMD_GANG,
                 auto_xgmi_domain: bool = False):
        self.store = store
        self.scheduler_name = scheduler_name
        self.auto_xgmi_domain = auto_xgmi_domain
        # structural-sync fingerprint per PCS: spec generation + child specs
        # (generations bump only on SPEC changes) + pod add/delete epoch (bumped
        # by the cluster watch). While only child STATUSES churn — the dominant
        # regime at 10k-pod scale, where a full G1-G3 resync costs ~0.5 s per
        # pass — the spec-sync phase is skipped entirely; status rollup, gang
        # termination and rolling update always run.
        self._sync_fp: Dict[str, tuple] = {}
        self.pod_epoch: Dict[str, int] = {}

    def bump_pod_epoch(self, namespace: str, pcs_name: str) -> None:
        key = f"{namespace}/{pcs_name}"
        self.pod_epoch[key] = self.pod_epoch.get(key, 0) + 1

    def invalidate_sync_fingerprints(self) -> None:
        """Force full structural resyncs (topology/CTB inputs changed — they feed
        PodGang constraint translation but are outside the fingerprint)."""
        self._sync_fp.clear()

    # ------------------------------------------------------------------ entry
    def reconcile(self, namespace: str, name: str) -> Result:
        """Top-level reconcile. Every mutation step runs under the StepRecorder
        (reconcileerrorrecorder.go parity): per-step ApiErrors that are not benign
        races are batched into status.lastErrors + Warning Events at flush, and
        escaping exceptions are recorded with their ERR_* code before re-raise."""
        rec = groveerr.StepRecorder(self.store, c.KIND_PCS, namespace, name)
        try:
            res = self._reconcile(namespace, name, rec)
            rec.flush()
            return res
        except groveerr.GroveError as e:
            rec.record(e.code, e.message)
            rec.flush()
            raise
        except Exception as e:
            rec.record(groveerr.ERR_RECONCILE, str(e))
            rec.flush()
            raise

    def _reconcile(self, namespace: str, name: str,
                   rec: groveerr.StepRecorder) -> Result:
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

        self._process_generation_hash(pcs, rec)
        pcs = self.store.get(c.KIND_PCS, namespace, name)
        key = f"{namespace}/{name}"
        child_sig = tuple(sorted(
            (q["metadata"]["name"], q["metadata"].get("generation", 0))
            for kind in (c.KIND_PCLQ, c.KIND_PCSG)
            for q in self.store.list(kind, namespace,
                                     {c.LABEL_PART_OF: name},
                                     copy_objects=False)))
        st = pcs.get("status") or {}
        prog = st.get("updateProgress") or {}
        fp = (pcs["metadata"].get("generation"),
              self.pod_epoch.get(key, 0), hash(child_sig),
              st.get("currentGenerationHash"),
              tuple(sorted(_currently_updating_indices(prog))),
              bool(prog) and not prog.get("updateEndedAt"))
        if self._sync_fp.get(key) != fp:
            res = self._sync_resources(pcs, rec)
            if not rec.errors and not rec.retry_needed:
                self._sync_fp[key] = fp
        else:
            res = Result.DONE
        term = self._gang_termination(pcs, rec)
        self._orchestrate_rolling_update(pcs, rec)
        self._reconcile_status(namespace, name, rec)
        if rec.retry_needed:
            # a step lost an optimistic race and was skipped — nothing else may
            # retrigger this PCS (PodGang events don't map back), so requeue
            return Result(requeue_after=0.05)
        if term is not None:
            return Result(requeue_after=term)
        return res

    # ------------------------------------------------------------------ delete
    def _reconcile_delete(self, pcs: Obj) -> Result:
        ns, name = pcs["metadata"].get("namespace"), pcs["metadata"]["name"]
        self._sync_fp.pop(f"{ns}/{name}", None)
        self.pod_epoch.pop(f"{ns}/{name}", None)
        rec = groveerr.StepRecorder(self.store, c.KIND_PCS, ns, name)
        sel = {c.LABEL_PART_OF: name}
        remaining = 0
        for kind in (c.KIND_PCLQ, c.KIND_PCSG):
            for obj in self.store.list(kind, ns, sel):
                remaining += 1
                with rec.step(groveerr.ERR_DELETE, benign=groveerr.BENIGN_DELETE,
                              detail=f"delete {kind} {obj['metadata']['name']}"):
                    self.store.delete(kind, ns, obj["metadata"]["name"])
        if remaining:
            rec.flush()
            return Result(requeue_after=0.02)
        for pg in self.store.list(c.KIND_PODGANG, ns, sel):
            with rec.step(groveerr.ERR_DELETE, benign=groveerr.BENIGN_DELETE,
                          detail=f"delete PodGang {pg['metadata']['name']}"):
                self.store.delete(c.KIND_PODGANG, ns, pg["metadata"]["name"])

        def rm(o: Obj) -> None:
            o["metadata"]["finalizers"] = [
                f for f in o["metadata"].get("finalizers", []) if f != c.FINALIZER_PCS]
        try:
            self.store.patch(c.KIND_PCS, ns, name, rm)
        except ApiError:
            pass
        return Result.DONE

    # ------------------------------------------------------------------ generation hash
    def _process_generation_hash(self, pcs: Obj,
                                  rec: groveerr.StepRecorder) -> None:
        """reconcilespec.go:72-158: persist template hash; on change, start update.
        updateProgress uses the published status schema (podcliqueset.go/CRD):
        updateStartedAt/updateEndedAt + currentlyUpdating as a LIST of
        {replicaIndex, updateStartedAt} (one at a time => <= 1 entry)."""
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
                    "currentlyUpdating": [],
                }
            s["currentGenerationHash"] = new_hash
        with rec.step(groveerr.ERR_UPDATE_STATUS, benign=groveerr.BENIGN_UPDATE,
                      detail="persist generation hash"):
            self.store.patch(c.KIND_PCS, ns, name, upd, status=True)

    # ------------------------------------------------------------------ spec sync
    def _sync_resources(self, pcs: Obj, rec: groveerr.StepRecorder) -> Result:
        ns = pcs["metadata"].get("namespace", "default")
        name = pcs["metadata"]["name"]
        replicas = int(pcs["spec"].get("replicas", 0))
        tmpl = pcs["spec"]["template"]
        sg_cfgs = tmpl.get("podCliqueScalingGroups") or []
        sg_members = {m for sg in sg_cfgs for m in (sg.get("cliqueNames") or [])}

        # ---- G1: RBAC + token secret + per-replica headless Service + HPAs
        self._ensure(builders.build_service_account(pcs), rec, groveerr.ERR_SYNC_RBAC)
        self._ensure(builders.build_role(pcs), rec, groveerr.ERR_SYNC_RBAC)
        self._ensure(builders.build_role_binding(pcs), rec, groveerr.ERR_SYNC_RBAC)
        self._ensure(builders.build_sa_token_secret(pcs), rec,
                     groveerr.ERR_SYNC_RBAC)
        existing_svcs = {s["metadata"]["name"] for s in self.store.list(
            "Service", ns, {c.LABEL_PART_OF: name,
                            c.LABEL_COMPONENT: c.COMPONENT_HEADLESS_SERVICE})}
        for r in range(replicas):
            svc_name = namegen.headless_service_name(name, r)
            if svc_name not in existing_svcs:
                self._ensure(builders.build_headless_service(pcs, r), rec,
                             groveerr.ERR_SYNC_SERVICE)
        for svc_name in existing_svcs:
            idx = svc_name.rsplit("-", 1)[-1]
            if idx.isdigit() and int(idx) >= replicas:
                with rec.step(groveerr.ERR_SYNC_SERVICE,
                              benign=groveerr.BENIGN_DELETE,
                              detail=f"GC service {svc_name}"):
                    self.store.delete("Service", ns, svc_name)
        self._sync_hpas(pcs, rec)

        # ---- G2: standalone PodCliques per replica (+ shared ResourceClaims)
        # template hashes are per-CLIQUE, identical across replicas — computing
        # them once per pass instead of per replica removes 5000 sha256 calls per
        # reconcile of a 5000-replica PCS
        clique_hash = {cl["name"]: pod_template_hash(
            cl["name"], cl["spec"].get("podSpec", {}),
            tmpl.get("priorityClassName", ""), cl.get("labels"),
            cl.get("annotations")) for cl in tmpl.get("cliques") or []}
        pclq_by_name = {q["metadata"]["name"]: q for q in self.store.list(
            c.KIND_PCLQ, ns, {c.LABEL_PART_OF: name,
                              c.LABEL_COMPONENT: c.COMPONENT_PCS_PODCLIQUE},
            copy_objects=False)}
        expected_pclqs: Set[str] = set()
        for r in range(replicas):
            claims = resourceclaims.pcs_claims_for_replica(
                self.store, pcs, r, self.auto_xgmi_domain)
            with rec.step(groveerr.ERR_SYNC_RESOURCE_CLAIM,
                          detail=f"ensure claims for replica {r}"):
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
                                cur=pclq_by_name.get(fqn), rec=rec,
                                new_hash=clique_hash[cl["name"]])
        # GC excess standalone PCLQs (scale-in / replica removal)
        for pclq in pclq_by_name.values():
            if pclq["metadata"]["name"] not in expected_pclqs:
                with rec.step(groveerr.ERR_SYNC_PODCLIQUE,
                              benign=groveerr.BENIGN_DELETE,
                              detail=f"GC PodClique {pclq['metadata']['name']}"):
                    self.store.delete(c.KIND_PCLQ, ns, pclq["metadata"]["name"])

        # ---- G3a: PCSGs per config per replica
        pcsg_names = {g["metadata"]["name"] for g in self.store.list(
            c.KIND_PCSG, ns, {c.LABEL_PART_OF: name}, copy_objects=False)}
        expected_pcsgs: Set[str] = set()
        for r in range(replicas):
            for sg in sg_cfgs:
                fqn = namegen.pcsg_name(name, r, sg["name"])
                expected_pcsgs.add(fqn)
                if fqn not in pcsg_names:
                    with rec.step(groveerr.ERR_SYNC_PCSG,
                                  benign=groveerr.BENIGN_CREATE,
                                  detail=f"create PCSG {fqn}"):
                        self.store.create(builders.build_pcsg(pcs, r, sg))
        for pcsg in self.store.list(c.KIND_PCSG, ns, {
                c.LABEL_PART_OF: name, c.LABEL_COMPONENT: c.COMPONENT_PCSG}):
            if pcsg["metadata"]["name"] not in expected_pcsgs:
                with rec.step(groveerr.ERR_SYNC_PCSG,
                              benign=groveerr.BENIGN_DELETE,
                              detail=f"GC PCSG {pcsg['metadata']['name']}"):
                    self.store.delete(c.KIND_PCSG, ns, pcsg["metadata"]["name"])

        # ---- G3b: PodGangs
        sync_podgangs(self.store, pcs, self.scheduler_name, rec)
        return Result.DONE

    def _sync_pclq(self, pcs: Obj, r: int, clique_tmpl: Obj, fqn: str, owner: Obj,
                   claim_refs=None, cur="__lookup__",
                   rec: Optional[groveerr.StepRecorder] = None,
                   new_hash: Optional[str] = None) -> None:
        ns = pcs["metadata"].get("namespace", "default")
        rec = rec or groveerr.StepRecorder(self.store, c.KIND_PCS, ns,
                                           pcs["metadata"]["name"])
        if cur == "__lookup__":
            cur = self.store.try_get(c.KIND_PCLQ, ns, fqn)
        if cur is None:
            obj = builders.build_podclique(pcs, r, clique_tmpl, owner)
            obj["spec"]["updateStrategy"] = (pcs["spec"].get("updateStrategy") or {}).get(
                "type", c.UPDATE_ROLLING_RECREATE)
            if claim_refs:
                obj["spec"]["resourceClaims"] = claim_refs
            with rec.step(groveerr.ERR_SYNC_PODCLIQUE,
                          benign=groveerr.BENIGN_CREATE,
                          detail=f"create PodClique {fqn}"):
                self.store.create(obj)
            return
        if cur["metadata"].get("deletionTimestamp"):
            return
        # propagate template changes only to replicas selected for update (GREP-393)
        if new_hash is None:
            new_hash = pod_template_hash(
                clique_tmpl["name"], clique_tmpl["spec"].get("podSpec", {}),
                pcs["spec"]["template"].get("priorityClassName", ""),
                clique_tmpl.get("labels"), clique_tmpl.get("annotations"))
        want_strategy = (pcs["spec"].get("updateStrategy") or {}).get(
            "type", c.UPDATE_ROLLING_RECREATE)
        if cur["spec"].get("updateStrategy") != want_strategy:
            # strategy transitions (OD9: OnDelete -> RollingRecreate) propagate
            # immediately and independently of template changes
            with rec.step(groveerr.ERR_SYNC_PODCLIQUE,
                          benign=groveerr.BENIGN_UPDATE,
                          detail=f"propagate updateStrategy to {fqn}"):
                self.store.patch(c.KIND_PCLQ, ns, fqn, lambda o: o["spec"].update(
                    updateStrategy=want_strategy), return_copy=False)
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
            with rec.step(groveerr.ERR_SYNC_PODCLIQUE,
                          benign=groveerr.BENIGN_UPDATE,
                          detail=f"propagate template to {fqn}"):
                self.store.patch(c.KIND_PCLQ, ns, fqn, upd)
        # HPA-aware replica preservation: never stomp replicas on scaled cliques —
        # only non-scaled fields drift-corrected here.

    def _replica_selected_for_update(self, pcs: Obj, r: int) -> bool:
        if (pcs["spec"].get("updateStrategy") or {}).get("type") == c.UPDATE_ON_DELETE:
            return True  # OnDelete: spec propagates immediately, pods wait for the user
        prog = (pcs.get("status") or {}).get("updateProgress")
        if prog is None:
            return True  # no update in flight → initial create path
        return r in _currently_updating_indices(prog)

    def _sync_hpas(self, pcs: Obj, rec: groveerr.StepRecorder) -> None:
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
                self._ensure(builders.build_hpa(pcs, kind, target, cfg), rec,
                             groveerr.ERR_SYNC_HPA)
        for hname in existing:
            if hname not in expected:
                with rec.step(groveerr.ERR_SYNC_HPA,
                              benign=groveerr.BENIGN_DELETE,
                              detail=f"GC HPA {hname}"):
                    self.store.delete("HorizontalPodAutoscaler", ns, hname)

    def _ensure(self, obj: Obj, rec: groveerr.StepRecorder, code: str) -> None:
        with rec.step(code, benign=groveerr.BENIGN_CREATE,
                      detail=f"create {obj.get('kind')} "
                             f"{obj['metadata'].get('name', '')}"):
            self.store.create(obj)

    # ------------------------------------------------------------------ gang termination
    def _gang_termination(self, pcs: Obj,
                          rec: groveerr.StepRecorder) -> Optional[float]:
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

        updating = _currently_updating_indices(
            (pcs.get("status") or {}).get("updateProgress") or {})
        for ridx, constituents in by_replica.items():
            # suppression: a replica mid-rolling-update dips below MinAvailable by
            # design; never gang-terminate it while it is the one being updated
            if ridx in updating:
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
                with rec.step(groveerr.ERR_GANG_TERMINATION,
                              benign=groveerr.BENIGN_UPDATE,
                              detail=f"mark PCSG {g['metadata']['name']}"):
                    self.store.patch(c.KIND_PCSG, ns, g["metadata"]["name"], mark,
                                     status=True)
            # DisruptionTarget on every PodGang of the doomed replica
            # (scheduler/api/core/v1alpha1/podgang.go:152-171 contract): schedulers
            # and drain tooling see the gang is being terminated by the operator.
            base_pg = namegen.base_podgang_name(name, ridx)
            for pg in self.store.list(c.KIND_PODGANG, ns, {c.LABEL_PART_OF: name},
                                      copy_objects=False):
                pg_name = pg["metadata"]["name"]
                if pg_name != base_pg and \
                        pg["metadata"].get("labels", {}).get(
                            c.LABEL_BASE_PODGANG) != base_pg:
                    continue

                def mark_disrupted(o: Obj) -> None:
                    cond.set_condition(o, c.PODGANG_COND_DISRUPTION_TARGET, True,
                                       "GangTerminated")
                    # reset the gang's lifecycle for the recycle: a recreated
                    # replica must re-run init → permit → ready from scratch. A
                    # stale Initialized=True would short-circuit the inline
                    # podReferences refill (try_complete_podgang) and leave the
                    # recreated pods schedule-gated forever (observed wedge).
                    cond.set_condition(o, c.PODGANG_COND_INITIALIZED, False,
                                       "GangTerminated")
                    cond.set_condition(o, c.PODGANG_COND_SCHEDULED, False,
                                       "GangTerminated")
                    cond.set_condition(o, c.PODGANG_COND_READY, False,
                                       "GangTerminated")
                    o.setdefault("status", {})["phase"] = "Pending"
                with rec.step(groveerr.ERR_GANG_TERMINATION,
                              benign=groveerr.BENIGN_UPDATE,
                              detail=f"mark DisruptionTarget on {pg_name}"):
                    self.store.patch(c.KIND_PODGANG, ns, pg_name,
                                     mark_disrupted, status=True)
            self.store.delete_collection(c.KIND_PCLQ, ns, {
                c.LABEL_PART_OF: name, c.LABEL_PCS_REPLICA_INDEX: str(ridx)})
            self.store.record_event(pcs, "Warning", "GangTerminated",
                                    f"gang-terminated replica {ridx}")
        return next_wait

    # ------------------------------------------------------------------ rolling update
    def _orchestrate_rolling_update(self, pcs: Obj,
                                    rec: groveerr.StepRecorder) -> None:
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

        updating = sorted(_currently_updating_indices(prog))
        if updating:
            r = updating[0]
            if not replica_updated(r):
                return  # still updating this replica (one at a time)

            def done(o: Obj) -> None:
                p = o["status"].get("updateProgress") or {}
                p["currentlyUpdating"] = [
                    e for e in p.get("currentlyUpdating") or []
                    if int(e.get("replicaIndex", -1)) != r]
                o["status"]["updateProgress"] = p
            with rec.step(groveerr.ERR_ROLLING_UPDATE,
                          benign=groveerr.BENIGN_UPDATE,
                          detail=f"complete update of replica {r}"):
                self.store.patch(c.KIND_PCS, ns, name, done, status=True)

        # completion is derived from ground truth (template hashes + pods), not a
        # bookkeeping list — survives status pruning on a real apiserver
        pending = [r for r in range(replicas) if not replica_updated(r)]
        if not pending:
            def finish(o: Obj) -> None:
                p = o["status"].get("updateProgress") or {}
                p["updateEndedAt"] = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
                p["currentlyUpdating"] = []
                o["status"]["updateProgress"] = p
            with rec.step(groveerr.ERR_ROLLING_UPDATE,
                          benign=groveerr.BENIGN_UPDATE,
                          detail="finish rolling update"):
                self.store.patch(c.KIND_PCS, ns, name, finish, status=True)
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
            cur = p.get("currentlyUpdating") or []
            if not any(int(e.get("replicaIndex", -1)) == target for e in cur):
                cur = cur + [{
                    "replicaIndex": target,
                    "updateStartedAt": time.strftime("%Y-%m-%dT%H:%M:%SZ",
                                                     time.gmtime()),
                }]
            p["currentlyUpdating"] = cur
            o["status"]["updateProgress"] = p
        with rec.step(groveerr.ERR_ROLLING_UPDATE, benign=groveerr.BENIGN_UPDATE,
                      detail=f"select replica {target} for update"):
            self.store.patch(c.KIND_PCS, ns, name, select, status=True)

    # ------------------------------------------------------------------ status
    def _reconcile_status(self, namespace: str, name: str,
                          rec: Optional[groveerr.StepRecorder] = None) -> None:
        rec = rec or groveerr.StepRecorder(self.store, c.KIND_PCS, namespace, name)
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

        # updated-replica + per-PCLQ/PCSG update counters, derived from template
        # hashes (printer-column contract: PCLQs-Updated/PCLQs-Total etc.)
        st = pcs.get("status") or {}
        prog = st.get("updateProgress")

        _hash_by_clique = {cl["name"]: pod_template_hash(
            cl["name"], cl["spec"].get("podSpec", {}),
            tmpl.get("priorityClassName", ""), cl.get("labels"),
            cl.get("annotations")) for cl in tmpl.get("cliques") or []}
        _hash_by_fqn: Dict[str, str] = {}

        def expected_hash_of(q: Obj) -> str:
            fqn = q["metadata"]["name"]
            h = _hash_by_fqn.get(fqn)
            if h is None:
                cl = builders.match_by_fqn_suffix(fqn, tmpl.get("cliques") or [])
                h = _hash_by_clique.get(cl["name"], "") if cl is not None else ""
                _hash_by_fqn[fqn] = h
            return h

        def pclq_hash_current(q: Obj) -> bool:
            return q["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH) \
                == expected_hash_of(q)

        all_pclq_by_replica: Dict[int, List[Obj]] = {}
        for q in pclqs:
            ridx = q["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX, "")
            if ridx.isdigit():
                all_pclq_by_replica.setdefault(int(ridx), []).append(q)
        updated = 0
        for r in range(replicas):
            qs = all_pclq_by_replica.get(r, [])
            if qs and all(pclq_hash_current(q) for q in qs):
                updated += 1
        total_pclqs = len(pclqs)
        updated_pclqs = sum(1 for q in pclqs if pclq_hash_current(q))
        pclqs_of_pcsg: Dict[str, List[Obj]] = {}
        for q in pclqs:
            sg = q["metadata"]["labels"].get(c.LABEL_PCSG)
            if sg:
                pclqs_of_pcsg.setdefault(sg, []).append(q)
        total_pcsgs = len(pcsgs)
        updated_pcsgs = sum(
            1 for g in pcsgs
            if all(pclq_hash_current(q)
                   for q in pclqs_of_pcsg.get(g["metadata"]["name"], [])))

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
            # a recycled gang that is Running again is no longer a disruption target
            if phase == "Running" and cond.condition_true(
                    pg, c.PODGANG_COND_DISRUPTION_TARGET):
                def clear_dt(o: Obj) -> None:
                    cond.set_condition(o, c.PODGANG_COND_DISRUPTION_TARGET, False,
                                       "Recovered")
                with rec.step(groveerr.ERR_SYNC_PODGANG,
                              benign=groveerr.BENIGN_UPDATE,
                              detail=f"clear DisruptionTarget on "
                                     f"{pg['metadata']['name']}"):
                    self.store.patch(c.KIND_PODGANG, ns, pg["metadata"]["name"],
                                     clear_dt, status=True)
            was = cond.condition_true(pg, c.PODGANG_COND_UNHEALTHY)
            if unhealthy != was and cond.condition_true(pg, c.PODGANG_COND_SCHEDULED):
                def flip_unhealthy(o: Obj, v=unhealthy) -> None:
                    cond.set_condition(o, c.PODGANG_COND_UNHEALTHY, v,
                                       "MinAvailableBreached" if v else "Recovered")
                with rec.step(groveerr.ERR_SYNC_PODGANG,
                              benign=groveerr.BENIGN_UPDATE,
                              detail=f"flip Unhealthy on {pg['metadata']['name']}"):
                    self.store.patch(c.KIND_PODGANG, ns, pg["metadata"]["name"],
                                     flip_unhealthy, status=True)

        def upd(o: Obj) -> None:
            s = o.setdefault("status", {})
            s["replicas"] = replicas
            s["availableReplicas"] = available
            s["updatedReplicas"] = updated
            s["observedGeneration"] = o["metadata"].get("generation")
            s["podGangStatuses"] = sorted(gang_statuses, key=lambda x: x["name"])
            s["hpaPodSelector"] = f"{c.LABEL_PART_OF}={name}"
            if s.get("updateProgress"):
                p = s["updateProgress"]
                p["totalPodCliquesCount"] = total_pclqs
                p["updatedPodCliquesCount"] = updated_pclqs
                p["totalPodCliqueScalingGroupsCount"] = total_pcsgs
                p["updatedPodCliqueScalingGroupsCount"] = updated_pcsgs
        with rec.step(groveerr.ERR_UPDATE_STATUS, benign=groveerr.BENIGN_UPDATE,
                      detail="write PCS status"):
            self.store.patch(c.KIND_PCS, namespace, name, upd, status=True)
