"""PodCliqueScalingGroup controller.

Behavior parity: operator/internal/controller/podcliquescalinggroup/ — member PodClique
creation per replica index (components/podclique/sync.go:55-199), scale-in deletion,
PCSG-replica-scoped gang recycle (sync.go:127), availability status + MinAvailableBreached /
GangTerminationInProgress conditions (reconcilestatus.go), PCSG-scoped rolling update
(components/rollingupdate.go). Fresh implementation.
"""
from __future__ import annotations

import logging
import time
from typing import Dict, List, Optional

from ..api import constants as c
from ..api import namegen
from ..api.defaulting import parse_duration_seconds
from ..kubecore.store import Store, Obj, ApiError
from ..utils import conditions as cond
from ..utils import errors as groveerr
from ..utils.hashing import pod_template_hash
from . import builders
from .manager import Result
from .podcliqueset import _iso_to_epoch, _currently_updating_indices
from . import resourceclaims

log = logging.getLogger("grove.pcsg")


class PCSGReconciler:
    def __init__(self, store: Store, scheduler_name: str = c.SCHEDULER_AMD_GANG,
                 auto_xgmi_domain: bool = False):
        self.store = store
        self.scheduler_name = scheduler_name
        self.auto_xgmi_domain = auto_xgmi_domain
        # structural-sync fingerprint (PCS reconciler pattern): member-PCLQ sync
        # is skipped while only statuses churn — PCSG/PCS spec generations, the
        # rolling-update state and member generations are the structural inputs
        self._sync_fp: Dict[str, tuple] = {}

    def reconcile(self, namespace: str, name: str) -> Result:
        pcsg = self.store.try_get(c.KIND_PCSG, namespace, name)
        if pcsg is None:
            return Result.DONE
        if pcsg["metadata"].get("deletionTimestamp"):
            return self._reconcile_delete(pcsg)
        pcs = self._find_pcs(pcsg)
        if pcs is None:
            return Result(requeue_after=0.1)
        rec = groveerr.StepRecorder(self.store, c.KIND_PCSG, namespace, name)
        key = f"{namespace}/{name}"
        members_sig = tuple(sorted(
            (q["metadata"]["name"], q["metadata"].get("generation", 0))
            for q in self._member_pclqs(pcsg)))
        st = pcs.get("status") or {}
        prog = st.get("updateProgress") or {}
        fp = (pcsg["metadata"].get("generation"),
              pcs["metadata"].get("generation"),
              st.get("currentGenerationHash"),
              tuple(sorted(_currently_updating_indices(prog))),
              hash(members_sig))
        if self._sync_fp.get(key) != fp:
            res = self._sync_member_pclqs(pcs, pcsg, rec)
            if not rec.errors and not rec.retry_needed:
                self._sync_fp[key] = fp
        else:
            res = Result.DONE
        recycle_wait = self._replica_recycle(pcs, pcsg, rec)
        self._reconcile_status(namespace, name)
        rec.flush()
        if rec.retry_needed:
            return Result(requeue_after=0.05)
        if recycle_wait is not None:
            return Result(requeue_after=recycle_wait)
        return res

    # ------------------------------------------------------------------ helpers
    def _find_pcs(self, pcsg: Obj) -> Optional[Obj]:
        pcs_name = pcsg["metadata"]["labels"].get(c.LABEL_PART_OF)
        if not pcs_name:
            return None
        return self.store.try_get(c.KIND_PCS, pcsg["metadata"].get("namespace"), pcs_name)

    def _member_pclqs(self, pcsg: Obj) -> List[Obj]:
        return self.store.list(c.KIND_PCLQ, pcsg["metadata"].get("namespace"),
                               {c.LABEL_PCSG: pcsg["metadata"]["name"]},
                               copy_objects=False)

    @staticmethod
    def _sg_config(pcs: Obj, pcsg: Obj) -> Optional[Obj]:
        pcs_replica = int(pcsg["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX, 0))
        sg_name = namegen.extract_scaling_group_name(
            pcsg["metadata"]["name"], pcs["metadata"]["name"], pcs_replica)
        for sg in pcs["spec"]["template"].get("podCliqueScalingGroups") or []:
            if sg["name"] == sg_name:
                return sg
        return None

    # ------------------------------------------------------------------ delete
    def _reconcile_delete(self, pcsg: Obj) -> Result:
        ns, name = pcsg["metadata"].get("namespace"), pcsg["metadata"]["name"]
        self._sync_fp.pop(f"{ns}/{name}", None)
        remaining = 0
        for q in self._member_pclqs(pcsg):
            remaining += 1
            try:
                self.store.delete(c.KIND_PCLQ, ns, q["metadata"]["name"])
            except ApiError:
                pass
        if remaining:
            return Result(requeue_after=0.02)

        def rm(o: Obj) -> None:
            o["metadata"]["finalizers"] = [
                f for f in o["metadata"].get("finalizers", []) if f != c.FINALIZER_PCSG]
        try:
            self.store.patch(c.KIND_PCSG, ns, name, rm)
        except ApiError:
            pass
        return Result.DONE

    # ------------------------------------------------------------------ spec
    def _sync_member_pclqs(self, pcs: Obj, pcsg: Obj,
                           rec: groveerr.StepRecorder) -> Result:
        ns = pcsg["metadata"].get("namespace", "default")
        sg_fqn = pcsg["metadata"]["name"]
        pcs_replica = int(pcsg["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX, 0))
        replicas = int(pcsg["spec"].get("replicas", 1))
        min_avail = int(pcsg["spec"].get("minAvailable", 1))
        member_names = pcsg["spec"].get("cliqueNames") or []
        cliques = {cl["name"]: cl for cl in pcs["spec"]["template"].get("cliques") or []}
        base_pg = namegen.base_podgang_name(pcs["metadata"]["name"], pcs_replica)

        sg_cfg = self._sg_config(pcs, pcsg) or {}
        sg_claims = resourceclaims.pcsg_claims(pcs, sg_cfg, sg_fqn, replicas)
        resourceclaims.ensure_claims(
            self.store, [(cl0, e) for (cl0, e, _j) in sg_claims])
        # xGMI-group claims are PCS-replica-scoped (created by the PCS reconciler);
        # member cliques in a group reference them too
        xgmi_groups = resourceclaims.effective_xgmi_groups(
            pcs, self.auto_xgmi_domain)
        existing = {q["metadata"]["name"]: q for q in self._member_pclqs(pcsg)}
        member_hash = {mn: pod_template_hash(
            mn, (cliques.get(mn) or {}).get("spec", {}).get("podSpec", {}),
            pcs["spec"]["template"].get("priorityClassName", ""),
            (cliques.get(mn) or {}).get("labels"),
            (cliques.get(mn) or {}).get("annotations"))
            for mn in member_names if mn in cliques}
        expected: set = set()
        for j in range(replicas):
            pg_name = namegen.podgang_name_for_pclq_in_pcsg(
                pcs["metadata"]["name"], pcs_replica, sg_fqn, min_avail, j)
            for mn in member_names:
                cl = cliques.get(mn)
                if cl is None:
                    continue
                fqn = namegen.podclique_name(sg_fqn, j, mn)
                expected.add(fqn)
                cur = existing.get(fqn)
                if cur is None:
                    obj = builders.build_podclique(
                        pcs, pcs_replica, cl, owner=pcsg,
                        pcsg_name=sg_fqn, pcsg_replica=j,
                        podgang_name=pg_name,
                        base_podgang_name=base_pg if j >= min_avail else None)
                    refs = resourceclaims.claim_refs_for_clique(
                        [(cl0, e) for (cl0, e, jj) in sg_claims
                         if jj is None or jj == j], mn)
                    g = xgmi_groups.get(mn)
                    if g is not None:
                        refs = refs + [{
                            "name": f"{resourceclaims.XGMI_TEMPLATE_NAME}-{g}",
                            "resourceClaimName":
                                f"{pcs['metadata']['name']}-{pcs_replica}-xgmi-{g}"}]
                    if refs:
                        obj["spec"]["resourceClaims"] = refs
                    obj["spec"]["updateStrategy"] = (
                        pcs["spec"].get("updateStrategy") or {}).get(
                        "type", c.UPDATE_ROLLING_RECREATE)
                    with rec.step(groveerr.ERR_SYNC_PODCLIQUE,
                                  benign=groveerr.BENIGN_CREATE,
                                  detail=f"create member PodClique {fqn}"):
                        self.store.create(obj)
                    continue
                if cur["metadata"].get("deletionTimestamp"):
                    continue
                want_strategy = (pcs["spec"].get("updateStrategy") or {}).get(
                    "type", c.UPDATE_ROLLING_RECREATE)
                if cur["spec"].get("updateStrategy") != want_strategy:
                    with rec.step(groveerr.ERR_SYNC_PODCLIQUE,
                                  benign=groveerr.BENIGN_UPDATE,
                                  detail=f"propagate updateStrategy to {fqn}"):
                        self.store.patch(
                            c.KIND_PCLQ, ns, fqn,
                            lambda o: o["spec"].update(
                                updateStrategy=want_strategy),
                            return_copy=False)
                new_hash = member_hash[mn]
                if cur["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH) != new_hash \
                        and self._replica_selected_for_update(pcs, pcs_replica):
                    def upd(o: Obj) -> None:
                        o["metadata"]["labels"][c.LABEL_POD_TEMPLATE_HASH] = new_hash
                        o["spec"]["podSpec"] = cl["spec"].get("podSpec", {})
                    with rec.step(groveerr.ERR_SYNC_PODCLIQUE,
                                  benign=groveerr.BENIGN_UPDATE,
                                  detail=f"propagate template to {fqn}"):
                        self.store.patch(c.KIND_PCLQ, ns, fqn, upd)
        # scale-in: delete member PCLQs beyond current replicas
        for fqn, q in existing.items():
            if fqn not in expected:
                with rec.step(groveerr.ERR_SYNC_PODCLIQUE,
                              benign=groveerr.BENIGN_DELETE,
                              detail=f"scale-in PodClique {fqn}"):
                    self.store.delete(c.KIND_PCLQ, ns, fqn)
        return Result.DONE

    @staticmethod
    def _replica_selected_for_update(pcs: Obj, r: int) -> bool:
        if (pcs["spec"].get("updateStrategy") or {}).get("type") == c.UPDATE_ON_DELETE:
            return True  # OnDelete: spec propagates immediately, pods wait for the user
        prog = (pcs.get("status") or {}).get("updateProgress")
        if prog is None:
            return True
        return r in _currently_updating_indices(prog)

    # ------------------------------------------------------------------ replica recycle
    def _replica_recycle(self, pcs: Obj, pcsg: Obj,
                         rec: groveerr.StepRecorder) -> Optional[float]:
        """PCSG-replica-scoped gang recycle (sync.go:127): a scaled replica whose member
        cliques breached MinAvailable past terminationDelay is deleted and recreated.
        Fresh cliques have everScheduled=False so the loop cannot re-fire (WasPCLQ-
        EverScheduled parity)."""
        ns = pcsg["metadata"].get("namespace", "default")
        delay = parse_duration_seconds(
            pcs["spec"]["template"].get("terminationDelay", "4h"))
        now = time.time()
        next_wait: Optional[float] = None
        by_replica: Dict[int, List[Obj]] = {}
        for q in self._member_pclqs(pcsg):
            js = q["metadata"]["labels"].get(c.LABEL_PCSG_REPLICA_INDEX, "")
            if js.isdigit():
                by_replica.setdefault(int(js), []).append(q)
        for j, qs in by_replica.items():
            breach_since: Optional[float] = None
            for q in qs:
                if not (q.get("status") or {}).get("everScheduled"):
                    breach_since = None
                    break
                bc = cond.get_condition(q, c.COND_MIN_AVAILABLE_BREACHED)
                if bc and bc.get("status") == "True":
                    ts = _iso_to_epoch(bc.get("lastTransitionTime", ""))
                    breach_since = ts if breach_since is None else min(breach_since, ts)
            if breach_since is None:
                continue
            remaining = delay - (now - breach_since)
            if remaining > 0:
                next_wait = remaining if next_wait is None else min(next_wait, remaining)
                continue
            log.info("PCSG %s/%s recycling replica %d", ns, pcsg["metadata"]["name"], j)
            self._reset_gang_for_recycle(pcs, pcsg, j, rec)
            for q in qs:
                with rec.step(groveerr.ERR_GANG_TERMINATION,
                              benign=groveerr.BENIGN_DELETE,
                              detail=f"recycle PodClique {q['metadata']['name']}"):
                    self.store.delete(c.KIND_PCLQ, ns, q["metadata"]["name"])
        return next_wait

    def _reset_gang_for_recycle(self, pcs: Obj, pcsg: Obj, j: int,
                                rec: groveerr.StepRecorder) -> None:
        """Reset the lifecycle conditions of the gang covering PCSG replica j before
        its cliques are deleted: a recycled replica re-runs init → permit → ready, and
        a stale Initialized=True would short-circuit the inline podReferences refill
        and leave the recreated pods schedule-gated (same wedge as PCS-scope gang
        termination)."""
        ns = pcsg["metadata"].get("namespace", "default")
        min_avail = int(pcsg["spec"].get("minAvailable", 1))
        if j >= min_avail:
            gang = namegen.scaled_podgang_name(pcsg["metadata"]["name"], j - min_avail)
        else:
            pcs_replica = int(
                pcsg["metadata"]["labels"].get(c.LABEL_PCS_REPLICA_INDEX, 0))
            gang = namegen.base_podgang_name(pcs["metadata"]["name"], pcs_replica)

        def reset(o: Obj) -> None:
            cond.set_condition(o, c.PODGANG_COND_INITIALIZED, False, "GangTerminated")
            cond.set_condition(o, c.PODGANG_COND_SCHEDULED, False, "GangTerminated")
            cond.set_condition(o, c.PODGANG_COND_READY, False, "GangTerminated")
            o.setdefault("status", {})["phase"] = "Pending"
        with rec.step(groveerr.ERR_GANG_TERMINATION,
                      benign=groveerr.BENIGN_UPDATE,
                      detail=f"reset gang {gang} for recycle"):
            self.store.patch(c.KIND_PODGANG, ns, gang, reset, status=True)

    # ------------------------------------------------------------------ status
    def _reconcile_status(self, namespace: str, name: str) -> None:
        pcsg = self.store.try_get(c.KIND_PCSG, namespace, name)
        if pcsg is None or pcsg["metadata"].get("deletionTimestamp"):
            return
        replicas = int(pcsg["spec"].get("replicas", 1))
        min_avail = int(pcsg["spec"].get("minAvailable", 1))
        members = pcsg["spec"].get("cliqueNames") or []
        by_replica: Dict[int, List[Obj]] = {}
        for q in self._member_pclqs(pcsg):
            js = q["metadata"]["labels"].get(c.LABEL_PCSG_REPLICA_INDEX, "")
            if js.isdigit():
                by_replica.setdefault(int(js), []).append(q)

        sched = avail = 0
        for j in range(replicas):
            qs = by_replica.get(j, [])
            if len(qs) < len(members):
                continue
            if all(int((q.get("status") or {}).get("scheduledReplicas", 0))
                   >= int(q["spec"].get("minAvailable", 1)) for q in qs):
                sched += 1
            if all(int((q.get("status") or {}).get("readyReplicas", 0))
                   >= int(q["spec"].get("minAvailable", 1)) for q in qs):
                avail += 1

        def upd(o: Obj) -> None:
            st = o.setdefault("status", {})
            st["replicas"] = replicas
            st["scheduledReplicas"] = sched
            st["availableReplicas"] = avail
            st["observedGeneration"] = o["metadata"].get("generation")
            st["selector"] = f"{c.LABEL_PCSG}={name}"
            pcs_name = o["metadata"].get("labels", {}).get(c.LABEL_PART_OF)
            pcs_obj = self.store.try_get(c.KIND_PCS, namespace, pcs_name) \
                if pcs_name else None
            if pcs_obj is not None:
                gh = (pcs_obj.get("status") or {}).get("currentGenerationHash")
                if gh:
                    st["currentPodCliqueSetGenerationHash"] = gh
            # updateProgress parity: active while member cliques carry mixed hashes
            member_qs = [q for qs in by_replica.values() for q in qs]
            hashes = {q["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH)
                      for q in member_qs}
            total_q = len(member_qs)
            updated_q = sum(
                1 for q in member_qs
                if int((q.get("status") or {}).get("updatedReplicas", 0))
                >= int(q["spec"].get("replicas", 1)))
            if len(hashes) > 1 or updated_q < total_q:
                prog = st.get("updateProgress") or {}
                if not prog.get("updateStartedAt"):
                    import time as _time
                    prog["updateStartedAt"] = _time.strftime(
                        "%Y-%m-%dT%H:%M:%SZ", _time.gmtime())
                prog["totalPodCliquesCount"] = total_q
                prog["updatedPodCliquesCount"] = updated_q
                prog.pop("updateEndedAt", None)
                st["updateProgress"] = prog
            elif st.get("updateProgress") and not st["updateProgress"].get(
                    "updateEndedAt"):
                import time as _time
                st["updateProgress"]["updateEndedAt"] = _time.strftime(
                    "%Y-%m-%dT%H:%M:%SZ", _time.gmtime())
                st["updateProgress"]["updatedPodCliquesCount"] = updated_q
            ever = bool(st.get("everAvailable")) or avail >= min_avail
            st["everAvailable"] = ever
            if ever and avail < min_avail:
                cond.set_condition(o, c.COND_MIN_AVAILABLE_BREACHED, True,
                                   c.REASON_INSUFFICIENT_AVAILABLE_PCSG_REPLICAS)
            else:
                changed = cond.set_condition(o, c.COND_MIN_AVAILABLE_BREACHED, False,
                                             c.REASON_SUFFICIENT_AVAILABLE_PCSG_REPLICAS)
                # recovery clears the PCS-level gang-termination latch
                if cond.condition_true(o, c.COND_GANG_TERMINATION_IN_PROGRESS) \
                        and avail >= min_avail:
                    cond.set_condition(o, c.COND_GANG_TERMINATION_IN_PROGRESS, False,
                                       c.REASON_GANG_TERMINATION_ACTIVE)
        # no-op fast path (see podclique._reconcile_status): skip the store
        # round-trip when the recomputed status is unchanged
        from ..kubecore.store import json_copy
        old_status = json_copy(pcsg.get("status") or {})
        upd(pcsg)
        if pcsg.get("status") == old_status:
            return
        try:
            self.store.patch(c.KIND_PCSG, namespace, name, upd, status=True)
        except ApiError:
            pass
