"""PodClique controller: the pod machine.

Behavior parity: operator/internal/controller/podclique/ — pod create/delete loop with
hole-filling indices (components/pod/syncflow.go:135-185, internal/index/tracker.go),
hierarchical gate removal (syncflow.go:271-424), deletion priority (deletionsort.go),
rolling pod update (rollingupdate.go), status + MinAvailableBreached / PodCliqueScheduled
conditions (reconcilestatus.go:40-282). Fresh MI355X-native implementation.
"""
from __future__ import annotations

import logging
import time
from typing import List, Optional, Tuple

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError
from ..utils import conditions as cond
from ..utils import errors as groveerr
from ..utils.indexing import available_indices
from ..utils.concurrent import run_concurrently_with_slow_start
from . import builders
from .manager import Result

log = logging.getLogger("grove.podclique")


class PodCliqueReconciler:
    def __init__(self, store: Store, scheduler_name: str = c.SCHEDULER_AMD_GANG):
        self.store = store
        self.scheduler_name = scheduler_name

    # ------------------------------------------------------------------ entry
    def reconcile(self, namespace: str, name: str) -> Result:
        # read-only view: _reconcile_spec/_reconcile_delete never mutate the PCLQ
        pclq = self.store.try_get(c.KIND_PCLQ, namespace, name, copy=False)
        if pclq is None:
            return Result.DONE
        if pclq["metadata"].get("deletionTimestamp"):
            return self._reconcile_delete(pclq)
        rec = groveerr.StepRecorder(self.store, c.KIND_PCLQ, namespace, name)
        res, pods = self._reconcile_spec(pclq, rec)
        self._reconcile_status(namespace, name, pods=pods)
        rec.flush()
        if rec.retry_needed:
            # a step (e.g. the ungate patch) lost an optimistic race and was
            # swallowed as benign — nothing may retrigger this clique, so requeue
            return Result(requeue_after=0.05)
        return res

    # ------------------------------------------------------------------ delete
    def _reconcile_delete(self, pclq: Obj) -> Result:
        ns, name = pclq["metadata"].get("namespace"), pclq["metadata"]["name"]
        pods = self._owned_pods(pclq)
        for p in pods:
            try:
                self.store.delete("Pod", ns, p["metadata"]["name"])
            except ApiError:
                pass
        fins = pclq["metadata"].get("finalizers") or []
        if c.FINALIZER_PCLQ in fins:
            def rm(o: Obj) -> None:
                o["metadata"]["finalizers"] = [
                    f for f in o["metadata"].get("finalizers", []) if f != c.FINALIZER_PCLQ]
            self.store.patch(c.KIND_PCLQ, ns, name, rm)
        return Result.DONE

    # ------------------------------------------------------------------ spec
    def _owned_pods(self, pclq: Obj) -> List[Obj]:
        return self.store.list(
            "Pod", pclq["metadata"].get("namespace"),
            {c.LABEL_PODCLIQUE: pclq["metadata"]["name"]}, copy_objects=False)

    def _reconcile_spec(self, pclq: Obj, rec: groveerr.StepRecorder
                        ) -> Tuple[Result, List[Obj]]:
        ns = pclq["metadata"].get("namespace")
        desired = int(pclq["spec"].get("replicas", 1))
        pods = self._owned_pods(pclq)
        tmpl_hash = pclq["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH, "")

        # Rolling update: delete outdated pods one-ready-at-a-time (rollingupdate.go).
        # Under OnDelete strategy the user deletes pods; outdated pods count as current
        # and are never proactively replaced (GREP-291 parity).
        strategy = pclq["spec"].get("updateStrategy", c.UPDATE_ROLLING_RECREATE)
        outdated = [p for p in pods
                    if p["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH) != tmpl_hash]
        current = [p for p in pods
                   if p["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH) == tmpl_hash]
        if strategy == c.UPDATE_ON_DELETE:
            current = pods
            outdated = []
        if outdated:
            # delete gated/unready outdated pods freely; ready outdated pods one at a time
            not_ready = [p for p in outdated if not cond.pod_is_ready(p)]
            ready = [p for p in outdated if cond.pod_is_ready(p)]
            victims = not_ready + ready[:1]
            for p in victims:
                with rec.step(groveerr.ERR_ROLLING_UPDATE,
                              benign=groveerr.BENIGN_DELETE,
                              detail=f"replace pod {p['metadata']['name']}"):
                    self.store.delete("Pod", ns, p["metadata"]["name"])
            pods = current
        n = len(pods)

        if n < desired:
            in_use = [int(p["metadata"]["labels"].get(c.LABEL_POD_INDEX, -1)) for p in pods]
            pcs = self._find_pcs(pclq)
            if pcs is None:
                return Result(requeue_after=0.1), pods
            num_pods = self._pcsg_template_num_pods(pcs, pclq)
            indices = available_indices([i for i in in_use if i >= 0], desired - n)
            if len(indices) <= 2:
                for idx in indices:
                    pod = builders.build_pod(pcs, pclq, idx, self.scheduler_name,
                                             num_pods)
                    pods.append(self.store.create(pod))
            else:
                # slow-start batches (1,2,4,...) — utils/concurrent.go parity: a
                # systematic create failure is found after one cheap attempt
                def mk(idx):
                    return lambda: pods.append(self.store.create(
                        builders.build_pod(pcs, pclq, idx, self.scheduler_name,
                                           num_pods)))
                errs = run_concurrently_with_slow_start(
                    [(f"create-pod-{i}", mk(i)) for i in indices])
                if errs:
                    raise errs[0]
        elif n > desired:
            victims2 = self._deletion_order(pods)[: n - desired]
            for p in victims2:
                with rec.step(groveerr.ERR_SYNC_PODS,
                              benign=groveerr.BENIGN_DELETE,
                              detail=f"scale-in pod {p['metadata']['name']}"):
                    self.store.delete("Pod", ns, p["metadata"]["name"])
            gone = {p["metadata"]["name"] for p in victims2}
            pods = [p for p in pods if p["metadata"]["name"] not in gone]

        # inline gang completion before gate removal: a freshly completed gang is
        # Initialized and its pods ungated within THIS pass (latency chain collapse)
        gang_name = pclq["metadata"]["labels"].get(c.LABEL_PODGANG)
        if gang_name:
            from .podgang_component import try_complete_podgang
            try_complete_podgang(self.store, ns, gang_name)
        # the maintained pod list (creates appended, deletes removed) saves one
        # store list per reconcile — a measured hotspot at 10k-pod scale; the
        # status pass reuses the same list
        self._remove_scheduling_gates(pclq, rec, pods=pods)
        return Result.DONE, pods

    @staticmethod
    def _deletion_order(pods: List[Obj]) -> List[Obj]:
        """Victim priority (deletionsort.go:103): gated → pending → unscheduled →
        not-ready → younger first."""
        def rank(p: Obj):
            gated = cond.pod_is_gated(p)
            scheduled = cond.pod_is_scheduled(p)
            ready = cond.pod_is_ready(p)
            phase = (p.get("status") or {}).get("phase", "Pending")
            return (
                0 if gated else 1,
                0 if phase == "Pending" else 1,
                0 if not scheduled else 1,
                0 if not ready else 1,
                # younger (later creation) first
                tuple(-ord(ch) for ch in p["metadata"].get("creationTimestamp", "")),
            )
        return sorted(pods, key=rank)

    def _find_pcs(self, pclq: Obj) -> Optional[Obj]:
        # zero-copy read: callers only read the PCS (template lookups, gen hash)
        pcs_name = pclq["metadata"]["labels"].get(c.LABEL_PART_OF)
        if not pcs_name:
            return None
        return self.store.try_get(c.KIND_PCS, pclq["metadata"].get("namespace"),
                                  pcs_name, copy=False)

    @staticmethod
    def _pcsg_template_num_pods(pcs: Obj, pclq: Obj) -> Optional[int]:
        sg_fqn = pclq["metadata"]["labels"].get(c.LABEL_PCSG)
        if not sg_fqn:
            return None
        tmpl = pcs["spec"]["template"]
        sg = builders.match_by_fqn_suffix(
            sg_fqn, tmpl.get("podCliqueScalingGroups") or [])
        if sg is not None:
            total = 0
            for cl in tmpl.get("cliques") or []:
                if cl["name"] in (sg.get("cliqueNames") or []):
                    total += int(cl.get("spec", {}).get("replicas", 1))
            return total
        return None

    # ------------------------------------------------------------------ gates
    def _remove_scheduling_gates(self, pclq: Obj, rec: groveerr.StepRecorder,
                                 pods: Optional[List[Obj]] = None) -> None:
        """Hierarchical gang admission (syncflow.go:271-424): ungate a pod only when
        (a) its name is in its PodGang's podReferences and (b) it has no base-podgang
        label (base gang → immediate) OR the base PodGang is fully scheduled."""
        ns = pclq["metadata"].get("namespace")
        podgang_name = pclq["metadata"]["labels"].get(c.LABEL_PODGANG)
        if not podgang_name:
            return
        podgang = self.store.try_get(c.KIND_PODGANG, ns, podgang_name, copy=False)
        if podgang is None:
            return
        refs = set()
        for group in (podgang.get("spec") or {}).get("podgroups") or []:
            for ref in group.get("podReferences") or []:
                refs.add(ref.get("name") if isinstance(ref, dict) else ref)

        base_name = pclq["metadata"]["labels"].get(c.LABEL_BASE_PODGANG)
        base_scheduled: Optional[bool] = None  # lazily computed

        refs_repaired = False
        for p in (pods if pods is not None else self._owned_pods(pclq)):
            gates = p.get("spec", {}).get("schedulingGates") or []
            if not any(g.get("name") == c.POD_GANG_SCHEDULING_GATE for g in gates):
                continue
            if p["metadata"]["name"] not in refs:
                # gated pod not in an Initialized gang's refs = the gang still
                # references a dead predecessor (pod replaced out-of-band) —
                # repair the refs inline and retry this clique
                if not refs_repaired and cond.condition_true(
                        podgang, c.PODGANG_COND_INITIALIZED):
                    from .podgang_component import try_complete_podgang
                    try_complete_podgang(self.store, ns, podgang_name, rec=rec,
                                         force=True)
                    refs_repaired = True
                    rec.retry_needed = True  # re-run with the repaired refs
                continue
            if base_name:
                if base_scheduled is None:
                    base_scheduled = self._is_base_podgang_scheduled(ns, base_name)
                if not base_scheduled:
                    continue

            def ungate(o: Obj) -> None:
                o["spec"]["schedulingGates"] = [
                    g for g in o["spec"].get("schedulingGates", [])
                    if g.get("name") != c.POD_GANG_SCHEDULING_GATE]
            with rec.step(groveerr.ERR_UNGATE_POD,
                          benign=groveerr.BENIGN_UPDATE,
                          detail=f"ungate pod {p['metadata']['name']}"):
                self.store.patch("Pod", ns, p["metadata"]["name"], ungate)

    def _is_base_podgang_scheduled(self, ns: Optional[str], base_name: str) -> bool:
        """Base gang 'scheduled' = every podGroup's PCLQ has scheduledReplicas >=
        minReplicas (syncflow.go:343-424)."""
        base = self.store.try_get(c.KIND_PODGANG, ns, base_name, copy=False)
        if base is None:
            return False
        groups = (base.get("spec") or {}).get("podgroups") or []
        if not groups:
            return False
        for group in groups:
            pclq = self.store.try_get(c.KIND_PCLQ, ns, group.get("name", ""),
                                      copy=False)
            if pclq is None:
                return False
            scheduled = int((pclq.get("status") or {}).get("scheduledReplicas", 0))
            if scheduled < int(group.get("minReplicas", 0)):
                return False
        return True

    # ------------------------------------------------------------------ status
    def _reconcile_status(self, namespace: str, name: str,
                          pods: Optional[List[Obj]] = None) -> None:
        pclq = self.store.try_get(c.KIND_PCLQ, namespace, name, copy=False)
        if pclq is None or pclq["metadata"].get("deletionTimestamp"):
            return
        if pods is None:
            pods = self._owned_pods(pclq)
        tmpl_hash = pclq["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH, "")
        n_total = len(pods)
        n_ready = sum(1 for p in pods if cond.pod_is_ready(p))
        n_sched = sum(1 for p in pods if cond.pod_is_scheduled(p))
        n_gated = sum(1 for p in pods if cond.pod_is_gated(p))
        n_updated = sum(1 for p in pods
                        if p["metadata"]["labels"].get(c.LABEL_POD_TEMPLATE_HASH) == tmpl_hash)
        # "starting" = scheduled but not yet ready counts toward breach tolerance
        n_starting = sum(1 for p in pods
                         if cond.pod_is_scheduled(p) and not cond.pod_is_ready(p)
                         and (p.get("status") or {}).get("phase") in ("Pending", "Running"))
        min_avail = int(pclq["spec"].get("minAvailable", 1))

        def upd(o: Obj) -> None:
            st = o.setdefault("status", {})
            st["replicas"] = n_total
            st["readyReplicas"] = n_ready
            st["scheduledReplicas"] = n_sched
            st["scheduleGatedReplicas"] = n_gated
            st["updatedReplicas"] = n_updated
            st["observedGeneration"] = o["metadata"].get("generation")
            st["currentPodTemplateHash"] = tmpl_hash
            pcs_for_hash = self._find_pcs(o)
            if pcs_for_hash is not None:
                gh = (pcs_for_hash.get("status") or {}).get("currentGenerationHash")
                if gh:
                    st["currentPodCliqueSetGenerationHash"] = gh
            st["hpaPodSelector"] = f"{c.LABEL_PODCLIQUE}={name}"
            # rolling-update progress (status.updateProgress parity): active while any
            # pod carries an outdated template hash
            if n_updated < n_total:
                prog = st.get("updateProgress") or {}
                if prog.get("podTemplateHash") != tmpl_hash:
                    prog = {"podTemplateHash": tmpl_hash,
                            "updateStartedAt": time.strftime(
                                "%Y-%m-%dT%H:%M:%SZ", time.gmtime())}
                prog.pop("updateEndedAt", None)
                st["updateProgress"] = prog
            elif st.get("updateProgress") and not st["updateProgress"].get(
                    "updateEndedAt"):
                st["updateProgress"]["updateEndedAt"] = time.strftime(
                    "%Y-%m-%dT%H:%M:%SZ", time.gmtime())
            # PodCliqueScheduled (reconcilestatus.go:282)
            if n_sched >= min_avail:
                cond.set_condition(o, c.COND_PODCLIQUE_SCHEDULED, True,
                                   c.REASON_SUFFICIENT_SCHEDULED_PODS)
            else:
                cond.set_condition(o, c.COND_PODCLIQUE_SCHEDULED, False,
                                   c.REASON_INSUFFICIENT_SCHEDULED_PODS)
            # Breach eligibility requires the clique to have been scheduled at least once
            # (reference WasPCLQEverScheduled gate, gangterminate.go:171-206): freshly
            # created cliques are never 'breached', which breaks gang-termination re-fire.
            ever_scheduled = bool(st.get("everScheduled")) or n_sched >= min_avail
            st["everScheduled"] = ever_scheduled
            # MinAvailableBreached (reconcilestatus.go:218-255): breached when ready-or-
            # starting pods, or scheduled pods, dropped below minAvailable.
            if ever_scheduled and (n_ready + n_starting) < min_avail:
                cond.set_condition(o, c.COND_MIN_AVAILABLE_BREACHED, True,
                                   c.REASON_INSUFFICIENT_READY_PODS)
            elif ever_scheduled and n_sched < min_avail:
                cond.set_condition(o, c.COND_MIN_AVAILABLE_BREACHED, True,
                                   c.REASON_INSUFFICIENT_SCHEDULED_PODS)
            else:
                cond.set_condition(o, c.COND_MIN_AVAILABLE_BREACHED, False,
                                   c.REASON_SUFFICIENT_READY_PODS)
        # No-op fast path: apply the status function to a status-only probe (the
        # stored object is immutable; only the status subtree is copied — the
        # podSpec, the object's bulk, is shared read-only) and skip the store
        # round-trip when nothing would change — the dominant case during churn
        # storms at 10k-pod scale.
        from ..kubecore.store import json_copy
        old_status = pclq.get("status") or {}
        probe = {"kind": pclq.get("kind"), "metadata": pclq["metadata"],
                 "spec": pclq["spec"], "status": json_copy(old_status)}
        upd(probe)
        if probe["status"] == old_status:
            return
        try:
            self.store.patch(c.KIND_PCLQ, namespace, name, upd, status=True,
                             return_copy=False)
        except ApiError:
            pass
