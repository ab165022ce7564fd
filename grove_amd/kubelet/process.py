"""Process kubelet — pods as real OS processes.

Container-runtime analog for single-machine deployments: each bound pod becomes a
subprocess running kubelet/podrunner.py with the pod's injected env (the GROVE_* env
contract from the pod spec), HIP_VISIBLE_DEVICES pinned to the scheduler-assigned GPU
ids, and startup ordering enforced by the real grove-initc waiter against the HTTP
apiserver — the closest in-pod lifecycle to the reference's init-container + container
flow this environment can run. Pod phases follow the process: Running on spawn,
Ready once the process reports startup-dependencies-satisfied (or immediately for
gangless pods), Succeeded/Failed on exit.
"""
from __future__ import annotations

import logging
import os
import subprocess
import sys
import threading
from typing import Dict, List, Optional

from ..api import constants as c
from ..kubecore.store import Store, Obj, ApiError
from ..controllers.manager import Result
from .gpunode import GPU_IDS_ANNOTATION

log = logging.getLogger("grove.processkubelet")


class ProcessKubelet:
    """Per-pod reconciler that spawns/reaps pod processes."""

    def __init__(self, store: Store, api_url: Optional[str] = None,
                 node_names: Optional[List[str]] = None):
        self.store = store
        self.api_url = api_url
        self.node_names = set(node_names) if node_names is not None else None
        self._procs: Dict[str, subprocess.Popen] = {}
        self._lock = threading.Lock()

    def handles(self, node: str) -> bool:
        return self.node_names is None or node in self.node_names

    def reconcile(self, namespace: str, name: str) -> Result:
        key = f"{namespace}/{name}"
        pod = self.store.try_get("Pod", namespace, name)
        if pod is None or pod["metadata"].get("deletionTimestamp"):
            self._kill(key)
            return Result.DONE
        node = pod.get("spec", {}).get("nodeName")
        if not node or not self.handles(node):
            return Result.DONE
        phase = (pod.get("status") or {}).get("phase", "Pending")
        with self._lock:
            proc = self._procs.get(key)
        if proc is None:
            if phase == "Pending":
                self._spawn(key, pod)
                return Result(requeue_after=0.05)
            return Result.DONE
        rc = proc.poll()
        if rc is None:
            # Still running. Readiness is granted on successful EXIT (this runner's
            # pods are payload-then-exit jobs); a long-running serving variant would
            # use a readiness probe here instead.
            return Result(requeue_after=0.05)
        with self._lock:
            self._procs.pop(key, None)
        self._finish(namespace, name, rc)
        return Result.DONE

    # ------------------------------------------------------------------ internals
    def _spawn(self, key: str, pod: Obj) -> None:
        env = dict(os.environ)
        for ctr in pod["spec"].get("containers", []):
            for ev in ctr.get("env", []) or []:
                if "value" in ev:
                    env[ev["name"]] = str(ev["value"])
        ns = pod["metadata"].get("namespace", "default")
        env["GROVE_NAMESPACE"] = ns
        env["GROVE_PODGANG_NAME"] = pod["metadata"].get("labels", {}).get(
            c.LABEL_PODGANG, "")
        if self.api_url:
            env["GROVE_API_SERVER"] = self.api_url
        ann = pod["metadata"].get("annotations") or {}
        if ann.get("grove.io/payload-shape"):
            env["GROVE_PAYLOAD_SHAPE"] = ann["grove.io/payload-shape"]
        if ann.get("grove.io/payload"):
            env["GROVE_PAYLOAD"] = ann["grove.io/payload"]
        gpu_ids = ann.get(GPU_IDS_ANNOTATION)
        if gpu_ids:
            env["HIP_VISIBLE_DEVICES"] = gpu_ids
        # startup deps: prefer the pod's own grove-initc init-container flags (what a
        # real kubelet would execute), falling back to the PCLQ's startsAfter list
        deps = []
        for ic in pod["spec"].get("initContainers") or []:
            if ic.get("name") == "grove-initc":
                deps = [a.split("=", 1)[1] for a in ic.get("args", [])
                        if a.startswith("--podcliques=")]
                break
        if not deps:
            pclq_name = pod["metadata"].get("labels", {}).get(c.LABEL_PODCLIQUE)
            pclq = (self.store.try_get(c.KIND_PCLQ, ns, pclq_name)
                    if pclq_name else None)
            for fqn in (pclq or {}).get("spec", {}).get("startsAfter") or []:
                dep = self.store.try_get(c.KIND_PCLQ, ns, fqn)
                min_avail = int((dep or {}).get("spec", {}).get("minAvailable", 1))
                deps.append(f"{fqn}:{min_avail}")
        if deps:
            env["GROVE_STARTS_AFTER"] = ",".join(deps)
        proc = subprocess.Popen(
            [sys.executable, "-m", "grove_amd.kubelet.podrunner"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
            text=True, cwd=os.path.dirname(os.path.dirname(
                os.path.dirname(os.path.abspath(__file__)))))
        with self._lock:
            self._procs[key] = proc
        ns_, name = key.split("/", 1)
        self._set_phase(ns_, name, "Running")

    def _set_phase(self, ns: str, name: str, phase: str) -> None:
        def upd(o: Obj) -> None:
            o.setdefault("status", {})["phase"] = phase
        try:
            self.store.patch("Pod", ns, name, upd, status=True)
        except ApiError:
            pass

    def _finish(self, ns: str, name: str, rc: int) -> None:
        def upd(o: Obj) -> None:
            st = o.setdefault("status", {})
            st["phase"] = "Succeeded" if rc == 0 else "Failed"
            if rc == 0:
                conds = st.setdefault("conditions", [])
                for want in ("ContainersReady", "Ready"):
                    for cd in conds:
                        if cd.get("type") == want:
                            cd["status"] = "True"
                            break
                    else:
                        conds.append({"type": want, "status": "True",
                                      "reason": "ProcessExited"})
        try:
            self.store.patch("Pod", ns, name, upd, status=True)
        except ApiError:
            pass

    def _kill(self, key: str) -> None:
        with self._lock:
            proc = self._procs.pop(key, None)
        if proc is not None and proc.poll() is None:
            proc.terminate()
            try:
                proc.wait(timeout=5)
            except subprocess.TimeoutExpired:
                proc.kill()

    def shutdown(self) -> None:
        with self._lock:
            keys = list(self._procs)
        for k in keys:
            self._kill(k)
