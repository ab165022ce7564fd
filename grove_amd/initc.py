"""grove-initc — the startup-ordering init waiter.

Parity source: operator/initc/ (cmd/opts/options.go:60, internal/wait.go:109-281): runs
as an init container, takes --podcliques=<fqn>:<minAvailable> flags plus the pod's
namespace/podgang identity (downward API in the reference; flags/env here), watches
pods carrying the grove.io/podgang label, and exits 0 only when every parent PodClique
has at least minAvailable Ready pods.

Two transports: --server http://... polls the HTTP apiserver (the in-pod deployment
shape); without --server it resolves an in-process Store via grove_amd.initc.attach()
(used by tests and the virtual kubelet, which embeds the same predicate).

CLI:
  python -m grove_amd.initc --namespace ns --podgang my-pcs-0 \
      --podcliques parent-a:2 --podcliques parent-b:1 [--server URL] [--timeout S]
"""
from __future__ import annotations

import argparse
import json
import sys
import time
import urllib.parse
import urllib.request
from typing import Dict, List, Optional, Tuple

from .api import constants as c

_attached_store = None


def attach(store) -> None:
    """Attach an in-process Store (test/virtual-kubelet transport)."""
    global _attached_store
    _attached_store = store


def parse_podcliques(values: List[str]) -> List[Tuple[str, int]]:
    out = []
    for v in values:
        if ":" not in v:
            raise ValueError(f"--podcliques expects <fqn>:<minAvailable>, got {v!r}")
        fqn, min_s = v.rsplit(":", 1)
        out.append((fqn, int(min_s)))
    return out


def _pods_via_http(server: str, namespace: str, podgang: str) -> List[dict]:
    sel = urllib.parse.quote(f"{c.LABEL_PODGANG}={podgang}")
    url = f"{server}/api/v1/namespaces/{namespace}/pods?labelSelector={sel}"
    with urllib.request.urlopen(url, timeout=5) as resp:
        return json.loads(resp.read()).get("items", [])


def _pods_via_store(namespace: str, podgang: str) -> List[dict]:
    if _attached_store is None:
        raise RuntimeError("no --server given and no in-process store attached")
    return _attached_store.list("Pod", namespace, {c.LABEL_PODGANG: podgang})


def ready_counts(pods: List[dict]) -> Dict[str, int]:
    counts: Dict[str, int] = {}
    for p in pods:
        pclq = (p.get("metadata", {}).get("labels") or {}).get(c.LABEL_PODCLIQUE)
        if not pclq:
            continue
        ready = any(cond.get("type") == "Ready" and cond.get("status") == "True"
                    for cond in (p.get("status") or {}).get("conditions") or [])
        if ready:
            counts[pclq] = counts.get(pclq, 0) + 1
    return counts


def wait_for_parents(namespace: str, podgang: str,
                     parents: List[Tuple[str, int]],
                     server: Optional[str] = None,
                     timeout: float = 0.0, poll: float = 0.5) -> bool:
    deadline = time.monotonic() + timeout if timeout > 0 else None
    while True:
        pods = (_pods_via_http(server, namespace, podgang) if server
                else _pods_via_store(namespace, podgang))
        counts = ready_counts(pods)
        if all(counts.get(fqn, 0) >= min_avail for fqn, min_avail in parents):
            return True
        if deadline is not None and time.monotonic() > deadline:
            return False
        time.sleep(poll)


def main(argv: Optional[List[str]] = None) -> int:
    ap = argparse.ArgumentParser(prog="grove-initc")
    ap.add_argument("--namespace", default="default")
    ap.add_argument("--podgang", required=True,
                    help="PodGang name this pod belongs to (grove.io/podgang label)")
    ap.add_argument("--podcliques", action="append", default=[],
                    help="<parent-pclq-fqn>:<minAvailable>; repeatable")
    ap.add_argument("--server", default=None, help="apiserver base URL")
    ap.add_argument("--timeout", type=float, default=0.0)
    ap.add_argument("--poll-interval", type=float, default=0.5)
    args = ap.parse_args(argv)

    parents = parse_podcliques(args.podcliques)
    if not parents:
        return 0
    ok = wait_for_parents(args.namespace, args.podgang, parents,
                          server=args.server, timeout=args.timeout,
                          poll=args.poll_interval)
    if ok:
        print("grove-initc: all startup dependencies satisfied")
        return 0
    print("grove-initc: timed out waiting for startup dependencies", file=sys.stderr)
    return 1


if __name__ == "__main__":
    sys.exit(main())
