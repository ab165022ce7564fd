"""Operator launcher CLI.

Parity source: operator/cmd/main.go:44 + cmd/cli/cli.go:64 (config-file launch options,
version) — `python -m grove_amd operator --config-file config.yaml` runs the full
control plane (store + HTTP apiserver + controllers + gang scheduler + virtual kubelet
or GPU node agent) until interrupted.
"""
from __future__ import annotations

import argparse
import logging
import signal
import sys
import time

from . import __version__


def cmd_operator(args) -> int:
    import sys as _sys
    _sys.setswitchinterval(0.0002)  # see bench.py: watch-chain handoff latency
    import os as _os0
    if _os0.environ.get("GROVE_GC_TUNE", "1") != "0":
        import gc as _gc
        _gc.collect()
        _gc.freeze()
        _gc.set_threshold(100000, 50, 50)  # keep gen0 GC off the reconcile path
    from .cluster import Cluster
    from .config import load_configuration
    from .kubecore.apiserver import ApiServer
    from .topology.agent import discover_node

    cfg = load_configuration(args.config_file)
    logging.basicConfig(
        level=getattr(logging, cfg.log_level.upper().replace("WARN", "WARNING")),
        format="%(asctime)s %(levelname)s %(name)s %(message)s")
    log = logging.getLogger("grove.main")

    cluster = Cluster(
        scheduler_name=cfg.default_scheduler,
        concurrent_syncs=cfg.concurrent_syncs("podCliqueSet"),
        controller_workers={name: cc.concurrent_syncs
                            for name, cc in cfg.controllers.items()},
        enable_authorizer=cfg.authorizer_enabled,
        auto_xgmi_domain=cfg.auto_xgmi_domain_enabled,
        use_native_scheduler=None,
    )
    snap = None
    if args.state_file:
        from .kubecore.persistence import SnapshotLoop, load as load_snapshot
        import os as _os
        if _os.path.exists(args.state_file):
            n = load_snapshot(cluster.store, args.state_file)
            log.info("restored %d objects from %s", n, args.state_file)
        snap = SnapshotLoop(cluster.store, args.state_file).start()

    if args.discover_node:
        node = discover_node()
        cluster.store.create(node)
        log.info("registered local node %s (%s GPUs)",
                 node["metadata"]["name"],
                 node["status"]["allocatable"].get("amd.com/gpu", 0))
    if args.virtual_nodes:
        cluster.add_virtual_nodes(args.virtual_nodes, gpus=args.virtual_gpus)
        log.info("registered %d virtual nodes (%d GPUs each)",
                 args.virtual_nodes, args.virtual_gpus)

    api = None
    if cfg.api_server.enabled or args.serve:
        import os as _os
        tokens = dict(cfg.api_server.tokens)
        agent_token = _os.environ.get("GROVE_AGENT_TOKEN")
        if agent_token:
            from .kubecore.identity import NODE_AGENT_USER
            tokens[agent_token] = NODE_AGENT_USER
        api = ApiServer(cluster.store, cfg.api_server.host, cfg.api_server.port,
                        metrics_fn=cluster.metrics_lines,
                        auth_tokens=tokens or None)
        api.start()
        log.info("apiserver listening on %s", api.url)

    cluster.start()
    log.info("grove-amd operator %s running (scheduler=%s, authorizer=%s, TAS=%s)",
             __version__, cfg.default_scheduler, cfg.authorizer_enabled,
             cfg.topology_aware_scheduling_enabled)

    stop = {"flag": False}

    def on_sig(_s, _f):
        stop["flag"] = True
    signal.signal(signal.SIGINT, on_sig)
    signal.signal(signal.SIGTERM, on_sig)
    while not stop["flag"]:
        time.sleep(0.2)
    log.info("shutting down")
    if snap is not None:
        snap.stop(final_snapshot=True)
    cluster.stop()
    if api is not None:
        api.stop()
    return 0


def cmd_agent(args) -> int:
    """Remote node agent: registers this machine as a Node with the operator's
    apiserver and runs its scheduled pods as OS processes (ProcessKubelet over the
    HTTP client) — the kubelet of the distributed deployment shape."""
    import time as _time

    from .kubecore.httpclient import HttpStoreClient
    from .kubecore.store import ApiError
    from .kubelet.process import ProcessKubelet
    from .topology.agent import discover_node
    from .kubelet.virtual import make_virtual_node

    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(levelname)s %(name)s %(message)s")
    log = logging.getLogger("grove.agent")
    import os as _os
    token = args.token or _os.environ.get("GROVE_AGENT_TOKEN")
    client = HttpStoreClient(args.server, token=token)
    if args.virtual_gpus is not None:
        node = make_virtual_node(args.node_name or "agent-node",
                                 gpus=args.virtual_gpus)
    else:
        node = discover_node(args.node_name)
    name = node["metadata"]["name"]
    node["metadata"].setdefault("annotations", {})["grove.io/external-kubelet"] = \
        "true"
    try:
        client.create(node)
        log.info("registered node %s", name)
    except ApiError as e:
        if e.reason != "AlreadyExists":
            raise
        log.info("node %s already registered", name)
    kubelet = ProcessKubelet(client, api_url=args.server, node_names=[name])
    stop = {"flag": False}

    def on_sig(_s, _f):
        stop["flag"] = True
    signal.signal(signal.SIGINT, on_sig)
    signal.signal(signal.SIGTERM, on_sig)
    log.info("agent serving node %s against %s", name, args.server)

    # watch-driven (kubelet informer analog): pod events for this node reconcile
    # immediately; a periodic full relist catches anything a dropped stream missed
    import queue as _queue
    import threading as _threading
    events: "_queue.Queue" = _queue.Queue()

    def pump() -> None:
        while not stop["flag"]:
            try:
                for ev, pod in client.watch_events(
                        "Pod", args.namespace, seed=True,
                        field_selector={"spec.nodeName": name}):
                    if stop["flag"]:
                        return
                    events.put(pod)
            except Exception as e:
                log.warning("pod watch stream ended (%s); reconnecting", e)
                _time.sleep(min(2.0, args.poll_interval * 5))
    _threading.Thread(target=pump, daemon=True).start()

    last_resync = 0.0
    while not stop["flag"]:
        try:
            pod = events.get(timeout=args.poll_interval)
            kubelet.reconcile(pod["metadata"].get("namespace", "default"),
                              pod["metadata"]["name"])
        except _queue.Empty:
            pass
        except Exception as e:
            log.warning("agent reconcile failed: %s", e)
        now = _time.monotonic()
        if now - last_resync > max(10.0, args.poll_interval * 50):
            last_resync = now
            try:
                for p in client.list("Pod", args.namespace,
                                     field_selector={"spec.nodeName": name}):
                    kubelet.reconcile(p["metadata"].get("namespace", "default"),
                                      p["metadata"]["name"])
            except Exception as e:
                log.warning("agent resync failed: %s", e)
    kubelet.shutdown()
    return 0


def cmd_get(args) -> int:
    import json
    import urllib.parse
    import urllib.request

    import yaml as _y

    from .kubecore.apiserver import PLURALS, CLUSTER_SCOPED_PLURALS
    plural = args.resource.lower()
    kind = PLURALS.get(plural)
    if kind is None:
        print(f"unknown resource {plural!r}; one of: {', '.join(sorted(PLURALS))}")
        return 1
    group = "grove.io/v1alpha1" if kind.startswith("PodClique") or         kind == "ClusterTopologyBinding" else         ("scheduler.grove.io/v1alpha1" if kind == "PodGang" else "v1")
    base = f"{args.server}/apis/{group}" if "/" in group else         f"{args.server}/api/{group}"
    if plural in CLUSTER_SCOPED_PLURALS:
        url = f"{base}/{plural}"
    else:
        url = f"{base}/namespaces/{args.namespace}/{plural}"
    if args.name:
        url += f"/{args.name}"
    elif args.selector:
        url += f"?labelSelector={urllib.parse.quote(args.selector)}"
    if getattr(args, "watch", False):
        # kubectl get -w analog: follow the ndjson watch stream
        sep = "&" if "?" in url else "?"
        with urllib.request.urlopen(url + f"{sep}watch=true") as r:
            try:
                for line in r:
                    ev = json.loads(line)
                    o = ev.get("object", {})
                    print(ev.get("type", "?"),
                          o.get("kind", kind),
                          o.get("metadata", {}).get("name", ""),
                          (o.get("status") or {}).get("phase", ""), flush=True)
            except KeyboardInterrupt:
                pass
        return 0
    with urllib.request.urlopen(url, timeout=10) as r:
        data = json.loads(r.read())
    items = data.get("items", [data] if args.name else [])
    if args.output == "json":
        print(json.dumps(items if not args.name else items[0], indent=2))
        return 0
    if args.output == "yaml":
        print(_y.safe_dump_all(items, sort_keys=False))
        return 0
    rows = []
    for o in items:
        st = o.get("status") or {}
        phase = st.get("phase") or ""
        ready = st.get("readyReplicas", st.get("availableReplicas", ""))
        rows.append((o["metadata"]["name"], phase, str(ready),
                     o["metadata"].get("creationTimestamp", "")))
    w = max([len(r[0]) for r in rows] + [4]) + 2
    print(f"{'NAME':<{w}}{'PHASE':<12}{'READY':<8}CREATED")
    for r in rows:
        print(f"{r[0]:<{w}}{r[1]:<12}{r[2]:<8}{r[3]}")
    return 0


def cmd_apply(args) -> int:
    import json
    import urllib.request

    import yaml as _y

    from .kubecore.apiserver import PLURALS
    kind_to_plural = {v: k for k, v in PLURALS.items()}
    n = 0
    with open(args.filename) as f:
        for doc in _y.safe_load_all(f):
            if not doc:
                continue
            plural = kind_to_plural.get(doc.get("kind"))
            group = doc.get("apiVersion", "v1")
            base = f"{args.server}/apis/{group}" if "/" in group else                 f"{args.server}/api/{group}"
            url = f"{base}/namespaces/{args.namespace}/{plural}"
            req = urllib.request.Request(
                url, data=json.dumps(doc).encode(), method="POST",
                headers={"Content-Type": "application/json"})
            try:
                urllib.request.urlopen(req, timeout=10)
                n += 1
                print(f"created {doc.get('kind')}/{doc['metadata']['name']}")
            except urllib.error.HTTPError as e:
                if e.code == 409:
                    # kubectl-apply semantics: exists -> strategic-merge PATCH
                    preq = urllib.request.Request(
                        f"{url}/{doc['metadata']['name']}",
                        data=json.dumps(doc).encode(), method="PATCH",
                        headers={"Content-Type":
                                 "application/strategic-merge-patch+json"})
                    try:
                        urllib.request.urlopen(preq, timeout=10)
                        n += 1
                        print(f"configured {doc.get('kind')}/"
                              f"{doc['metadata']['name']}")
                        continue
                    except urllib.error.HTTPError as e2:
                        e = e2
                print(f"error {e.code} for {doc.get('kind')}/"
                      f"{doc['metadata'].get('name')}: {e.read().decode()[:200]}")
    return 0 if n else 1


def cmd_install_crds(args) -> int:
    from .api.crds import install_crds, write_crds
    if args.output_dir:
        for p in write_crds(args.output_dir):
            print("wrote", p)
    if args.server:
        print(f"applied {install_crds(args.server)} CRDs to {args.server}")
    if not args.output_dir and not args.server:
        from .api.crds import render_all
        import yaml as _y
        print(_y.safe_dump_all(render_all(), sort_keys=False))
    return 0


def cmd_version(_args) -> int:
    print(f"grove-amd {__version__}")
    return 0


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="grove-amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    op = sub.add_parser("operator", help="run the control plane")
    op.add_argument("--config-file", default=None)
    op.add_argument("--serve", action="store_true",
                    help="serve the HTTP apiserver even if config disables it")
    op.add_argument("--discover-node", action="store_true",
                    help="register this machine as a Node via the topology agent")
    op.add_argument("--virtual-nodes", type=int, default=0)
    op.add_argument("--virtual-gpus", type=int, default=8)
    op.add_argument("--state-file", default=None,
                    help="persist/restore the store to this snapshot file")
    op.set_defaults(fn=cmd_operator)

    get = sub.add_parser("get", help="list resources from a running apiserver")
    get.add_argument("resource", help="plural, e.g. podcliquesets, pods, podgangs")
    get.add_argument("name", nargs="?", default=None)
    get.add_argument("--server", default="http://127.0.0.1:8081")
    get.add_argument("-n", "--namespace", default="default")
    get.add_argument("-l", "--selector", default=None)
    get.add_argument("-o", "--output", choices=["wide", "json", "yaml"],
                     default="wide")
    get.add_argument("-w", "--watch", action="store_true",
                     help="stream ADDED/MODIFIED/DELETED events (ndjson watch)")
    get.set_defaults(fn=cmd_get)

    ap_cmd = sub.add_parser("apply", help="apply a manifest file to the apiserver")
    ap_cmd.add_argument("-f", "--filename", required=True)
    ap_cmd.add_argument("--server", default="http://127.0.0.1:8081")
    ap_cmd.add_argument("-n", "--namespace", default="default")
    ap_cmd.set_defaults(fn=cmd_apply)

    agent = sub.add_parser("agent", help="remote node agent (kubelet) for a GPU node")
    agent.add_argument("--server", required=True)
    agent.add_argument("--node-name", default=None)
    agent.add_argument("--namespace", default=None,
                       help="restrict to one namespace (default: all)")
    agent.add_argument("--virtual-gpus", type=int, default=None,
                       help="register a virtual node instead of probing hardware")
    agent.add_argument("--poll-interval", type=float, default=0.2)
    agent.add_argument("--token", default=None,
                       help="bearer token for the apiserver "
                            "(default: $GROVE_AGENT_TOKEN)")
    agent.set_defaults(fn=cmd_agent)

    crds = sub.add_parser("install-crds", help="render or apply the CRDs")
    crds.add_argument("--server", default=None, help="apiserver URL to POST CRDs to")
    crds.add_argument("--output-dir", default=None, help="write CRD YAML files here")
    crds.set_defaults(fn=cmd_install_crds)

    ver = sub.add_parser("version")
    ver.set_defaults(fn=cmd_version)

    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
