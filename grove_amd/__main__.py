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
        api = ApiServer(cluster.store, cfg.api_server.host, cfg.api_server.port)
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
