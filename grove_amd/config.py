"""OperatorConfiguration — file-based operator config with defaults + validation.

Parity source: operator/api/config/v1alpha1/types.go:119-319 ({client QPS/burst, leader
election, servers, per-controller concurrentSyncs, authorizer, topology-aware
scheduling, network auto-domain flag, scheduler profiles}), defaults.go and
api/config/validation/. NVIDIA's autoMNNVLEnabled becomes autoXGMIDomainEnabled.
"""
from __future__ import annotations

import dataclasses
from typing import Dict, List, Optional

import yaml

from .api import constants as c
from .kubecore.store import invalid

ALLOWED_SCHEDULERS = (c.SCHEDULER_AMD_GANG, c.SCHEDULER_DEFAULT, "lpx-scheduler")


@dataclasses.dataclass
class ServerConfig:
    host: str = "127.0.0.1"
    port: int = 8081
    enabled: bool = False
    # bearer token -> user identity (the static-token authenticator analog);
    # the GROVE_AGENT_TOKEN env var adds a node-agent token at launch.
    tokens: Dict[str, str] = dataclasses.field(default_factory=dict)


@dataclasses.dataclass
class ControllerConfig:
    concurrent_syncs: int = 4


@dataclasses.dataclass
class OperatorConfiguration:
    client_qps: float = 100.0
    client_burst: int = 150
    leader_election_enabled: bool = False
    api_server: ServerConfig = dataclasses.field(default_factory=ServerConfig)
    metrics_server: ServerConfig = dataclasses.field(
        default_factory=lambda: ServerConfig(port=8082))
    controllers: Dict[str, ControllerConfig] = dataclasses.field(default_factory=dict)
    authorizer_enabled: bool = True
    authorizer_exempt_users: List[str] = dataclasses.field(default_factory=list)
    topology_aware_scheduling_enabled: bool = True
    auto_xgmi_domain_enabled: bool = False
    default_scheduler: str = c.SCHEDULER_AMD_GANG
    scheduler_profiles: List[str] = dataclasses.field(
        default_factory=lambda: [c.SCHEDULER_AMD_GANG, c.SCHEDULER_DEFAULT])
    log_level: str = "info"

    def concurrent_syncs(self, controller: str) -> int:
        cc = self.controllers.get(controller)
        return cc.concurrent_syncs if cc else 4


def default_configuration() -> OperatorConfiguration:
    return OperatorConfiguration()


def load_configuration(path: Optional[str]) -> OperatorConfiguration:
    cfg = default_configuration()
    if path is None:
        return cfg
    with open(path) as f:
        raw = yaml.safe_load(f) or {}
    client = raw.get("client") or {}
    cfg.client_qps = float(client.get("qps", cfg.client_qps))
    cfg.client_burst = int(client.get("burst", cfg.client_burst))
    le = raw.get("leaderElection") or {}
    cfg.leader_election_enabled = bool(le.get("enabled", False))
    servers = raw.get("servers") or {}
    api = servers.get("api") or {}
    cfg.api_server = ServerConfig(api.get("host", "127.0.0.1"),
                                  int(api.get("port", 8081)),
                                  bool(api.get("enabled", False)),
                                  {t["token"]: t["user"]
                                   for t in (api.get("tokens") or [])})
    met = servers.get("metrics") or {}
    cfg.metrics_server = ServerConfig(met.get("host", "127.0.0.1"),
                                      int(met.get("port", 8082)),
                                      bool(met.get("enabled", False)))
    for name, cc in (raw.get("controllers") or {}).items():
        cfg.controllers[name] = ControllerConfig(
            int((cc or {}).get("concurrentSyncs", 4)))
    authz = raw.get("authorizer") or {}
    cfg.authorizer_enabled = bool(authz.get("enabled", True))
    cfg.authorizer_exempt_users = list(authz.get("exemptServiceAccounts", []))
    tas = raw.get("topologyAwareScheduling") or {}
    cfg.topology_aware_scheduling_enabled = bool(tas.get("enabled", True))
    net = raw.get("network") or {}
    cfg.auto_xgmi_domain_enabled = bool(net.get("autoXGMIDomainEnabled", False))
    sched = raw.get("scheduler") or {}
    cfg.default_scheduler = sched.get("default", c.SCHEDULER_AMD_GANG)
    cfg.scheduler_profiles = list(sched.get("profiles", cfg.scheduler_profiles))
    cfg.log_level = raw.get("logLevel", "info")
    validate_configuration(cfg)
    return cfg


def validate_configuration(cfg: OperatorConfiguration) -> None:
    if cfg.client_qps <= 0 or cfg.client_burst <= 0:
        raise invalid("client qps/burst must be positive")
    if cfg.default_scheduler not in ALLOWED_SCHEDULERS:
        raise invalid(f"scheduler.default must be one of {ALLOWED_SCHEDULERS}")
    for p in cfg.scheduler_profiles:
        if p not in ALLOWED_SCHEDULERS:
            raise invalid(f"unknown scheduler profile {p!r}")
    if cfg.default_scheduler not in cfg.scheduler_profiles:
        raise invalid("scheduler.default must be listed in scheduler.profiles")
    for name, cc in cfg.controllers.items():
        if cc.concurrent_syncs < 1:
            raise invalid(f"controllers[{name}].concurrentSyncs must be >= 1")
    if cfg.log_level not in ("debug", "info", "warn", "error"):
        raise invalid(f"unknown logLevel {cfg.log_level!r}")
