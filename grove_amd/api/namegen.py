"""Deterministic resource naming — byte-compatible with the reference scheme.

Parity source: /root/reference/operator/api/common/namegen.go:33-130.
"""
from __future__ import annotations

from . import constants


def headless_service_name(pcs_name: str, replica: int) -> str:
    return f"{pcs_name}-{replica}"


def headless_service_address(pcs_name: str, replica: int, namespace: str) -> str:
    return f"{headless_service_name(pcs_name, replica)}.{namespace}.svc.cluster.local"


def pod_role_name(pcs_name: str) -> str:
    return f"{constants.GROUP}:pcs:{pcs_name}"


def pod_role_binding_name(pcs_name: str) -> str:
    return f"{constants.GROUP}:pcs:{pcs_name}"


def pod_service_account_name(pcs_name: str) -> str:
    return pcs_name


def initc_sa_token_secret_name(pcs_name: str) -> str:
    return f"{pcs_name}-ic-sat"


def podclique_name(owner_name: str, owner_replica: int, clique_template_name: str) -> str:
    """PCLQ FQN: '<owner>-<replica>-<clique>' (namegen.go:79). Owner is the PCS for
    standalone cliques and the PCSG FQN+replica for scaling-group members."""
    return f"{owner_name}-{owner_replica}-{clique_template_name}"


def pcsg_name(pcs_name: str, pcs_replica: int, scaling_group_name: str) -> str:
    return f"{pcs_name}-{pcs_replica}-{scaling_group_name}"


def base_podgang_name(pcs_name: str, pcs_replica: int) -> str:
    return f"{pcs_name}-{pcs_replica}"


def scaled_podgang_name(pcsg_fqn: str, scaled_index: int) -> str:
    return f"{pcsg_fqn}-{scaled_index}"


def podgang_name_for_pclq_in_pcsg(pcs_name: str, pcs_replica: int, pcsg_fqn: str,
                                  pcsg_min_available: int, pcsg_replica: int) -> str:
    """PCSG replicas [0, minAvailable) belong to the base PodGang; replicas >= minAvailable
    each get a scaled PodGang with 0-based index (namegen.go:109-127)."""
    if pcsg_replica < pcsg_min_available:
        return base_podgang_name(pcs_name, pcs_replica)
    return scaled_podgang_name(pcsg_fqn, pcsg_replica - pcsg_min_available)


def extract_scaling_group_name(pcsg_fqn: str, pcs_name: str, pcs_replica: int) -> str:
    prefix = f"{pcs_name}-{pcs_replica}-"
    if not pcsg_fqn.startswith(prefix):
        raise ValueError(f"PCSG FQN {pcsg_fqn!r} does not start with {prefix!r}")
    return pcsg_fqn[len(prefix):]


def pod_hostname(pclq_name: str, pod_index: int) -> str:
    return f"{pclq_name}-{pod_index}"
