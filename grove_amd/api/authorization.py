"""Managed-resource protection — the authorizer webhook.

Parity source: operator/internal/webhook/admission/pcs/authorization/handler.go:39 and
config/v1alpha1/types.go:293-303: an optional validating admission that rejects
mutations of grove-managed resources (app.kubernetes.io/managed-by=grove-operator) by
identities other than the operator (plus an exempt list), with the
grove.io/disable-managed-resource-protection annotation on the owning PodCliqueSet as
the escape hatch.
"""
from __future__ import annotations

from typing import Iterable, Optional

from . import constants as c
from ..kubecore.identity import current_user, OPERATOR_USER, NODE_AGENT_USER
from ..kubecore.store import Store, Obj, forbidden

PROTECTED_KINDS = (c.KIND_PCLQ, c.KIND_PCSG, c.KIND_PODGANG, "Pod", "Service",
                   "Secret", "ServiceAccount", "Role", "RoleBinding",
                   "HorizontalPodAutoscaler")


class Authorizer:
    def __init__(self, store: Store, exempt_users: Optional[Iterable[str]] = None):
        self.store = store
        # Node agents (the kubelet analog) must write pod/node status; the
        # reference exempts system components the same way (types.go:293-303).
        self.exempt = {OPERATOR_USER, NODE_AGENT_USER, *(exempt_users or ())}

    def register(self) -> None:
        for kind in PROTECTED_KINDS:
            self.store.register_validator(kind, self._validate)
            self.store.register_delete_validator(
                kind, lambda obj: self._validate(obj, obj))

    def _validate(self, obj: Obj, old: Optional[Obj]) -> None:
        if old is None:
            return  # creations are validated by ownership elsewhere
        labels = old.get("metadata", {}).get("labels") or {}
        if labels.get(c.LABEL_MANAGED_BY) != c.LABEL_MANAGED_BY_VALUE:
            return
        user = current_user()
        if user in self.exempt:
            return
        pcs_name = labels.get(c.LABEL_PART_OF)
        if pcs_name:
            pcs = self.store.try_get(c.KIND_PCS,
                                     old["metadata"].get("namespace"), pcs_name)
            ann = (pcs or {}).get("metadata", {}).get("annotations") or {}
            if ann.get(c.ANNOTATION_DISABLE_MANAGED_RESOURCE_PROTECTION) == "true":
                return
        raise forbidden(
            f"user {user!r} may not modify grove-managed "
            f"{old.get('kind')} {old['metadata'].get('name')!r} "
            f"(set {c.ANNOTATION_DISABLE_MANAGED_RESOURCE_PROTECTION}=true on the "
            f"owning PodCliqueSet to override)")
