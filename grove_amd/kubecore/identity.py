"""Request identity for admission — the userInfo analog of admission.Request.

The reference's authorizer webhook sees the requester's identity from the apiserver
(admission/pcs/authorization/handler.go:39). The in-process store has no HTTP auth, so
identity is a thread-local: operator/controller threads run as the operator service
account by default; tests and external callers switch identity with `as_user(...)`.
"""
from __future__ import annotations

import threading
from contextlib import contextmanager

OPERATOR_USER = "system:serviceaccount:grove-system:grove-operator"
ANONYMOUS_USER = "system:anonymous"
NODE_AGENT_USER = "system:node:grove-agent"

_local = threading.local()


def current_user() -> str:
    return getattr(_local, "user", OPERATOR_USER)


@contextmanager
def as_user(name: str):
    prev = getattr(_local, "user", None)
    _local.user = name
    try:
        yield
    finally:
        if prev is None:
            del _local.user
        else:
            _local.user = prev
