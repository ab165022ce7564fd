"""Webhook/API TLS bootstrap — self-provisioned CA + serving cert.

Parity source: operator/internal/controller/cert/cert.go:50-86 ('auto' mode
self-provisions CA+cert via the cert-controller rotator into a Secret and gates
webhook readiness; 'manual' expects external certs). Here 'auto' shells out to the
system openssl (no python cryptography package in this image), stores the PEMs in a
kube-style Secret in the store, and hands uvicorn the files for a TLS apiserver.
"""
from __future__ import annotations

import os
import subprocess
import tempfile
from typing import Dict, Tuple

from .store import Store, ApiError

CERT_SECRET_NAME = "grove-operator-tls"


def generate_self_signed(common_name: str = "grove-amd-apiserver",
                         days: int = 365) -> Dict[str, str]:
    """Returns {'ca.crt','tls.crt','tls.key'} PEM strings (CA == serving cert for the
    single-process deployment; SANs cover localhost)."""
    with tempfile.TemporaryDirectory() as td:
        key = os.path.join(td, "tls.key")
        crt = os.path.join(td, "tls.crt")
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
             "-keyout", key, "-out", crt, "-days", str(days),
             "-subj", f"/CN={common_name}",
             "-addext", "subjectAltName=DNS:localhost,IP:127.0.0.1"],
            check=True, capture_output=True)
        with open(crt) as f:
            crt_pem = f.read()
        with open(key) as f:
            key_pem = f.read()
    return {"ca.crt": crt_pem, "tls.crt": crt_pem, "tls.key": key_pem}


def ensure_cert_secret(store: Store, namespace: str = "grove-system",
                       mode: str = "auto") -> Dict[str, str]:
    """auto: create (or reuse) the TLS Secret; manual: require it to exist."""
    cur = store.try_get("Secret", namespace, CERT_SECRET_NAME)
    if cur is not None and cur.get("data", {}).get("tls.crt"):
        return cur["data"]
    if mode == "manual":
        raise ApiError(500, "CertsMissing",
                       f"manual cert mode: Secret {namespace}/{CERT_SECRET_NAME} "
                       f"with tls.crt/tls.key required")
    data = generate_self_signed()
    secret = {
        "apiVersion": "v1", "kind": "Secret",
        "metadata": {"name": CERT_SECRET_NAME, "namespace": namespace,
                     "labels": {"app.kubernetes.io/managed-by": "grove-operator"}},
        "type": "kubernetes.io/tls",
        "data": data,
    }
    try:
        store.create(secret)
    except ApiError as e:
        if e.reason != "AlreadyExists":
            raise
    return data


def write_cert_files(data: Dict[str, str], directory: str) -> Tuple[str, str]:
    os.makedirs(directory, exist_ok=True)
    crt = os.path.join(directory, "tls.crt")
    key = os.path.join(directory, "tls.key")
    with open(crt, "w") as f:
        f.write(data["tls.crt"])
    with open(key, "w") as f:
        f.write(data["tls.key"])
    os.chmod(key, 0o600)
    return crt, key
