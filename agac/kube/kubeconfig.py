"""kubeconfig parsing → RestConfig.

The equivalent of clientcmd.BuildConfigFromFlags (reference
``cmd/controller/controller.go:50``): resolves the current (or named)
context into server URL, auth material (bearer token or client certs,
inline base64 ``*-data`` fields written to temp files) and CA verification.
In-cluster config reads the standard service-account mount.
"""

from __future__ import annotations

import base64
import os
import tempfile
from dataclasses import dataclass
from typing import Optional

import yaml

SERVICE_ACCOUNT_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


@dataclass
class RestConfig:
    host: str
    token: Optional[str] = None
    client_cert: Optional[str] = None  # file path
    client_key: Optional[str] = None  # file path
    ca_cert: Optional[str] = None  # file path; None = system trust
    insecure_skip_tls_verify: bool = False

    @property
    def verify(self):
        if self.insecure_skip_tls_verify:
            return False
        return self.ca_cert if self.ca_cert else True

    @property
    def cert(self):
        if self.client_cert and self.client_key:
            return (self.client_cert, self.client_key)
        return None


def _materialize(data_b64: Optional[str], path: Optional[str], suffix: str) -> Optional[str]:
    """Inline ``*-data`` wins over file paths, matching client-go."""
    if data_b64:
        f = tempfile.NamedTemporaryFile(
            prefix="agac-kubeconfig-", suffix=suffix, delete=False
        )
        f.write(base64.b64decode(data_b64))
        f.close()
        return f.name
    return path


def load_kubeconfig(path: str, context: Optional[str] = None) -> RestConfig:
    with open(path) as f:
        config = yaml.safe_load(f) or {}

    context_name = context or config.get("current-context")
    if not context_name:
        raise ValueError(f"{path}: no current-context and no context given")
    contexts = {c["name"]: c["context"] for c in config.get("contexts", [])}
    if context_name not in contexts:
        raise ValueError(f"{path}: context {context_name!r} not found")
    ctx = contexts[context_name]

    clusters = {c["name"]: c["cluster"] for c in config.get("clusters", [])}
    users = {u["name"]: u.get("user", {}) for u in config.get("users", [])}
    cluster = clusters.get(ctx.get("cluster"))
    if cluster is None:
        raise ValueError(f"{path}: cluster {ctx.get('cluster')!r} not found")
    user = users.get(ctx.get("user"), {})

    token = user.get("token")
    token_file = user.get("tokenFile")
    if token is None and token_file:
        with open(token_file) as f:
            token = f.read().strip()

    return RestConfig(
        host=cluster["server"].rstrip("/"),
        token=token,
        client_cert=_materialize(
            user.get("client-certificate-data"), user.get("client-certificate"), ".crt"
        ),
        client_key=_materialize(
            user.get("client-key-data"), user.get("client-key"), ".key"
        ),
        ca_cert=_materialize(
            cluster.get("certificate-authority-data"),
            cluster.get("certificate-authority"),
            ".ca.crt",
        ),
        insecure_skip_tls_verify=bool(cluster.get("insecure-skip-tls-verify", False)),
    )


def in_cluster_config() -> RestConfig:
    """rest.InClusterConfig: service-account token + CA + KUBERNETES_SERVICE env."""
    host = os.environ.get("KUBERNETES_SERVICE_HOST")
    port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
    token_path = os.path.join(SERVICE_ACCOUNT_DIR, "token")
    ca_path = os.path.join(SERVICE_ACCOUNT_DIR, "ca.crt")
    if not host or not os.path.exists(token_path):
        raise RuntimeError("not running in-cluster (no service account mount)")
    with open(token_path) as f:
        token = f.read().strip()
    return RestConfig(
        host=f"https://{host}:{port}",
        token=token,
        ca_cert=ca_path if os.path.exists(ca_path) else None,
    )


def build_config(master_url: str = "", kubeconfig: str = "") -> RestConfig:
    """clientcmd.BuildConfigFromFlags: kubeconfig file if given (master URL
    overrides its server), else in-cluster."""
    if kubeconfig:
        config = load_kubeconfig(kubeconfig)
        if master_url:
            config.host = master_url.rstrip("/")
        return config
    if master_url:
        return RestConfig(host=master_url.rstrip("/"))
    return in_cluster_config()
