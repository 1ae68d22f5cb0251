"""Cluster connection config: kubeconfig + in-cluster service account.

The role of the reference's ``ctrl.GetConfigOrDie()`` (cmd/main.go:70): find
apiserver coordinates and credentials from, in order,

1. an explicit server URL (flags),
2. the in-cluster service-account mount
   (``/var/run/secrets/kubernetes.io/serviceaccount``),
3. ``$KUBECONFIG`` / ``~/.kube/config`` (current-context; bearer token or
   client-certificate auth, cluster CA, insecure-skip-tls-verify).
"""
from __future__ import annotations

import base64
import os
import tempfile
from dataclasses import dataclass
from pathlib import Path
from typing import Optional

import yaml

SA_DIR = Path("/var/run/secrets/kubernetes.io/serviceaccount")


class ConfigError(Exception):
    pass


@dataclass
class ClusterConfig:
    server: str
    token: Optional[str] = None
    ca_cert_path: Optional[str] = None
    client_cert_path: Optional[str] = None
    client_key_path: Optional[str] = None
    verify: bool = True

    def make_client(self):
        from .http import HttpClient

        client = HttpClient(
            self.server,
            token=self.token,
            verify=self.verify,
            ca_cert=self.ca_cert_path,
        )
        client.client_cert = (self.client_cert_path, self.client_key_path)
        return client


def in_cluster_config() -> Optional[ClusterConfig]:
    host = os.environ.get("KUBERNETES_SERVICE_HOST")
    token_file = SA_DIR / "token"
    if not host or not token_file.exists():
        return None
    port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
    ca = SA_DIR / "ca.crt"
    return ClusterConfig(
        server=f"https://{host}:{port}",
        token=token_file.read_text().strip(),
        ca_cert_path=str(ca) if ca.exists() else None,
    )


def _materialize(data_b64: Optional[str], path: Optional[str], suffix: str) -> Optional[str]:
    """kubeconfig allows inline base64 ``*-data`` or file paths."""
    if path:
        return path
    if not data_b64:
        return None
    f = tempfile.NamedTemporaryFile(
        mode="wb", suffix=suffix, prefix="amkube-", delete=False
    )
    f.write(base64.b64decode(data_b64))
    f.close()
    return f.name


def load_kubeconfig(path: Optional[str] = None, context: Optional[str] = None) -> ClusterConfig:
    cfg_path = path or os.environ.get("KUBECONFIG") or os.path.expanduser("~/.kube/config")
    try:
        doc = yaml.safe_load(Path(cfg_path).read_text())
    except OSError as e:
        raise ConfigError(f"cannot read kubeconfig {cfg_path}: {e}") from e
    if not isinstance(doc, dict):
        raise ConfigError(f"invalid kubeconfig {cfg_path}")

    ctx_name = context or doc.get("current-context")
    contexts = {c.get("name"): c.get("context", {}) for c in doc.get("contexts", []) or []}
    if ctx_name not in contexts:
        raise ConfigError(f"context {ctx_name!r} not found in {cfg_path}")
    ctx = contexts[ctx_name]

    clusters = {c.get("name"): c.get("cluster", {}) for c in doc.get("clusters", []) or []}
    users = {u.get("name"): u.get("user", {}) for u in doc.get("users", []) or []}
    cluster = clusters.get(ctx.get("cluster"))
    if cluster is None:
        raise ConfigError(f"cluster {ctx.get('cluster')!r} not found in {cfg_path}")
    user = users.get(ctx.get("user"), {})

    server = cluster.get("server", "")
    if not server:
        raise ConfigError(f"cluster {ctx.get('cluster')!r} has no server")

    return ClusterConfig(
        server=server,
        token=user.get("token"),
        ca_cert_path=_materialize(
            cluster.get("certificate-authority-data"),
            cluster.get("certificate-authority"),
            ".crt",
        ),
        client_cert_path=_materialize(
            user.get("client-certificate-data"), user.get("client-certificate"), ".crt"
        ),
        client_key_path=_materialize(
            user.get("client-key-data"), user.get("client-key"), ".key"
        ),
        verify=not cluster.get("insecure-skip-tls-verify", False),
    )


def get_config(server: str = "", token: str = "", insecure: bool = False,
               kubeconfig: Optional[str] = None) -> ClusterConfig:
    """GetConfigOrDie-style resolution (explicit > in-cluster > kubeconfig)."""
    if server:
        return ClusterConfig(server=server, token=token or None, verify=not insecure)
    in_cluster = in_cluster_config()
    if in_cluster is not None:
        return in_cluster
    return load_kubeconfig(kubeconfig)
