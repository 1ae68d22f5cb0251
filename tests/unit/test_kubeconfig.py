"""Cluster config resolution tests (GetConfigOrDie equivalent)."""
import base64
from pathlib import Path

import pytest
import yaml

from active_monitor_amd.kube.config import (
    ClusterConfig,
    ConfigError,
    get_config,
    load_kubeconfig,
)


def write_kubeconfig(tmp_path, user):
    doc = {
        "apiVersion": "v1",
        "kind": "Config",
        "current-context": "dev",
        "contexts": [{"name": "dev", "context": {"cluster": "c1", "user": "u1"}}],
        "clusters": [{"name": "c1", "cluster": {
            "server": "https://api.example:6443",
            "certificate-authority-data": base64.b64encode(b"CA PEM").decode(),
        }}],
        "users": [{"name": "u1", "user": user}],
    }
    p = tmp_path / "config"
    p.write_text(yaml.safe_dump(doc))
    return p


def test_token_auth(tmp_path):
    p = write_kubeconfig(tmp_path, {"token": "sekret"})
    cfg = load_kubeconfig(str(p))
    assert cfg.server == "https://api.example:6443"
    assert cfg.token == "sekret"
    assert cfg.verify is True
    assert Path(cfg.ca_cert_path).read_bytes() == b"CA PEM"


def test_client_cert_auth(tmp_path):
    p = write_kubeconfig(tmp_path, {
        "client-certificate-data": base64.b64encode(b"CERT").decode(),
        "client-key-data": base64.b64encode(b"KEY").decode(),
    })
    cfg = load_kubeconfig(str(p))
    assert Path(cfg.client_cert_path).read_bytes() == b"CERT"
    assert Path(cfg.client_key_path).read_bytes() == b"KEY"
    client = cfg.make_client()
    assert client.client_cert[0] == cfg.client_cert_path


def test_missing_context_errors(tmp_path):
    p = write_kubeconfig(tmp_path, {"token": "x"})
    with pytest.raises(ConfigError, match="context 'nope' not found"):
        load_kubeconfig(str(p), context="nope")


def test_missing_file_errors(tmp_path):
    with pytest.raises(ConfigError, match="cannot read kubeconfig"):
        load_kubeconfig(str(tmp_path / "absent"))


def test_explicit_server_wins(tmp_path, monkeypatch):
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    cfg = get_config(server="http://127.0.0.1:8001", token="t", insecure=True)
    assert cfg == ClusterConfig(server="http://127.0.0.1:8001", token="t", verify=False)


def test_kubeconfig_fallback(tmp_path, monkeypatch):
    monkeypatch.delenv("KUBERNETES_SERVICE_HOST", raising=False)
    p = write_kubeconfig(tmp_path, {"token": "from-file"})
    monkeypatch.setenv("KUBECONFIG", str(p))
    cfg = get_config()
    assert cfg.token == "from-file"
