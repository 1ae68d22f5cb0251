"""Artifact store tests (reference: internal/store/store_test.go)."""
import http.server
import ssl
import threading

import pytest

from active_monitor_amd.api import ArtifactLocation, FileArtifact, URLArtifact
from active_monitor_amd.store import (
    FileReader,
    InlineReader,
    URLReader,
    get_artifact_reader,
)
from active_monitor_amd.store.artifacts import ArtifactReadError


def test_inline_reader_returns_content():
    r = get_artifact_reader(ArtifactLocation(inline="hello: world"))
    assert isinstance(r, InlineReader)
    assert r.read() == b"hello: world"


def test_inline_reader_empty_errors():
    with pytest.raises(ArtifactReadError, match="InlineArtifact does not exist"):
        InlineReader("")
    with pytest.raises(ArtifactReadError, match="InlineArtifact does not exist"):
        InlineReader(None)


def test_unknown_location_errors():
    # reference: store.go:21 ("unknown artifact location")
    with pytest.raises(ArtifactReadError, match="unknown artifact location"):
        get_artifact_reader(ArtifactLocation())
    with pytest.raises(ArtifactReadError, match="unknown artifact location"):
        get_artifact_reader(None)


def test_file_source_unknown_in_strict_mode():
    # reference parity: File is declared but unimplemented (store.go:15-22)
    loc = ArtifactLocation(file=FileArtifact(path="/tmp/x.yaml"))
    with pytest.raises(ArtifactReadError, match="unknown artifact location"):
        get_artifact_reader(loc, allow_file=False)


def test_file_reader_reads(tmp_path):
    p = tmp_path / "wf.yaml"
    p.write_text("kind: Workflow\n")
    r = get_artifact_reader(ArtifactLocation(file=FileArtifact(path=str(p))))
    assert isinstance(r, FileReader)
    assert r.read() == b"kind: Workflow\n"


def test_file_reader_missing_path_errors():
    with pytest.raises(ArtifactReadError):
        FileReader("")
    with pytest.raises(ArtifactReadError):
        FileReader("/nonexistent/path.yaml").read()


class _Handler(http.server.BaseHTTPRequestHandler):
    payload = b"apiVersion: argoproj.io/v1alpha1\nkind: Workflow\n"

    def do_GET(self):
        if self.path == "/ok":
            self.send_response(200)
            self.send_header("Content-Type", "text/yaml")
            self.end_headers()
            self.wfile.write(self.payload)
        else:
            self.send_response(404)
            self.end_headers()

    def log_message(self, *a):
        pass


@pytest.fixture
def http_server():
    srv = http.server.HTTPServer(("127.0.0.1", 0), _Handler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{srv.server_address[1]}"
    srv.shutdown()


def test_url_reader_ok(http_server):
    r = URLReader(URLArtifact(path=http_server + "/ok"))
    assert r.read() == _Handler.payload


def test_url_reader_non_200_status(http_server):
    # reference: url.go:46-49 ("status code <n>")
    with pytest.raises(ArtifactReadError, match="status code 404"):
        URLReader(URLArtifact(path=http_server + "/missing")).read()


def test_url_reader_nil_artifact_errors():
    with pytest.raises(ArtifactReadError, match="URLArtifact cannot be empty"):
        URLReader(None)


def test_url_reader_tls_verify_matrix(tmp_path):
    """TLS verification on by default; verifyCert=false skips it
    (reference: url.go:30-38, store_test.go:100-183)."""
    import datetime
    cryptography = pytest.importorskip("cryptography", reason="no self-signed cert tooling")
    from cryptography import x509
    from cryptography.hazmat.primitives import hashes, serialization
    from cryptography.hazmat.primitives.asymmetric import rsa
    from cryptography.x509.oid import NameOID

    key = rsa.generate_private_key(public_exponent=65537, key_size=2048)
    name = x509.Name([x509.NameAttribute(NameOID.COMMON_NAME, "127.0.0.1")])
    now = datetime.datetime.utcnow()
    cert = (
        x509.CertificateBuilder()
        .subject_name(name).issuer_name(name).public_key(key.public_key())
        .serial_number(x509.random_serial_number())
        .not_valid_before(now).not_valid_after(now + datetime.timedelta(days=1))
        .add_extension(
            x509.SubjectAlternativeName([x509.IPAddress(__import__("ipaddress").ip_address("127.0.0.1"))]),
            critical=False,
        )
        .sign(key, hashes.SHA256())
    )
    certfile = tmp_path / "cert.pem"
    keyfile = tmp_path / "key.pem"
    certfile.write_bytes(cert.public_bytes(serialization.Encoding.PEM))
    keyfile.write_bytes(
        key.private_bytes(
            serialization.Encoding.PEM,
            serialization.PrivateFormat.TraditionalOpenSSL,
            serialization.NoEncryption(),
        )
    )

    srv = http.server.HTTPServer(("127.0.0.1", 0), _Handler)
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(str(certfile), str(keyfile))
    srv.socket = ctx.wrap_socket(srv.socket, server_side=True)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        url = f"https://127.0.0.1:{srv.server_address[1]}/ok"
        # default: verification on → self-signed cert rejected
        with pytest.raises(ArtifactReadError):
            URLReader(URLArtifact(path=url)).read()
        # verifyCert=false → succeeds
        assert URLReader(URLArtifact(path=url, verify_cert=False)).read() == _Handler.payload
    finally:
        srv.shutdown()
