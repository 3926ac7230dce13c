"""TLS certificate management (reference ``cdn-proto/src/crypto/tls.rs`` +
``build.rs``): per-boot leaf certificates generated from a CA, a generated
local testing CA (the reference pins one at build time via build.rs:40-57),
and root stores.  SAN/SNI is the fixed name "espresso" for wire parity with
the reference (tls.rs:63-71, tcp_tls.rs:91-95).

Certificates are produced with the system ``openssl`` CLI (the reference's
scripts/gen-ca.bash does the same); the ssl stdlib handles the handshake.
"""

from __future__ import annotations

import ssl
import subprocess
import tempfile
from pathlib import Path
from typing import Optional, Tuple

CERT_NAME = "espresso"  # fixed SAN/SNI (reference tls.rs:63-71)


def _run(args, **kw):
    subprocess.run(args, check=True, capture_output=True, **kw)


def generate_ca(dir_path: str) -> Tuple[str, str]:
    """Create a root CA (cert, key) under dir_path; returns the paths
    (the reference's scripts/gen-ca.bash equivalent)."""
    d = Path(dir_path)
    d.mkdir(parents=True, exist_ok=True)
    ca_key = str(d / "ca.key")
    ca_cert = str(d / "ca.crt")
    _run(["openssl", "ecparam", "-genkey", "-name", "prime256v1", "-out", ca_key])
    _run([
        "openssl", "req", "-x509", "-new", "-key", ca_key, "-days", "3650",
        "-subj", "/CN=pushcdn-local-ca", "-out", ca_cert,
    ])
    return ca_cert, ca_key


def generate_cert_from_ca(
    ca_cert_path: str, ca_key_path: str, out_dir: str
) -> Tuple[str, str]:
    """Per-boot leaf cert signed by the CA, SAN=espresso
    (reference tls.rs:52-93). Returns (cert_path, key_path)."""
    d = Path(out_dir)
    d.mkdir(parents=True, exist_ok=True)
    key = str(d / "leaf.key")
    csr = str(d / "leaf.csr")
    cert = str(d / "leaf.crt")
    ext = str(d / "leaf.ext")
    _run(["openssl", "ecparam", "-genkey", "-name", "prime256v1", "-out", key])
    _run(["openssl", "req", "-new", "-key", key, "-subj", f"/CN={CERT_NAME}", "-out", csr])
    Path(ext).write_text(f"subjectAltName=DNS:{CERT_NAME}\n")
    _run([
        "openssl", "x509", "-req", "-in", csr, "-CA", ca_cert_path, "-CAkey", ca_key_path,
        "-CAcreateserial", "-days", "30", "-extfile", ext, "-out", cert,
    ])
    return cert, key


_LOCAL_CA: Optional[Tuple[str, str]] = None


def local_ca() -> Tuple[str, str]:
    """Process-wide local testing CA (the reference bakes one in at build
    time; we generate once per process — same trust model for local runs)."""
    global _LOCAL_CA
    if _LOCAL_CA is None:
        d = tempfile.mkdtemp(prefix="pushcdn-ca-")
        _LOCAL_CA = generate_ca(d)
    return _LOCAL_CA


def load_ca(ca_cert_path: Optional[str], ca_key_path: Optional[str]) -> Tuple[str, str]:
    """Use provided CA paths, falling back to the local testing CA
    (reference tls.rs:100-126)."""
    if ca_cert_path and ca_key_path:
        return ca_cert_path, ca_key_path
    return local_ca()


def server_context(ca_cert_path: Optional[str], ca_key_path: Optional[str]) -> ssl.SSLContext:
    ca_cert, ca_key = load_ca(ca_cert_path, ca_key_path)
    leaf_dir = tempfile.mkdtemp(prefix="pushcdn-leaf-")
    cert, key = generate_cert_from_ca(ca_cert, ca_key, leaf_dir)
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
    ctx.load_cert_chain(cert, key)
    return ctx


def client_context(use_local_authority: bool, ca_cert_path: Optional[str] = None) -> ssl.SSLContext:
    ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_CLIENT)
    ctx.check_hostname = True
    if use_local_authority:
        ctx.load_verify_locations(local_ca()[0])
    elif ca_cert_path:
        ctx.load_verify_locations(ca_cert_path)
    else:
        ctx.load_default_certs()
    return ctx
