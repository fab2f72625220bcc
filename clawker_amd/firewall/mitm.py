"""TLS MITM material for path-scoped HTTPS rules.

Reference: controlplane/firewall/certs.go — ECDSA P-256 CA (EnsureCA),
per-domain MITM leaf certs for Envoy's MITM filter chains, rotate. Here
the proxy is the gateway (firewall/gateway.py); this module mints the CA
and per-domain leaves with the openssl CLI (the python `cryptography`
package is not in this image) and caches them on disk.

Sandboxes trust the CA through a combined bundle the engine stages into
each rundir (SSL_CERT_FILE / CURL_CA_BUNDLE / NODE_EXTRA_CA_CERTS).
"""
from __future__ import annotations

import os
import subprocess
import threading
from pathlib import Path

from .. import consts
from ..errors import ClawkerError
from ..logger import get as get_logger

log = get_logger("mitm")

_lock = threading.RLock()


def mitm_dir() -> Path:
    d = consts.config_dir() / "mitm"
    d.mkdir(parents=True, exist_ok=True)
    return d


def _openssl(*args: str, input_bytes: bytes = b"") -> bytes:
    r = subprocess.run(["openssl", *args], input=input_bytes,
                       capture_output=True, timeout=30)
    if r.returncode != 0:
        raise ClawkerError(f"openssl {' '.join(args[:2])}: "
                           f"{r.stderr.decode(errors='replace')[-300:]}")
    return r.stdout


def ensure_ca() -> tuple[Path, Path]:
    """Idempotent MITM CA (cert, key)."""
    d = mitm_dir()
    crt, key = d / "ca.crt", d / "ca.key"
    with _lock:
        if crt.is_file() and key.is_file():
            return crt, key
        _openssl("req", "-x509", "-newkey", "ec", "-pkeyopt",
                 "ec_paramgen_curve:P-256", "-keyout", str(key), "-out",
                 str(crt), "-nodes", "-subj", "/CN=clawker MITM CA",
                 "-days", "3650")
        key.chmod(0o600)
        log.info("mitm_ca_created")
    return crt, key


def rotate_ca() -> None:
    """Reference: FirewallRotateCA — new CA; cached leaves dropped."""
    d = mitm_dir()
    for p in d.glob("*"):
        p.unlink()
    ensure_ca()


def leaf_for(domain: str) -> tuple[Path, Path]:
    """Per-domain leaf signed by the CA (cached). Returns (cert, key)."""
    d = mitm_dir()
    safe = domain.replace("*", "_wild_").replace("/", "_")
    crt, key = d / f"{safe}.crt", d / f"{safe}.key"
    import fcntl
    with _lock, open(d / ".mint.lock", "w") as lockf:
        # cross-PROCESS serialization too: cpd and a CLI minting the
        # same domain concurrently would race openssl's file writes
        fcntl.flock(lockf, fcntl.LOCK_EX)
        if crt.is_file() and key.is_file():
            return crt, key
        ca_crt, ca_key = ensure_ca()
        _openssl("req", "-newkey", "ec", "-pkeyopt",
                 "ec_paramgen_curve:P-256", "-keyout", str(key), "-out",
                 str(d / f"{safe}.csr"), "-nodes", "-subj", f"/CN={domain}")
        ext = d / f"{safe}.ext"
        san = domain if not domain.startswith("*.") else domain
        ext.write_text(f"subjectAltName=DNS:{san}\n"
                       "basicConstraints=CA:FALSE\n"
                       "keyUsage=digitalSignature,keyEncipherment\n"
                       "extendedKeyUsage=serverAuth\n")
        _openssl("x509", "-req", "-in", str(d / f"{safe}.csr"), "-CA",
                 str(ca_crt), "-CAkey", str(ca_key), "-CAcreateserial",
                 "-out", str(crt), "-days", "825", "-extfile", str(ext))
        key.chmod(0o600)
        log.info("mitm_leaf_minted", domain=domain)
    return crt, key


def combined_trust_bundle() -> Path:
    """System CA bundle + the MITM CA, for in-sandbox TLS clients."""
    d = mitm_dir()
    bundle = d / "trust-bundle.crt"
    ca_crt, _ = ensure_ca()
    system = Path("/etc/ssl/certs/ca-certificates.crt")
    with _lock:
        parts = []
        if system.is_file():
            parts.append(system.read_text())
        parts.append(ca_crt.read_text())
        content = "\n".join(parts)
        # atomic replace: concurrent readers (TLS clients loading the
        # bundle mid-rewrite) must never see a torn file
        try:
            if bundle.read_text() == content:
                return bundle
        except OSError:
            pass
        tmp = bundle.with_suffix(".tmp")
        tmp.write_text(content)
        os.chmod(tmp, 0o644)
        tmp.replace(bundle)
    return bundle
