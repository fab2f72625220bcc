"""Agent identity + trust material.

Reference: internal/auth + controlplane/auth — the CLI is the root of
trust (EnsureAuthMaterial mints a CA + keys at `project init`;
MintAgentCert issues per-agent leaf certs; clawkerd presents a single-use
assertion JWT at register — agent_bootstrap.go:79).

Single-node redesign: all control links are root-owned Unix sockets, so
X.509 is replaced by an HMAC scheme with the same roles:
  * a root key (EnsureAuthMaterial) in the config dir, 0600
  * a per-agent bootstrap token minted at sandbox create and installed at
    <rundir>/bootstrap/token (visible in-sandbox at
    /run/clawker/bootstrap/token) — the agent's identity proof towards
    host services (hostproxy credential routes verify it)
  * verification derives the expected token from (root key, sandbox name)
    — stateless, like cert verification against the CA
"""
from __future__ import annotations

import hashlib
import hmac
import os
import secrets
from pathlib import Path

from . import consts
from .logger import get as get_logger

log = get_logger("auth")

ROOT_KEY_NAME = "auth-root.key"


def ensure_auth_material() -> bytes:
    """Idempotent root-key creation (reference: auth_material.go:64)."""
    path = consts.config_dir() / ROOT_KEY_NAME
    if path.is_file():
        return bytes.fromhex(path.read_text().strip())
    path.parent.mkdir(parents=True, exist_ok=True)
    key = secrets.token_bytes(32)
    try:
        fd = os.open(path, os.O_WRONLY | os.O_CREAT | os.O_EXCL, 0o600)
    except FileExistsError:
        # concurrent creator won the O_EXCL race: use ITS key (two
        # divergent root keys would invalidate half the fleet's tokens)
        return bytes.fromhex(path.read_text().strip())
    try:
        os.write(fd, key.hex().encode())
    finally:
        os.close(fd)
    log.info("auth_material_created")
    return key


def rotate_auth_material() -> bytes:
    """New root key; existing agent tokens become invalid (reference:
    RotateAuthMaterial / FirewallRotateCA semantics)."""
    path = consts.config_dir() / ROOT_KEY_NAME
    path.unlink(missing_ok=True)
    return ensure_auth_material()


def mint_agent_token(sandbox_name: str) -> str:
    key = ensure_auth_material()
    mac = hmac.new(key, f"agent:{sandbox_name}".encode(), hashlib.sha256)
    return f"{sandbox_name}:{mac.hexdigest()}"


def verify_agent_token(token: str) -> str | None:
    """Returns the sandbox name if valid, else None. Constant-time."""
    name, _, mac_hex = token.rpartition(":")
    if not name or not mac_hex:
        return None
    expected = mint_agent_token(name).rpartition(":")[2]
    if hmac.compare_digest(expected, mac_hex):
        return name
    return None


def install_bootstrap(rundir: Path, sandbox_name: str) -> None:
    """Write bootstrap material into the sandbox rundir (reference:
    InstallAgentBootstrapMaterial tar-streaming cert+assertion into the
    container's writable layer, agent_bootstrap.go:209)."""
    bdir = rundir / "bootstrap"
    bdir.mkdir(parents=True, exist_ok=True)
    os.chmod(bdir, 0o755)
    token_path = bdir / "token"
    token_path.write_text(mint_agent_token(sandbox_name) + "\n")
    os.chmod(token_path, 0o644)   # in-sandbox agent user must read it
