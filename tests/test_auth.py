"""Agent identity material tests (reference: internal/auth cert minting +
assertion flow, adapted to the HMAC scheme)."""
import json

from clawker_amd import auth


def test_root_key_idempotent_and_private(isolated_env):
    k1 = auth.ensure_auth_material()
    k2 = auth.ensure_auth_material()
    assert k1 == k2 and len(k1) == 32
    from clawker_amd import consts
    path = consts.config_dir() / auth.ROOT_KEY_NAME
    assert (path.stat().st_mode & 0o777) == 0o600


def test_token_mint_verify_and_rotate(isolated_env):
    t = auth.mint_agent_token("clawker.p.a")
    assert auth.verify_agent_token(t) == "clawker.p.a"
    # tampered token rejected
    assert auth.verify_agent_token(t[:-2] + "00") is None
    assert auth.verify_agent_token("garbage") is None
    assert auth.verify_agent_token("") is None
    # name swap rejected (token bound to identity)
    other = auth.mint_agent_token("clawker.p.b")
    head = t.rpartition(":")[0]
    mac_b = other.rpartition(":")[2]
    assert auth.verify_agent_token(f"{head}:{mac_b}") is None
    # rotation invalidates old tokens
    auth.rotate_auth_material()
    assert auth.verify_agent_token(t) is None


def test_bootstrap_installed_at_create(isolated_env, tmp_path):
    from clawker_amd.engine import Engine, SandboxSpec
    eng = Engine()
    try:
        info = eng.create(SandboxSpec(name="clawker.authtest.a", cmd=["true"]))
        token = (info.rundir / "bootstrap" / "token").read_text().strip()
        assert auth.verify_agent_token(token) == "clawker.authtest.a"
    finally:
        for i in eng.list():
            eng.remove(i.name, force=True)
        eng.close()


def test_hostproxy_git_credential_requires_token(isolated_env):
    from clawker_amd.hostproxy import HostProxyManager
    mgr = HostProxyManager()
    mgr.ensure_running()
    try:
        status, body = mgr.request("POST", "/git/credential",
                                   b"protocol=https\nhost=github.com\n\n")
        assert status == 403
        assert b"invalid agent token" in body
    finally:
        mgr.stop()
