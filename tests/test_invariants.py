"""Invariant pinning tests (reference: persisted_values_test.go, ABI size
asserts mirrored Go-side, Dockerfile layer-ordering pin, import-boundary
tests — SURVEY.md §4.7)."""
import re
import struct
import subprocess
from pathlib import Path

from clawker_amd import consts

REPO = Path(__file__).resolve().parent.parent


def test_persisted_values_pinned():
    """These values live in user state on disk / in sandbox images; moving
    them is a migration, not an edit."""
    assert consts.SANDBOX_NAME_PREFIX == "clawker."
    assert consts.MANAGED_LABEL == "dev.clawker.managed"
    assert consts.PROJECT_FILE_NAME == ".clawker.yaml"
    assert consts.PROJECT_DIR_NAME == ".clawker"
    assert consts.SETTINGS_BASENAME == "settings.yaml"
    assert consts.REGISTRY_BASENAME == "registry.yaml"
    assert consts.EGRESS_RULES_BASENAME == "egress-rules.yaml"
    assert consts.ROUTE_IDENTITIES_BASENAME == "route-identities.yaml"
    assert consts.CKD_SOCK_NAME == "ctl.sock"
    assert consts.HOSTPROXY_PORT == 18374
    assert consts.MI355X_HBM_BYTES == 288 * 2**30


def test_wire_format_pinned():
    """Frame format is shared with native/common/util.hpp — 4-byte BE
    length + JSON. A change here must change both sides."""
    import io
    import socket

    from clawker_amd.engine import wire
    a, b = socket.socketpair()
    wire.send_frame(a, {"t": "x"})
    raw = b.recv(100)
    assert raw[:4] == struct.pack(">I", len(raw) - 4)
    assert raw[4:] == b'{"t": "x"}'.replace(b": ", b":")
    a.close()
    b.close()


def test_ckd_wire_interop():
    """The C++ side parses/emits the same frames (golden handshake against
    a real ckd is covered by engine tests; here: pin the spec JSON keys the
    native side reads — parse_spec in ckrt.cpp + main() in ckd.cpp)."""
    from clawker_amd.engine.spec import SandboxSpec
    import json
    spec = json.loads(SandboxSpec(name="clawker.x.y", rundir="/r",
                                  lowerdirs=["/"], upper="/u", work="/w",
                                  merged="/m").to_json())
    for key in ("name", "backend", "rundir", "rootfs", "hostname", "netns",
                "tty", "autostart", "mounts", "devices", "cgroup", "env",
                "user", "workdir", "cmd", "paths", "services"):
        assert key in spec, f"spec key {key} missing (native side reads it)"
    for key in ("lowerdirs", "upper", "work", "merged"):
        assert key in spec["rootfs"]
    for key in ("mem_bytes", "pids", "device_allow_only"):
        assert key in spec["cgroup"]


def test_import_boundaries():
    """Only iostreams/tui/monitor-render import rich (reference:
    tui/import_boundary_test.go — only iostreams imports lipgloss)."""
    allowed = {"iostreams.py", "dashboard.py", "components.py", "stats.py"}
    offenders = []
    for p in (REPO / "clawker_amd").rglob("*.py"):
        if p.name in allowed or "/cli/" in str(p):
            continue   # cli renders tables directly (presentation layer)
        text = p.read_text()
        if re.search(r"^\s*(from rich|import rich)", text, re.M):
            offenders.append(str(p))
    assert offenders == [], offenders


def test_no_docker_no_cuda_anywhere():
    """This is an MI355X-native framework: no docker SDK calls, no CUDA,
    no nvidia toolkit, no hipify shims (BASELINE.json north star)."""
    banned = [r"\bimport docker\b", r"nvidia-container", r"\bcuda(rt|_runtime)\b",
              r"hipify", r"__HIP_PLATFORM_NVIDIA__"]
    offenders = []
    for p in list((REPO / "clawker_amd").rglob("*.py")) + list((REPO / "native").rglob("*.[ch]pp")):
        if p.name == "__init__.py":
            continue   # the package docstring names what we DON'T use
        text = p.read_text(errors="replace")
        for pat in banned:
            if re.search(pat, text):
                offenders.append(f"{p}: {pat}")
    assert offenders == [], offenders


def test_native_sources_compile_warning_clean():
    """Keep the native build free of -Wall -Wextra errors (and catch any
    bitrot in the Makefile)."""
    r = subprocess.run(["make", "-n", "native"], cwd=REPO,
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stderr


def test_admin_ops_client_daemon_coverage():
    """Every op the CPClient can send is handled by cpd, and cpd's
    handled set is what the client + CLI rely on (reference:
    TestAdminMethodScopes_CoversAllRPCs — no RPC falls through)."""
    daemon_src = (REPO / "clawker_amd/controlplane/daemon.py").read_text()
    handled = set(re.findall(r'op == "([a-z_]+)"', daemon_src)) | \
        set(re.findall(r'req\.get\("op"\) == "([a-z_]+)"', daemon_src))
    client_src = (REPO / "clawker_amd/controlplane/client.py").read_text()
    sent = set(re.findall(r'"op": "([a-z_]+)"', client_src))
    missing = sent - handled
    assert missing == set(), f"client sends unhandled ops: {missing}"
    # the firewall attach path used by the orchestrator is present
    for required in ("fw_attach", "fw_detach", "fw_add_rules", "fw_bootstrap",
                     "fw_status",
                     "events", "events_follow", "bypass", "reload_policy",
                     "agents", "status", "ping", "shutdown"):
        assert required in handled, f"cpd lost op {required}"
