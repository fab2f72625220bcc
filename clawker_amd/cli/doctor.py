"""clawker doctor — host capability diagnosis (the SOS/assist analog:
the reference ships WatchSOS + CLI-assisted recovery for boot failures;
on a single node the common failure is missing host capabilities, so the
assist is a diagnosis command)."""
from __future__ import annotations

import os
import shutil
import subprocess
from pathlib import Path

import click

from .root import Ctx, cli, pass_factory


def _check(name, ok, detail=""):
    return {"check": name, "ok": bool(ok), "detail": detail}


def run_checks() -> list[dict]:
    checks = []
    checks.append(_check("root", os.geteuid() == 0,
                         "sandboxes need root (rootful node design)"))
    r = subprocess.run(["unshare", "-pmf", "true"], capture_output=True)
    checks.append(_check(
        "namespaces", r.returncode == 0,
        "full ns backend available" if r.returncode == 0 else
        f"proc backend fallback ({r.stderr.decode().strip()[:80]})"))
    overlay_ok = False
    if r.returncode == 0:
        import tempfile
        with tempfile.TemporaryDirectory() as td:
            for d in ("l", "u", "w", "m"):
                os.mkdir(f"{td}/{d}")
            m = subprocess.run(
                ["mount", "-t", "overlay", "overlay", "-o",
                 f"lowerdir={td}/l,upperdir={td}/u,workdir={td}/w", f"{td}/m"],
                capture_output=True)
            overlay_ok = m.returncode == 0
            if overlay_ok:
                subprocess.run(["umount", f"{td}/m"], capture_output=True)
    checks.append(_check("overlayfs", overlay_ok,
                         "" if overlay_ok else "image builds unavailable"))
    cg = Path("/sys/fs/cgroup")
    v2 = (cg / "cgroup.controllers").exists()
    v1 = (cg / "memory").exists()
    checks.append(_check("cgroups", v1 or v2, "v2 unified" if v2 else
                         "v1 hybrid" if v1 else "no limits available"))
    # kernel-side device enforcement: v1 devices controller, or the BPF
    # device program on a cgroup2 hierarchy (probe does a real
    # load+attach+deny check — native/tests/devbpf_probe.cpp)
    from ..engine.engine import native_bin_dir
    dev_v1 = (cg / "devices").exists()
    cg2_root = (cg if v2 else cg / "unified"
                if (cg / "unified" / "cgroup.controllers").exists() else None)
    bpf_detail = ""
    bpf_ok = False
    if dev_v1:
        bpf_ok, bpf_detail = True, "v1 devices controller"
    elif cg2_root is not None:
        import subprocess as _sp
        probe = native_bin_dir() / "devbpf_probe"
        scratch = cg2_root / "clawker-doctor"
        try:
            scratch.mkdir(exist_ok=True)
            r = _sp.run([str(probe), str(cg2_root)], capture_output=True,
                        text=True, timeout=15)
            bpf_ok = r.returncode == 0
            bpf_detail = ("BPF device program enforced" if bpf_ok else
                          "bpf() unavailable: /dev construction only")
        except (OSError, _sp.TimeoutExpired) as e:
            bpf_detail = f"probe failed: {e}"
        finally:
            try:
                scratch.rmdir()
            except OSError:
                pass
    checks.append(_check("device enforcement", bpf_ok, bpf_detail))
    from ..engine.engine import native_bin_dir
    bins = {b: (native_bin_dir() / b).is_file() for b in ("ckrt", "ckd", "ckgw")}
    checks.append(_check("native runtime", all(bins.values()),
                         ", ".join(f"{k}={'ok' if v else 'MISSING'}"
                                   for k, v in bins.items())))
    try:
        from .. import _native  # noqa: F401
        checks.append(_check("_native extension", True))
    except ImportError:
        checks.append(_check("_native extension", False, "run `make pymod`"))
    checks.append(_check("openssl", shutil.which("openssl") is not None,
                         "needed for TLS MITM path rules"))
    checks.append(_check("git", shutil.which("git") is not None,
                         "needed for worktree fan-out"))
    kfd = Path("/dev/kfd").exists()
    dri = sorted(Path("/dev/dri").glob("renderD*")) if Path("/dev/dri").is_dir() else []
    from ..gpu import GPUInventory
    inv = GPUInventory.detect()
    checks.append(_check(
        "amdgpu devices", bool(inv.devices),
        f"{len(inv)} GPU(s): kfd={'ok' if kfd else 'missing'}, "
        f"render nodes={[p.name for p in dri]}" if dri or kfd else
        "no /dev/kfd or /dev/dri (CPU-only host)"))
    if inv.devices:
        d = inv.get(0)
        checks.append(_check(
            "MI355X HBM", d.vram_total >= 280 * 2**30,
            f"{d.vram_total / 2**30:.0f} GiB on GPU 0"))
    from ..controlplane.client import CPClient
    cp = CPClient(auto_start=False)
    checks.append(_check("control plane", True,
                         "running" if cp.running() else "stopped (starts on demand)"))
    return checks


@cli.command("doctor")
@click.option("--format", "fmt", default="")
@click.option("--collect", "collect_dir", is_flag=False, flag_value=".",
              default=None,
              help="write an SOS diagnostic tarball (checks, sandbox "
                   "forensics, CP events, GPU ledger) to DIR (default .)")
@pass_factory
def doctor_cmd(ctx: Ctx, fmt, collect_dir):
    """Diagnose host capabilities for sandboxing + GPU pinning."""
    import json
    f = ctx.factory
    if collect_dir is not None:
        from ..sos import collect_bundle
        out = collect_bundle(Path(collect_dir))
        f.io.print(str(out))
        return
    checks = run_checks()
    if fmt == "json":
        f.io.print(json.dumps(checks, indent=1))
        return
    worst = 0
    for c in checks:
        mark = "[green]✓[/green]" if c["ok"] else "[red]✗[/red]"
        detail = f" [dim]{c['detail']}[/dim]" if c["detail"] else ""
        f.io.print(f" {mark} {c['check']}{detail}")
        if not c["ok"] and c["check"] in ("root", "native runtime"):
            worst = 1
    if worst:
        from ..errors import ExitError
        raise ExitError(worst)
