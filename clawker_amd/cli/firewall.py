"""Firewall verbs (reference: internal/cmd/firewall — status/list/add/
remove/reload/enable/disable/bypass backed by the 13 AdminService RPCs;
here backed by the rules store + control-plane daemon)."""
from __future__ import annotations

import json

import click

from ..config.schema import EgressRule
from ..errors import ClawkerError
from ..firewall import EgressRulesStore, IdentityAllocator
from .root import Ctx, cli, pass_factory


@cli.group("firewall")
def firewall_group():
    """Deny-by-default egress policy."""


def _store() -> EgressRulesStore:
    return EgressRulesStore()


@firewall_group.command("list")
@click.option("--format", "fmt", default="")
@pass_factory
def fw_list(ctx: Ctx, fmt):
    """Current egress rules with sticky identities."""
    f = ctx.factory
    rules = _store().list()
    idents = IdentityAllocator()
    if fmt == "json":
        from ..storage.store import to_plain
        f.io.print(json.dumps([to_plain(r) for r in rules], indent=1))
        return
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for c in ("DST", "PROTO", "PORT", "IDENTITY", "PATHS", "DENY-PATHS"):
        t.add_column(c)
    for r in rules:
        ident = idents.get(r.dst)
        t.add_row(r.dst, r.proto, str(r.port), str(ident or "-"),
                  ",".join(r.paths) or "-", ",".join(r.deny_paths) or "-")
    f.io.print(t)


@firewall_group.command("add")
@click.argument("dst")
@click.option("--proto", type=click.Choice(["tls", "http", "tcp", "udp", "ssh"]),
              default="tls", show_default=True)
@click.option("--port", type=int, default=443, show_default=True)
@click.option("--path", "paths", multiple=True, help="allow-only path prefix ('~' = regex)")
@click.option("--deny-path", "deny_paths", multiple=True)
@pass_factory
def fw_add(ctx: Ctx, dst, proto, port, paths, deny_paths):
    """Allow egress to DST (domain, *.wildcard, or IP)."""
    f = ctx.factory
    rule = EgressRule(dst=dst, proto=proto, port=port,
                      paths=list(paths), deny_paths=list(deny_paths))
    changed = _store().add([rule])
    IdentityAllocator().allocate(dst)
    _reload_running(f)
    f.io.success(f"{'added' if changed else 'already present'}: {rule.key()}")
    if proto == "udp":
        f.io.warn(
            "generic UDP has no egress datapath (the sandbox netns has no "
            "uplink and only DNS is relayed) — this rule documents intent "
            "but UDP traffic to the destination remains blocked")


@firewall_group.command("remove")
@click.argument("dst")
@pass_factory
def fw_remove(ctx: Ctx, dst):
    """Remove all rules for DST and reload gateways."""
    f = ctx.factory
    if not _store().remove(dst):
        raise ClawkerError(f"no rule for: {dst}")
    _reload_running(f)
    f.io.success(f"removed: {dst}")


@firewall_group.command("status")
@pass_factory
def fw_status(ctx: Ctx):
    """Gateway attachments, bypass state, rule count."""
    f = ctx.factory
    rules = _store().list()
    infos = f.engine().list()
    backend = f.engine().backend
    out = {
        "backend": backend,
        "enforcement": "netns+gateway+mitm" if backend == "ns" else
                       "unavailable (proc backend: no netns on this host)",
        "rules": len(rules),
        "running_sandboxes": len([i for i in infos if i.state == "running"]),
    }
    try:
        cp = f.controlplane()
        cp.auto_start = False
        if cp.running():
            out.update(cp.request({"op": "fw_status"}))
            out.pop("ok", None)
    except Exception:
        pass
    f.io.print(json.dumps(out, indent=1))


@firewall_group.command("reload")
@pass_factory
def fw_reload(ctx: Ctx):
    """Push current rules to every running sandbox's gateway."""
    f = ctx.factory
    n = _reload_running(f)
    f.io.success(f"reloaded policy on {n} gateway(s)")


firewall_group.add_command(fw_reload, "refresh")   # reference alias


@firewall_group.command("rotate-ca")
@pass_factory
def fw_rotate_ca(ctx: Ctx):
    """Mint a new TLS-MITM CA and drop cached per-domain leaves
    (reference: FirewallRotateCA). Running sandboxes keep trusting the
    old CA until restarted (their trust bundle is baked at create);
    gateways pick up the new CA on the next leaf mint."""
    f = ctx.factory
    from ..firewall import mitm
    mitm.rotate_ca()
    n = _reload_running(f)
    f.io.success(f"rotated MITM CA; {n} gateway(s) reloaded "
                 "(restart sandboxes to refresh their trust bundles)")


@firewall_group.command("bypass")
@click.option("--minutes", type=int, default=15, show_default=True)
@pass_factory
def fw_bypass(ctx: Ctx, minutes):
    """Temporarily allow all egress (dead-man capped at settings
    firewall.bypass_max_s; reference: FirewallBypass 1h cap)."""
    f = ctx.factory
    cap = f.config().settings.firewall.bypass_max_s
    secs = min(minutes * 60, cap)
    cp = f.controlplane()
    cp.bypass(secs)
    f.io.warn(f"firewall BYPASSED for {secs}s (auto-restore)")
    if f.io.is_stdout_tty():
        # live countdown until the dead-man restores (reference:
        # bypass_dash.go tui.RunDashboard precedent); ctrl-c restores NOW
        import time as _t
        from rich.live import Live
        end = _t.monotonic() + secs
        try:
            with Live(console=f.io.console, refresh_per_second=4) as live:
                while True:
                    left = end - _t.monotonic()
                    if left <= 0:
                        break
                    live.update(
                        f"[yellow]⚠ all egress OPEN[/yellow] — restoring "
                        f"policy in [bold]{int(left // 60)}:"
                        f"{int(left % 60):02d}[/bold]  (ctrl-c: restore now)")
                    _t.sleep(0.25)
            f.io.success("policy restored")
        except KeyboardInterrupt:
            cp.bypass(0)
            f.io.success("policy restored early")


@firewall_group.command("enable")
@click.argument("sandbox")
@pass_factory
def fw_enable(ctx: Ctx, sandbox):
    """Attach the policy gateway to a running sandbox (reference:
    FirewallEnable)."""
    f = ctx.factory
    from ..cmdutil import resolve_sandbox_name
    name = resolve_sandbox_name(f, sandbox)
    info = f.engine().inspect(name)
    f.controlplane().request({"op": "fw_attach", "sandbox": name,
                              "rundir": str(info.rundir)})
    f.io.success(f"firewall gateway attached: {name}")


@firewall_group.command("disable")
@click.argument("sandbox")
@pass_factory
def fw_disable(ctx: Ctx, sandbox):
    """Detach the policy gateway (the netns still has no uplink — this
    only stops the allow-listed paths; reference: FirewallDisable)."""
    f = ctx.factory
    from ..cmdutil import resolve_sandbox_name
    name = resolve_sandbox_name(f, sandbox)
    f.controlplane().request({"op": "fw_detach", "sandbox": name, "sticky": True})
    f.io.success(f"firewall gateway detached: {name}")


@firewall_group.command("resolve")
@click.argument("hostname")
@pass_factory
def fw_resolve(ctx: Ctx, hostname):
    """Resolve a hostname the way the policy resolver would
    (reference: FirewallResolveHostname)."""
    f = ctx.factory
    from ..firewall.gateway import GatewayManager
    mgr = GatewayManager()
    ips = mgr._resolve(hostname)
    rule = _store().match_domain(hostname) or _store().match_domain(
        hostname, proto="http", port=80)
    f.io.print(json.dumps({
        "hostname": hostname, "ips": ips,
        "policy": "allowed" if rule else "denied (no rule)",
    }))


def _reload_running(f) -> int:
    """Signal running sandbox gateways to re-read policy (the gateway polls
    the policy file; CP push lands in controlplane/daemon)."""
    try:
        cp = f.controlplane()
        return cp.reload_policy()
    except Exception:
        return 0
