"""Container (sandbox) verbs + Docker-style top-level aliases.

Reference: internal/cmd/container/* (20 subverbs) and the top-level alias
table in internal/cmd/root/aliases.go.
"""
from __future__ import annotations

import json
import signal as _signal
import sys
import time

import click

from ..cmdutil import Factory, format_age, resolve_sandbox_name
from ..errors import ClawkerError, ExitError
from ..orchestrator import RunOptions
from . import _attach
from .root import Ctx, cli, pass_factory


@cli.group("container")
def container_group():
    """Manage agent sandboxes."""


def _run_opts_flags(fn):
    fn = click.option("--agent", "-a", default="agent", show_default=True,
                      help="agent name (sandbox = clawker.<project>.<agent>)")(fn)
    fn = click.option("--image", default="", help="image (@ = project image, hostfs = host rootfs)")(fn)
    fn = click.option("--gpus", type=int, default=None, help="GPUs to pin (default: project gpu.count)")(fn)
    fn = click.option("--gpu-index", "gpu_indices", type=int, multiple=True,
                      help="pin specific GPU indices")(fn)
    fn = click.option("--hbm-gb", type=int, default=None, help="HBM budget per GPU (GB)")(fn)
    fn = click.option("--env", "-e", "env_kv", multiple=True, help="KEY=VALUE")(fn)
    fn = click.option("--workdir", "-w", default="", help="working directory inside the sandbox")(fn)
    fn = click.option("--user", "-u", default=None, help="user (name or uid:gid)")(fn)
    fn = click.option("--workspace-mode", type=click.Choice(["", "bind", "snapshot"]),
                      default="", help="workspace strategy override")(fn)
    fn = click.option("--worktree", default="", metavar="BRANCH[:BASE]",
                      help="run in a git worktree for BRANCH")(fn)
    fn = click.option("--firewall/--no-firewall", "firewall", default=None,
                      help="override project security.firewall")(fn)
    fn = click.option("--memory", "-m", "mem", default="", help="memory limit (e.g. 8g)")(fn)
    fn = click.option("--pids-limit", type=int, default=4096, show_default=True)(fn)
    fn = click.option("--label", "-l", "labels_kv", multiple=True, help="KEY=VALUE label")(fn)
    fn = click.option("--no-host-services", is_flag=True,
                      help="skip hostproxy + ssh/gpg agent bridges")(fn)
    fn = click.option("--restart", "restart_policy", default="no",
                      metavar="no|on-failure[:N]", show_default=True)(fn)
    fn = click.option("-v", "--volume", "volumes_kv", multiple=True,
                      metavar="SRC:DST[:ro]",
                      help="bind mount a host path (or named volume) into the sandbox")(fn)
    fn = click.option("--name", "name_override", default="",
                      help="full sandbox name (default: clawker.<project>.<agent>)")(fn)
    return fn


def _parse_kv(pairs) -> dict:
    out = {}
    for p in pairs:
        if "=" not in p:
            raise ClawkerError(f"expected KEY=VALUE, got: {p}")
        k, v = p.split("=", 1)
        out[k] = v
    return out


def _parse_mem(s: str) -> int:
    if not s:
        return 0
    mult = {"k": 1024, "m": 1024**2, "g": 1024**3, "t": 1024**4}
    if s[-1].lower() in mult:
        return int(float(s[:-1]) * mult[s[-1].lower()])
    return int(s)


def _parse_volumes(f: Factory, specs) -> list:
    """-v SRC:DST[:ro] — SRC is a host path (absolute/relative) or a
    named volume (created on demand)."""
    from pathlib import Path as _P
    from ..engine.spec import Mount
    out = []
    for sp in specs:
        parts = sp.split(":")
        if len(parts) < 2:
            raise ClawkerError(f"-v expects SRC:DST[:ro], got: {sp}")
        src, dst = parts[0], parts[1]
        ro = len(parts) > 2 and parts[2] == "ro"
        if not dst.startswith("/"):
            raise ClawkerError(f"-v destination must be absolute: {dst}")
        if src.startswith("/") or src.startswith(".") or src.startswith("~"):
            src_path = _P(src).expanduser().resolve()
            if not src_path.exists():
                raise ClawkerError(f"-v source missing: {src_path}")
            out.append(Mount(src=str(src_path), dst=dst, ro=ro))
        else:
            vol_path, _ = f.engine().ensure_volume(f"clawker.user.{src}", {})
            out.append(Mount(src=str(vol_path), dst=dst, ro=ro))
    return out


def _build_opts(f: Factory, agent, image, gpus, gpu_indices, hbm_gb, env_kv,
                workdir, user, workspace_mode, worktree, firewall, mem,
                pids_limit, labels_kv, cmd, tty, autostart,
                no_host_services=False, volumes_kv=(), name_override="") -> RunOptions:
    cfg = f.config()
    if image == "@":
        image = cfg.image_name()
    opts = RunOptions(
        agent=agent, image=image, cmd=list(cmd), tty=tty,
        gpus=gpus, gpu_indices=list(gpu_indices) or None, hbm_gb=hbm_gb,
        env=_parse_kv(env_kv), user=user, workdir=workdir,
        workspace_mode=workspace_mode, firewall=firewall,
        mem_bytes=_parse_mem(mem), pids_max=pids_limit,
        labels=_parse_kv(labels_kv), autostart=autostart,
        host_services=not no_host_services,
        mounts=_parse_volumes(f, volumes_kv), name=name_override)
    if worktree:
        from ..project.worktrees import ensure_worktree
        wt = ensure_worktree(f.config(require_project=True), worktree)
        opts.workspace = wt.path
        if not opts.agent or opts.agent == "agent":
            opts.agent = wt.safe_name
    _guard_home_workspace(f, opts, cfg)
    return opts


def _guard_home_workspace(f: Factory, opts: RunOptions, cfg) -> None:
    """Mounting $HOME or / as the workspace hands the agent the whole
    blast radius; require explicit confirmation (reference: the
    home-mount safety prompt, container/shared/safety.go)."""
    from pathlib import Path as _P
    ws = opts.workspace or (cfg.workspace_path() if cfg.project_root else None)
    if ws is None:
        return
    try:
        ws = _P(ws).resolve()
        home = _P.home().resolve()
    except OSError:
        return
    if ws == _P("/") or ws == home or ws in home.parents:
        if f.io.can_prompt():
            if f.prompter().confirm(
                    f"workspace {ws} contains your entire home/root — "
                    "mount it into the sandbox anyway?", default=False):
                return
        raise ClawkerError(
            f"refusing to mount {ws} as the workspace (it contains your "
            "home directory); use a project subdirectory, or confirm "
            "interactively")


def _boot_and_wait(f: Factory, name: str, interactive: bool, tty: bool,
                   detach: bool, rm: bool) -> int:
    """start -> attach -> init/boot plans -> agent_ready -> stream/wait
    (reference run.go attachThenStart ordering: attach precedes CMD release
    so no output is lost)."""
    orch = f.orchestrator()
    orch.start(name)
    client = orch.client(name)
    try:
        hello = client.hello()
        from ..controlplane.plans import run_boot_plans
        run_boot_plans(f, name, client, hello)
        client.attach()
        client.agent_ready()
        if detach:
            f.io.eprint(name)
            return 0
        code = _attach.stream(client, interactive=interactive, tty_mode=tty)
        if code == -2:
            f.io.eprint("detached")
            return 0
        if code < 0:
            code = orch.engine.wait(name, timeout_s=30)
        return code
    finally:
        client.close()
        if rm and not detach:
            try:
                orch.teardown(name, force=True)
            except ClawkerError:
                pass


@cli.command("run", context_settings={"ignore_unknown_options": True})
@_run_opts_flags
@click.option("-i", "--interactive", is_flag=True, help="keep stdin open")
@click.option("-t", "--tty", is_flag=True, help="allocate a pseudo-TTY")
@click.option("-d", "--detach", is_flag=True, help="run in background")
@click.option("--rm", "rm_after", is_flag=True, help="remove sandbox on exit")
@click.argument("cmd", nargs=-1, type=click.UNPROCESSED)
@pass_factory
def run_cmd(ctx: Ctx, agent, image, gpus, gpu_indices, hbm_gb, env_kv, workdir,
            user, workspace_mode, worktree, firewall, mem, pids_limit,
            labels_kv, no_host_services, restart_policy, volumes_kv, name_override, interactive, tty, detach, rm_after, cmd):
    """Create and start an agent sandbox (alias of `container run`).

    CMD may start with an image reference: `clawker run @ -- <cmd>` runs the
    project image; with no CMD the harness default runs."""
    f = ctx.factory
    cmd = list(cmd)
    if cmd and cmd[0] in ("@",) or (cmd and not cmd[0].startswith("-") and
                                    f.engine().images.exists(cmd[0])):
        image = cmd[0]
        cmd = cmd[1:]
    if image == "@":
        image = f.config(require_project=True).image_name()
    opts = _build_opts(f, agent, image, gpus, gpu_indices, hbm_gb, env_kv,
                       workdir, user, workspace_mode, worktree, firewall, mem,
                       pids_limit, labels_kv, cmd, tty, autostart=False,
                       no_host_services=no_host_services,
                       volumes_kv=volumes_kv, name_override=name_override)
    opts.restart = restart_policy
    orch = f.orchestrator()
    info = orch.create(opts)
    code = _boot_and_wait(f, info.name, interactive, tty, detach, rm_after)
    if code != 0:
        raise ExitError(code)


container_group.add_command(run_cmd, "run")


@cli.command("create")
@_run_opts_flags
@click.option("-t", "--tty", is_flag=True)
@click.argument("cmd", nargs=-1, type=click.UNPROCESSED)
@pass_factory
def create_cmd(ctx: Ctx, agent, image, gpus, gpu_indices, hbm_gb, env_kv,
               workdir, user, workspace_mode, worktree, firewall, mem,
               pids_limit, labels_kv, no_host_services, restart_policy, volumes_kv, name_override, tty, cmd):
    """Create a sandbox without starting it."""
    f = ctx.factory
    opts = _build_opts(f, agent, image, gpus, gpu_indices, hbm_gb, env_kv,
                       workdir, user, workspace_mode, worktree, firewall, mem,
                       pids_limit, labels_kv, list(cmd), tty, autostart=False,
                       no_host_services=no_host_services,
                       volumes_kv=volumes_kv, name_override=name_override)
    opts.restart = restart_policy
    info = f.orchestrator().create(opts)
    f.io.print(info.name)


container_group.add_command(create_cmd, "create")


@cli.command("start")
@click.option("-a", "--attach", "do_attach", is_flag=True, help="attach to console")
@click.option("-i", "--interactive", is_flag=True)
@click.argument("names", nargs=-1, required=True)
@pass_factory
def start_cmd(ctx: Ctx, do_attach, interactive, names):
    """Start created/stopped sandboxes."""
    f = ctx.factory
    last = 0
    for n in names:
        name = resolve_sandbox_name(f, n)
        if do_attach and len(names) == 1:
            info = f.engine().inspect(name)
            tty = False
            try:
                spec = json.loads((info.rundir / "spec.json").read_text())
                tty = bool(spec.get("tty"))
            except (OSError, ValueError):
                pass
            last = _boot_and_wait(f, name, interactive, tty, detach=False, rm=False)
        else:
            _boot_and_wait(f, name, interactive=False, tty=False, detach=True, rm=False)
            f.io.print(name)
    if last:
        raise ExitError(last)


container_group.add_command(start_cmd, "start")


@cli.command("stop")
@click.option("-t", "--time", "timeout", type=float, default=10.0, show_default=True)
@click.argument("names", nargs=-1, required=True)
@pass_factory
def stop_cmd(ctx: Ctx, timeout, names):
    """Stop running sandboxes (SIGTERM, then SIGKILL after --time)."""
    f = ctx.factory
    for n in names:
        name = resolve_sandbox_name(f, n)
        f.engine().stop(name, timeout_s=timeout)
        f.io.print(name)


container_group.add_command(stop_cmd, "stop")


@cli.command("kill")
@click.option("-s", "--signal", "sig", default="KILL", show_default=True)
@click.argument("names", nargs=-1, required=True)
@pass_factory
def kill_cmd(ctx: Ctx, sig, names):
    """Send a signal to sandboxes."""
    f = ctx.factory
    signum = getattr(_signal, f"SIG{sig.upper().removeprefix('SIG')}", None)
    if signum is None:
        raise ClawkerError(f"unknown signal: {sig}")
    for n in names:
        f.engine().kill(resolve_sandbox_name(f, n), signum)
        f.io.print(n)


container_group.add_command(kill_cmd, "kill")


@cli.command("rm")
@click.option("-f", "--force", is_flag=True, help="stop first if running")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def rm_cmd(ctx: Ctx, force, names):
    """Remove sandboxes (and release their GPUs)."""
    f = ctx.factory
    for n in names:
        name = resolve_sandbox_name(f, n)
        f.orchestrator().teardown(name, force=force)
        f.io.print(name)


container_group.add_command(rm_cmd, "rm")


@cli.command("ps")
@click.option("-a", "--all", "show_all", is_flag=True, help="include stopped")
@click.option("-q", "--quiet", is_flag=True, help="names only")
@click.option("--format", "fmt", default="", help="'json' for machine output")
@click.option("--filter", "filters", multiple=True, help="label=KEY=VALUE or project=P")
@pass_factory
def ps_cmd(ctx: Ctx, show_all, quiet, fmt, filters):
    """List sandboxes (alias: container ls)."""
    f = ctx.factory
    project = None
    label_filters = {}
    for flt in filters:
        if flt.startswith("project="):
            project = flt.split("=", 1)[1]
        elif flt.startswith("label="):
            kv = flt.split("=", 2)
            if len(kv) == 3:
                label_filters[kv[1]] = kv[2]
    infos = f.engine().list(project=project, all_states=True,
                            label_filters=label_filters or None)
    if not show_all:    # paused sandboxes stay visible (docker parity)
        infos = [i for i in infos if i.state in ("running", "paused")]
    if quiet:
        for i in infos:
            f.io.print(i.name)
        return
    if fmt == "json":
        f.io.print(json.dumps([{
            "name": i.name, "project": i.project, "agent": i.agent,
            "image": i.image, "state": i.state, "exit_code": i.exit_code,
            "pid": i.pid, "gpus": i.gpus, "created": i.created,
        } for i in infos], indent=1))
        return
    from rich.table import Table
    t = Table(box=None, pad_edge=False)
    for col in ("NAME", "IMAGE", "STATE", "GPUS", "AGE", "AGENT"):
        t.add_column(col)
    now = time.time()
    for i in infos:
        gpus = ",".join(map(str, i.gpus)) or "-"
        t.add_row(i.name, i.image, i.state, gpus,
                  format_age(now - i.created), i.agent)
    f.io.print(t)


container_group.add_command(ps_cmd, "ls")
container_group.add_command(ps_cmd, "ps")


@cli.command("exec")
@click.option("-u", "--user", default="", help="run as user")
@click.option("-w", "--workdir", default="")
@click.option("-e", "--env", "env_kv", multiple=True)
@click.option("-i", "--interactive", is_flag=True, help="keep stdin open")
@click.option("-t", "--tty", is_flag=True, help="allocate a pseudo-TTY")
@click.option("-d", "--detach", is_flag=True, help="run in background")
@click.argument("name")
@click.argument("cmd", nargs=-1, required=True, type=click.UNPROCESSED)
@pass_factory
def exec_cmd(ctx: Ctx, user, workdir, env_kv, interactive, tty, detach, name, cmd):
    """Run a command in a running sandbox (-it for an interactive shell)."""
    f = ctx.factory
    sb = resolve_sandbox_name(f, name)
    if not user:
        # docker semantics: exec runs as the sandbox's configured user
        # unless -u overrides (plans/admin paths send explicit stages)
        try:
            with f.engine().client(sb, timeout=5) as _c:
                user = _c.hello().get("user", "") or ""
        except Exception:
            user = ""
    if detach:
        stage = {"argv": list(cmd)}
        if user:
            stage["user"] = user
        if workdir:
            stage["cwd"] = workdir
        with f.engine().client(sb) as c:
            eid = c.exec_start([stage], env=_parse_kv(env_kv) or None)
        f.io.print(eid)
        return
    if tty:
        code = _exec_interactive(f, sb, list(cmd), user, workdir,
                                 _parse_kv(env_kv) or None)
        if code != 0:
            raise ExitError(code)
        return
    # docker semantics: only -i consumes stdin. Slurping whenever stdin
    # is merely non-tty blocks forever under a parent that keeps the pipe
    # open without writing (observed: exec inside a CI bash wrapper).
    stdin = b""
    if interactive and not sys.stdin.isatty():
        stdin = sys.stdin.buffer.read()
    with f.engine().client(sb) as c:
        stage = {"argv": list(cmd)}
        if user:
            stage["user"] = user
        if workdir:
            stage["cwd"] = workdir
        code, out, err = c.exec([stage], stdin=stdin, env=_parse_kv(env_kv) or None)
    sys.stdout.buffer.write(out)
    sys.stderr.buffer.write(err)
    if code != 0:
        raise ExitError(code)


def _exec_interactive(f: Factory, sb: str, argv, user, workdir, env) -> int:
    """Interactive pty exec: raw-mode pump like attach, but frame-scoped
    to the exec id."""
    import os
    import select
    import termios
    import tty as _tty
    from ..engine import wire as _wire
    with f.engine().client(sb) as c:
        eid = c.exec_start_tty(argv, user=user, cwd=workdir, env=env)
        stdin_fd = sys.stdin.fileno()
        saved = None
        if sys.stdin.isatty():
            saved = termios.tcgetattr(stdin_fd)
            _tty.setraw(stdin_fd)
            try:
                sz = os.get_terminal_size(sys.stdout.fileno())
                c.exec_resize(eid, sz.lines, sz.columns)
            except OSError:
                pass
        code = -1
        try:
            sock = c.sock
            sock.settimeout(None)
            while True:
                rlist = [sock] + ([stdin_fd] if saved is not None or interactive_stdin() else [])
                ready, _, _ = select.select(rlist, [], [])
                if stdin_fd in ready:
                    data = os.read(stdin_fd, 65536)
                    if not data:
                        break
                    c.exec_stdin(eid, data)
                if sock in ready:
                    fr = _wire.recv_frame(sock)
                    if fr is None:
                        break
                    t = fr.get("t")
                    if t == "out" and fr.get("id") == eid:
                        sys.stdout.buffer.write(_wire.unb64(fr.get("data", "")))
                        sys.stdout.buffer.flush()
                    elif t == "done" and fr.get("id") == eid:
                        code = int(fr.get("code", -1))
                        break
        finally:
            if saved is not None:
                termios.tcsetattr(stdin_fd, termios.TCSADRAIN, saved)
        return code


def interactive_stdin() -> bool:
    try:
        return not sys.stdin.closed
    except Exception:
        return False


container_group.add_command(exec_cmd, "exec")


@cli.command("logs")
@click.option("-f", "--follow", is_flag=True)
@click.option("-n", "--tail", "tail_n", type=int, default=0, help="last N lines")
@click.argument("name")
@pass_factory
def logs_cmd(ctx: Ctx, follow, tail_n, name):
    """Print a sandbox's console log."""
    f = ctx.factory
    sb = resolve_sandbox_name(f, name)
    data = f.engine().logs(sb)
    if tail_n > 0:
        data = b"\n".join(data.splitlines()[-tail_n:]) + (b"\n" if data else b"")
    sys.stdout.buffer.write(data)
    sys.stdout.buffer.flush()
    if follow:
        info = f.engine().inspect(sb)
        path = info.rundir / "console.log"
        pos = len(data)
        while True:
            info = f.engine().inspect(sb)
            try:
                with open(path, "rb") as fh:
                    fh.seek(pos)
                    chunk = fh.read()
                    if chunk:
                        sys.stdout.buffer.write(chunk)
                        sys.stdout.buffer.flush()
                        pos += len(chunk)
            except OSError:
                pass
            if info.state != "running":
                break
            time.sleep(0.2)


container_group.add_command(logs_cmd, "logs")


@cli.command("attach")
@click.option("--no-stdin", is_flag=True)
@click.argument("name")
@pass_factory
def attach_cmd(ctx: Ctx, no_stdin, name):
    """Attach the terminal to a running sandbox's console."""
    f = ctx.factory
    sb = resolve_sandbox_name(f, name)
    info = f.engine().inspect(sb)
    tty = False
    try:
        spec = json.loads((info.rundir / "spec.json").read_text())
        tty = bool(spec.get("tty"))
    except (OSError, ValueError):
        pass
    with f.engine().client(sb) as c:
        c.attach()
        code = _attach.stream(c, interactive=not no_stdin, tty_mode=tty)
    if code > 0:
        raise ExitError(code)


container_group.add_command(attach_cmd, "attach")


@cli.command("wait")
@click.option("--timeout", "timeout_s", type=float, default=None,
              help="give up after SECONDS (exit 124, like timeout(1))")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def wait_cmd(ctx: Ctx, timeout_s, names):
    """Block until sandboxes exit; print their exit codes."""
    from ..errors import EngineError
    f = ctx.factory
    last = 0
    deadline = None if timeout_s is None else time.monotonic() + timeout_s
    for n in names:
        remaining = None if deadline is None else deadline - time.monotonic()
        try:
            code = f.engine().wait(resolve_sandbox_name(f, n),
                                   timeout_s=remaining)
        except EngineError:
            raise ExitError(124)
        f.io.print(str(code))
        last = code
    if last:
        raise ExitError(last)


container_group.add_command(wait_cmd, "wait")


@cli.command("inspect")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def inspect_cmd(ctx: Ctx, names):
    """Low-level sandbox details (JSON)."""
    f = ctx.factory
    out = []
    for n in names:
        name = resolve_sandbox_name(f, n)
        i = f.engine().inspect(name)
        d = {
            "name": i.name, "project": i.project, "agent": i.agent,
            "image": i.image, "state": i.state, "exit_code": i.exit_code,
            "pid": i.pid, "gpus": i.gpus, "labels": i.labels,
            "created": i.created, "rundir": str(i.rundir),
            "statedir": str(i.statedir),
        }
        try:
            d["spec"] = json.loads((i.rundir / "spec.json").read_text())
        except (OSError, ValueError):
            pass
        out.append(d)
    f.io.print(json.dumps(out, indent=1))


container_group.add_command(inspect_cmd, "inspect")


@cli.command("top")
@click.argument("name")
@pass_factory
def top_cmd(ctx: Ctx, name):
    """Processes running in a sandbox."""
    f = ctx.factory
    sb = resolve_sandbox_name(f, name)
    code, out, err = f.engine().exec(sb, ["/bin/ps", "-ef"])
    sys.stdout.buffer.write(out or err)
    if code != 0:
        raise ExitError(code)


container_group.add_command(top_cmd, "top")


@container_group.command("prune")
@click.option("-f", "--force", is_flag=True)
@pass_factory
def prune_cmd(ctx: Ctx, force):
    """Remove all stopped sandboxes."""
    f = ctx.factory
    removed = 0
    for i in f.engine().list():
        if i.state != "running":
            f.orchestrator().teardown(i.name, force=True)
            f.io.print(i.name)
            removed += 1
    f.io.eprint(f"removed {removed} sandbox(es)")


@cli.command("cp")
@click.argument("src")
@click.argument("dst")
@pass_factory
def cp_cmd(ctx: Ctx, src, dst):
    """Copy files between the host and a running sandbox
    (SANDBOX:PATH <-> PATH), streamed as tar through the control socket."""
    import io
    import tarfile
    from pathlib import Path as _P
    f = ctx.factory

    def split(ref):
        if ":" in ref and not ref.startswith("/") and not ref.startswith("."):
            name, _, path = ref.partition(":")
            return resolve_sandbox_name(f, name), path
        return None, ref

    src_sb, src_path = split(src)
    dst_sb, dst_path = split(dst)
    if (src_sb is None) == (dst_sb is None):
        raise ClawkerError("exactly one of SRC/DST must be SANDBOX:PATH")
    if src_sb:
        parent = str(_P(src_path).parent) or "/"
        base = _P(src_path).name
        code, out, err = f.engine().exec(
            src_sb, ["tar", "-C", parent, "-cf", "-", base])
        if code != 0:
            raise ClawkerError(f"tar in sandbox failed: {err.decode()[-200:]}")
        with tarfile.open(fileobj=io.BytesIO(out)) as tf:
            dest = _P(dst_path)
            dest.mkdir(parents=True, exist_ok=True)
            # the archive was produced INSIDE the sandbox: treat it as
            # hostile — no absolute/.. members, no links escaping dest
            for m in tf.getmembers():
                parts = _P(m.name).parts
                if m.name.startswith("/") or ".." in parts:
                    raise ClawkerError(f"refusing unsafe tar member {m.name!r}")
                if (m.issym() or m.islnk()) and (
                        m.linkname.startswith("/")
                        or ".." in _P(m.linkname).parts):
                    raise ClawkerError(
                        f"refusing unsafe link {m.name!r} -> {m.linkname!r}")
            tf.extractall(dest)
        f.io.success(f"copied {src} -> {dst}")
    else:
        sp = _P(src_path)
        if not sp.exists():
            raise ClawkerError(f"no such file: {sp}")
        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w") as tf:
            tf.add(str(sp), arcname=sp.name)
        import shlex
        q = shlex.quote(dst_path)
        code, out, err = f.engine().exec(
            dst_sb, ["/bin/sh", "-c", f"mkdir -p {q} && tar -C {q} -xf -"],
            stdin=buf.getvalue())
        if code != 0:
            raise ClawkerError(f"tar extract failed: {err.decode()[-200:]}")
        f.io.success(f"copied {src} -> {dst}")


container_group.add_command(cp_cmd, "cp")


@cli.command("restart")
@click.option("-t", "--time", "timeout", type=float, default=10.0, show_default=True)
@click.argument("names", nargs=-1, required=True)
@pass_factory
def restart_cmd(ctx: Ctx, timeout, names):
    """Stop then start sandboxes (agent CMD released immediately)."""
    f = ctx.factory
    for n in names:
        name = resolve_sandbox_name(f, n)
        f.engine().stop(name, timeout_s=timeout)
        _boot_and_wait(f, name, interactive=False, tty=False, detach=True, rm=False)
        f.io.print(name)


container_group.add_command(restart_cmd, "restart")


@container_group.command("commit")
@click.option("-m", "--message", default="", help="commit annotation label")
@click.argument("name")
@click.argument("image")
@pass_factory
def commit_cmd(ctx: Ctx, message, name, image):
    """Snapshot a sandbox's writable layer as a new image (docker commit
    analog: agent-installed tools become a reusable layer)."""
    f = ctx.factory
    sb = resolve_sandbox_name(f, name)
    from ..engine.build import commit_sandbox
    meta = commit_sandbox(f.engine(), sb, image, message=message)
    f.io.print(f"{meta.name}  layers={len(meta.layers)}")


@container_group.command("pause")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def pause_cmd(ctx: Ctx, names):
    """Freeze a sandbox's workload (SIGSTOP the tree under its init;
    the supervisor stays live so status/unpause keep working)."""
    f = ctx.factory
    for n in names:
        sb = resolve_sandbox_name(f, n)
        f.engine().pause(sb)
        f.io.print(sb)


@container_group.command("unpause")
@click.argument("names", nargs=-1, required=True)
@pass_factory
def unpause_cmd(ctx: Ctx, names):
    """Thaw a paused sandbox (SIGCONT)."""
    f = ctx.factory
    for n in names:
        sb = resolve_sandbox_name(f, n)
        f.engine().unpause(sb)
        f.io.print(sb)


@container_group.command("update")
@click.option("--memory", "-m", "mem", default="", help="new memory limit (e.g. 8g)")
@click.option("--pids-limit", type=int, default=0)
@click.argument("name")
@pass_factory
def update_cmd(ctx: Ctx, mem, pids_limit, name):
    """Update resource limits of a RUNNING sandbox (live cgroup rewrite)."""
    from pathlib import Path as _P
    f = ctx.factory
    sb = resolve_sandbox_name(f, name)
    info = f.engine().inspect(sb)
    if info.state != "running":
        raise ClawkerError(f"sandbox not running: {sb}")
    cg_root = _P("/sys/fs/cgroup")
    changed = []
    if mem:
        nbytes = _parse_mem(mem)
        for p in (cg_root / "memory" / "clawker" / sb / "memory.limit_in_bytes",
                  cg_root / "clawker" / sb / "memory.max"):
            if p.parent.is_dir():
                p.write_text(str(nbytes))
                changed.append(f"memory={mem}")
                break
        else:
            raise ClawkerError(
                "no memory cgroup for this sandbox (started without -m; "
                "restart with --memory to enable live updates)")
    if pids_limit:
        for p in (cg_root / "pids" / "clawker" / sb / "pids.max",
                  cg_root / "clawker" / sb / "pids.max"):
            if p.parent.is_dir():
                p.write_text(str(pids_limit))
                changed.append(f"pids={pids_limit}")
                break
    if not changed:
        raise ClawkerError("nothing to update (pass --memory/--pids-limit)")
    f.io.success(f"updated {sb}: {', '.join(changed)}")


@container_group.command("diff")
@click.argument("name")
@pass_factory
def diff_cmd(ctx: Ctx, name):
    """Changed files vs the image (the sandbox's copy-on-write upper)."""
    from pathlib import Path as _P
    f = ctx.factory
    sb = resolve_sandbox_name(f, name)
    info = f.engine().inspect(sb)
    upper = info.statedir / "upper"
    if not upper.is_dir():
        raise ClawkerError(f"no writable layer for {sb} (proc backend?)")
    skip_prefixes = ("etc/hostname", "etc/hosts", "etc/resolv.conf",
                     "var/lib/clawker", "run/", "tmp/")
    try:
        lowers = f.engine().images.lowerdirs_for(info.image)
    except Exception:
        lowers = ["/"]
    rows = []
    for p in sorted(upper.rglob("*")):
        rel = str(p.relative_to(upper))
        if any(rel.startswith(s) for s in skip_prefixes):
            continue
        # overlay whiteouts: 0:0 char devices mark deletions
        try:
            st = p.lstat()
        except OSError:
            continue
        import stat as _stat
        if _stat.S_ISCHR(st.st_mode) and st.st_rdev == 0:
            rows.append(("D", rel))
        elif p.is_dir():
            continue
        else:
            # A if the path exists in no lower layer, C otherwise
            # (docker diff A/C/D fidelity)
            mark = "A"
            for low in lowers:
                if (_P(low) / rel).exists():
                    mark = "C"
                    break
            rows.append((mark, rel))
    for mark, rel in rows:
        f.io.print(f"{mark} /{rel}")
