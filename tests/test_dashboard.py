"""Dashboard/stats rendering (text snapshot via rich console capture)."""
from clawker_amd.monitor.sampler import GpuSample
from clawker_amd.monitor.stats import StatsSnapshot, render_stats
from clawker_amd.tui.dashboard import render_dashboard


def _snap():
    return StatsSnapshot(
        ts=1700000000.0,
        sandboxes=[{"name": "clawker.p.a", "state": "running", "pid": 42,
                    "gpus": [0], "agent": "a", "project": "p",
                    "mem_bytes": 512 * 2**20, "pids": 7,
                    "tail": "training step 18 loss=0.42"}],
        gpus=[GpuSample(index=0, render_minor=128, busy_pct=87.0,
                        vram_used=120 * 2**30, vram_total=288 * 2**30,
                        power_w=980.0, temp_junction_c=74.0, sclk_mhz=2100.0)],
        allocations={0: "clawker.p.a"},
        events=[{"ts": 1700000000.0, "sandbox": "clawker.p.a",
                 "action": "deny", "dst": "evil.example", "proto": "tls"}])


def _render(obj) -> str:
    from rich.console import Console
    import io
    buf = io.StringIO()
    Console(file=buf, width=120, force_terminal=False).print(obj)
    return buf.getvalue()


def test_render_dashboard_panes():
    out = _render(render_dashboard(_snap()))
    assert "MI355X GPUs" in out
    assert "clawker.p.a" in out
    assert "87%" in out and "980W" in out and "74C" in out
    assert "120.0G" in out
    assert "training step 18" in out
    assert "egress decisions" in out and "evil.example" in out


def test_render_stats_tables():
    out = _render(render_stats(_snap()))
    assert "clawker.p.a" in out and "512M" in out and "2100MHz" in out


def test_exporter_event_counters(isolated_env):
    """cp-events.jsonl decisions surface as monotonic Prometheus counters
    (the netlogger->OTLP lane, re-landed on the scrape endpoint)."""
    import json as _json
    from clawker_amd.controlplane.daemon import EventLog, events_path
    from clawker_amd.monitor.exporter import _EventCounters
    log = EventLog(events_path())
    log.emit("cp_ready")
    log.emit("egress_decision", sandbox="clawker.p.a", action="deny",
             proto="tls", dst="evil.example")
    log.emit("egress_decision", sandbox="clawker.p.a", action="deny",
             proto="tls", dst="evil.example")
    log.emit("egress_decision", sandbox="clawker.p.a", action="allow",
             proto="dns", dst="ok.example")
    c = _EventCounters()
    c.update()
    text = "\n".join(c.lines())
    assert 'clawker_cp_events_total{event="egress_decision"} 3' in text
    assert ('clawker_egress_decisions_total{sandbox="clawker.p.a",'
            'action="deny",proto="tls"} 2') in text
    # incremental: a new event adds without recounting the file
    log.emit("egress_decision", sandbox="clawker.p.a", action="deny",
             proto="tls", dst="evil.example")
    c.update()
    assert ('action="deny",proto="tls"} 3') in "\n".join(c.lines())
    # rotation resets the offset without crashing
    events_path().replace(events_path().with_suffix(".jsonl.1"))
    log2 = EventLog(events_path())
    log2.emit("cp_ready")
    c.update()
    assert 'clawker_cp_events_total{event="cp_ready"} 2' in "\n".join(c.lines())


def test_tui_components_render():
    """Shared component library: progress tree marks, kv panel, table
    (reference: internal/tui reusable components)."""
    from rich.console import Console
    import io as _io
    from clawker_amd.tui.components import ProgressSteps, kv_panel, plain_table
    steps = ProgressSteps(["base", "harness"])
    steps.start("base")
    steps.add("seed configs", parent="harness")
    steps.done("base", detail="cached")
    steps.fail("seed configs", detail="exit 1")
    steps.skip("harness")
    assert steps.failed
    con = Console(file=_io.StringIO(), force_terminal=False, width=80)
    con.print(steps)
    out = con.file.getvalue()
    assert "✓ base" in out and "cached" in out
    assert "✗ seed configs" in out and "- harness" in out
    con = Console(file=_io.StringIO(), force_terminal=False, width=80)
    con.print(kv_panel("sandbox", {"gpus": [0, 1], "fw": True, "err": None}))
    con.print(plain_table(("A", "B"), [(1, None), ("x", ["y", "z"])]))
    out = con.file.getvalue()
    assert "sandbox" in out and "0, 1" in out and "yes" in out
    assert "y, z" in out


def test_stats_json_format(isolated_env):
    import json as _json
    from click.testing import CliRunner
    from clawker_amd.cli.root import cli
    r = CliRunner().invoke(cli, ["stats", "--format", "json"])
    if r.exception is not None and not isinstance(r.exception, SystemExit):
        raise r.exception
    d = _json.loads(r.output)
    assert set(d) == {"gpus", "allocations", "sandboxes"}
