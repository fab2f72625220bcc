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
