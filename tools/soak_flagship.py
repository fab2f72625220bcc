#!/usr/bin/env python3
"""Flagship-config soak: N concurrent agent loops in the PRODUCTION
posture — ns backend, enforcing firewall (netns + gateway + DNS), and
the unprivileged materialized `agent` user — sustained for R rounds.

This is the configuration VERDICT r01 flagged as dark: every loop is a
full cold start whose agent (non-root) resolves an allowed domain,
fetches through the gateway, verifies a denied domain fails, writes to
its workspace, and exits. Reports per-loop cold-start latency and
whole-run throughput; asserts zero leaks.

Usage: python tools/soak_flagship.py [concurrency] [rounds] [out.json]
"""
from __future__ import annotations

import http.server
import json
import os
import statistics
import sys
import threading
import time
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def setup_dirs() -> Path:
    base = Path(os.environ.get("TMPDIR", "/tmp")) / "clawker-flagship-soak"
    for var, sub in [
        ("CLAWKER_CONFIG_DIR", "config"), ("CLAWKER_DATA_DIR", "data"),
        ("CLAWKER_STATE_DIR", "state"), ("CLAWKER_RUNTIME_DIR", "run"),
        ("CLAWKER_IMAGE_DIR", "images"), ("CLAWKER_SANDBOX_DIR", "sandboxes"),
        ("CLAWKER_VOLUME_DIR", "volumes"),
    ]:
        d = base / sub
        d.mkdir(parents=True, exist_ok=True)
        os.environ[var] = str(d)
    return base


AGENT = r"""
import json, os, socket, urllib.request
assert os.getuid() != 0, "agent must not be root"
out = {"uid": os.getuid()}
with urllib.request.urlopen("http://allowed.test:%PORT%/ok", timeout=15) as r:
    out["allowed"] = r.status
try:
    urllib.request.urlopen("http://denied.test:%PORT%/", timeout=15)
    out["denied"] = "LEAK"
except Exception:
    out["denied"] = "blocked"
open("/workspace/out-%TAG%.json", "w").write(json.dumps(out))
print("RESULT " + json.dumps(out), flush=True)
"""


def main() -> int:
    conc = int(sys.argv[1]) if len(sys.argv) > 1 else 8
    rounds = int(sys.argv[2]) if len(sys.argv) > 2 else 10
    out_path = sys.argv[3] if len(sys.argv) > 3 else "/tmp/flagship-soak.json"
    setup_dirs()
    os.environ["CLAWKER_DNS_STATIC"] = ("allowed.test=127.0.0.1,"
                                        "denied.test=127.0.0.1")

    class H(http.server.BaseHTTPRequestHandler):
        protocol_version = "HTTP/1.1"   # keep-alive: no reconnect storm

        def do_GET(self):
            body = b"OK"
            self.send_response(200)
            self.send_header("Content-Length", "2")
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    class Srv(http.server.ThreadingHTTPServer):
        request_queue_size = 128        # 8-way bursts overflow the
                                        # default backlog of 5 -> resets

    srv = Srv(("127.0.0.1", 0), H)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    port = srv.server_address[1]

    ws = Path(os.environ["CLAWKER_STATE_DIR"]) / "ws"
    ws.mkdir(exist_ok=True)
    (ws / ".clawker.yaml").write_text("project: flagship\n")
    os.chown(ws, 1000, 1000)

    from clawker_amd.config import load_config
    from clawker_amd.config.schema import EgressRule
    from clawker_amd.firewall import EgressRulesStore
    from clawker_amd.orchestrator import Orchestrator, RunOptions

    EgressRulesStore().add(
        [EgressRule(dst="allowed.test", proto="http", port=port)])
    orch = Orchestrator(load_config(ws))
    assert orch.engine.backend == "ns", "flagship soak needs the ns backend"

    lats: list[float] = []
    failures: list[str] = []
    lock = threading.Lock()

    def loop(tag: str) -> None:
        name = f"clawker.flagship.{tag}"
        t0 = time.perf_counter()
        try:
            info = orch.run(RunOptions(
                agent=tag, name=name, autostart=False, firewall=True,
                cmd=["python3", "-c",
                     AGENT.replace("%PORT%", str(port)).replace("%TAG%", tag)]))
            rundir = info.rundir
            deadline = time.monotonic() + 15
            while time.monotonic() < deadline:
                if (rundir / "egress.sock").exists():
                    break
                time.sleep(0.01)
            with orch.client(name) as c:
                c.agent_ready()
            t1 = time.perf_counter()
            code = orch.engine.wait(name, timeout_s=60)
            logs = orch.engine.logs(name).decode()
            assert code == 0, logs[-300:]
            if "RESULT " not in logs:
                raise AssertionError(f"no RESULT line: {logs[-300:]!r}")
            res = json.loads(logs.split("RESULT ", 1)[1].splitlines()[0])
            assert res["uid"] != 0 and res["allowed"] == 200
            assert res["denied"] == "blocked"
            assert (ws / f"out-{tag}.json").exists()
            with lock:
                lats.append((t1 - t0) * 1000)
        except Exception as e:  # noqa: BLE001
            with lock:
                failures.append(f"{tag}: {e}")
        finally:
            try:
                orch.teardown(name, force=True)
            except Exception:
                pass

    t_start = time.perf_counter()
    for r in range(rounds):
        ts = [threading.Thread(target=loop, args=(f"r{r}w{w}",))
              for w in range(conc)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        done = (r + 1) * conc
        print(f"round {r + 1}/{rounds}: {done} loops, "
              f"{len(failures)} failures", flush=True)
    wall = time.perf_counter() - t_start

    leaks = {
        "sandbox_rows": len(orch.engine.list()),
        "gpu_allocs": sum(1 for _ in getattr(orch.allocator, "ledger", [])
                          ) if hasattr(orch.allocator, "ledger") else 0,
        "rundirs": len(list((Path(os.environ["CLAWKER_RUNTIME_DIR"])
                             / "sandboxes").glob("clawker.flagship.*"))),
    }
    from clawker_amd.controlplane.client import CPClient
    CPClient(auto_start=False).stop()
    orch.close()
    srv.shutdown()

    result = {
        "config": "ns backend + enforcing firewall + unprivileged agent "
                  "(uid 1000) + MITM trust env + workspace-owner uid",
        "concurrency": conc,
        "rounds": rounds,
        "loops_total": conc * rounds,
        "failures": failures[:10],
        "failure_count": len(failures),
        "cold_start_ms": {
            "p50": round(statistics.median(lats), 2) if lats else None,
            "p95": round(sorted(lats)[max(0, int(len(lats) * .95) - 1)], 2)
                   if lats else None,
            "max": round(max(lats), 2) if lats else None,
        },
        "loops_per_min": round(conc * rounds / wall * 60, 1),
        "wall_s": round(wall, 1),
        "leaks": leaks,
    }
    Path(out_path).write_text(json.dumps(result, indent=1))
    print(json.dumps(result))
    return 0 if not failures and leaks["sandbox_rows"] == 0 else 1


if __name__ == "__main__":
    sys.exit(main())
