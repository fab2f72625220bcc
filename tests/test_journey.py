"""Full user-journey integration test: the README quickstart end to end
(init -> build -> run @ -> exec/cp/stats -> fleet -> cleanup)."""
import json
import os
import subprocess
import sys
import time
from pathlib import Path

import pytest

from conftest import requires_isolation

pytestmark = requires_isolation

REPO = Path(__file__).resolve().parent.parent


def test_full_user_journey(isolated_env, tmp_path):
    proj = tmp_path / "journey"
    proj.mkdir()
    (proj / "app.py").write_text("print('hello from workspace')\n")
    env = dict(os.environ, PYTHONPATH=str(REPO))

    def clawker(*args, timeout=180, check=True):
        r = subprocess.run([sys.executable, "-m", "clawker_amd", *args],
                           capture_output=True, text=True, timeout=timeout,
                           cwd=str(proj), env=env)
        if check:
            assert r.returncode == 0, (args, r.stdout[-800:], r.stderr[-800:])
        return r

    # 1. init with the echo harness + python stack preset
    clawker("init", "--yes", "--name", "journey", "--harness", "echo",
            "--preset", "python", "--vcs", "github")
    # 2. doctor is happy
    r = clawker("doctor", "--format", "json")
    assert all(c["ok"] for c in json.loads(r.stdout)
               if c["check"] in ("root", "namespaces", "overlayfs", "native runtime"))
    # 3. build the project image (base + harness overlay layers)
    r = clawker("build", timeout=300)
    assert "clawker-journey:echo" in r.stdout
    r = clawker("image", "ls", "--format", "json")
    names = [m["name"] for m in json.loads(r.stdout)]
    assert {"clawker-journey:base", "clawker-journey:echo",
            "clawker-journey:default"} <= set(names)
    # 4. run the project image detached with the firewall on
    clawker("run", "-d", "--agent", "main", "@", "--",
            "/bin/sh", "-c", "python3 /workspace/app.py; sleep 45")
    r = clawker("ps", "--format", "json")
    rows = json.loads(r.stdout)
    assert any(x["name"] == "clawker.journey.main" and
               x["image"] == "clawker-journey:echo" for x in rows)
    # image provides the plan scripts; boot hooks ran
    r = clawker("exec", "main", "--", "/bin/cat", "/tmp/clawker-hooks.log")
    assert "post-init hook executed" in r.stdout
    assert "pre-run hook executed" in r.stdout
    # workspace visible
    deadline = time.monotonic() + 15
    while time.monotonic() < deadline:
        if "hello from workspace" in clawker("logs", "main").stdout:
            break
        time.sleep(0.2)
    assert "hello from workspace" in clawker("logs", "main").stdout
    # 5. firewall status + rules from the init VCS merge
    r = clawker("firewall", "list", "--format", "json")
    assert any(x["dst"] == "github.com" for x in json.loads(r.stdout))
    # 6. copy a file out
    clawker("cp", "main:/etc/clawker/egress-floor.yaml", str(proj / "copied"))
    assert (proj / "copied" / "egress-floor.yaml").exists()
    # 7. stats snapshot includes the sandbox
    r = clawker("stats", "--no-stream")
    assert "clawker.journey.main" in r.stdout
    # 8. control plane knows the agent
    r = clawker("controlplane", "agents")
    assert "clawker.journey.main" in r.stdout
    # 9. teardown
    clawker("rm", "-f", "main")
    r = clawker("ps", "-a", "--format", "json")
    assert json.loads(r.stdout) == []
    clawker("controlplane", "down")
