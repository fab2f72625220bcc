"""Fleet: N parallel agent loops over git worktrees with 1:1 GPU pinning.

The BASELINE.json north star (configs 3-5): "worktree/workspace fan-out
schedules N concurrent agent loops with 1:1 GPU affinity". Reference
building blocks: worktree add + run --worktree; there the
fan-out was manual — here `clawker fleet up -n 8` is first-class: each
agent gets its own worktree branch, its own sandbox, and its own MI355X
from the allocator (exclusive), with the live dashboard on top.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field

from .config import Config
from .errors import ClawkerError
from .logger import get as get_logger
from .orchestrator import Orchestrator, RunOptions
from .project.worktrees import is_git_repo, setup_worktree

log = get_logger("fleet")


@dataclass
class FleetOptions:
    count: int = 2
    branch_prefix: str = "agent"
    gpus_per_agent: int | None = None    # None => project gpu.count
    cmd: list[str] = field(default_factory=list)
    env: dict = field(default_factory=dict)
    image: str = ""
    firewall: bool | None = None
    use_worktrees: bool | None = None    # None => auto (git repo present)
    base: str = ""                       # worktree start point
    prompt_file: str = ""                # one-shot prompt handed to each agent


@dataclass
class FleetMember:
    agent: str
    sandbox: str
    branch: str = ""
    gpus: list[int] = field(default_factory=list)
    state: str = "created"
    exit_code: int | None = None


class Fleet:
    def __init__(self, cfg: Config, orch: Orchestrator | None = None):
        self.cfg = cfg
        self.orch = orch or Orchestrator(cfg)

    def up(self, opts: FleetOptions) -> list[FleetMember]:
        """Create + start N agent sandboxes; returns members. On partial
        failure, already-started members keep running (caller decides)."""
        use_wt = opts.use_worktrees
        if use_wt is None:
            use_wt = (self.cfg.project_root is not None
                      and is_git_repo(self.cfg.project_root))
        members: list[FleetMember] = []
        prompt_cmd: list[str] = []
        if opts.prompt_file and not opts.cmd:
            from .bundle import load_harness
            harness = load_harness(self.cfg.project.agent.harness,
                                   self.cfg.project_root)
            if not harness.prompt_cmd:
                raise ClawkerError(
                    f"harness '{harness.name}' declares no prompt_cmd")
            prompt_cmd = [a.replace("@PROMPT_FILE@", "/run/clawker/prompt.md")
                          for a in harness.prompt_cmd]
        for i in range(opts.count):
            agent = f"{opts.branch_prefix}{i}"
            ropts = RunOptions(
                agent=agent, cmd=list(opts.cmd) or list(prompt_cmd),
                env=dict(opts.env),
                image=opts.image, gpus=opts.gpus_per_agent,
                firewall=opts.firewall, autostart=False)
            branch = ""
            if use_wt:
                wt = setup_worktree(self.cfg, f"{opts.branch_prefix}/{i}", opts.base)
                ropts.workspace = wt.path
                branch = wt.branch
            else:
                # disposable copies so parallel agents never collide
                ropts.workspace_mode = "snapshot"
            info = self.orch.create(ropts)
            if opts.prompt_file:
                import shutil as _sh
                _sh.copy2(opts.prompt_file, info.rundir / "prompt.md")
            self.orch.start(info.name)
            # drive init/boot plans then release the CMD (reference: the CP
            # Executor path — fan-out members get the same boot contract)
            from .controlplane.plans import drive_boot
            with self.orch.client(info.name) as c:
                drive_boot(c, c.hello())
                c.agent_ready()
            members.append(FleetMember(
                agent=agent, sandbox=info.name, branch=branch, gpus=info.gpus,
                state=info.state))
            log.info("fleet_member_up", sandbox=info.name, gpus=info.gpus)
        return members

    def status(self, branch_prefix: str = "agent") -> list[FleetMember]:
        out = []
        for info in self.orch.engine.list(project=self.cfg.project_slug):
            if not info.agent.startswith(branch_prefix):
                continue
            out.append(FleetMember(
                agent=info.agent, sandbox=info.name, gpus=info.gpus,
                state=info.state, exit_code=info.exit_code))
        return out

    def wait(self, members: list[FleetMember], timeout_s: float = 3600.0) -> list[FleetMember]:
        deadline = time.monotonic() + timeout_s
        for m in members:
            remaining = deadline - time.monotonic()
            if remaining <= 0:
                raise ClawkerError("fleet wait timed out")
            m.exit_code = self.orch.engine.wait(m.sandbox, timeout_s=remaining)
            m.state = "exited"
        return members

    def down(self, branch_prefix: str = "agent", remove: bool = True) -> int:
        n = 0
        for m in self.status(branch_prefix):
            if remove:
                self.orch.teardown(m.sandbox, force=True)
            else:
                self.orch.engine.stop(m.sandbox)
            n += 1
        return n
