"""Init/Boot plans: the staged commands the control plane drives through a
sandbox's ckd before releasing the agent CMD.

Reference: controlplane/agent/init_steps.go:67 (one-time InitPlan:
config seed-apply, gitconfig filter, git-credentials, ssh known_hosts,
post_init, AgentInitialized marker) and boot_steps.go:52 (every-start
BootPlan: pre_run, AgentReady releases the CMD). Step progress events are
emitted so the CLI can render a boot banner (reference: clawkerd
progress.go parsing command_ids back into TTY step banners).

Plan scripts live at fixed in-image paths baked by the bundler:
  /etc/clawker/seed-apply.sh   (InitPlan, optional)
  /etc/clawker/post-init.sh    (InitPlan, optional)
  /etc/clawker/pre-run.sh      (BootPlan, optional)
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Callable

from ..engine import CkdClient
from ..errors import ClawkerError
from ..logger import get as get_logger

log = get_logger("plans")

SEED_APPLY = "/etc/clawker/seed-apply.sh"
POST_INIT = "/etc/clawker/post-init.sh"
PRE_RUN = "/etc/clawker/pre-run.sh"


@dataclass
class Step:
    name: str
    argv: list[str]
    user: str = ""          # "" = root (init steps run privileged)
    required: bool = False  # required steps abort the plan on failure
    skip_if_missing: str = ""   # skip when this in-sandbox path is absent


def _exists_in_sandbox(client: CkdClient, path: str) -> bool:
    import shlex
    code, _, _ = client.exec(
        [{"argv": ["/bin/sh", "-c", f"test -e {shlex.quote(path)}"]}])
    return code == 0


def run_plan(client: CkdClient, steps: list[Step],
             on_step: Callable[[str, int], None] | None = None,
             on_start: Callable[[str], None] | None = None) -> None:
    for step in steps:
        if step.skip_if_missing and not _exists_in_sandbox(client, step.skip_if_missing):
            continue
        if on_start:
            on_start(step.name)
        stage: dict = {"argv": step.argv}
        if step.user:
            stage["user"] = step.user
        code, out, err = client.exec([stage])
        if on_step:
            on_step(step.name, code)
        log.info("plan_step_done", step=step.name, code=code)
        if code != 0 and step.required:
            raise ClawkerError(
                f"init step '{step.name}' failed ({code}): "
                f"{(err or out)[-400:].decode(errors='replace')}")


def init_plan() -> list[Step]:
    return [
        Step("seed-apply", ["/bin/sh", SEED_APPLY], skip_if_missing=SEED_APPLY),
        Step("post-init", ["/bin/sh", POST_INIT], skip_if_missing=POST_INIT),
    ]


def boot_plan() -> list[Step]:
    return [
        Step("pre-run", ["/bin/sh", PRE_RUN], skip_if_missing=PRE_RUN),
    ]


def drive_boot(client: CkdClient, hello: dict,
               on_step: Callable[[str, int], None] | None = None,
               on_start: Callable[[str], None] | None = None) -> None:
    """Drive InitPlan (first boot only) then BootPlan, matching the
    reference's CP dial flow (SURVEY.md §3.1 lower half)."""
    if not hello.get("initialized"):
        run_plan(client, init_plan(), on_step, on_start)
        client.agent_initialized()
    run_plan(client, boot_plan(), on_step, on_start)


def run_boot_plans(factory, name: str, client: CkdClient, hello: dict,
                   quiet: bool = False) -> None:
    def on_start(step: str) -> None:
        if not quiet:
            factory.io.eprint(f"  [dim]»[/dim] {step} ...")

    def on_step(step: str, code: int) -> None:
        if not quiet:
            mark = "[green]✓[/green]" if code == 0 else "[yellow]![/yellow]"
            factory.io.eprint(f"  {mark} {step}")

    drive_boot(client, hello, None if quiet else on_step,
               None if quiet else on_start)
