"""Sandbox spec: the contract between the Python engine and the native
runtime (ckrt/ckd read this as JSON; see native/ckrt/ckrt.cpp parse_spec).
Analog of the reference's ContainerCreateOptions -> docker HostConfig
translation (container_create.go:2254 buildContainerConfigs)."""
from __future__ import annotations

import dataclasses
import json
from dataclasses import dataclass, field
from pathlib import Path


@dataclass
class Mount:
    src: str = ""
    dst: str = ""
    ro: bool = False
    type: str = "bind"        # bind | tmpfs
    opts: str = ""


@dataclass
class Device:
    path: str = ""


@dataclass
class SandboxSpec:
    name: str = ""
    backend: str = "ns"        # ns (namespaces+overlay) | proc (no-namespace host)
    rundir: str = ""
    lowerdirs: list[str] = field(default_factory=list)   # top-most first
    upper: str = ""
    work: str = ""
    merged: str = ""
    hostname: str = "clawker"
    netns: bool = True
    tty: bool = False
    autostart: bool = False    # True: ckd spawns CMD immediately (no CP gate)
    mounts: list[Mount] = field(default_factory=list)
    devices: list[Device] = field(default_factory=list)
    mem_bytes: int = 0
    pids_max: int = 4096
    device_allow_only: bool = True
    env: dict[str, str] = field(default_factory=dict)
    user: str = ""             # "", "root", "name", or "uid:gid"
    # preferred ids when a named user must be materialized into the
    # overlay upper (ns backend): the workspace owner's ids keep
    # bind-mounted files writable without idmap mounts
    uid_hint: int = 0
    gid_hint: int = 0
    workdir: str = "/"
    cmd: list[str] = field(default_factory=list)
    labels: dict[str, str] = field(default_factory=dict)
    # ckd path contract: where the control socket/console/ready live from
    # ckd's point of view, and where the one-time init marker persists
    paths: dict[str, str] = field(default_factory=dict)
    # auxiliary in-sandbox daemons ckd supervises (e.g. ckgw gateway shims)
    services: list[dict] = field(default_factory=list)
    # restart policy (reference: docker-style on-failure:N)
    restart_policy: str = "no"       # no | on-failure
    restart_max: int = 3

    def to_json(self) -> str:
        d = {
            "name": self.name,
            "backend": self.backend,
            "rundir": self.rundir,
            "paths": self.paths,
            "services": self.services,
            "restart": {"policy": self.restart_policy, "max": self.restart_max},
            "rootfs": {
                "lowerdirs": self.lowerdirs,
                "upper": self.upper,
                "work": self.work,
                "merged": self.merged,
            },
            "hostname": self.hostname,
            "netns": self.netns,
            "tty": self.tty,
            "autostart": self.autostart,
            "mounts": [dataclasses.asdict(m) for m in self.mounts],
            "devices": [dataclasses.asdict(d) for d in self.devices],
            "cgroup": {
                "mem_bytes": self.mem_bytes,
                "pids": self.pids_max,
                "device_allow_only": self.device_allow_only,
            },
            "env": self.env,
            "user": self.user,
            "workdir": self.workdir,
            "cmd": self.cmd,
            "labels": self.labels,
        }
        return json.dumps(d, indent=1)

    def write(self, path: Path) -> None:
        path.write_text(self.to_json())
