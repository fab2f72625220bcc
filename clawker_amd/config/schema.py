"""Project + Settings schemas.

Reference: internal/config/schema.go (Project :14 — build :58, agent :162,
workspace :231, security/egress :270-305; Settings :379 — control_plane
:405, firewall :424, logging :457, monitoring :490).

MI355X-first additions: a ``gpu`` section on both Project (what an agent
requests) and Settings (node inventory policy) — device passthrough count,
HBM budget, pinning policy. This replaces nothing in the reference; it is
the BASELINE.json north-star surface (per-container GPU cgroup pinning
sized for 8 devices / 288 GB HBM each).
"""
from __future__ import annotations

from dataclasses import dataclass, field


# ------------------------------------------------------------------ project -

@dataclass
class BuildConfig:
    """Image build inputs (reference: schema.go build section)."""
    image: str = ""                 # explicit image name; "" => clawker-<project>
    base: str = "hostfs"            # base image: hostfs | <image name>
    stacks: list[str] = field(default_factory=list, metadata={"merge": "union"})
    packages: list[str] = field(default_factory=list, metadata={"merge": "union"})
    env: dict = field(default_factory=dict)
    steps: list[str] = field(default_factory=list)   # extra Buildfile RUN steps
    user: str = "agent"             # in-sandbox user baked at build time


@dataclass
class AgentConfig:
    """What runs inside the sandbox (reference: schema.go agent section)."""
    harness: str = "claude"
    cmd: list[str] = field(default_factory=list)      # override harness CMD
    env: dict = field(default_factory=dict)
    env_file: str = ""
    managed_prompt: str = ""        # path to briefing file baked into image
    workdir: str = ""               # default: workspace mount point


@dataclass
class WorkspaceConfig:
    """Workspace strategy (reference: schema.go workspace :231;
    internal/workspace bind vs snapshot)."""
    mode: str = "bind"              # bind | snapshot
    path: str = ""                  # default: project root
    mount: str = "/workspace"       # in-sandbox mount point
    share_volume: bool = True       # shared scratch volume across agents


@dataclass
class EgressRule:
    """One egress allow rule (reference: schema.go egress rules :270;
    firewall rules_store.go RuleKey dst:proto:port)."""
    dst: str = ""                   # domain (wildcards: *.example.com) or IP
    proto: str = "tls"              # tls | http | tcp | udp
    port: int = 443
    paths: list[str] = field(default_factory=list, metadata={"merge": "union"})
    # path rules: allow-only these URL paths (MITM mode); "~" prefix = regex
    deny_paths: list[str] = field(default_factory=list, metadata={"merge": "union"})

    def key(self) -> str:
        return f"{self.dst}:{self.proto}:{self.port}"


@dataclass
class SecurityConfig:
    """Per-project security posture (reference: schema.go security :305)."""
    firewall: bool = True           # deny-by-default egress enforcement
    egress: list[EgressRule] = field(default_factory=list, metadata={"merge": "union"})
    add_domains: list[str] = field(default_factory=list, metadata={"merge": "union"})
    mount_docker_socket: bool = False   # kept for parity; no docker here
    raw_sockets: bool = False           # SOCK_RAW (ICMP) allowed?


@dataclass
class GPUConfig:
    """MI355X allocation request for this project's agents."""
    count: int = 0                  # GPUs per agent sandbox (0 = CPU-only)
    hbm_gb: int = 0                 # HBM budget per GPU (0 = whole 288 GB)
    exclusive: bool = True          # 1:1 pinning (no GPU sharing)
    prefer_xgmi_adjacent: bool = True   # >1 GPU: pick xGMI-adjacent set


@dataclass
class Project:
    """clawker.yaml (project layer)."""
    version: int = 1
    project: str = ""               # slug; default: dir name
    build: BuildConfig = field(default_factory=BuildConfig)
    agent: AgentConfig = field(default_factory=AgentConfig)
    workspace: WorkspaceConfig = field(default_factory=WorkspaceConfig)
    security: SecurityConfig = field(default_factory=SecurityConfig)
    gpu: GPUConfig = field(default_factory=GPUConfig)


# ----------------------------------------------------------------- settings -

@dataclass
class ControlPlaneSettings:
    """CP daemon knobs (reference: schema.go control_plane :405)."""
    auto_start: bool = True
    drain_to_zero: bool = True      # CP self-stops when last agent exits
    drain_grace_s: int = 30
    # a gated (autostart=False) sandbox whose starting client vanished
    # gets its Init/Boot plans driven by the CP after this grace
    # (reference: the CP-side executor model, init_steps.go:67)
    orphan_grace_s: int = 10


@dataclass
class FirewallSettings:
    """(reference: schema.go firewall :424)"""
    enable: bool = True
    dns_upstream: list[str] = field(default_factory=lambda: ["1.1.1.2", "1.0.0.2"])
    bypass_max_s: int = 3600        # dead-man cap on `firewall bypass`
    event_rate_limit: int = 640     # events/s per sandbox (burst 64)
    event_burst: int = 64


@dataclass
class LoggingSettings:
    """(reference: schema.go logging :457)"""
    level: str = "info"
    file_enabled: bool = True
    max_size_mb: int = 10
    max_backups: int = 3


@dataclass
class MonitoringSettings:
    """(reference: schema.go monitoring :490). MI355X: sampler cadence for
    the zero-spawn sysfs GPU sampler + Prometheus exporter."""
    enable: bool = True
    sample_interval_ms: int = 1000
    prometheus_port: int = 19090
    retention_hours: int = 72


@dataclass
class NodeGPUSettings:
    """Node-level GPU inventory policy (MI355X-new; no reference analog)."""
    devices: int = 0                # 0 = autodetect
    hbm_gb_per_device: int = 288
    reserve: list[int] = field(default_factory=list)    # indices never allocated
    # HBM budget enforcement: cgroups cannot cap VRAM, so the CP samples
    # per-sandbox drm fdinfo and enforces the budget itself
    # (docs/security.md): "kill" stops the sandbox at >100%, "warn" only
    # emits events, "off" disables the watchdog
    hbm_enforce: str = "kill"


@dataclass
class BundlerSettings:
    """Version-resolver endpoints (reference: bundler/versions.go npm +
    github-release resolvers). Empty = air-gapped; harness pins apply."""
    npm_registry: str = ""          # e.g. http://mirror:4873
    github_api: str = ""            # e.g. https://api.github.com


@dataclass
class Settings:
    """settings.yaml (global layer)."""
    version: int = 1
    bundler: BundlerSettings = field(default_factory=BundlerSettings)
    control_plane: ControlPlaneSettings = field(default_factory=ControlPlaneSettings)
    firewall: FirewallSettings = field(default_factory=FirewallSettings)
    logging: LoggingSettings = field(default_factory=LoggingSettings)
    monitoring: MonitoringSettings = field(default_factory=MonitoringSettings)
    gpu: NodeGPUSettings = field(default_factory=NodeGPUSettings)
    aliases: dict = field(default_factory=dict)
