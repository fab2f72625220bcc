"""Cross-cutting constants (reference: internal/consts/consts.go).

Every persisted name, path, label and env var lives here so tests can pin
them (the reference keeps a persisted_values_test for the same reason).
"""
from __future__ import annotations

import os
from pathlib import Path

APP_NAME = "clawker"

# ---------------------------------------------------------------- paths ----
# XDG-style directories; CLAWKER_* env overrides exist so tests (and the
# sandboxed GPU box) can fully isolate state (reference: internal/testenv).


def _xdg(env: str, default: str) -> Path:
    v = os.environ.get(env)
    return Path(v) if v else Path.home() / default


def config_dir() -> Path:
    v = os.environ.get("CLAWKER_CONFIG_DIR")
    if v:
        return Path(v)
    return _xdg("XDG_CONFIG_HOME", ".config") / APP_NAME


def data_dir() -> Path:
    v = os.environ.get("CLAWKER_DATA_DIR")
    if v:
        return Path(v)
    return _xdg("XDG_DATA_HOME", ".local/share") / APP_NAME


def state_dir() -> Path:
    v = os.environ.get("CLAWKER_STATE_DIR")
    if v:
        return Path(v)
    return _xdg("XDG_STATE_HOME", ".local/state") / APP_NAME


def runtime_dir() -> Path:
    """Per-boot runtime state: sandbox dirs, control sockets, pidfiles.

    /run/clawker on a real node; overridable for tests and unprivileged runs.
    """
    v = os.environ.get("CLAWKER_RUNTIME_DIR")
    if v:
        return Path(v)
    if os.geteuid() == 0 and os.access("/run", os.W_OK):
        return Path("/run") / APP_NAME
    return Path(os.environ.get("TMPDIR", "/tmp")) / f"{APP_NAME}-run-{os.geteuid()}"


def image_store_dir() -> Path:
    """Content-addressed overlay layer store."""
    v = os.environ.get("CLAWKER_IMAGE_DIR")
    if v:
        return Path(v)
    return data_dir() / "images"


def sandbox_store_dir() -> Path:
    """Per-sandbox persistent dirs (upper layers, volumes)."""
    v = os.environ.get("CLAWKER_SANDBOX_DIR")
    if v:
        return Path(v)
    return data_dir() / "sandboxes"


def volume_store_dir() -> Path:
    v = os.environ.get("CLAWKER_VOLUME_DIR")
    if v:
        return Path(v)
    return data_dir() / "volumes"


def log_dir() -> Path:
    return state_dir() / "logs"


# project-level config file names (reference: internal/storage/discover.go —
# `.clawker/` dir wins over `.clawker.yaml` dotfile)
PROJECT_DIR_NAME = ".clawker"
PROJECT_FILE_NAME = ".clawker.yaml"
PROJECT_CONFIG_BASENAME = "clawker.yaml"
PROJECT_LOCAL_BASENAME = "clawker.local.yaml"
SETTINGS_BASENAME = "settings.yaml"
REGISTRY_BASENAME = "registry.yaml"
EGRESS_RULES_BASENAME = "egress-rules.yaml"
ROUTE_IDENTITIES_BASENAME = "route-identities.yaml"
IGNORE_FILE_NAME = ".clawkerignore"

# ---------------------------------------------------------------- naming ----
# sandbox name: clawker.<project>.<agent>  (reference: internal/docker/names.go)
SANDBOX_NAME_PREFIX = "clawker."
MANAGED_LABEL = "dev.clawker.managed"
PROJECT_LABEL = "dev.clawker.project"
AGENT_LABEL = "dev.clawker.agent"
HARNESS_LABEL = "dev.clawker.harness"
GPU_LABEL = "dev.clawker.gpu"
TEST_LABEL = "dev.clawker.test"

# ---------------------------------------------------------------- env -------
# container runtime env contract (reference: internal/docker/env.go:64,
# consts.go:813-838). The MI355X build adds the GPU vars.
ENV_PROJECT = "CLAWKER_PROJECT"
ENV_AGENT = "CLAWKER_AGENT"
ENV_WORKSPACE_MODE = "CLAWKER_WORKSPACE_MODE"
ENV_WORKSPACE_SOURCE = "CLAWKER_WORKSPACE_SOURCE"
ENV_USER = "CLAWKER_USER"
ENV_VERSION = "CLAWKER_VERSION"
ENV_FIREWALL = "CLAWKER_FIREWALL"
ENV_CP_SOCK = "CLAWKER_CP_SOCK"
ENV_GPU_INDEX = "CLAWKER_GPU"
ENV_ROCR_VISIBLE = "ROCR_VISIBLE_DEVICES"
ENV_HIP_VISIBLE = "HIP_VISIBLE_DEVICES"

# ---------------------------------------------------------------- sockets ---
# All control links are Unix sockets in runtime_dir() (single-node appliance;
# the reference used mTLS gRPC because links crossed a docker network).
CKD_SOCK_NAME = "ctl.sock"          # per-sandbox ckd control socket
CP_ADMIN_SOCK = "cp-admin.sock"     # CLI -> control-plane daemon
CP_EVENTS_SOCK = "cp-events.sock"   # pub/sub event stream
HOSTPROXY_SOCK = "hostproxy.sock"   # in-sandbox -> host services
DNSD_SOCK = "dnsd.sock"             # sandbox DNS stub -> policy resolver
EGRESSD_SOCK = "egressd.sock"       # sandbox TCP shim -> policy gateway
HOSTPROXY_PORT = 18374              # loopback HTTP fallback (reference port)

# ---------------------------------------------------------------- devices ---
KFD_DEV = "/dev/kfd"
DRI_DIR = "/dev/dri"
# amdgpu char device majors: kfd is dynamic (misc, major 10), renderD* is drm
# major 226; resolved at runtime from stat(2) — these are fallbacks.
DRM_MAJOR = 226
MISC_MAJOR = 10

# ---------------------------------------------------------------- limits ----
MI355X_HBM_BYTES = 288 * 1024**3     # 288 GB HBM3E per GPU
MI355X_GPUS_PER_NODE = 8

# ---------------------------------------------------------------- files -----
READY_FILE = "/var/run/clawker/ready"       # in-sandbox (image HEALTHCHECK analog)
BOOTSTRAP_DIR = "/run/clawker/bootstrap"    # in-sandbox bootstrap material
INIT_MARKER = "/var/lib/clawker/initialized"  # one-time InitPlan marker
