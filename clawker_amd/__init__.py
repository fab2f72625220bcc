"""clawker-amd: an MI355X-native agent-in-container orchestrator.

A from-scratch rebuild of the capabilities of schmitthub/clawker (reference:
Go CLI + Docker + eBPF) designed for one dedicated rootful
8xMI355X ROCm node:

- its own native container runtime (``native/ckrt``, C++): namespaces +
  overlayfs-over-hostfs images + cgroup limits + amdgpu device rules --
  no Docker daemon, no nvidia-container-toolkit, no multi-runtime dispatch;
- a PID-1 supervisor per sandbox (``native/ckd``, C++) with a framed control
  socket (the reference's clawkerd gRPC Session analog, clawkerd/session.go);
- deny-by-default egress *by construction*: every sandbox lives in a network
  namespace with no uplink; allowed egress flows through a userspace policy
  gateway (DNS + SNI-aware TCP) over Unix sockets (the reference's
  eBPF + Envoy + CoreDNS datapath, controlplane/firewall/);
- 1:1 GPU pinning for agent fan-out: /dev/kfd + /dev/dri/renderD<N>
  passthrough, device-cgroup rules, ROCR_VISIBLE_DEVICES, HBM budgets;
- zero-spawn ROCm telemetry: a native sysfs/KFD sampler (``_native`` ext)
  feeding live TUI panes and a Prometheus exporter (replacing the
  reference's `docker stats` loop and compose monitoring stack).
"""

__version__ = "0.2.0"
