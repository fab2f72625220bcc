"""MI355X GPU inventory via sysfs (no rocm-smi process spawn).

No reference analog (the reference orchestrates CPU containers via Docker);
this is the BASELINE.json north-star surface: enumerate the node's amdgpu
devices (/sys/class/drm/renderD*), their VRAM (288 GB HBM3E on MI355X),
PCI identity and xGMI links, so the allocator can hand out 1:1 pinned
devices and the spec builder can emit /dev/kfd + /dev/dri/renderD<N>
passthrough.
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field
from pathlib import Path

AMD_VENDOR = "0x1002"
DRM_CLASS = Path("/sys/class/drm")
KFD_NODES = Path("/sys/class/kfd/kfd/topology/nodes")


@dataclass
class GPUDevice:
    index: int                  # logical index (render minor order)
    render_minor: int           # /dev/dri/renderD<minor>
    card: int                   # /dev/dri/card<N> (may be -1)
    pci_bus: str = ""
    vram_total: int = 0
    unique_id: str = ""
    device_id: str = ""
    xgmi_peers: list[int] = field(default_factory=list)   # logical indices

    @property
    def render_path(self) -> str:
        return f"/dev/dri/renderD{self.render_minor}"

    @property
    def card_path(self) -> str | None:
        return f"/dev/dri/card{self.card}" if self.card >= 0 else None

    def device_paths(self) -> list[str]:
        out = [self.render_path]
        if self.card_path:
            out.append(self.card_path)
        return out


def _read(p: Path) -> str:
    try:
        return p.read_text().strip()
    except OSError:
        return ""


class GPUInventory:
    """Snapshot of the node's AMD GPUs. ``GPUInventory.detect()``."""

    def __init__(self, devices: list[GPUDevice]):
        self.devices = devices

    @classmethod
    def detect(cls, drm_class: Path = DRM_CLASS) -> "GPUInventory":
        devs: list[GPUDevice] = []
        if not drm_class.is_dir():
            return cls(devs)
        renders = sorted(
            (p for p in drm_class.iterdir() if re.fullmatch(r"renderD\d+", p.name)),
            key=lambda p: int(p.name[7:]))
        idx = 0
        for r in renders:
            dev = r / "device"
            if _read(dev / "vendor") != AMD_VENDOR:
                continue
            minor = int(r.name[7:])
            # containerized hosts expose the full host sysfs but only a
            # subset of /dev/dri nodes — a GPU without an accessible render
            # node is unusable, so it is not inventory
            if not Path(f"/dev/dri/renderD{minor}").exists():
                continue
            # find the cardN sharing this PCI device
            card = -1
            pci = ""
            try:
                pci = (dev.resolve()).name          # e.g. 0000:0c:00.0
            except OSError:
                pass
            for c in drm_class.iterdir():
                if re.fullmatch(r"card\d+", c.name):
                    try:
                        if (c / "device").resolve().name == pci:
                            card = int(c.name[4:])
                            break
                    except OSError:
                        continue
            vram = _read(dev / "mem_info_vram_total")
            devs.append(GPUDevice(
                index=idx, render_minor=minor, card=card, pci_bus=pci,
                vram_total=int(vram) if vram.isdigit() else 0,
                unique_id=_read(dev / "unique_id"),
                device_id=_read(dev / "device")))
            idx += 1
        inv = cls(devs)
        inv._detect_xgmi()
        return inv

    def _detect_xgmi(self) -> None:
        """xGMI adjacency from KFD topology io_links (type XGMI/IOLINK)."""
        if not KFD_NODES.is_dir():
            return
        # map KFD node -> render minor (gpu nodes expose drm_render_minor)
        kfd_to_dev: dict[int, GPUDevice] = {}
        node_of_minor: dict[int, int] = {}
        for node in KFD_NODES.iterdir():
            props = _read(node / "properties")
            m = re.search(r"drm_render_minor (\d+)", props)
            if not m:
                continue
            minor = int(m.group(1))
            for d in self.devices:
                if d.render_minor == minor:
                    kfd_to_dev[int(node.name)] = d
                    node_of_minor[minor] = int(node.name)
        for node_id, dev in kfd_to_dev.items():
            links_dir = KFD_NODES / str(node_id) / "io_links"
            if not links_dir.is_dir():
                continue
            for link in links_dir.iterdir():
                lp = _read(link / "properties")
                tm = re.search(r"type (\d+)", lp)
                nm = re.search(r"node_to (\d+)", lp)
                if not nm:
                    continue
                # type 2 == XGMI in kfd_topology (11 == PCIe)
                if tm and tm.group(1) == "2":
                    peer = kfd_to_dev.get(int(nm.group(1)))
                    if peer is not None and peer.index != dev.index:
                        dev.xgmi_peers.append(peer.index)

    def __len__(self) -> int:
        return len(self.devices)

    def get(self, index: int) -> GPUDevice:
        for d in self.devices:
            if d.index == index:
                return d
        raise KeyError(f"no GPU with index {index}")

    def xgmi_adjacent_set(self, count: int, available: list[int]) -> list[int] | None:
        """Pick `count` mutually-xGMI-connected GPUs from `available`
        (for multi-GPU sandboxes; on one MI355X node all 8 are linked,
        so this matters for partial availability)."""
        avail = [self.get(i) for i in available]
        for d in avail:
            group = [d.index] + [p for p in d.xgmi_peers if p in available]
            if len(group) >= count:
                return group[:count]
        return None
