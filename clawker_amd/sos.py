"""SOS diagnostic bundle (reference: internal/controlplane WatchSOS +
CLI-assisted recovery, cmd.go SOS hooks). When something is wrong on a
node, `clawker doctor --collect` gathers everything a human (or an
agent) needs to debug it into one tarball: host capability checks,
sandbox inventory + per-sandbox status/exit/audit/console tails, control
plane status + recent events, GPU inventory/allocations, and the
clawker log tail. Nothing secret is included: credentials, tokens and
MITM keys are explicitly excluded."""
from __future__ import annotations

import dataclasses
import io
import json
import tarfile
import time
from pathlib import Path

from . import consts
from .logger import get as get_logger

log = get_logger("sos")

# per-sandbox rundir files worth shipping (never auth/ or *.key)
_RUNDIR_FILES = ("status.json", "exit.json", "spec.json", "audit.jsonl")
_TAIL_BYTES = 64 * 1024


def _tail(path: Path, n: int = _TAIL_BYTES) -> bytes:
    try:
        data = path.read_bytes()
        return data[-n:]
    except OSError:
        return b""


def collect_bundle(dest: Path | None = None) -> Path:
    """Write clawker-sos-<ts>.tar.gz under dest (default CWD); returns
    the tarball path."""
    ts = time.strftime("%Y%m%d-%H%M%S")
    dest = Path(dest) if dest else Path.cwd()
    dest.mkdir(parents=True, exist_ok=True)
    out = dest / f"clawker-sos-{ts}.tar.gz"

    def add_bytes(tar: tarfile.TarFile, name: str, data: bytes) -> None:
        info = tarfile.TarInfo(name)
        info.size = len(data)
        info.mtime = int(time.time())
        tar.addfile(info, io.BytesIO(data))

    def add_json(tar: tarfile.TarFile, name: str, obj) -> None:
        add_bytes(tar, name, json.dumps(obj, indent=1, default=str).encode())

    with tarfile.open(out, "w:gz") as tar:
        # host capability checks (doctor)
        try:
            from .cli.doctor import run_checks
            add_json(tar, "doctor.json", run_checks())
        except Exception as e:
            add_bytes(tar, "doctor.error", str(e).encode())

        # sandbox inventory + per-sandbox forensics
        try:
            from .engine import Engine
            eng = Engine()
            infos = eng.list()
            add_json(tar, "sandboxes.json",
                     [dataclasses.asdict(i) for i in infos])
            for i in infos:
                base = f"sandboxes/{i.name}"
                rundir = Path(i.rundir)
                for fn in _RUNDIR_FILES:
                    p = rundir / fn
                    if p.is_file():
                        add_bytes(tar, f"{base}/{fn}", _tail(p))
                add_bytes(tar, f"{base}/console.log",
                          eng.logs(i.name)[-_TAIL_BYTES:])
            eng.close()
        except Exception as e:
            add_bytes(tar, "sandboxes.error", str(e).encode())

        # control plane status + recent events (without starting it)
        try:
            from .controlplane.client import CPClient
            cp = CPClient(auto_start=False)
            if cp.running():
                add_json(tar, "controlplane/status.json", cp.status())
                add_json(tar, "controlplane/events.json", cp.events(500))
            else:
                add_bytes(tar, "controlplane/status.json",
                          b'{"running": false}')
        except Exception as e:
            add_bytes(tar, "controlplane/error", str(e).encode())
        from .controlplane.daemon import events_path
        ev = events_path()
        if ev.is_file():
            add_bytes(tar, "controlplane/cp-events.jsonl", _tail(ev))

        # GPU inventory + allocation ledger
        try:
            from .gpu import GPUAllocator, GPUInventory
            inv = GPUInventory.detect()
            add_json(tar, "gpu/inventory.json",
                     [dataclasses.asdict(d) for d in inv.devices])
            if inv.devices:
                add_json(tar, "gpu/allocations.json",
                         GPUAllocator(inv).allocations())
        except Exception as e:
            add_bytes(tar, "gpu/error", str(e).encode())

        # host log tail
        logf = consts.state_dir() / "logs" / "clawker.log"
        if logf.is_file():
            add_bytes(tar, "clawker.log", _tail(logf))

        add_json(tar, "meta.json", {
            "ts": ts, "version": __import__("clawker_amd").__version__,
            "data_dir": str(consts.data_dir()),
            "runtime_dir": str(consts.runtime_dir()),
        })
    log.info("sos_bundle", path=str(out))
    return out
