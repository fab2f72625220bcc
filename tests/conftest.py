import os
import sys
from pathlib import Path

import pytest

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires a real MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session", autouse=True)
def _ensure_native_built():
    """Most tests drive real sandboxes; build the native runtime once."""
    import subprocess
    root = Path(__file__).resolve().parent.parent
    missing = [b for b in ("ckrt", "ckd", "ckgw")
               if not (root / "native" / "bin" / b).exists()]
    if missing or not list(root.glob("clawker_amd/_native*.so")):
        subprocess.run(["make", "native", "pymod", "-j4"], cwd=root,
                       check=True, capture_output=True)


@pytest.fixture
def isolated_env(tmp_path, monkeypatch):
    """Fully isolated clawker state dirs (reference: internal/testenv)."""
    for var, sub in [
        ("CLAWKER_CONFIG_DIR", "config"),
        ("CLAWKER_DATA_DIR", "data"),
        ("CLAWKER_STATE_DIR", "state"),
        ("CLAWKER_RUNTIME_DIR", "run"),
        ("CLAWKER_IMAGE_DIR", "images"),
        ("CLAWKER_SANDBOX_DIR", "sandboxes"),
        ("CLAWKER_VOLUME_DIR", "volumes"),
    ]:
        d = tmp_path / sub
        d.mkdir(exist_ok=True)
        monkeypatch.setenv(var, str(d))
    yield tmp_path
    # the image store may have mounted a layer tmpfs (overlayfs nesting
    # rule); unmount so pytest can clean its tmp dir
    import subprocess
    subprocess.run(["umount", "-l", str(tmp_path / "images" / "layers")],
                   capture_output=True)


def _isolation_available() -> bool:
    """Can we create real sandboxes here (root + namespaces + overlayfs)?"""
    if os.geteuid() != 0:
        return False
    import subprocess
    r = subprocess.run(["unshare", "-pmf", "true"], capture_output=True)
    return r.returncode == 0


ISOLATION = _isolation_available()

requires_isolation = pytest.mark.skipif(
    not ISOLATION, reason="needs root + namespace support for real sandboxes")
