from .engine import Engine, SandboxInfo  # noqa: F401
from .spec import SandboxSpec, Mount, Device  # noqa: F401
from .ckd_client import CkdClient  # noqa: F401
