from .manager import HostProxyManager  # noqa: F401
