from .schema import (  # noqa: F401
    AgentConfig,
    BuildConfig,
    EgressRule,
    FirewallSettings,
    GPUConfig,
    LoggingSettings,
    MonitoringSettings,
    Project,
    SecurityConfig,
    Settings,
    WorkspaceConfig,
)
from .config import Config, load_config, load_settings  # noqa: F401
