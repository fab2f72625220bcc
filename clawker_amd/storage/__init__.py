from .store import (  # noqa: F401
    Layer,
    MergeResult,
    Store,
    dataclass_defaults,
    discover_project_layers,
    materialize,
    merge_layers,
    to_plain,
)
