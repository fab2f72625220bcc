from .registry import ProjectEntry, ProjectRegistry, WorktreeEntry  # noqa: F401
