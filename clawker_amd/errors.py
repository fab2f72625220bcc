"""Typed error hierarchy (reference: internal/cmdutil typed errors +
pkg/whail/errors.go DockerError user formatting)."""
from __future__ import annotations


class ClawkerError(Exception):
    """Base class: carries a user-facing message and an exit code."""

    exit_code = 1

    def user_message(self) -> str:
        return str(self)


class FlagError(ClawkerError):
    """Bad CLI usage; usage help should be shown."""
    exit_code = 2


class SilentError(ClawkerError):
    """Error already rendered; just exit non-zero."""
    exit_code = 1

    def user_message(self) -> str:
        return ""


class ExitError(ClawkerError):
    """Propagate the agent process's exit code (bash convention 128+sig)."""

    def __init__(self, code: int, message: str = ""):
        super().__init__(message)
        self.exit_code = code


class NotFoundError(ClawkerError):
    """Managed resource not found (a caller cannot distinguish 'absent' from
    'unmanaged' — reference: whail managed-label filter, engine.go:152)."""
    exit_code = 1


class ConflictError(ClawkerError):
    exit_code = 1


class EngineError(ClawkerError):
    """Sandbox runtime failure with context for the user."""

    def __init__(self, op: str, detail: str):
        super().__init__(f"{op}: {detail}")
        self.op = op
        self.detail = detail


class CPSOSError(ClawkerError):
    """Control-plane boot failure that the CLI can assist with
    (reference: controlplane/manager CPSOSError)."""

    def __init__(self, reason: str, assist: str):
        super().__init__(reason)
        self.assist = assist
