"""Error types.

Mirrors the reference's status-code idiom (src/common/error/src/status_code.rs)
with a flat Python exception hierarchy instead of per-crate snafu enums.
"""

from __future__ import annotations


class GreptimeError(Exception):
    """Base error; every engine error carries a status code name."""

    code = "Internal"

    def __init__(self, msg: str = "", *, cause: Exception | None = None):
        super().__init__(msg)
        self.cause = cause


class InvalidArguments(GreptimeError):
    code = "InvalidArguments"


class TableNotFound(GreptimeError):
    code = "TableNotFound"


class TableAlreadyExists(GreptimeError):
    code = "TableAlreadyExists"


class RegionNotFound(GreptimeError):
    code = "RegionNotFound"


class InvalidSyntax(GreptimeError):
    code = "InvalidSyntax"


class PlanQuery(GreptimeError):
    code = "PlanQuery"


class EngineExecuteQuery(GreptimeError):
    code = "EngineExecuteQuery"


class StorageUnavailable(GreptimeError):
    code = "StorageUnavailable"


class NativeExtensionMissing(GreptimeError):
    """Raised when a GPU is present but the HIP extension failed to load.

    We never silently fall back to eager PyTorch on a GPU box: the HIP path
    must be the one that runs (or fail loudly).
    """

    code = "Internal"


class RegionFenced(GreptimeError):
    """Write rejected: region downgraded (migration write fence)."""


class QueryCancelled(GreptimeError):
    """Query terminated by KILL (reference: process manager cancellation)."""
