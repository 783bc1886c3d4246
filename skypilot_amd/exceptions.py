"""Typed error hierarchy (reference: sky/exceptions.py)."""
from __future__ import annotations


class SkyAmdError(Exception):
    """Base class for all framework errors."""


class PermissionDeniedError(SkyAmdError):
    """RBAC: the requesting user's role does not allow this operation."""


class ClusterNotUpError(SkyAmdError):
    pass


class ClusterDoesNotExist(SkyAmdError):
    pass


class ResourcesUnavailableError(SkyAmdError):
    """No feasible placement on the pool (reference:
    sky/exceptions.py ResourcesUnavailableError)."""


class ResourcesMismatchError(SkyAmdError):
    pass


class TaskValidationError(SkyAmdError):
    pass


class CommandError(SkyAmdError):
    def __init__(self, returncode: int, command: str, error_msg: str = ""):
        self.returncode = returncode
        self.command = command
        self.error_msg = error_msg
        super().__init__(
            f"command failed (exit {returncode}): {command[:200]} "
            f"{error_msg[:400]}")


class JobNotFoundError(SkyAmdError):
    pass


class ServeError(SkyAmdError):
    pass


class ManagedJobError(SkyAmdError):
    pass


class ApiServerError(SkyAmdError):
    pass


class RequestCancelled(SkyAmdError):
    pass
