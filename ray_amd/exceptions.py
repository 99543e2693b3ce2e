"""Public exception types (reference: python/ray/exceptions.py)."""
from __future__ import annotations

import traceback


class RayError(Exception):
    """Base class for ray_amd errors."""


class RayTaskError(RayError):
    """A task raised an exception; re-raised at `ray.get`.

    Mirrors the reference's RayTaskError: carries the remote traceback
    and the original cause when it could be serialized.
    """

    def __init__(self, function_name="", traceback_str="", cause=None):
        self.function_name = function_name
        self.traceback_str = traceback_str
        self.cause = cause
        super().__init__(
            f"task {function_name} failed:\n{traceback_str}"
            if traceback_str
            else f"task {function_name} failed"
        )

    def as_instanceof_cause(self):
        return self

    @staticmethod
    def from_exception(function_name: str, exc: BaseException) -> "RayTaskError":
        tb = "".join(traceback.format_exception(type(exc), exc, exc.__traceback__))
        return RayTaskError(function_name, tb, exc)


class RayActorError(RayError):
    """The actor died before or during this call."""

    def __init__(self, msg="actor died", actor_id=None):
        self.actor_id = actor_id
        super().__init__(msg)


class ActorDiedError(RayActorError):
    pass


class ActorUnavailableError(RayActorError):
    pass


class WorkerCrashedError(RayError):
    pass


class ObjectLostError(RayError):
    def __init__(self, object_id_hex=""):
        super().__init__(f"object {object_id_hex} is lost")


class ObjectFreedError(ObjectLostError):
    pass


class GetTimeoutError(RayError, TimeoutError):
    pass


class TaskCancelledError(RayError):
    pass


class RuntimeEnvSetupError(RayError):
    pass


class RaySystemError(RayError):
    pass


class OutOfMemoryError(RayError):
    pass
