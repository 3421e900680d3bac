"""Exception types, API-compatible with the reference's ray.exceptions
(reference: python/ray/exceptions.py)."""
from __future__ import annotations

import traceback


class RayError(Exception):
    """Base class for runtime errors."""


class RayTaskError(RayError):
    """Wraps an exception raised inside a remote task/actor method.

    Re-raised at the `ray.get` site. `cause` carries the original exception
    (when picklable); `traceback_str` the remote traceback text.
    """

    def __init__(self, function_name="", traceback_str="", cause=None):
        self.function_name = function_name
        self.traceback_str = traceback_str
        self.cause = cause
        super().__init__(
            f"{type(cause).__name__ if cause else 'Error'} in {function_name}()\n"
            f"{traceback_str}"
        )

    @classmethod
    def from_exception(cls, exc: BaseException, function_name: str):
        tb = "".join(traceback.format_exception(type(exc), exc, exc.__traceback__))
        return cls(function_name=function_name, traceback_str=tb, cause=exc)

    def as_instanceof_cause(self):
        """Return an exception that is also an instance of the cause's type."""
        cause = self.cause
        if cause is None or isinstance(cause, RayTaskError):
            return self
        try:
            cls = type(
                "RayTaskError(" + type(cause).__name__ + ")",
                (RayTaskError, type(cause)),
                {},
            )
            instance = cls.__new__(cls)
            RayTaskError.__init__(
                instance, self.function_name, self.traceback_str, cause
            )
            return instance
        except TypeError:
            return self


class RayActorError(RayError):
    """The actor died (creation failed, crashed, or was killed)."""

    def __init__(self, message="The actor died unexpectedly before finishing this task."):
        super().__init__(message)


class ActorDiedError(RayActorError):
    pass


class ActorUnavailableError(RayActorError):
    pass


class GetTimeoutError(RayError, TimeoutError):
    """ray.get timed out."""


class TaskCancelledError(RayError):
    def __init__(self, task_id=None):
        self.task_id = task_id
        super().__init__("This task or its dependency was cancelled")


class ObjectLostError(RayError):
    def __init__(self, object_ref_hex=""):
        self.object_ref_hex = object_ref_hex
        super().__init__(f"Object {object_ref_hex} is lost")


class ObjectStoreFullError(RayError):
    pass


class OutOfMemoryError(RayError):
    pass


class RuntimeEnvSetupError(RayError):
    pass


class OwnerDiedError(ObjectLostError):
    pass


class RaySystemError(RayError):
    pass


class CrossLanguageError(RayError):
    pass


class PendingCallsLimitExceeded(RayError):
    """Raised when an actor handle with max_pending_calls set already has
    that many calls outstanding (parity: ray.exceptions
    .PendingCallsLimitExceeded)."""
