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


class WorkerCrashedError(RayError):
    """The worker executing a task died unexpectedly (parity:
    reference exceptions.py WorkerCrashedError)."""


class LocalRayletDiedError(RayError):
    """This node's raylet died while the worker depended on it."""


class NodeDiedError(RayError):
    """The node a task/object lived on died."""


class TaskUnschedulableError(RayError):
    """The task can never be scheduled (infeasible resources after
    cluster-shape changes, or scheduling constraints)."""

    def __init__(self, error_message: str = ""):
        self.error_message = error_message
        super().__init__(error_message)


class ActorUnschedulableError(TaskUnschedulableError):
    """The actor can never be scheduled."""


class TaskPlacementGroupRemoved(RayError):
    """The placement group a queued task targeted was removed."""


class ActorPlacementGroupRemoved(RayError):
    """The placement group a pending actor targeted was removed."""


class ActorAlreadyExistsError(RayError):
    """A named actor with this name already exists
    (get_if_exists=False)."""


class AsyncioActorExit(RayError):
    """Internal marker used to unwind an async actor on exit_actor()."""


class ObjectFreedError(ObjectLostError):
    """The object was explicitly freed via ray.internal.free."""


class ObjectFetchTimedOutError(ObjectLostError):
    """Fetching the object from its holder timed out."""


class ObjectReconstructionFailedError(ObjectLostError):
    """Lineage reconstruction could not recompute the object."""


class ObjectReconstructionFailedMaxAttemptsExceededError(
        ObjectReconstructionFailedError):
    """Reconstruction gave up after max_retries resubmissions."""


class ObjectReconstructionFailedLineageEvictedError(
        ObjectReconstructionFailedError):
    """Reconstruction impossible: the lineage buffer evicted the
    producing task (see the bounded lineage cache in
    _private/worker.py)."""


class ReferenceCountingAssertionError(ObjectLostError, AssertionError):
    """The object was deleted while this process still held a
    reference — a reference-protocol invariant was violated."""


class ObjectRefStreamEndOfStreamError(RayError):
    """Internal: a streaming generator's stream is exhausted (surfaced
    as StopIteration/StopAsyncIteration to user code)."""


class OutOfDiskError(RayError):
    """The object store spill directory ran out of disk."""


class OufOfBandObjectRefSerializationException(RayError):
    """An ObjectRef was pickled outside ray serialization (name kept
    with the reference's spelling, typo included)."""


class PlasmaObjectNotAvailable(RayError):
    """The requested object bytes are not available in the local store."""


class UnserializableException(RayError):
    """The exception raised by user code could not be pickled; this
    carries its string form instead."""


class UserCodeException(RayError):
    """Wrapper marking that the failure originated in user code, not
    the runtime."""


class RpcError(RayError):
    """A low-level RPC failed (parity: reference RpcError; the
    transport's own error type lives in _private/protocol.py)."""

    def __init__(self, message: str = "", rpc_code: int = None):
        self.rpc_code = rpc_code
        super().__init__(message)


class RayChannelError(RaySystemError):
    """Compiled-DAG channel failure (parity: experimental.channel
    errors): the peer actor died or the channel was closed."""

    def __init__(self, message: str = ""):
        super().__init__(message)


class RayChannelTimeoutError(RayChannelError, TimeoutError):
    """Compiled-DAG channel read/write timed out."""


class RayCgraphCapacityExceeded(RaySystemError):
    """A compiled-graph channel's buffer capacity was exceeded."""


class AuthenticationError(RayError):
    """Cluster authentication failed (no auth layer in this air-gapped
    single-tenant build; exists for API parity)."""
