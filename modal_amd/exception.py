"""Exception hierarchy for modal_amd.

Mirrors the user-visible error taxonomy of the reference SDK
(/root/reference/py/modal/exception.py) so that code written against the
reference keeps working: ``Error`` is the catch-all base, with typed
subclasses for auth, invalid usage, missing resources, timeouts, and
remote execution failure.
"""

from __future__ import annotations


class Error(Exception):
    """Base class for all framework errors."""


class RemoteError(Error):
    """An error on the "server" (scheduler/worker) side."""


class FunctionTimeoutError(TimeoutError, Error):
    """A function invocation exceeded its configured timeout."""


class SandboxTimeoutError(TimeoutError, Error):
    """A sandbox exceeded its configured timeout."""


class SandboxTerminatedError(Error):
    """The sandbox was terminated before the operation completed."""


class TimeoutError(TimeoutError, Error):  # noqa: A001 - parity with reference name
    """Generic operation timeout."""


class OutputExpiredError(Error):
    """The requested output has expired and is no longer available."""


class AuthError(Error):
    """Credentials are missing or invalid (kept for API parity; local runs rarely hit it)."""


class ConnectionError(Error):  # noqa: A001
    """Could not reach the scheduler/worker plane."""


class InvalidError(Error):
    """User constructed or used an object incorrectly."""


class VersionError(Error):
    """Version requirement not met."""


class NotFoundError(Error):
    """Referenced a resource that does not exist."""


class AlreadyExistsError(Error):
    """Tried to create a resource that already exists."""


class ExecutionError(Error):
    """Internal error during execution of a function."""


class DeserializationError(Error):
    """Failed to deserialize a payload (version skew, missing modules, ...)."""


class SerializationError(Error):
    """Failed to serialize a payload."""


class RequestSizeError(Error):
    """A request payload exceeded size limits."""


class DeprecationError(UserWarning):
    """Deprecated API use (warning category, raised as error in strict mode)."""


class PendingDeprecationError(UserWarning):
    """Soon-to-be-deprecated API use."""


class ServerWarning(UserWarning):
    """Warning issued by the scheduler."""


class InternalFailure(Error):
    """Retryable internal failure: the input should be rescheduled.

    Parity: the reference retries inputs that fail with
    GENERIC_STATUS_INTERNAL_FAILURE up to 8 times without counting against the
    user retry policy (/root/reference/py/modal/_functions.py:106).
    """


class ClientClosed(Error):
    """Operation attempted on a closed client/runtime."""


class InputCancellation(BaseException):
    """Raised inside user code when its input is cancelled.

    BaseException (not Error) so that bare ``except Exception`` in user code
    does not swallow a cancellation; parity with the reference's
    modal.exception.InputCancellation semantics.
    """


class ModuleNotMountable(Error):
    """A Python module could not be packaged for a worker."""


import queue as _stdlib_queue


class QueueEmptyError(Error, _stdlib_queue.Empty):
    """Non-blocking Queue.get on an empty queue.

    Subclasses stdlib ``queue.Empty`` for drop-in parity: the reference
    raises ``queue.Empty`` (reference queue.py), so user code ported from it
    catching ``queue.Empty`` keeps working."""


class QueueFullError(Error, _stdlib_queue.Full):
    """Queue partition is at capacity (5,000 items per partition).

    Subclasses stdlib ``queue.Full`` for drop-in parity with the reference."""
