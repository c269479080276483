"""Typed errors (analog of reference python/framework/errors_impl.py); the
pybind layer surfaces C++ Status as RuntimeError("Error(N): msg") which
session.run re-raises as the matching class here."""
import re

OK = 0
CANCELLED = 1
UNKNOWN = 2
INVALID_ARGUMENT = 3
DEADLINE_EXCEEDED = 4
NOT_FOUND = 5
ALREADY_EXISTS = 6
PERMISSION_DENIED = 7
RESOURCE_EXHAUSTED = 8
FAILED_PRECONDITION = 9
ABORTED = 10
OUT_OF_RANGE = 11
UNIMPLEMENTED = 12
INTERNAL = 13
UNAVAILABLE = 14
DATA_LOSS = 15


class OpError(Exception):
    def __init__(self, message, error_code=UNKNOWN):
        super().__init__(message)
        self.message = message
        self.error_code = error_code


class CancelledError(OpError):
    pass


class UnknownError(OpError):
    pass


class InvalidArgumentError(OpError):
    pass


class DeadlineExceededError(OpError):
    pass


class NotFoundError(OpError):
    pass


class AlreadyExistsError(OpError):
    pass


class PermissionDeniedError(OpError):
    pass


class ResourceExhaustedError(OpError):
    pass


class FailedPreconditionError(OpError):
    pass


class AbortedError(OpError):
    pass


class OutOfRangeError(OpError):
    pass


class UnimplementedError(OpError):
    pass


class InternalError(OpError):
    pass


class UnavailableError(OpError):
    pass


class DataLossError(OpError):
    pass



_CODE_TO_CLASS = {
    CANCELLED: CancelledError,
    UNKNOWN: UnknownError,
    INVALID_ARGUMENT: InvalidArgumentError,
    DEADLINE_EXCEEDED: DeadlineExceededError,
    NOT_FOUND: NotFoundError,
    ALREADY_EXISTS: AlreadyExistsError,
    PERMISSION_DENIED: PermissionDeniedError,
    RESOURCE_EXHAUSTED: ResourceExhaustedError,
    FAILED_PRECONDITION: FailedPreconditionError,
    ABORTED: AbortedError,
    OUT_OF_RANGE: OutOfRangeError,
    UNIMPLEMENTED: UnimplementedError,
    INTERNAL: InternalError,
    UNAVAILABLE: UnavailableError,
    DATA_LOSS: DataLossError,
}


def raise_from_message(msg):
    m = re.match(r'Error\((\d+)\): (.*)', msg, re.S)
    if m:
        code = int(m.group(1))
        cls = _CODE_TO_CLASS.get(code, UnknownError)
        raise cls(m.group(2), code)
    raise UnknownError(msg)
