# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Error hierarchy for mlrun_amd.

Mirrors the error surface of the reference (mlrun/errors.py) so user code
catching e.g. ``MLRunNotFoundError`` keeps working, but is written fresh
for the MI355X-native framework.
"""


class MLRunBaseError(Exception):
    """Base for all framework errors."""


class MLRunBadRequestError(MLRunBaseError):
    pass


class MLRunInvalidArgumentError(MLRunBadRequestError, ValueError):
    """Invalid user input (maps to HTTP 400, like the reference's
    MLRunInvalidArgumentError -> BadRequest)."""


class MLRunNotFoundError(MLRunBaseError):
    pass


class MLRunConflictError(MLRunBaseError):
    pass


class MLRunAccessDeniedError(MLRunBaseError):
    pass


class MLRunRuntimeError(MLRunBaseError, RuntimeError):
    pass


class MLRunTimeoutError(MLRunBaseError, TimeoutError):
    pass


class MLRunPreconditionFailedError(MLRunBaseError):
    pass


class MLRunMissingDependencyError(MLRunBaseError, ImportError):
    """An optional integration's package is not installed."""


class MLRunIncompatibleVersionError(MLRunBaseError):
    pass


class MLRunTaskCancelledError(MLRunBaseError):
    pass


class MLRunHTTPError(MLRunBaseError):
    """An error that carries an HTTP status code (service <-> client)."""

    def __init__(self, message: str = "", status_code: int = 500):
        super().__init__(message)
        self.status_code = status_code


class MLRunGPUError(MLRunRuntimeError):
    """Raised when a GPU-native op is requested but the HIP extension or a
    GPU device is unavailable.  GPU ops must fail loudly rather than fall
    back silently to eager CPU execution."""


STATUS_ERRORS = {
    400: MLRunBadRequestError,
    403: MLRunAccessDeniedError,
    404: MLRunNotFoundError,
    409: MLRunConflictError,
    412: MLRunPreconditionFailedError,
}


def err_for_status(status_code: int, message: str = "") -> MLRunBaseError:
    cls = STATUS_ERRORS.get(status_code)
    if cls is not None:
        return cls(message)
    return MLRunHTTPError(message, status_code=status_code)


def err_to_status(exc: Exception) -> int:
    for code, cls in STATUS_ERRORS.items():
        if isinstance(exc, cls):
            return code
    if isinstance(exc, MLRunHTTPError):
        return exc.status_code
    return 500
