"""Typed exception hierarchy + structured error responses
(reference src/utils/exceptions.py:21-411)."""

from __future__ import annotations

import enum
import time
from typing import Any


class ErrorCode(enum.Enum):
    VALIDATION_ERROR = "VALIDATION_ERROR"
    AUTH_ERROR = "AUTH_ERROR"
    RATE_LIMITED = "RATE_LIMITED"
    NOT_FOUND = "NOT_FOUND"
    SERVICE_ERROR = "SERVICE_ERROR"
    PROCESSING_ERROR = "PROCESSING_ERROR"
    GPU_ERROR = "GPU_ERROR"
    SYSTEM_ERROR = "SYSTEM_ERROR"


class SentioException(Exception):
    code = ErrorCode.SYSTEM_ERROR
    status = 500

    def __init__(self, message: str, details: dict[str, Any] | None = None):
        super().__init__(message)
        self.message = message
        self.details = details or {}

    def to_dict(self) -> dict[str, Any]:
        return {
            "error": self.code.value,
            "message": self.message,
            "details": self.details,
            "timestamp": time.time(),
        }


class ValidationException(SentioException):
    code = ErrorCode.VALIDATION_ERROR
    status = 422


class AuthException(SentioException):
    code = ErrorCode.AUTH_ERROR
    status = 401


class RateLimitException(SentioException):
    code = ErrorCode.RATE_LIMITED
    status = 429


class NotFoundException(SentioException):
    code = ErrorCode.NOT_FOUND
    status = 404


class ServiceException(SentioException):
    code = ErrorCode.SERVICE_ERROR
    status = 503


class ProcessingException(SentioException):
    code = ErrorCode.PROCESSING_ERROR
    status = 500


class GPUException(SentioException):
    """HIP/device failures — the engine-level analogue of the reference's
    remote-service errors."""

    code = ErrorCode.GPU_ERROR
    status = 503
