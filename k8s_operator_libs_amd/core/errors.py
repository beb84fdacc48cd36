"""Typed Kubernetes API errors (apimachinery ``k8s.io/apimachinery/pkg/api/errors`` analogue)."""

from __future__ import annotations


class ApiError(Exception):
    """Base class; carries an HTTP-style status code."""

    code = 500

    def __init__(self, message: str = "") -> None:
        super().__init__(message or self.__class__.__name__)
        self.message = message


class NotFoundError(ApiError):
    code = 404


class AlreadyExistsError(ApiError):
    code = 409


class ConflictError(ApiError):
    """Optimistic-concurrency failure (stale resourceVersion)."""

    code = 409


class BadRequestError(ApiError):
    code = 400


class InvalidError(ApiError):
    """Schema-invalid object (HTTP 422, status reason ``Invalid``) — what a
    real apiserver returns when a custom resource violates its CRD's
    structural openAPIV3Schema."""

    code = 422


class GoneError(ApiError):
    """Watch resume window expired (HTTP 410, status reason ``Expired``) —
    the client must relist and re-watch from the fresh resourceVersion."""

    code = 410


def is_not_found(exc: BaseException) -> bool:
    return isinstance(exc, NotFoundError)


def is_conflict(exc: BaseException) -> bool:
    return isinstance(exc, ConflictError)
