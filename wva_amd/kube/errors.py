"""Kubernetes API error classes (apimachinery apierrors analog)."""


class KubeError(Exception):
    """Base class; retryable unless a subclass says otherwise."""

    retryable = True


class NotFoundError(KubeError):
    retryable = False


class InvalidError(KubeError):
    retryable = False


class ForbiddenError(KubeError):
    retryable = False


class ConflictError(KubeError):
    """resourceVersion conflict on update; retry with a fresh read."""

    retryable = True
