"""Kubernetes API error classes — the apimachinery `apierrors` analog
(the reference relies on client-go's IsConflict/IsNotFound/IsGone
predicates inside its retry helpers, internal/utils/utils.go:58-104)."""


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


class GoneError(KubeError):
    """410 Gone: the requested resourceVersion is older than the server's
    retained watch history (apimachinery's 'Expired').  Not retryable
    in-place — the caller must re-list and restart the watch from the
    fresh list resourceVersion."""

    retryable = False
