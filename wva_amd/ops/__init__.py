"""Native batched queue-solver ops.

Loads the in-tree extension ``wva_amd._queue_native`` (built by setup.py via
hipcc for gfx950).  Policy:

- on a machine with a GPU, a missing/broken native extension is a hard
  error — the HIP path must be the path that actually runs;
- on CPU-only machines the pure-Python analyzer fallback is allowed (it is
  the semantic reference the native code is tested against).
"""

from __future__ import annotations

_native = None
_native_err: Exception | None = None

try:  # pragma: no cover - import side effect
    import torch  # noqa: F401  (the extension links against libtorch)

    from wva_amd import _queue_native as _native  # type: ignore
except Exception as e:  # pragma: no cover
    _native_err = e

# torch-free CPU binding (pybind11/numpy/OpenMP) — what the slim
# controller image builds; same host solver (csrc/queue_host.h)
_native_cpu = None
_native_cpu_err: Exception | None = None
try:  # pragma: no cover - import side effect
    from wva_amd import _queue_native_cpu as _native_cpu  # type: ignore
except Exception as e:  # pragma: no cover
    _native_cpu_err = e

_fallback_warned = False


def warn_if_pure_python_fallback() -> None:
    """One-time log when sizing runs pure-Python despite a broken extension.

    A missing-by-design extension (dev checkout, never built) is silent; a
    present-but-unloadable one (e.g. the libgomp1 case from the r01 image)
    is surfaced so the degradation is diagnosable (advisor r01 low).
    """
    global _fallback_warned
    if _fallback_warned:
        return
    _fallback_warned = True
    err = _native_err if _native_err is not None else _native_cpu_err
    if err is not None:
        import logging

        logging.getLogger("wva_amd.ops").warning(
            "native sizing extension failed to import (%s: %s); "
            "falling back to the pure-Python solver",
            type(err).__name__, err,
        )


def native_available() -> bool:
    return _native is not None


def native_cpu_available() -> bool:
    return _native_cpu is not None


def get_native_cpu():
    return _native_cpu


def get_native():
    """Return the native module, failing loudly when a GPU is present."""
    if _native is not None:
        return _native
    import torch

    if torch.cuda.is_available():
        raise RuntimeError(
            "wva_amd._queue_native is not importable on a GPU machine — "
            "build it with `PYTORCH_ROCM_ARCH=gfx950 python setup.py "
            f"build_ext --inplace` (import error: {_native_err})"
        )
    return None


from .batched import BatchedAllocationSolver, solve_problems  # noqa: E402

__all__ = [
    "native_available",
    "native_cpu_available",
    "get_native",
    "get_native_cpu",
    "BatchedAllocationSolver",
    "solve_problems",
]
