"""Monotone bisection with boundary-region classification.

Parity with /root/reference/pkg/analyzer/utils.go:26-70 — including the exact
boundary semantics that drive "no feasible allocation" decisions upstream:

- returns (x, -1) == *below region* when the target is under the reachable
  range of a monotonically increasing function (or over, for decreasing);
- returns (x, +1) == *above region* when the target exceeds the range (the
  caller then uses x == x_max);
- returns (x, 0) when found within relative tolerance.

Unlike the reference there is no package-global model: ``eval_fn`` is any
callable, typically a closure over a model instance.
"""

from __future__ import annotations

from typing import Callable, Tuple

TOLERANCE = 1e-6
MAX_ITERATIONS = 100

BelowRegion = -1
InRegion = 0
AboveRegion = 1


class EvalError(RuntimeError):
    """The search function could not be evaluated (e.g. invalid model)."""


def within_tolerance(x: float, value: float, tolerance: float) -> bool:
    if x == value:
        return True
    if value == 0 or tolerance < 0:
        return False
    return abs((x - value) / value) <= tolerance


def binary_search(
    x_min: float,
    x_max: float,
    y_target: float,
    eval_fn: Callable[[float], float],
    *,
    tolerance: float = TOLERANCE,
    max_iterations: int = MAX_ITERATIONS,
) -> Tuple[float, int]:
    """Find x* in [x_min, x_max] with eval_fn(x*) ~= y_target.

    ``eval_fn`` must be monotone over the range; it may raise
    :class:`EvalError` (or any exception) to abort the search.
    Returns ``(x_star, indicator)`` with indicator in {-1, 0, +1}.
    """
    if x_min > x_max:
        raise ValueError(f"invalid range [{x_min}, {x_max}]")

    y_bounds = []
    for x in (x_min, x_max):
        y = eval_fn(x)
        if within_tolerance(y, y_target, tolerance):
            return x, InRegion
        y_bounds.append(y)

    if within_tolerance(y_bounds[0], y_bounds[1], tolerance):
        # flat function: direction detection would be decided by rounding
        # noise (e.g. a batch-1 queue's ITL curve); classify by value only
        if y_target > max(y_bounds):
            return x_max, AboveRegion
        return x_min, BelowRegion

    increasing = y_bounds[0] < y_bounds[1]
    if (increasing and y_target < y_bounds[0]) or (not increasing and y_target > y_bounds[0]):
        return x_min, BelowRegion
    if (increasing and y_target > y_bounds[1]) or (not increasing and y_target < y_bounds[1]):
        return x_max, AboveRegion

    x_star = x_min
    for _ in range(max_iterations):
        x_star = 0.5 * (x_min + x_max)
        y_star = eval_fn(x_star)
        if within_tolerance(y_star, y_target, tolerance):
            break
        if (increasing and y_target < y_star) or (not increasing and y_target > y_star):
            x_max = x_star
        else:
            x_min = x_star
    return x_star, InRegion
