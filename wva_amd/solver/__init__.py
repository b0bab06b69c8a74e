"""Global min-cost (replica x accelerator-type) assignment (layer L4).

Parity with /root/reference/pkg/solver/: unlimited-mode separable argmin
(the production path), greedy limited mode with regret-based reordering and
all four saturation policies, and a timing Optimizer wrapper — here the
solve wall-clock is also exported as a Prometheus histogram because it is
the headline BASELINE metric.
"""

from .solver import Solver
from .optimizer import Optimizer
from .manager import Manager

__all__ = ["Solver", "Optimizer", "Manager"]
