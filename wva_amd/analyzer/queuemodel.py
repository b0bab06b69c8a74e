"""Base queueing model and the classic finite-buffer M/M/1/K model.

Behavioral parity with /root/reference/pkg/analyzer/queuemodel.go and
mm1kmodel.go, expressed as an ordinary class hierarchy (the reference
emulates virtual dispatch with function-pointer fields).
"""

from __future__ import annotations

import numpy as np


class QueueModel:
    """Abstract single-queue model solved for a (lambda, mu) pair.

    Subclasses override :meth:`compute_rho`, :meth:`rho_max` and
    :meth:`_compute_statistics`.

    Validity semantics mirror the reference (queuemodel.go:27-37): ``rho`` is
    computed *before* statistics — for state-dependent models this reads the
    probabilities of the previous solve (initially zero), which is part of the
    de-facto contract.
    """

    def __init__(self) -> None:
        self.lam: float = 0.0
        self.mu: float = 0.0
        self.rho: float = 0.0
        self.avg_resp_time: float = 0.0
        self.avg_wait_time: float = 0.0
        self.avg_serv_time: float = 0.0
        self.avg_num_in_system: float = 0.0
        self.avg_queue_length: float = 0.0
        self.is_valid: bool = False

    # -- overridables -------------------------------------------------------
    def compute_rho(self) -> float:
        raise NotImplementedError

    def rho_max(self) -> float:
        raise NotImplementedError

    def _compute_statistics(self) -> None:
        raise NotImplementedError

    # -- public API ---------------------------------------------------------
    def solve(self, lam: float, mu: float) -> None:
        self.lam = float(lam)
        self.mu = float(mu)
        self.rho = self.compute_rho()
        if self.rho < 0 or self.rho >= self.rho_max() or lam < 0 or mu <= 0:
            self.is_valid = False
        else:
            self.is_valid = True
            self._compute_statistics()

    def __repr__(self) -> str:  # diagnostic, mirrors String()
        s = f"isValid={self.is_valid}; lambda={self.lam}; mu={self.mu}; rho={self.rho}; "
        if self.is_valid:
            s += (
                f"T={self.avg_resp_time}; W={self.avg_wait_time}; X={self.avg_serv_time}; "
                f"N={self.avg_num_in_system}; Q={self.avg_queue_length}; "
            )
        return f"{type(self).__name__}: {s}"


class MM1KModel(QueueModel):
    """Finite-buffer M/M/1/K queue with geometric state probabilities.

    Parity with mm1kmodel.go: throughput = lambda * (1 - p[K]), response time
    by Little's law, service time 1/mu, waiting time clamped at zero.
    """

    def __init__(self, K: int) -> None:
        super().__init__()
        if K < 1:
            raise ValueError(f"invalid occupancy bound K={K}")
        self.K = int(K)
        self.p = np.zeros(self.K + 1, dtype=np.float64)
        self.sum_p: float = 0.0
        self.throughput: float = 0.0

    def compute_rho(self) -> float:
        if self.lam == self.mu:
            return 1.0
        return self.lam / self.mu if self.mu != 0 else 0.0

    def rho_max(self) -> float:
        return float(self.K)

    def _compute_probabilities(self) -> None:
        K, rho = self.K, float(self.rho)
        if rho == 1.0:
            p0 = 1.0 / (K + 1)
        else:
            p0 = (1.0 - rho) / (1.0 - rho ** (K + 1))
        self.p = p0 * np.power(rho, np.arange(K + 1, dtype=np.float64))
        self.sum_p = float(self.p.sum())

    def _compute_statistics(self) -> None:
        if not self.is_valid:
            return
        self._compute_probabilities()
        idx = np.arange(self.K + 1, dtype=np.float64)
        self.avg_num_in_system = float((idx * self.p).sum())
        self.throughput = self.lam * (1.0 - float(self.p[self.K]))
        self.avg_resp_time = self.avg_num_in_system / self.throughput
        self.avg_serv_time = 1.0 / self.mu
        self.avg_wait_time = max(self.avg_resp_time - self.avg_serv_time, 0.0)
        self.avg_queue_length = self.throughput * self.avg_wait_time

    def get_probabilities(self) -> np.ndarray:
        return self.p
