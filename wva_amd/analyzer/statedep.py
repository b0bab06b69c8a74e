"""M/M/1 queue with state-dependent service rates (birth-death chain).

Parity with /root/reference/pkg/analyzer/mm1modelstatedependent.go, with one
deliberate improvement: the product-form state probabilities

    p[n+1] = p[n] * lambda / mu(n)

are computed in **log space** and normalized with a softmax instead of the
reference's MaxFloat-rescale loops (mm1modelstatedependent.go:70-116).  The
normalized distribution is mathematically identical; log space cannot
overflow/underflow for any K or utilization, so no rescale machinery is
needed.  The same formulation is used by the native batched solver
(wva_amd/csrc/queue_solver.*) — cumulative sums are embarrassingly parallel.
"""

from __future__ import annotations

from typing import Sequence

import numpy as np

from .queuemodel import MM1KModel


class MM1ModelStateDependent(MM1KModel):
    """Birth-death chain: constant arrival rate, service rate mu(n) taken from
    ``serv_rate[n]`` for n < len(serv_rate) and flat beyond the batch limit.
    """

    def __init__(self, K: int, serv_rate: Sequence[float]) -> None:
        super().__init__(K)
        sr = np.asarray(serv_rate, dtype=np.float64)
        if sr.ndim != 1 or sr.size < 1:
            raise ValueError("serv_rate must be a non-empty 1-D sequence")
        self.serv_rate = sr
        self.avg_num_in_servers: float = 0.0
        # log of the per-state service rate over states 0..K-1 (flat tail)
        n = sr.size
        ext = np.empty(self.K, dtype=np.float64)
        ext[: min(n, self.K)] = sr[: self.K]
        if self.K > n:
            ext[n:] = sr[-1]
        with np.errstate(divide="ignore"):
            self._log_mu = np.log(ext)

    def compute_rho(self) -> float:
        # utilization = P[server busy]; reads the previous solve's p[0]
        # (initially zero -> rho=1), matching the reference's call ordering.
        return 1.0 - float(self.p[0])

    def _compute_probabilities(self) -> None:
        # log p[n] = n*log(lambda) - sum_{i<n} log(mu(i))   (up to a constant)
        K = self.K
        logp = np.empty(K + 1, dtype=np.float64)
        logp[0] = 0.0
        with np.errstate(divide="ignore", invalid="ignore"):
            logp[1:] = np.arange(1, K + 1, dtype=np.float64) * np.log(self.lam) - np.cumsum(self._log_mu)
        # softmax normalization
        m = np.max(logp)
        if not np.isfinite(m):
            # lambda == 0: all mass at state 0
            self.p = np.zeros(K + 1, dtype=np.float64)
            self.p[0] = 1.0
        else:
            ex = np.exp(logp - m)
            self.p = ex / ex.sum()
        self.sum_p = float(self.p.sum())
        self.rho = self.compute_rho()

    def _compute_statistics(self) -> None:
        if not self.is_valid:
            return
        self._compute_probabilities()
        num = int(self.serv_rate.size)
        idx = np.arange(self.K + 1, dtype=np.float64)
        weighted = idx * self.p
        self.avg_num_in_system = float(weighted.sum())
        if num <= self.K:
            # E[#in service] = sum_{i<=num} i p[i] + num * P[occupancy > num]
            self.avg_num_in_servers = float(weighted[: num + 1].sum()) + (
                1.0 - float(self.p[: num + 1].sum())
            ) * num
        else:
            self.avg_num_in_servers = self.avg_num_in_system
        self.throughput = self.lam * (1.0 - float(self.p[self.K]))
        if self.throughput == 0.0:
            # idle queue: no departures, latencies are zero by convention
            self.avg_resp_time = self.avg_serv_time = self.avg_wait_time = 0.0
            self.avg_queue_length = 0.0
            return
        self.avg_resp_time = self.avg_num_in_system / self.throughput
        self.avg_serv_time = self.avg_num_in_servers / self.throughput
        self.avg_wait_time = max(self.avg_resp_time - self.avg_serv_time, 0.0)
        self.avg_queue_length = self.throughput * self.avg_wait_time

    def get_avg_num_in_servers(self) -> float:
        return self.avg_num_in_servers
