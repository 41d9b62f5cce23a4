"""Heterogeneity injection.

Capability parity with the reference's seeded fake-heterogeneity generator
(reference: scaelum/stimulator/stimulator.py:4-24): deterministic per-rank
slowdown factors used to demonstrate the load balancer on homogeneous
8xMI355X nodes. Factor ranges follow the reference's distributions
(memory [1,3), network [1,2), compute [1,4)).
"""

from __future__ import annotations

import numpy as np


class Stimulator:
    def __init__(self, num_workers: int, seed: int = 1024):
        rng = np.random.RandomState(seed)
        self._num_workers = num_workers
        self.memory_factors = 1.0 + 2.0 * rng.rand(num_workers)
        self.network_factors = 1.0 + 1.0 * rng.rand(num_workers)
        self.compute_factors = 1.0 + 3.0 * rng.rand(num_workers)

    @property
    def num_workers(self) -> int:
        return self._num_workers

    def compute_factor(self, rank: int) -> float:
        return float(self.compute_factors[rank])

    def memory_factor(self, rank: int) -> float:
        return float(self.memory_factors[rank])

    def network_factor(self, rank: int) -> float:
        return float(self.network_factors[rank])

    def scale_benchmark(self, results: dict) -> dict:
        """Scale a {rank: {'time': t, 'avai_mem': m}} benchmark result in place,
        mimicking the reference's STIMULATE-gated scaling
        (reference: scaelum/dynamics/benchmarker.py:126-129)."""
        for rank, rec in results.items():
            rec["time"] = rec["time"] * self.compute_factor(rank)
            rec["avai_mem"] = rec["avai_mem"] / self.memory_factor(rank)
        return results
