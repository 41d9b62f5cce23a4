"""Worker descriptor + pool.

Capability parity with the reference's Worker/WorkerManager
(reference: scaelum/dynamics/worker.py:8-97, worker_manager.py:7-79) minus
the RPC server plumbing: in the SPMD world a "worker" is one rank = one GPU.
The reference's latent bugs (never-set _env_config, property-called-as-
function in assign_model_to_worker) are not reproduced.
"""

from __future__ import annotations

import uuid as _uuid
from dataclasses import dataclass, field


@dataclass
class Worker:
    rank: int
    name: str = ""
    uuid: str = field(default_factory=lambda: _uuid.uuid4().hex)
    model_config: tuple | None = None  # (start, end) layer range
    extra_config: dict = field(default_factory=dict)  # StageModule kwargs (slowdown, mem_limit, ...)
    order: int | None = None
    benchmark_time: float | None = None
    avai_mem: float | None = None

    def __post_init__(self):
        if not self.name:
            self.name = f"worker{self.rank}"

    def to_dict(self) -> dict:
        return {
            "rank": self.rank, "name": self.name, "uuid": self.uuid,
            "model_config": list(self.model_config) if self.model_config else None,
            "extra_config": dict(self.extra_config), "order": self.order,
            "benchmark_time": self.benchmark_time, "avai_mem": self.avai_mem,
        }

    @classmethod
    def from_dict(cls, d: dict) -> "Worker":
        w = cls(rank=d["rank"], name=d.get("name", ""), uuid=d.get("uuid", _uuid.uuid4().hex))
        mc = d.get("model_config")
        w.model_config = tuple(mc) if mc else None
        w.extra_config = dict(d.get("extra_config", {}))
        w.order = d.get("order")
        w.benchmark_time = d.get("benchmark_time")
        w.avai_mem = d.get("avai_mem")
        return w


class WorkerManager:
    """Ordered pool of workers (one per rank)."""

    def __init__(self, workers: list[Worker] | None = None):
        self.workers: list[Worker] = list(workers or [])

    @classmethod
    def from_world(cls, world_size: int, extra_configs: list[dict] | dict | None = None) -> "WorkerManager":
        workers = []
        for r in range(world_size):
            if isinstance(extra_configs, list):
                extra = extra_configs[r] if r < len(extra_configs) else {}
            else:
                extra = dict(extra_configs or {})
            workers.append(Worker(rank=r, extra_config=dict(extra)))
        return cls(workers)

    @classmethod
    def load_worker_pool_from_config(cls, worker_cfgs: list[dict]) -> "WorkerManager":
        """(reference: worker_manager.py:31-34)"""
        return cls([Worker(rank=i, extra_config=dict(c)) for i, c in enumerate(worker_cfgs)])

    def __len__(self):
        return len(self.workers)

    def __iter__(self):
        return iter(self.workers)

    def get_worker_by_rank(self, rank: int) -> Worker:
        for w in self.workers:
            if w.rank == rank:
                return w
        raise KeyError(f"no worker with rank {rank}")

    def add_worker(self, worker: Worker):
        if any(w.rank == worker.rank for w in self.workers):
            raise ValueError(f"rank {worker.rank} already in pool")
        self.workers.append(worker)

    def remove_worker(self, rank: int):
        self.workers = [w for w in self.workers if w.rank != rank]

    def assign_model_to_worker(self, rank: int, layer_range: tuple, order: int | None = None):
        w = self.get_worker_by_rank(rank)
        w.model_config = tuple(layer_range)
        if order is not None:
            w.order = order

    def reset_rank_by_order(self):
        """Sort pool by allocator-assigned order (reference:
        worker_manager.py:62-64). Ranks are physical (one per GPU) and do
        not change; only the pipeline order does."""
        self.workers.sort(key=lambda w: (w.order if w.order is not None else w.rank))

    def pipeline_order(self) -> list[Worker]:
        return sorted(
            [w for w in self.workers if w.model_config is not None],
            key=lambda w: (w.order if w.order is not None else w.rank),
        )
