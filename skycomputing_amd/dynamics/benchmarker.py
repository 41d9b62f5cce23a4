"""Device and model benchmarking.

MI355X-native redesign of the reference benchmarkers
(reference: scaelum/dynamics/benchmarker.py:30-201):

  * DeviceBenchmarker — SPMD self-benchmark: EVERY rank builds the probe
    locally and times it on its own GPU (no 134 MB probe-tensor broadcast,
    SURVEY.md C2); scalar results + free HBM (hipMemGetInfo via
    torch.cuda.mem_get_info) are all-gathered over the gloo control plane.
    The default probe is a BERT encoder triplet at the training geometry
    (fwd+bwd) instead of the reference's Conv2d stack — the probe must
    measure the throughput class the allocator schedules.
  * ModelBenchmarker — analytic per-layer flops/memory from the Estimator,
    with construction cached per unique layer config (generalizing the
    reference's BERT [0..3]+[-2..] tiling shortcut, benchmarker.py:163-200,
    to any repeated-layer model).
"""

from __future__ import annotations

import json
import os

import torch

from ..builder import build_layer, build_module_from_cfg
from ..stimulator import Stimulator
from .estimator import Estimator
from .worker import WorkerManager


def default_bert_probe_cfg(hidden: int = 1024, heads: int | None = None, intermediate: int | None = None):
    heads = heads or max(1, hidden // 64)
    intermediate = intermediate or 4 * hidden
    bc = dict(hidden_size=hidden, num_attention_heads=heads, intermediate_size=intermediate,
              hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0)
    return [
        dict(layer_type="BertLayer_Head", config=bc),
        dict(layer_type="BertLayer_Body", config=bc),
        dict(layer_type="BertLayer_Tail", config=bc),
    ]


class DeviceBenchmarker:
    def __init__(
        self,
        comm,
        probe_layer_cfgs: list[dict] | None = None,
        batch_size: int = 32,
        seq_len: int = 128,
        hidden: int = 1024,
        iterations: int = 10,
        dtype: torch.dtype | None = None,
        stimulator: Stimulator | None = None,
    ):
        self.comm = comm
        self.probe_layer_cfgs = probe_layer_cfgs or default_bert_probe_cfg(hidden)
        self.batch_size = batch_size
        self.seq_len = seq_len
        self.hidden = hidden
        self.iterations = iterations
        self.dtype = dtype
        self.stimulator = stimulator
        if self.stimulator is None and os.environ.get("STIMULATE") == "1":
            self.stimulator = Stimulator(comm.world_size)

    class _ProbeGen:
        """Hidden-state shaped probe input (hidden, ext_mask)."""

        def __init__(self, batch, seq, hidden, dtype, device):
            self.shape = (batch, seq, hidden)
            self.dtype = dtype
            self.device = device

        def generate(self):
            h = torch.randn(self.shape, dtype=self.dtype, device=self.device, requires_grad=True)
            mask = torch.zeros(self.shape[0], 1, 1, self.shape[1], dtype=self.dtype, device=self.device)
            return h, mask

    def local_benchmark(self) -> dict:
        device = self.comm.device
        dtype = self.dtype or (torch.bfloat16 if device.type == "cuda" else torch.float32)
        stage = build_module_from_cfg(
            self.probe_layer_cfgs, device=device, dtype=dtype, record_forward_time=False
        )
        gen = self._ProbeGen(self.batch_size, self.seq_len, self.hidden, dtype, device)
        t = Estimator.benchmark_speed(
            stage, gen, iterations=self.iterations, warmup=2, backward=True, device=device
        )
        mem = stage.detect_mem()
        del stage
        if device.type == "cuda":
            torch.cuda.empty_cache()
        return {"time": t, "avai_mem": float(mem)}

    def benchmark(self, worker_manager: WorkerManager | None = None) -> dict:
        """All ranks call this collectively. Returns {rank: {'time','avai_mem'}}."""
        mine = self.local_benchmark()
        gathered = self.comm.all_gather_object(mine)
        results = {r: dict(rec) for r, rec in enumerate(gathered)}
        if self.stimulator is not None:
            results = self.stimulator.scale_benchmark(results)
        # injected per-worker compute slowdown also scales benchmark time,
        # so the allocator sees the simulated heterogeneity
        if worker_manager is not None:
            for w in worker_manager:
                sd = float(w.extra_config.get("slowdown", 0.0))
                if sd > 0 and w.rank in results:
                    results[w.rank]["time"] *= 1.0 + sd
                ml = w.extra_config.get("mem_limit")
                if ml is not None and w.rank in results:
                    results[w.rank]["avai_mem"] = min(results[w.rank]["avai_mem"], float(ml))
                w.benchmark_time = results[w.rank]["time"]
                w.avai_mem = results[w.rank]["avai_mem"]
        return results


class ModelBenchmarker:
    def __init__(self, layer_cfgs: list[dict], batch_size: int = 32, seq_len: int = 128):
        self.layer_cfgs = layer_cfgs
        self.batch_size = batch_size
        self.seq_len = seq_len

    @staticmethod
    def _sig(cfg: dict) -> str:
        return json.dumps(cfg, sort_keys=True, default=str)

    def benchmark(self) -> dict:
        """Per-layer analytic costs: {'flops': [...], 'mem': [...]} with one
        construction per unique layer config."""
        cache: dict[str, tuple[float, float]] = {}
        flops, mem = [], []
        for cfg in self.layer_cfgs:
            s = self._sig(cfg)
            if s not in cache:
                layer = build_layer(cfg)
                f = Estimator.layer_flops(layer, self.batch_size, self.seq_len)
                m = Estimator.layer_mem_bytes(layer, self.batch_size, self.seq_len)
                cache[s] = (f, m)
                del layer
            f, m = cache[s]
            flops.append(f)
            mem.append(m)
        return {"flops": flops, "mem": mem}
