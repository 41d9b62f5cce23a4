"""hipGraph-captured training step.

The 160-layer BERT step issues ~14k kernel launches; at ~3 us host cost
each, the eager step is partly host-bound (measured ~45 ms/iter of launch
gaps, profiles/r01_notes.md). With static shapes the whole
tick -> zero-grads -> forward -> loss -> backward -> SGD sequence is
captured ONCE into a hipGraph and replayed per iteration.

Graph-safety notes:
  * dropout masks vary across replays via the device-side RNG step counter
    (ops.functions.rng_tick, captured as the first node);
  * gradients are static buffers (zeroed in-graph with _foreach_zero_);
  * FusedSGD's multi-tensor descriptor table is built during warmup so no
    allocation happens inside capture;
  * stage timing (HIP events + host sync) must be off.

Currently used for the single-stage (1 GPU) topology; per-microbatch stage
graphs for the multi-rank pipeline are the planned extension.
"""

from __future__ import annotations

import torch

from ..ops import functions as F


class GraphedTrainStep:
    def __init__(self, stage, optimizer, loss_fn, sample_inputs, sample_labels,
                 warmup_iters: int = 3):
        assert torch.cuda.is_available()
        assert not stage.record_forward_time, "disable stage timing for graph capture"
        self.stage = stage
        self.optimizer = optimizer
        self.loss_fn = loss_fn
        dev = stage.device

        self.static_inputs = [
            t.to(dev).clone() if torch.is_tensor(t) else t for t in sample_inputs
        ]
        self.static_labels = sample_labels.to(dev).clone()

        F.rng_state()  # allocate before any capture

        def one_step():
            F.rng_tick()
            out = stage(*self.static_inputs)
            loss = loss_fn(out, self.static_labels)
            loss.backward()
            optimizer.step()
            return loss

        # warmup on a side stream (cudnn/hipblaslt heuristics, allocator)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                self._drop_grads()
                one_step()
        torch.cuda.current_stream().wait_stream(side)

        # Capture with grads set to None: AccumulateGrad then STEALS each
        # computed grad tensor (no zero-fill kernels, no += adds) and the
        # graph pool reallocates the same addresses every replay. Two
        # graphs over one pool: g1 = tick+forward+loss+backward; the SGD
        # descriptor is built BETWEEN captures (host allocs are illegal
        # inside), then g2 = the fused SGD step.
        self._drop_grads()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            F.rng_tick()
            out = stage(*self.static_inputs)
            self.static_loss = loss_fn(out, self.static_labels)
            self.static_loss.backward()
        optimizer.step()  # uncaptured: builds+caches the descriptor table
        self.graph2 = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph2, pool=self.graph.pool()):
            optimizer.step()

    def _drop_grads(self):
        for p in self.stage.parameters():
            p.grad = None

    def step(self, inputs, labels) -> float:
        for buf, t in zip(self.static_inputs, inputs):
            if torch.is_tensor(t):
                buf.copy_(t, non_blocking=True)
        self.static_labels.copy_(labels, non_blocking=True)
        self.graph.replay()
        self.graph2.replay()
        return self.static_loss

    def loss_value(self) -> float:
        return float(self.static_loss.detach().cpu())
