"""Per-microbatch stage graphs for the multi-rank pipeline.

At N ranks the eager GPipe step still issues every kernel launch from the
host (~28 per BERT sub-layer); with 1/N of the model per rank the GPU work
shrinks N-fold but the host launch stream does not — the pipeline goes
host-bound. This executor captures each microbatch's stage forward and
backward into hipGraphs ONCE (plus a zero-grads graph and an SGD graph) and
replays them; only the RCCL P2P hops and the tiny schedule logic stay on
the host.

Capture layout: one shared memory pool, captures in exactly the replay
order (fwd 0..M-1, bwd 0..M-1, step) so pool addresses are reproduced on
every replay. P2P moves through STATIC buffers: recvs land in
comm.recv_tensors_into targets, sends read the stage's stable output/grad
tensors from the pool.

Unlike the 1-GPU whole-step capture (graph.GraphedTrainStep), gradients
here must ACCUMULATE across microbatches, so grads are materialized before
capture and a zero-grads graph runs at each iteration start.
"""

from __future__ import annotations

import torch

from ..ops import functions as F


class GraphedPipelineStep:
    def __init__(self, engine, optimizer, num_microbatches: int,
                 sample_inputs, sample_labels, warmup_iters: int = 2):
        assert torch.cuda.is_available()
        assert engine.stage is not None
        assert not engine.stage.record_forward_time
        self.engine = engine
        self.comm = engine.comm
        self.opt = optimizer
        self.M = M = num_microbatches
        dev = self.comm.device

        eng = engine
        self.is_first, self.is_last = eng.is_first, eng.is_last

        mb_inputs = eng._split(sample_inputs, M) if eng.is_first else [None] * M
        mb_labels = eng._split(sample_labels, M) if eng.is_last else [None] * M

        # ---- eager warmup iterations (also perform the channel meta
        # handshake so static recv shapes are known) ----
        for _ in range(warmup_iters):
            self.opt.zero_grad(set_to_none=True)
            eng.run_iteration(sample_inputs, sample_labels,
                              num_microbatches=M, schedule="gpipe")
            self.opt.step()

        # static input-side buffers
        if eng.is_first:
            self.static_in = [
                [t.to(dev).clone() if torch.is_tensor(t) else t for t in _tup(mb_inputs[m])]
                for m in range(M)
            ]
        else:
            metas = self.comm.cached_recv_meta(eng.prev_rank, "fwd")
            assert metas is not None
            self.static_in = [
                [
                    torch.empty(shape, dtype=dt, device=dev).requires_grad_(rq)
                    for (shape, dt, rq) in metas
                ]
                for m in range(M)
            ]
        if eng.is_last:
            self.static_labels = [mb_labels[m].to(dev).clone() for m in range(M)]

        F.rng_state()
        torch.cuda.synchronize()

        # ---- capture, in replay order, over ONE pool ----
        self.g_fwd = [torch.cuda.CUDAGraph() for _ in range(M)]
        self.g_bwd = [torch.cuda.CUDAGraph() for _ in range(M)]
        pool = None
        self.saved = []
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            with torch.cuda.graph(self.g_fwd[0]):
                F.rng_tick()
                out0 = _tup(eng.stage(*self.static_in[0]))
            pool = self.g_fwd[0].pool()
            self.saved.append(out0)
            for m in range(1, M):
                with torch.cuda.graph(self.g_fwd[m], pool=pool):
                    self.saved.append(_tup(eng.stage(*self.static_in[m])))

            # backward captures; loss accumulates into a static scalar
            self.static_loss = torch.zeros((), device=dev)
            self.static_grads_in = [None] * M
            for m in range(M):
                out = self.saved[m]
                if eng.is_last:
                    with torch.cuda.graph(self.g_bwd[m], pool=pool):
                        logits = out[0] if len(out) == 1 else out
                        loss = eng.loss_fn(logits, self.static_labels[m])
                        (loss / M).backward()
                        self.static_loss += loss.detach() / M
                else:
                    metas = [
                        (tuple(t.shape), t.dtype, False)
                        for t in out
                        if torch.is_tensor(t) and t.requires_grad
                    ]
                    gin = [
                        torch.empty(shape, dtype=dt, device=dev)
                        for (shape, dt, _rq) in metas
                    ]
                    self.static_grads_in[m] = gin
                    outs_req = [t for t in out if torch.is_tensor(t) and t.requires_grad]
                    with torch.cuda.graph(self.g_bwd[m], pool=pool):
                        torch.autograd.backward(outs_req, gin)

            # optimizer graph (plan already built during warmup)
            self.g_opt = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_opt, pool=pool):
                self.opt.step()
        torch.cuda.current_stream().wait_stream(side)

        # every tensor that backward ACCUMULATES into must be zeroed per
        # iteration: param grads + received-input leaf grads
        zero_list = [p.grad for p in eng.parameters() if p.grad is not None]
        for m in range(M):
            for t in self.static_in[m]:
                if torch.is_tensor(t) and t.requires_grad and t.grad is not None:
                    zero_list.append(t.grad)
        self._zero_list = zero_list
        self.g_zero = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_zero, pool=pool):
            torch._foreach_zero_(self._zero_list)
            self.static_loss.zero_()

    def step(self, inputs, labels) -> float | None:
        eng = self.engine
        M = self.M
        self.g_zero.replay()
        mb_inputs = eng._split(inputs, M) if eng.is_first else [None] * M
        mb_labels = eng._split(labels, M) if eng.is_last else [None] * M
        for m in range(M):
            if eng.is_first:
                for buf, t in zip(self.static_in[m], _tup(mb_inputs[m])):
                    if torch.is_tensor(t):
                        buf.data.copy_(t, non_blocking=True)
            else:
                self.comm.recv_tensors_into(self.static_in[m], eng.prev_rank)
            if eng.is_last:
                self.static_labels[m].copy_(mb_labels[m], non_blocking=True)
            self.g_fwd[m].replay()
            if not eng.is_last:
                self.comm.send_tensors(list(self.saved[m]), eng.next_rank, "fwd")
        for m in range(M):
            if not eng.is_last:
                self.comm.recv_tensors_into(self.static_grads_in[m], eng.next_rank)
            self.g_bwd[m].replay()
            if not eng.is_first:
                in_grads = [
                    t.grad for t in self.static_in[m]
                    if torch.is_tensor(t) and t.requires_grad
                ]
                self.comm.send_tensors(in_grads, eng.prev_rank, "bwd")
        self.g_opt.replay()
        loss = float(self.static_loss.detach().cpu()) if eng.is_last else None
        return eng._broadcast_loss(loss if eng.is_last else None)


def _tup(x):
    if isinstance(x, (tuple, list)):
        return tuple(x)
    return (x,)
