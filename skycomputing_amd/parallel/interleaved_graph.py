"""Per-chunk hipGraph capture for the interleaved pipeline.

Same rationale as parallel/static_exec.py: with v chunks per rank the
per-event GPU work shrinks (world*v)-fold but eager launch traffic does
not, so each (chunk, microbatch) forward and backward is captured ONCE
and replayed; only the P2P hops, local-boundary handoffs and the event
loop stay on the host.

Capture discipline (mirrors GraphedPipelineStep):
  * captures happen in exactly the replay order (the engine's pipelined
    event order) over ONE shared pool;
  * P2P moves through static buffers (recv_tensors_into targets; sends
    read the stable pool tensors);
  * LOCAL chunk boundaries are zero-copy: the consumer's static input IS
    the producer's static output, re-tagged as a leaf via detach().
    requires_grad_() (shared storage), so the producer's backward reads
    the consumer-written .grad tensor directly;
  * grads accumulate across microbatches into materialized param grads,
    zeroed by a dedicated graph at iteration start.

Scope: requires every rank's schedule to interleave identically (the
deterministic event list) and static shapes. bench.py uses it via
--virtual-stages when graphs are enabled, with an eager fallback.
"""

from __future__ import annotations

import torch

from ..ops import functions as F


class GraphedInterleavedStep:
    def __init__(self, engine, optimizer, num_microbatches: int,
                 sample_inputs, sample_labels, warmup_iters: int = 2):
        assert torch.cuda.is_available()
        from .pipeline import PipelineEngine

        self.engine = eng = engine
        self.comm = comm = engine.comm
        self.opt = optimizer
        self.M = M = num_microbatches
        dev = comm.device
        me = comm.rank
        owner = eng.owner
        S = eng.S
        self.first_owner, self.last_owner = owner[0], owner[-1]

        # ---- eager warmup: channel handshakes (serialized + pipelined
        # orders), optimizer plan, shape discovery ----
        for _ in range(max(2, warmup_iters)):
            self.opt.zero_grad(set_to_none=True)
            eng.run_iteration(sample_inputs, sample_labels, num_microbatches=M)
            self.opt.step()

        my_events, inbound_seq = eng._schedule(M)
        self.my_events = my_events
        self.inbound_seq = inbound_seq

        mb_inputs = (PipelineEngine._split(sample_inputs, M)
                     if me == self.first_owner else [None] * M)
        mb_labels = (PipelineEngine._split(sample_labels, M)
                     if me == self.last_owner else [None] * M)
        if me == self.last_owner:
            self.static_labels = [lb.to(dev).clone() for lb in mb_labels]

        # ---- static inputs for every owned (chunk, mb): remote recvs get
        # fresh buffers from the cached channel meta; local boundaries are
        # aliased after the producer's capture; first chunk copies host
        # batches in ----
        self.static_in: dict = {}
        for s in eng.chunks:
            for m in range(M):
                if s == 0:
                    raw = mb_inputs[m]
                    raw = list(raw) if isinstance(raw, (tuple, list)) else [raw]
                    self.static_in[(s, m)] = [
                        t.to(dev).clone() if torch.is_tensor(t) else t
                        for t in raw
                    ]
                elif owner[s - 1] != me:
                    metas = comm.cached_recv_meta(owner[s - 1], f"if{s}m{M}")
                    assert metas is not None, f"channel if{s} not handshaken"
                    self.static_in[(s, m)] = [
                        torch.empty(shape, dtype=dt, device=dev).requires_grad_(rq)
                        for (shape, dt, rq) in metas
                    ]
                # local boundaries filled during capture (aliased)

        F.rng_state()
        torch.cuda.synchronize()

        # ---- capture in replay (pipelined event) order over one pool ----
        self.g: dict = {}          # (kind, s, m) -> CUDAGraph
        self.saved: dict = {}      # (s, m) -> output tuple (pool tensors)
        self.static_grads_in: dict = {}   # (s, m) -> grad buffers (remote)
        self.static_loss = torch.zeros((), device=dev)
        pool = None
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            first = True
            for ev in my_events:
                kind, s, m = ev
                graph = torch.cuda.CUDAGraph()
                if kind == "F":
                    if s > 0 and owner[s - 1] == me:
                        # zero-copy alias of the producer's pool outputs
                        self.static_in[(s, m)] = [
                            (t.detach().requires_grad_(
                                bool(t.is_floating_point() and t.requires_grad))
                             if torch.is_tensor(t) else t)
                            for t in self.saved[(s - 1, m)]
                        ]
                    ins = self.static_in[(s, m)]
                    if first:
                        with torch.cuda.graph(graph):
                            F.rng_tick()
                            out = _tup(eng.chunks[s](*ins))
                        pool = graph.pool()
                        first = False
                    else:
                        with torch.cuda.graph(graph, pool=pool):
                            out = _tup(eng.chunks[s](*ins))
                    self.saved[(s, m)] = out
                else:
                    out = self.saved[(s, m)]
                    if s == S - 1:
                        with torch.cuda.graph(graph, pool=pool):
                            logits = out[0] if len(out) == 1 else out
                            loss = eng.loss_fn(logits, self.static_labels[m])
                            (loss / M).backward()
                            self.static_loss += loss.detach() / M
                    else:
                        outs_req = [t for t in out
                                    if torch.is_tensor(t) and t.requires_grad]
                        if owner[s + 1] == me:
                            # grads live in the consumer's aliased leaves
                            gin = [t.grad for t in self.static_in[(s + 1, m)]
                                   if torch.is_tensor(t) and t.requires_grad]
                        else:
                            gin = [torch.empty_like(t) for t in outs_req]
                            self.static_grads_in[(s, m)] = gin
                        assert all(g is not None for g in gin), (s, m)
                        with torch.cuda.graph(graph, pool=pool):
                            torch.autograd.backward(outs_req, gin)
                self.g[ev] = graph

            self.g_opt = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.g_opt, pool=pool):
                self.opt.step()
        torch.cuda.current_stream().wait_stream(side)

        # zero graph: param grads + every leaf-input grad that backward
        # accumulates into
        zero_list = [p.grad for p in eng.parameters() if p.grad is not None]
        for key, ins in self.static_in.items():
            for t in ins:
                if torch.is_tensor(t) and t.requires_grad and t.grad is not None:
                    zero_list.append(t.grad)
        self._zero_list = zero_list
        self.g_zero = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_zero, pool=pool):
            torch._foreach_zero_(self._zero_list)
            self.static_loss.zero_()

    def step(self, inputs, labels) -> float | None:
        from .pipeline import PipelineEngine

        eng, comm, M = self.engine, self.comm, self.M
        me = comm.rank
        owner = eng.owner
        S = eng.S
        self.g_zero.replay()
        mb_inputs = (PipelineEngine._split(inputs, M)
                     if me == self.first_owner else [None] * M)
        if me == self.last_owner:
            for m, lb in enumerate(PipelineEngine._split(labels, M)):
                self.static_labels[m].copy_(lb, non_blocking=True)

        fifo = {p: {"i": 0, "stash": set()} for p in self.inbound_seq}

        def drain_until(peer, want, bufs_for):
            st = fifo[peer]
            if want in st["stash"]:
                st["stash"].discard(want)
                return
            seq = self.inbound_seq[peer]
            while True:
                ch, msgid = seq[st["i"]]
                st["i"] += 1
                comm.recv_tensors_into(bufs_for(msgid), peer)
                if msgid == want:
                    return
                st["stash"].add(msgid)

        def recv_target(msgid):
            kind, s, m = msgid
            if kind == "F":
                return self.static_in[(s, m)]
            return self.static_grads_in[(s, m)]

        for ev in self.my_events:
            kind, s, m = ev
            if kind == "F":
                if s == 0:
                    for buf, t in zip(self.static_in[(s, m)], _tup(mb_inputs[m])):
                        if torch.is_tensor(t):
                            buf.data.copy_(t, non_blocking=True)
                elif owner[s - 1] != me:
                    drain_until(owner[s - 1], ("F", s, m), recv_target)
                self.g[ev].replay()
                if s < S - 1 and owner[s + 1] != me:
                    comm.send_tensors(
                        [t for t in self.saved[(s, m)] if torch.is_tensor(t)],
                        owner[s + 1], f"if{s + 1}m{M}",
                    )
            else:
                if s < S - 1 and owner[s + 1] != me:
                    drain_until(owner[s + 1], ("B", s, m), recv_target)
                self.g[ev].replay()
                if s > 0 and owner[s - 1] != me:
                    in_grads = [t.grad for t in self.static_in[(s, m)]
                                if torch.is_tensor(t) and t.requires_grad]
                    comm.send_tensors(in_grads, owner[s - 1], f"ib{s - 1}m{M}")
        self.g_opt.replay()
        loss = (float(self.static_loss.detach().cpu())
                if me == self.last_owner else None)
        return PipelineEngine._broadcast_loss(eng, loss)


def _tup(x):
    if isinstance(x, (tuple, list)):
        return tuple(x)
    return (x,)
