"""Interleaved virtual-stage pipeline (each rank holds v model chunks).

With one contiguous chunk per rank (PipelineEngine) the GPipe bubble at
N stages and M microbatches is (N-1)/(M+N-1) — 47% at N=8, M=8 on the
bench workload (profiles/scaling_expectations.md). Giving each rank v
interleaved chunks multiplies the stage count but divides the per-stage
work, cutting the bubble roughly v-fold for the same M (the reference has
no microbatching at all — this is scaling groundwork beyond its scope,
see ROADMAP.md §5).

Execution model: every rank derives the SAME global schedule from a
discrete-event simulation (unit-time events, backward-priority — the
1F1B-style order emerges from the priority rule) and executes its own
events in that order. All sends are non-blocking isends (handles retained
until iteration end); receives block. Because each rank's blocking recv
only waits for a peer event that the simulation placed earlier, and every
earlier send was posted without blocking, the execution is deadlock-free
by construction. The first iteration runs the fully serialized order so
channel-meta handshakes (control-plane rendezvous) never cross.

Chunk boundaries detach activations even between two chunks on the SAME
rank — gradients always flow through explicit backward events, so local
and remote boundaries have identical semantics.
"""

from __future__ import annotations

import torch

from ..builder import build_module_from_cfg
from .pipeline import PartitionPlan, PipelineEngine


def build_interleaved_plan(num_layers: int, world: int, v: int) -> PartitionPlan:
    """Even split of ``num_layers`` into world*v contiguous chunks; chunk c
    belongs to rank c % world (rank r owns chunks r, world+r, ...)."""
    S = world * v
    base, rem = divmod(num_layers, S)
    bounds = [0]
    for i in range(S):
        bounds.append(bounds[-1] + base + (1 if i < rem else 0))
    return PartitionPlan(
        stage_ranks=[s % world for s in range(S)],
        ranges=[(bounds[s], bounds[s + 1]) for s in range(S)],
    )


def interleaved_schedule(owner: list[int], M: int, serialized: bool = False):
    """Per-rank ordered event lists [('F'|'B', stage, mb), ...] from a
    unit-time simulation. Deps: F(s,m) <- F(s-1,m); B(s,m) <- F(s,m) and
    B(s+1,m). Priority: backward first, then lower microbatch, then higher
    stage (drain deeper chunks first). ``serialized`` produces the strict
    one-event-at-a-time global order (used for the handshake iteration)."""
    S = len(owner)
    events = [("F", s, m) for s in range(S) for m in range(M)]
    events += [("B", s, m) for s in range(S) for m in range(M)]

    def deps(ev):
        kind, s, m = ev
        if kind == "F":
            return [("F", s - 1, m)] if s > 0 else []
        d = [("F", s, m)]
        if s < S - 1:
            d.append(("B", s + 1, m))
        return d

    # critical-path list scheduling: prefer the event with the longest
    # remaining dependency chain (simulated bubbles at N=8, M=8: v=2
    # 37% -> 30%, v=3 33% -> 23% vs the naive backward-first rule;
    # profiles/scaling_expectations.md)
    cp_cache: dict = {}

    def cp(ev):
        got = cp_cache.get(ev)
        if got is not None:
            return got
        kind, s, m = ev
        succ = []
        if kind == "F":
            if s < S - 1:
                succ.append(("F", s + 1, m))
            succ.append(("B", s, m))
        elif s > 0:
            succ.append(("B", s - 1, m))
        val = 1 + max((cp(x) for x in succ), default=0)
        cp_cache[ev] = val
        return val

    def prio(ev):
        if serialized:
            # the handshake iteration uses BLOCKING sends (gloo
            # rendezvous): keep the producer-consumer-adjacent order so a
            # send's receiver reaches its recv next, never forming
            # blocking-send chains. Performance is irrelevant here.
            kind, s, m = ev
            return (0 if kind == "B" else 1, m, -s)
        return (-cp(ev), ev[2], -ev[1])

    done_round: dict = {}
    pending = set(events)
    order: dict[int, list] = {r: [] for r in set(owner)}
    rnd = 0
    while pending:
        rnd += 1
        fired = []
        busy = set()
        for ev in sorted(pending, key=prio):
            r = owner[ev[1]]
            if r in busy:
                continue
            if all(done_round.get(d, rnd) < rnd for d in deps(ev)):
                fired.append(ev)
                busy.add(r)
                if serialized:
                    break
        if not fired:  # should not happen: the dep graph is acyclic
            raise RuntimeError("interleaved schedule stalled")
        for ev in fired:
            done_round[ev] = rnd
            pending.discard(ev)
            order[owner[ev[1]]].append(ev)
    return order


class InterleavedPipelineEngine:
    """Multi-chunk-per-rank pipeline execution over the same comm substrate
    as PipelineEngine. Eager execution (no stage-graph capture yet)."""

    def __init__(self, comm, layer_cfgs, plan: PartitionPlan, loss_fn=None,
                 dtype=None, stage_kwargs=None):
        self.comm = comm
        self.layer_cfgs = layer_cfgs
        self.plan = plan
        self.loss_fn = loss_fn
        self.dtype = dtype
        self.owner = list(plan.stage_ranks)
        self.S = len(self.owner)
        kw = dict(stage_kwargs or {})
        self.chunks = {}
        for s, (a, b) in enumerate(plan.ranges):
            if self.owner[s] == comm.rank:
                self.chunks[s] = build_module_from_cfg(
                    layer_cfgs[a:b], dtype=dtype, **kw
                )
        self._warmed: set = set()  # microbatch counts already handshaken
        self._orders: dict = {}

    def parameters(self):
        out = []
        for c in self.chunks.values():
            out.extend(c.parameters())
        return out

    def train(self, mode: bool = True):
        for c in self.chunks.values():
            c.train(mode)
        return self

    def eval(self):
        return self.train(False)

    def _schedule(self, M: int):
        """Full per-rank event lists + my inbound message sequence per peer
        (each peer's sends to me, in that peer's execution order — messages
        between a pair are untagged FIFO, so the receiver must drain them
        in the sender's order and stash out-of-order arrivals)."""
        warmed = M in self._warmed
        key = (M, warmed)
        if key not in self._orders:
            full = interleaved_schedule(self.owner, M,
                                        serialized=not warmed)
            me = self.comm.rank
            inbound: dict = {}
            for peer, evs in full.items():
                if peer == me:
                    continue
                seq = []
                for kind, s, m in evs:
                    if kind == "F" and s < self.S - 1 and self.owner[s + 1] == me:
                        seq.append((f"if{s + 1}m{M}", ("F", s + 1, m)))
                    elif kind == "B" and s > 0 and self.owner[s - 1] == me:
                        seq.append((f"ib{s - 1}m{M}", ("B", s - 1, m)))
                if seq:
                    inbound[peer] = seq
            self._orders[key] = (full.get(me, []), inbound)
        return self._orders[key]


    def run_iteration(self, inputs=None, labels=None,
                      num_microbatches: int = 1,
                      schedule: str | None = None):
        """Same signature as PipelineEngine.run_iteration; ``schedule`` is
        accepted for Runner compatibility and ignored — the interleaved
        engine's event order IS its schedule."""
        data = inputs
        comm, S, M = self.comm, self.S, num_microbatches
        me = comm.rank
        first_owner, last_owner = self.owner[0], self.owner[-1]
        mb_inputs = PipelineEngine._split(data, M) if me == first_owner else None
        mb_labels = PipelineEngine._split(labels, M) if me == last_owner else None

        held: dict = {}       # (s, m) -> (inputs, outputs)
        mailbox: dict = {}    # (s, m) -> activations for local next chunk
        gradbox: dict = {}    # (s, m) -> grads from local next chunk
        pending_sends = []
        total_loss = 0.0

        warmed = M in self._warmed
        my_events, inbound_seq = self._schedule(M)
        fifo = {p: {"i": 0, "stash": {}} for p in inbound_seq}

        def fetch(peer, want):
            """Blocking-receive from ``peer`` in ITS send order until the
            message consumed by event ``want`` arrives (stash the rest)."""
            st = fifo[peer]
            if want in st["stash"]:
                return st["stash"].pop(want)
            seq = inbound_seq[peer]
            while True:
                ch, msgid = seq[st["i"]]
                st["i"] += 1
                bufs = comm.recv_tensors(peer, ch)
                if msgid == want:
                    return bufs
                st["stash"][msgid] = bufs

        for ev in my_events:
            kind, s, m = ev
            if kind == "F":
                if s == 0:
                    raw = mb_inputs[m]
                    inputs = list(raw) if isinstance(raw, (tuple, list)) else [raw]
                elif self.owner[s - 1] == me:
                    # local boundary: fresh leaves with the producer's flags
                    inputs = [
                        (t.detach().requires_grad_(rq) if torch.is_tensor(t) else t)
                        for t, rq in mailbox.pop((s - 1, m))
                    ]
                else:
                    inputs = fetch(self.owner[s - 1], ("F", s, m))
                out = self.chunks[s](*inputs)
                out = list(out) if isinstance(out, (tuple, list)) else [out]
                held[(s, m)] = (inputs, out)
                if s < S - 1:
                    if self.owner[s + 1] == me:
                        mailbox[(s, m)] = [
                            (t, bool(torch.is_tensor(t) and t.requires_grad))
                            for t in out
                        ]
                    else:
                        pending_sends.extend(
                            comm.isend_tensors(
                                [t for t in out if torch.is_tensor(t)],
                                self.owner[s + 1], f"if{s + 1}m{M}",
                                blocking=not warmed,
                            )
                        )
            else:  # backward
                inputs, out = held.pop((s, m))
                if s == S - 1:
                    logits = out[0] if len(out) == 1 else out
                    loss = self.loss_fn(logits, mb_labels[m].to(logits.device))
                    (loss / M).backward()
                    total_loss += float(loss.detach()) / M
                else:
                    req = [t for t in out if torch.is_tensor(t) and t.requires_grad]
                    if self.owner[s + 1] == me:
                        grads = gradbox.pop((s, m))
                    else:
                        grads = fetch(self.owner[s + 1], ("B", s, m))
                    torch.autograd.backward(req, grads)
                if s > 0:
                    # count mirrors the receiver's requires_grad outputs
                    in_grads = [
                        t.grad for t in inputs
                        if torch.is_tensor(t) and t.requires_grad
                    ]
                    if self.owner[s - 1] == me:
                        gradbox[(s - 1, m)] = in_grads
                    else:
                        pending_sends.extend(
                            comm.isend_tensors(
                                in_grads, self.owner[s - 1], f"ib{s - 1}m{M}",
                                blocking=not warmed,
                            )
                        )

        for w in pending_sends:
            w.wait()
        self._warmed.add(M)
        loss_val = total_loss if me == last_owner else None
        return PipelineEngine._broadcast_loss(self, loss_val)

    def evaluate_batch(self, data, labels=None):
        """Forward-only pass through the chunk chain (whole batch, no
        microbatching); returns logits on the last-chunk owner, None
        elsewhere. Eval uses its own always-handshaking channels — output
        requires_grad flags differ from training, so the channels must not
        be shared (same rule as PipelineEngine's fwd_eval key)."""
        comm, S = self.comm, self.S
        me = comm.rank
        for c in self.chunks.values():
            c.eval()
        try:
            with torch.no_grad():
                cur = None
                for s in range(S):
                    if self.owner[s] != me:
                        continue
                    if s == 0:
                        ins = list(data) if isinstance(data, (tuple, list)) else [data]
                    elif self.owner[s - 1] == me:
                        ins = cur
                    else:
                        ins = comm.recv_tensors(self.owner[s - 1], f"iev{s}")
                    out = self.chunks[s](*ins)
                    cur = list(out) if isinstance(out, (tuple, list)) else [out]
                    if s < S - 1 and self.owner[s + 1] != me:
                        comm.send_tensors(
                            [t for t in cur if torch.is_tensor(t)],
                            self.owner[s + 1], f"iev{s + 1}",
                        )
        finally:
            for c in self.chunks.values():
                c.train()
        if me == self.owner[-1]:
            return cur[0]
        return None
