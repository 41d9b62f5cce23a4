"""SPMD pipeline engine.

MI355X-native replacement for the reference's host-orchestrated RRef chain
(reference: scaelum/model/rpc_model.py:16-63 forward chaining;
scaelum/runner/runner.py:127-139 dist_autograd backward). Every rank runs
the SAME program: builds its own stage slice from the broadcast partition
table (replacing remote module construction over RPC, rpc_module.py:83-89),
moves activations/grads with RCCL P2P over xGMI, and drives backward by
hand (explicit grad transport at stage boundaries replaces dist_autograd).

Schedules:
  * ``sequential`` — one whole batch per hop, matching the reference's
    per-batch chain semantics (rpc_model.py:44-55): the pipeline is only as
    fast as the sum of stages.
  * ``gpipe``    — M microbatches, all-forward then all-backward; fills the
    pipeline so K homogeneous stages approach K-fold throughput. This is a
    deliberate capability extension over the reference (which has no
    microbatching) — blocking P2P is deadlock-free here
    because no fwd and bwd hop between a rank pair can cross.
  * ``1f1b``     — one-forward-one-backward: same bubble as GPipe but peak
    activation memory ~ num_stages microbatches instead of M. Steady-state
    traffic crosses in both directions between neighbors, so the crossing
    pairs are posted as single batch_isend_irecv calls.

Gradient scaling: each microbatch loss is divided by M, so gradients equal
the full-batch gradient (mean-reduction losses).
"""

from __future__ import annotations

import os

from dataclasses import dataclass, field

import torch

from ..builder import StageModule, build_module_from_cfg
from .comm import CommContext


@dataclass
class PartitionPlan:
    """Who runs which contiguous layer range, in pipeline order.

    ``stage_ranks[s]`` is the global rank executing stage ``s``;
    ``ranges[s]`` its [start, end) slice of the full layer-config list.
    Produced by the allocator on rank 0 and broadcast over the control
    plane (replacing the reference's worker.model_config writes +
    reset_rank_by_order, allocator.py:154-179).
    """

    stage_ranks: list = field(default_factory=list)
    ranges: list = field(default_factory=list)

    def __post_init__(self):
        assert len(self.stage_ranks) == len(self.ranges)

    @property
    def num_stages(self) -> int:
        return len(self.stage_ranks)

    def stage_of_rank(self, rank: int) -> int | None:
        try:
            return self.stage_ranks.index(rank)
        except ValueError:
            return None

    def to_dict(self):
        return {"stage_ranks": list(self.stage_ranks), "ranges": [list(r) for r in self.ranges]}

    @classmethod
    def from_dict(cls, d):
        return cls(stage_ranks=list(d["stage_ranks"]), ranges=[tuple(r) for r in d["ranges"]])


def _to_tuple(x):
    if isinstance(x, (tuple, list)):
        return tuple(x)
    return (x,)


class PipelineEngine:
    def __init__(
        self,
        comm: CommContext,
        layer_cfgs: list[dict],
        plan: PartitionPlan,
        loss_fn=None,
        dtype: torch.dtype | None = None,
        stage_kwargs: dict | None = None,
    ):
        self.comm = comm
        self.layer_cfgs = layer_cfgs
        self.loss_fn = loss_fn
        self.dtype = dtype
        self.stage_kwargs = dict(stage_kwargs or {})
        self.stage: StageModule | None = None
        self.plan: PartitionPlan | None = None
        self.apply_plan(plan)

    # ---------------- partition management ----------------

    def apply_plan(self, plan: PartitionPlan):
        self.plan = plan
        self.stage_idx = plan.stage_of_rank(self.comm.rank)
        self.comm.reset_channels()
        if self.stage_idx is None:
            self.stage = None
            return
        start, end = plan.ranges[self.stage_idx]
        kw = dict(self.stage_kwargs)
        self.stage = build_module_from_cfg(
            self.layer_cfgs[start:end], dtype=self.dtype, **kw
        )

    @property
    def is_first(self) -> bool:
        return self.stage_idx == 0

    @property
    def is_last(self) -> bool:
        return self.stage_idx == self.plan.num_stages - 1

    @property
    def prev_rank(self) -> int | None:
        if self.stage_idx is None or self.is_first:
            return None
        return self.plan.stage_ranks[self.stage_idx - 1]

    @property
    def next_rank(self) -> int | None:
        if self.stage_idx is None or self.is_last:
            return None
        return self.plan.stage_ranks[self.stage_idx + 1]

    def parameters(self):
        return [] if self.stage is None else list(self.stage.parameters())

    def train(self, mode: bool = True):
        if self.stage is not None:
            self.stage.train(mode)

    def eval(self):
        self.train(False)

    # ---------------- micro-step primitives ----------------

    def _forward_microbatch(self, inputs, key: str = "fwd"):
        """Run one microbatch through this stage; returns (held_inputs, outputs).

        ``key`` separates channel metadata per context: eval-mode outputs
        carry different requires_grad flags than training, so they must not
        share the cached "fwd" channel (a shared key desynchronizes the
        control-plane handshake)."""
        if self.is_first:
            args = _to_tuple(inputs)
            held = ()
        else:
            args = tuple(self.comm.recv_tensors(self.prev_rank, key))
            held = args
        out = _to_tuple(self.stage(*args))
        if not self.is_last:
            self.comm.send_tensors(list(out), self.next_rank, key)
        return held, out

    def _backward_microbatch(self, held, out, labels=None, num_microbatches=1):
        """Backward for one microbatch; returns the (scaled) loss on the
        last stage, None elsewhere."""
        loss = None
        if self.is_last:
            logits = out[0] if len(out) == 1 else out
            loss = self.loss_fn(logits, labels.to(logits.device))
            (loss / num_microbatches).backward()
        else:
            outs_req = [t for t in out if torch.is_tensor(t) and t.requires_grad]
            grads = self.comm.recv_tensors(self.next_rank, "bwd")
            assert len(grads) == len(outs_req), (
                f"stage {self.stage_idx}: got {len(grads)} grads for {len(outs_req)} outputs"
            )
            torch.autograd.backward(outs_req, grads)
        if not self.is_first:
            in_grads = [t.grad for t in held if torch.is_tensor(t) and t.requires_grad]
            self.comm.send_tensors(in_grads, self.prev_rank, "bwd")
            for t in held:
                if torch.is_tensor(t):
                    t.grad = None
        return loss

    # ---------------- schedules ----------------

    def run_iteration(
        self,
        inputs=None,
        labels=None,
        num_microbatches: int = 1,
        schedule: str = "gpipe",
    ) -> float | None:
        """One optimizer-step's worth of forward+backward. Returns the mean
        loss (as float) on every rank (broadcast from the last stage)."""
        if self.stage_idx is None:
            # idle rank: still participates in the loss broadcast
            return self._broadcast_loss(None)
        M = num_microbatches if schedule != "sequential" else 1
        mb_inputs = self._split(inputs, M) if self.is_first else [None] * M
        mb_labels = self._split(labels, M) if self.is_last else [None] * M

        if schedule == "1f1b" and M > 1 and self.plan.num_stages > 1:
            total = self._run_1f1b(mb_inputs, mb_labels, M)
        elif (schedule == "gpipe" and M > 1 and self.plan.num_stages > 1
              and os.environ.get("SKY_NO_OVERLAP") != "1"):
            total = self._run_gpipe_overlapped(mb_inputs, mb_labels, M)
        else:
            saved = []
            for m in range(M):
                saved.append(self._forward_microbatch(mb_inputs[m]))
            total = 0.0
            for m in range(M):
                held, out = saved[m]
                loss = self._backward_microbatch(held, out, mb_labels[m], M)
                if loss is not None:
                    total += float(loss.detach())
            saved.clear()
        return self._broadcast_loss(total / M if self.is_last else None)

    def _run_gpipe_overlapped(self, mb_inputs, mb_labels, M: int) -> float:
        """GPipe with comm/compute overlap: the next
        microbatch's activation/grad irecv is pre-posted while the current
        one computes, and boundary sends are non-blocking (handles waited
        at phase end). Falls back to blocking hops on the first iteration
        of each channel (meta handshake)."""
        send_works: list = []
        saved = []
        pending = None
        for m in range(M):
            if self.is_first:
                args, held = _to_tuple(mb_inputs[m]), ()
            else:
                if pending is not None:
                    args = tuple(self.comm.wait_irecv(pending))
                else:
                    args = tuple(self.comm.recv_tensors(self.prev_rank, "fwd"))
                held = args
                pending = (self.comm.irecv_tensors(self.prev_rank, "fwd")
                           if m + 1 < M else None)
            out = _to_tuple(self.stage(*args))
            if not self.is_last:
                send_works += self.comm.send_tensors_async(
                    list(out), self.next_rank, "fwd")
            saved.append((held, out))
        for w in send_works:
            w.wait()
        send_works = []
        total = 0.0
        pending = None
        for m in range(M):
            held, out = saved[m]
            if self.is_last:
                logits = out[0] if len(out) == 1 else out
                loss = self.loss_fn(logits, mb_labels[m].to(logits.device))
                (loss / M).backward()
                total += float(loss.detach())
            else:
                outs_req = [t for t in out if torch.is_tensor(t) and t.requires_grad]
                if pending is not None:
                    grads = self.comm.wait_irecv(pending)
                else:
                    grads = self.comm.recv_tensors(self.next_rank, "bwd")
                assert len(grads) == len(outs_req), (
                    f"stage {self.stage_idx}: got {len(grads)} grads for "
                    f"{len(outs_req)} outputs")
                pending = (self.comm.irecv_tensors(self.next_rank, "bwd")
                           if m + 1 < M else None)
                torch.autograd.backward(outs_req, grads)
            if not self.is_first:
                in_grads = [t.grad for t in held
                            if torch.is_tensor(t) and t.requires_grad]
                send_works += self.comm.send_tensors_async(
                    in_grads, self.prev_rank, "bwd")
                for t in held:
                    if torch.is_tensor(t):
                        t.grad = None
            saved[m] = None
        for w in send_works:
            w.wait()
        return total

    # ---------------- 1F1B ----------------
    # One-forward-one-backward: peak activation memory ~ num_stages
    # microbatches instead of M. Steady-state traffic crosses in both
    # directions between neighbor ranks, so the two crossing pairs
    # (send-fwd + recv-bwd with next; send-bwd + recv-fwd with prev) are
    # posted as single batched P2P calls (comm.fused_send_recv).

    def _grad_specs(self, out):
        return [
            (tuple(t.shape), t.dtype, False)
            for t in out
            if torch.is_tensor(t) and t.requires_grad
        ]

    def _run_1f1b(self, mb_inputs, mb_labels, M: int) -> float:
        from collections import deque

        N = self.plan.num_stages
        s = self.stage_idx
        warmup = min(N - 1 - s, M)
        saved = deque()
        total = 0.0
        fwd_i = 0
        bwd_i = 0

        # ---- warmup: forward-only phase (plain blocking P2P is safe:
        # all traffic flows downstream) ----
        for _ in range(warmup):
            saved.append(self._forward_microbatch(mb_inputs[fwd_i]))
            fwd_i += 1

        pending_input = None  # fwd input received by a fused call
        remaining = M - warmup
        for i in range(remaining):
            # forward one microbatch
            if self.is_first:
                held, out = (), _to_tuple(self.stage(*_to_tuple(mb_inputs[fwd_i])))
            else:
                if pending_input is None:
                    args = tuple(self.comm.recv_tensors(self.prev_rank, "fwd"))
                else:
                    args = tuple(pending_input)
                    pending_input = None
                held, out = args, _to_tuple(self.stage(*args))
            fwd_i += 1
            saved.append((held, out))

            # send fwd output + receive bwd grads (fused with next rank)
            held_b, out_b = saved.popleft()
            if self.is_last:
                logits = out_b[0] if len(out_b) == 1 else out_b
                loss = self.loss_fn(logits, mb_labels[bwd_i].to(logits.device))
                (loss / M).backward()
                total += float(loss.detach())
            else:
                (grads,) = self.comm.fused_send_recv(
                    sends=[(list(out), self.next_rank)],
                    recvs=[(self._grad_specs(out_b), self.next_rank)],
                )
                outs_req = [t for t in out_b if torch.is_tensor(t) and t.requires_grad]
                torch.autograd.backward(outs_req, grads)

            # send bwd grads + receive next fwd input (fused with prev rank)
            if not self.is_first:
                in_grads = [t.grad for t in held_b if torch.is_tensor(t) and t.requires_grad]
                want_fwd = fwd_i < M and i + 1 < remaining
                fwd_meta = self.comm.cached_recv_meta(self.prev_rank, "fwd") if want_fwd else None
                res = self.comm.fused_send_recv(
                    sends=[(in_grads, self.prev_rank)],
                    recvs=[(fwd_meta, self.prev_rank)] if fwd_meta else [],
                )
                if fwd_meta:
                    pending_input = res[0]
                for t in held_b:
                    if torch.is_tensor(t):
                        t.grad = None
            bwd_i += 1

        # ---- cooldown: drain remaining backwards. NOTE: must use the
        # fused (meta-less) transport — the steady-state fused ops never
        # exchanged channel metadata, so the handshake-based plain path
        # would desynchronize the control plane. ----
        while saved:
            held_b, out_b = saved.popleft()
            if self.is_last:
                logits = out_b[0] if len(out_b) == 1 else out_b
                loss = self.loss_fn(logits, mb_labels[bwd_i].to(logits.device))
                (loss / M).backward()
                total += float(loss.detach())
            else:
                (grads,) = self.comm.fused_send_recv(
                    sends=[], recvs=[(self._grad_specs(out_b), self.next_rank)]
                )
                outs_req = [t for t in out_b if torch.is_tensor(t) and t.requires_grad]
                torch.autograd.backward(outs_req, grads)
            if not self.is_first:
                in_grads = [t.grad for t in held_b if torch.is_tensor(t) and t.requires_grad]
                self.comm.fused_send_recv(sends=[(in_grads, self.prev_rank)], recvs=[])
                for t in held_b:
                    if torch.is_tensor(t):
                        t.grad = None
            bwd_i += 1
        return total

    def evaluate_batch(self, inputs=None, labels=None):
        """Forward-only pass; returns logits on the last stage."""
        if self.stage_idx is None:
            return None
        with torch.no_grad():
            _, out = self._forward_microbatch(inputs, key="fwd_eval")
        return out[0] if self.is_last else None

    def _broadcast_loss(self, loss_val):
        src = self.plan.stage_ranks[-1] if self.plan is not None and self.plan.num_stages else 0
        return self.comm.broadcast_object(loss_val, src=src)

    @staticmethod
    def _split(x, M: int):
        if M == 1:
            return [x]
        if torch.is_tensor(x):
            assert x.shape[0] % M == 0, (
                f"batch {x.shape[0]} not divisible by {M} microbatches "
                "(P2P channel metadata assumes equal microbatch shapes)"
            )
            return list(torch.chunk(x, M, dim=0))
        if isinstance(x, (tuple, list)):
            parts = [PipelineEngine._split(t, M) for t in x]
            return [type(x)(p[m] for p in parts) for m in range(M)]
        raise TypeError(f"cannot split {type(x)}")
