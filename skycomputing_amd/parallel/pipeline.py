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

Gradient scaling: each microbatch loss is divided by M, so gradients equal
the full-batch gradient (mean-reduction losses).
"""

from __future__ import annotations

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

    def _forward_microbatch(self, inputs):
        """Run one microbatch through this stage; returns (held_inputs, outputs)."""
        if self.is_first:
            args = _to_tuple(inputs)
            held = ()
        else:
            args = tuple(self.comm.recv_tensors(self.prev_rank, "fwd"))
            held = args
        out = _to_tuple(self.stage(*args))
        if not self.is_last:
            self.comm.send_tensors(list(out), self.next_rank, "fwd")
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

    def evaluate_batch(self, inputs=None, labels=None):
        """Forward-only pass; returns logits on the last stage."""
        if self.stage_idx is None:
            return None
        with torch.no_grad():
            _, out = self._forward_microbatch(inputs)
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
