"""Layer -> device allocation (the load balancer).

Capability parity with the reference Allocator's three strategies
(reference: scaelum/dynamics/allocator.py:12-439): even / dynamic / optimal
over CONTIGUOUS layer ranges, driven by measured per-device speed + free
memory and per-layer flops + memory.

MI355X-native redesign of "optimal": the reference solves a PuLP MIP with a
300 s budget and a 20 % optimality gap (allocator.py:25,109-132). Here the
same problem — partition L layers into contiguous chunks, assign chunks to
an ORDERED CHOICE of devices, minimize max_d dt_d * flops(chunk_d) subject
to per-device memory — is solved EXACTLY: binary search on the bottleneck
time T with a memoized feasibility search over (layer-prefix, device-subset)
states. With <= 8 devices per node (2^8 subsets) this is milliseconds, not
minutes, and has no optimality gap. Devices may receive empty chunks (they
drop out of the pipeline).

Outputs a parallel.PartitionPlan (stage order + ranges), replacing the
reference's worker.model_config writes + reset_rank_by_order
(allocator.py:154-179).
"""

from __future__ import annotations

from bisect import bisect_right

from ..parallel.pipeline import PartitionPlan


class AllocationError(RuntimeError):
    pass


class Allocator:
    def __init__(
        self,
        layer_flops: list[float],
        layer_mem: list[float],
        workers: list[dict],
        boundary_cost: list[float] | None = None,
        comm_weight: float = 0.0,
    ):
        """``workers``: [{'rank': r, 'time': dt, 'avai_mem': bytes}, ...] —
        dt is the probe time (relative speed; higher = slower).
        ``boundary_cost``: optional per-cut payload estimate (len L-1),
        weighted by ``comm_weight``, for comm-aware balancing (penalizes
        e.g. Body->Tail cuts whose hop carries the 4x intermediate tensor,
        SURVEY.md §2c C4). It enters ``_chunk_time`` and therefore the
        dynamic allocator's objective and ``plan_cost``; the exact optimal
        solver's binary-search extension step needs monotone chunk cost, so
        it optimizes pure compute time and then, when cut costs are
        configured, runs a boundary-shift local search (``refine_plan``)
        over its exact device order to move cuts off expensive edges."""
        self.L = len(layer_flops)
        assert len(layer_mem) == self.L
        self.flops = [float(f) for f in layer_flops]
        self.mem = [float(m) for m in layer_mem]
        self.workers = sorted((dict(w) for w in workers), key=lambda w: w["rank"])
        self.W = len(self.workers)
        self.boundary_cost = boundary_cost
        self.comm_weight = comm_weight
        # prefix sums
        self.F = [0.0]
        self.M = [0.0]
        for f, m in zip(self.flops, self.mem):
            self.F.append(self.F[-1] + f)
            self.M.append(self.M[-1] + m)

    # ---------------- helpers ----------------

    def _chunk_time(self, w: dict, a: int, b: int) -> float:
        t = w["time"] * (self.F[b] - self.F[a])
        if self.boundary_cost is not None and self.comm_weight > 0:
            if a > 0:
                t += self.comm_weight * self.boundary_cost[a - 1]
            if b < self.L:
                t += self.comm_weight * self.boundary_cost[b - 1]
        return t

    def _chunk_mem_ok(self, w: dict, a: int, b: int) -> bool:
        return (self.M[b] - self.M[a]) <= w["avai_mem"]

    def _plan_from_bounds(self, bounds: list[int], rank_order: list[int]) -> PartitionPlan:
        """bounds: len W+1 cut positions; rank_order: device rank per chunk."""
        stage_ranks, ranges = [], []
        for i in range(len(rank_order)):
            a, b = bounds[i], bounds[i + 1]
            if b > a:
                stage_ranks.append(rank_order[i])
                ranges.append((a, b))
        return PartitionPlan(stage_ranks=stage_ranks, ranges=ranges)

    def plan_cost(self, plan: PartitionPlan) -> float:
        """max stage time under the benchmark speeds (diagnostic)."""
        by_rank = {w["rank"]: w for w in self.workers}
        return max(
            self._chunk_time(by_rank[r], a, b)
            for r, (a, b) in zip(plan.stage_ranks, plan.ranges)
        )

    # ---------------- even ----------------

    def even_allocate(self) -> PartitionPlan:
        """Floor division + remainder-to-front split
        (reference: allocator.py:259-280)."""
        base, rem = divmod(self.L, self.W)
        bounds = [0]
        for i in range(self.W):
            bounds.append(bounds[-1] + base + (1 if i < rem else 0))
        return self._plan_from_bounds(bounds, [w["rank"] for w in self.workers])

    # ---------------- dynamic ----------------

    def dynamic_allocate(self, break_iter: int = 1000) -> PartitionPlan:
        """Even init -> memory repair -> iterative single-layer boundary
        shifts toward minimizing the bottleneck stage
        (reference: allocator.py:181-257,295-439)."""
        base, rem = divmod(self.L, self.W)
        bounds = [0]
        for i in range(self.W):
            bounds.append(bounds[-1] + base + (1 if i < rem else 0))
        ws = self.workers
        bounds = self._repair_memory(bounds, ws)
        bounds = self._balance_bounds(bounds, ws, break_iter)
        return self._plan_from_bounds(bounds, [w["rank"] for w in ws])

    def _balance_bounds(self, bounds: list[int], ws: list[dict], break_iter: int) -> list[int]:
        """Iterative single-layer boundary shifts toward minimizing the
        bottleneck chunk (under the full ``_chunk_time`` objective, so cut
        costs count when configured)."""
        n = len(ws)
        for _ in range(break_iter):
            times = [self._chunk_time(ws[i], bounds[i], bounds[i + 1]) for i in range(n)]
            worst = max(range(n), key=lambda i: times[i])
            improved = False
            # try shedding one layer from the bottleneck to either neighbor
            for di in (-1, 1):
                j = worst + di
                if not (0 <= j < n):
                    continue
                nb = list(bounds)
                if nb[worst + 1] - nb[worst] <= 0:
                    continue
                if di == -1:
                    # give the first layer of `worst` to the left neighbor
                    nb[worst] += 1
                else:
                    nb[worst + 1] -= 1
                if not self._chunk_mem_ok(ws[j], nb[j], nb[j + 1]):
                    continue
                new_pair = max(
                    self._chunk_time(ws[worst], nb[worst], nb[worst + 1]),
                    self._chunk_time(ws[j], nb[j], nb[j + 1]),
                )
                if new_pair < times[worst] - 1e-12:
                    bounds = nb
                    improved = True
                    break
            if not improved:
                break
        return bounds

    def refine_plan(self, plan: PartitionPlan, break_iter: int = 1000) -> PartitionPlan:
        """Boundary-shift local search over a plan's FIXED stage order —
        improves cut placement under the full objective (incl. comm costs)
        without changing which devices participate or their order. Never
        returns a worse plan than the input."""
        by_rank = {w["rank"]: w for w in self.workers}
        ws = [by_rank[r] for r in plan.stage_ranks]
        bounds = [plan.ranges[0][0]] + [b for _a, b in plan.ranges]
        bounds = self._balance_bounds(bounds, ws, break_iter)
        refined = self._plan_from_bounds(bounds, [w["rank"] for w in ws])
        return refined if self.plan_cost(refined) <= self.plan_cost(plan) else plan

    def _repair_memory(self, bounds: list[int], ws: list[dict]) -> list[int]:
        """Shift layers off over-memory chunks (reference:
        allocator.py:370-439); raises AllocationError when infeasible."""
        if sum(w["avai_mem"] for w in ws) < self.M[self.L] - 1e-9:
            raise AllocationError("memory allocation failed: total memory insufficient")
        for _ in range(10 * self.L + 10):
            over = None
            for i in range(self.W):
                if not self._chunk_mem_ok(ws[i], bounds[i], bounds[i + 1]):
                    over = i
                    break
            if over is None:
                return bounds
            moved = False
            for di in (-1, 1):
                j = over + di
                if not (0 <= j < self.W):
                    continue
                nb = list(bounds)
                if di == -1:
                    nb[over] += 1
                else:
                    nb[over + 1] -= 1
                if nb[over + 1] < nb[over]:
                    continue
                if self._chunk_mem_ok(ws[j], nb[j], nb[j + 1]) or not self._chunk_mem_ok(
                    ws[j], bounds[j], bounds[j + 1]
                ):
                    # move if the receiver stays (or already was) the problem;
                    # progress is guaranteed by the total-memory check above
                    bounds = nb
                    moved = True
                    break
            if not moved:
                raise AllocationError("memory allocation failed: no feasible shift")
        raise AllocationError("memory allocation failed: repair did not converge")

    # ---------------- optimal (exact) ----------------

    def _max_extend(self, p: int, w: dict, T: float) -> int:
        """Largest j >= p with dt*(F[j]-F[p]) <= T and M[j]-M[p] <= mem."""
        dt = w["time"]
        if dt <= 0:
            j_f = self.L
        else:
            j_f = bisect_right(self.F, self.F[p] + T / dt) - 1
        j_m = bisect_right(self.M, self.M[p] + w["avai_mem"] * (1 + 1e-12)) - 1
        return min(j_f, j_m)

    def _feasible(self, T: float, reconstruct: bool = False):
        full = (1 << self.W) - 1
        memo: dict[tuple[int, int], bool] = {}

        def go(p: int, mask: int) -> bool:
            if p == self.L:
                return True
            key = (p, mask)
            if key in memo:
                return memo[key]
            ok = False
            m = mask
            while m:
                d = (m & -m).bit_length() - 1
                m &= m - 1
                j = self._max_extend(p, self.workers[d], T)
                if j > p and go(j, mask & ~(1 << d)):
                    ok = True
                    break
            memo[key] = ok
            return ok

        if not reconstruct:
            return go(0, full)
        # rebuild the chosen chain
        if not go(0, full):
            return None
        p, mask = 0, full
        chunks = []
        while p < self.L:
            for d in range(self.W):
                if not (mask >> d) & 1:
                    continue
                j = self._max_extend(p, self.workers[d], T)
                if j > p and go(j, mask & ~(1 << d)):
                    chunks.append((self.workers[d]["rank"], p, j))
                    p, mask = j, mask & ~(1 << d)
                    break
            else:
                return None
        return chunks

    def optimal_allocate(self, tol: float = 1e-7) -> PartitionPlan:
        """Exact min-max bottleneck partition (replaces the reference MIP,
        allocator.py:25-179)."""
        fastest = min(w["time"] for w in self.workers)
        hi = max(w["time"] for w in self.workers) * self.F[self.L]
        lo = fastest * self.F[self.L] / self.W  # perfect-split lower bound
        lo = min(lo, hi)
        if not self._feasible(hi * (1 + 1e-9)):
            raise AllocationError("optimal allocation infeasible (memory)")
        for _ in range(80):
            mid = 0.5 * (lo + hi)
            if self._feasible(mid):
                hi = mid
            else:
                lo = mid
            if hi - lo <= tol * max(hi, 1e-30):
                break
        chunks = self._feasible(hi * (1 + 1e-9), reconstruct=True)
        if chunks is None:
            raise AllocationError("optimal allocation reconstruction failed")
        stage_ranks = [c[0] for c in chunks]
        ranges = [(c[1], c[2]) for c in chunks]
        plan = PartitionPlan(stage_ranks=stage_ranks, ranges=ranges)
        if self.boundary_cost is not None and self.comm_weight > 0:
            plan = self.refine_plan(plan)
        return plan

    # ---------------- interleaved (virtual stages) ----------------

    def device_cost(self, plan: PartitionPlan) -> float:
        """max over devices of the SUM of their chunks' times — the
        bottleneck metric for interleaved plans, where one device executes
        several chunks per microbatch wave."""
        by_rank = {w["rank"]: w for w in self.workers}
        totals: dict = {}
        for r, (a, b) in zip(plan.stage_ranks, plan.ranges):
            totals[r] = totals.get(r, 0.0) + self._chunk_time(by_rank[r], a, b)
        return max(totals.values())

    def interleaved_allocate(self, v: int = 2) -> PartitionPlan:
        """Heterogeneity-aware interleaved plan: v chunks per device in
        round-robin order (d0..dk, d0..dk, ...), sized so each device's
        TOTAL flops matches the exact min-max solution's share.

        The per-device budget comes from ``optimal_allocate`` (merging a
        device's chunks never changes its total, so the merged solver's
        shares are optimal for the summed-bottleneck objective too); cuts
        then split each share into v pieces along the round-robin device
        cycle. Raises AllocationError if a device's chunks exceed its
        memory (per-chunk memory tracks flops closely for uniform layer
        stacks; heavily skewed memory profiles should fall back to
        ``build_interleaved_plan``'s even split)."""
        if v < 1:
            raise ValueError("v must be >= 1")
        base = self.optimal_allocate()
        if v == 1:
            return base
        order = list(base.stage_ranks)
        by_rank = {w["rank"]: w for w in self.workers}
        share = {
            r: self.F[b] - self.F[a]
            for r, (a, b) in zip(base.stage_ranks, base.ranges)
        }
        # candidate starting points — analytic share-proportional cuts and
        # the plain even split, over both the optimal-order cycle and the
        # rank-order cycle — each polished by the device-cost local search;
        # keep the best (the search only accepts improvements, so the
        # result never loses to the plain even interleave)
        def analytic_bounds(cycle):
            S = len(cycle)
            bs = [0]
            target = 0.0
            for i, r in enumerate(cycle[:-1]):
                target += share.get(r, self.F[self.L] / len(share)) / v
                j = bisect_right(self.F, target)
                j = max(bs[-1], min(j - 1, self.L - (S - 1 - i)))
                bs.append(j)
            bs.append(self.L)
            return bs

        def even_bounds(cycle):
            S = len(cycle)
            base_c, rem_c = divmod(self.L, S)
            bs = [0]
            for i in range(S):
                bs.append(bs[-1] + base_c + (1 if i < rem_c else 0))
            return bs

        cycles = [order * v]
        rank_cycle = [w["rank"] for w in self.workers if w["rank"] in share] * v
        if rank_cycle != cycles[0]:
            cycles.append(rank_cycle)
        plans = []
        for cycle in cycles:
            for bs in (analytic_bounds(cycle), even_bounds(cycle)):
                bb = self._balance_device_bounds(bs, cycle)
                plans.append(self._plan_from_bounds(bb, cycle))
        plan = min(plans, key=self.device_cost)
        mem_used: dict = {}
        for r, (a, b) in zip(plan.stage_ranks, plan.ranges):
            mem_used[r] = mem_used.get(r, 0.0) + (self.M[b] - self.M[a])
        for r, used in mem_used.items():
            if used > by_rank[r]["avai_mem"] * (1 + 1e-9):
                raise AllocationError(
                    f"interleaved allocation exceeds rank {r} memory"
                )
        return plan

    def _balance_device_bounds(self, bounds: list[int], cycle: list[int],
                               break_iter: int = 2000) -> list[int]:
        """Single-layer boundary shifts minimizing the per-DEVICE summed
        bottleneck (repairs the layer-granularity rounding of the analytic
        interleaved cuts; a one-layer shift moves flops between the two
        devices adjacent in the cycle)."""
        n = len(cycle)
        by_rank = {w["rank"]: w for w in self.workers}

        def totals(bs):
            t: dict = {}
            for i in range(n):
                r = cycle[i]
                t[r] = t.get(r, 0.0) + self._chunk_time(by_rank[r], bs[i], bs[i + 1])
            return t

        for _ in range(break_iter):
            t = totals(bounds)
            worst = max(t, key=t.get)
            improved = False
            for i in range(n):
                if improved:
                    break
                if cycle[i] != worst:
                    continue
                for di in (-1, 1):
                    j = i + di
                    if not (0 <= j < n) or cycle[j] == worst:
                        continue
                    nb = list(bounds)
                    if nb[i + 1] - nb[i] <= 0:
                        continue
                    if di == -1:
                        nb[i] += 1      # first layer of i -> chunk j
                    else:
                        nb[i + 1] -= 1  # last layer of i -> chunk j
                    nt = totals(nb)
                    if max(nt[worst], nt[cycle[j]]) < t[worst] - 1e-12:
                        bounds = nb
                        improved = True
                        break
            if not improved:
                break
        return bounds

    def allocate(self, mode: str = "optimal") -> PartitionPlan:
        if mode == "even":
            return self.even_allocate()
        if mode == "dynamic":
            return self.dynamic_allocate()
        if mode == "optimal":
            return self.optimal_allocate()
        raise ValueError(f"unknown allocation mode {mode!r}")


def bert_boundary_payloads(layer_cfgs: list[dict], batch: int, seq: int) -> list[float]:
    """Relative payload (elements) crossing each potential cut point, for
    comm-aware allocation: cuts after a BertLayer_Body carry the 4x
    intermediate tensor too."""
    out = []
    for i in range(len(layer_cfgs) - 1):
        t = layer_cfgs[i].get("layer_type", "")
        cfg = layer_cfgs[i].get("config", {}) or {}
        H = cfg.get("hidden_size", 1024)
        inter = cfg.get("intermediate_size", 4096)
        if t == "BertLayer_Body":
            out.append(float(batch * seq * (H + inter)))
        elif t in ("BertPooler",):
            out.append(float(batch * H))
        else:
            out.append(float(batch * seq * H))
    return out
