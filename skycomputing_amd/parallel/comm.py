"""Process-per-GPU communication substrate (RCCL over xGMI).

MI355X-native replacement for the reference's TensorPipe RPC world
(reference: experiment/launch.py:37-46 init; scaelum/model/rpc_module.py
RRef forwarding; scaelum/utils.py:27-33 rpc_sync helpers). Design:

  * one OS process per GPU, SPMD: ``torch.distributed`` with the "nccl"
    backend (= RCCL on ROCm) for device-resident tensor traffic;
  * a side gloo group is the CONTROL PLANE: partition-table broadcast,
    benchmark gathers, checkpoint object traffic, and the one-time shape/
    dtype handshake for each P2P channel (replacing RPC control traffic,
    SURVEY.md §2c C2/C3/C9);
  * stage-boundary activation/grad hops are RCCL point-to-point send/recv,
    device-to-device over single xGMI links — no CPU staging (the reference
    round-tripped hops through host memory, module_wrapper.py:172-175).

On CPU (tests) the same code runs with gloo as the tensor backend.
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist

_CTX: "CommContext | None" = None


class _RetainingWork:
    """Wraps a dist work handle, keeping the sent buffer alive until
    wait()."""

    def __init__(self, work, buf):
        self._work = work
        self._buf = buf

    def wait(self):
        self._work.wait()
        self._buf = None


class CommContext:
    def __init__(self, backend: str, device: torch.device, ctrl_group):
        self.backend = backend
        self.device = device
        self.ctrl = ctrl_group
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        self._meta_cache_send: dict = {}
        self._meta_cache_recv: dict = {}

    # ---------------- control plane (gloo, object-based) ----------------

    def broadcast_object(self, obj, src: int = 0):
        buf = [obj if self.rank == src else None]
        dist.broadcast_object_list(buf, src=src, group=self.ctrl)
        return buf[0]

    def all_gather_object(self, obj) -> list:
        out = [None] * self.world_size
        dist.all_gather_object(out, obj, group=self.ctrl)
        return out

    def gather_object(self, obj, dst: int = 0) -> list | None:
        out = [None] * self.world_size if self.rank == dst else None
        dist.gather_object(obj, out, dst=dst, group=self.ctrl)
        return out

    def send_object(self, obj, dst: int):
        dist.send_object_list([obj], dst=dst, group=self.ctrl)

    def recv_object(self, src: int):
        buf = [None]
        dist.recv_object_list(buf, src=src, group=self.ctrl)
        return buf[0]

    def barrier(self):
        if self.backend == "nccl":
            dist.barrier(device_ids=[self.device.index])
        else:
            dist.barrier()

    # ---------------- tensor plane (RCCL P2P) ----------------

    def send_tensors(self, tensors: list[torch.Tensor], dst: int, key: str):
        """Send a tuple of tensors to ``dst``. The first send on ``key``
        performs a one-time meta handshake over the control plane; later
        sends assume static shapes (revalidated cheaply sender-side)."""
        metas = [
            (tuple(t.shape), t.dtype, bool(t.requires_grad)) for t in tensors
        ]
        cached = self._meta_cache_send.get((dst, key))
        if cached != metas:
            self.send_object(("meta", metas), dst)
            self._meta_cache_send[(dst, key)] = metas
        for t in tensors:
            dist.send(t.detach().contiguous(), dst=dst)

    def recv_tensors(self, src: int, key: str) -> list[torch.Tensor]:
        """Receive a tuple of tensors from ``src`` into freshly allocated
        device buffers; float tensors flagged requires_grad arrive as leaf
        tensors ready for autograd."""
        metas = self._meta_cache_recv.get((src, key))
        if metas is None:
            tag, metas = self.recv_object(src)
            assert tag == "meta"
            self._meta_cache_recv[(src, key)] = metas
        out = []
        for shape, dtype, req in metas:
            buf = torch.empty(shape, dtype=dtype, device=self.device)
            dist.recv(buf, src=src)
            if req:
                buf.requires_grad_(True)
            out.append(buf)
        return out

    def cached_recv_meta(self, src: int, key: str):
        return self._meta_cache_recv.get((src, key))

    def isend_tensors(self, tensors: list, dst: int, key: str,
                      blocking: bool = False):
        """Non-blocking sends (the interleaved engine's transport): returns
        work handles that also retain the sent buffers until wait(). The
        channel meta must already be cached (do one ``blocking=True``
        iteration first — it delegates to the handshaking send_tensors)."""
        if blocking:
            self.send_tensors(tensors, dst, key)
            return []
        metas = [
            (tuple(t.shape), t.dtype, bool(t.requires_grad)) for t in tensors
        ]
        cached = self._meta_cache_send.get((dst, key))
        assert cached == metas, (
            f"channel ({dst},{key}) meta changed or missing; run a blocking "
            "handshake iteration first"
        )
        works = []
        for t in tensors:
            buf = t.detach().contiguous()
            works.append(_RetainingWork(dist.isend(buf, dst=dst), buf))
        return works

    def irecv_tensors(self, src: int, key: str):
        """Pre-post non-blocking receives for one microbatch hop so the
        xGMI link fills while the current microbatch computes (SURVEY §2c
        C4 overlap). Requires cached channel meta — returns None before the
        first (handshaking) blocking recv on this channel."""
        metas = self._meta_cache_recv.get((src, key))
        if metas is None:
            return None
        bufs, works = [], []
        for shape, dtype, req in metas:
            b = torch.empty(shape, dtype=dtype, device=self.device)
            works.append(dist.irecv(b, src=src))
            bufs.append((b, req))
        return bufs, works

    def wait_irecv(self, posted) -> list[torch.Tensor]:
        bufs, works = posted
        for w in works:
            w.wait()
        out = []
        for b, req in bufs:
            if req:
                b.requires_grad_(True)
            out.append(b)
        return out

    def send_tensors_async(self, tensors: list, dst: int, key: str) -> list:
        """Non-blocking sends when the channel meta is already cached (the
        common steady state); falls back to the handshaking blocking send
        on first use. Returns work handles to wait at phase end."""
        metas = [
            (tuple(t.shape), t.dtype, bool(t.requires_grad)) for t in tensors
        ]
        if self._meta_cache_send.get((dst, key)) != metas:
            self.send_tensors(tensors, dst, key)
            return []
        works = []
        for t in tensors:
            buf = t.detach().contiguous()
            works.append(_RetainingWork(dist.isend(buf, dst=dst), buf))
        return works

    def recv_tensors_into(self, bufs: list, src: int):
        """Receive into preallocated (static) buffers — the graphed
        pipeline executor's transport (no allocation, no meta traffic)."""
        for b in bufs:
            if torch.is_tensor(b):
                dist.recv(b.detach(), src=src)

    def fused_send_recv(self, sends, recvs):
        """Post sends and recvs as ONE batch_isend_irecv — required when
        traffic crosses in both directions between a rank pair (1F1B steady
        state), where sequential blocking rendezvous ops can deadlock.

        sends: [(tensor_list, dst)], recvs: [(spec_list, src)] with specs
        (shape, dtype, requires_grad). Returns one buffer list per recv.
        """
        ops = []
        for tensors, dst in sends:
            for t in tensors:
                ops.append(dist.P2POp(dist.isend, t.detach().contiguous(), dst))
        out = []
        for specs, src in recvs:
            bufs = []
            for shape, dtype, req in specs:
                b = torch.empty(shape, dtype=dtype, device=self.device)
                bufs.append(b)
                ops.append(dist.P2POp(dist.irecv, b, src))
            out.append(bufs)
        if ops:
            for r in dist.batch_isend_irecv(ops):
                r.wait()
        for (specs, _src), bufs in zip(recvs, out):
            for (shape, dtype, req), b in zip(specs, bufs):
                if req:
                    b.requires_grad_(True)
        return out

    def reset_channels(self):
        """Drop cached channel metadata (call after re-allocation changes
        the partition and therefore the boundary payloads).

        NOTE: sender caches are keyed by (peer, key); both sides must reset
        together — the allocator does this right after broadcasting a new
        partition table."""
        self._meta_cache_send.clear()
        self._meta_cache_recv.clear()


def init_distributed(
    backend: str | None = None,
    timeout_s: int = 600,
    device: torch.device | None = None,
) -> CommContext:
    """Initialize the SPMD world from torchrun-style env vars.

    Degenerate single-process worlds (no RANK in env) are initialized with a
    file store so the same code path runs in ``bench.py`` at N=1.
    """
    global _CTX
    if _CTX is not None:
        return _CTX
    rank = int(os.environ.get("RANK", 0))
    world_size = int(os.environ.get("WORLD_SIZE", 1))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if device is None:
        if torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK", rank % max(torch.cuda.device_count(), 1)))
            device = torch.device("cuda", local_rank)
            torch.cuda.set_device(device)
        else:
            device = torch.device("cpu")
    elif device.type == "cuda":
        torch.cuda.set_device(device)

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        os.environ.setdefault("RANK", str(rank))
        os.environ.setdefault("WORLD_SIZE", str(world_size))
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world_size,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    ctrl = dist.new_group(backend="gloo", timeout=datetime.timedelta(seconds=timeout_s))
    _CTX = CommContext(backend, device, ctrl)
    return _CTX


def get_comm() -> CommContext:
    if _CTX is None:
        raise RuntimeError("call init_distributed() first")
    return _CTX


def destroy():
    global _CTX
    if dist.is_initialized():
        dist.destroy_process_group()
    _CTX = None
