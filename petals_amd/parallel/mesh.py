"""LocalMesh: the RCCL-over-xGMI activation hand-off tier for co-located
servers.

The MI355X-native replacement for TCP `rpc_push` between servers that share
one node (the reference always pays CPU serialize + TCP per hop,
/root/reference/src/petals/server/handler.py:310-350): each server process
owns one GPU and one rank of a shared torch.distributed process group
(backend "nccl" == RCCL on ROCm); activation tensors move rank->rank over the
xGMI point-to-point links without ever touching the host. The TCP swarm path
remains the cross-node fallback and carries the (tiny) step metadata even for
mesh transfers.

Ordering. RCCL p2p has no tags: both endpoints of a directed pair MUST issue
their send/recv ops in the same order. The control plane assigns a per
directed-pair monotonic *ticket* to every transfer: the sender issues sends in
ticket order (tickets are assigned in enqueue order under a lock and a single
comm thread issues FIFO), and the receiver holds posted recvs back until all
lower tickets from that src have been issued. This makes concurrent sessions
over the same pair safe.

Failure domain. A mesh spans ONE node; if a member dies the RCCL communicator
is broken and every member's server falls back to TCP for subsequent steps
(`mark_broken`). Cross-node fault tolerance is unchanged (TCP + DHT).
"""

from __future__ import annotations

import collections
import logging
import threading
import time
from concurrent.futures import Future
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)


class MeshError(Exception):
    pass


class _SendEntry:
    """One (possibly pre-announced) outbound transfer."""

    __slots__ = ("ticket", "tensor", "ready_event", "fut", "committed")

    def __init__(self):
        self.ticket = -1
        self.tensor: Optional[torch.Tensor] = None
        self.ready_event = None
        self.fut: Future = Future()
        self.committed = False


class MeshRecvHandle:
    """A pending inbound transfer; resolves to a device tensor once the RCCL
    recv has completed (and its stream has been synchronized)."""

    __slots__ = ("future", "shape", "dtype", "src")

    def __init__(self, shape, dtype, src):
        self.future: Future = Future()
        self.shape = tuple(shape)
        self.dtype = dtype
        self.src = src

    def result(self, timeout: Optional[float] = None) -> torch.Tensor:
        return self.future.result(timeout)


class LocalMesh:
    """Ticketed p2p transport over one torch.distributed process group.

    All public methods are thread-safe; RCCL ops are issued by a single comm
    thread on a dedicated stream, so they never interleave with collectives or
    compute issued by other threads on other streams/communicators.
    """

    def __init__(
        self,
        mesh_id: str,
        rank: int,
        world: int,
        device: Optional[torch.device] = None,
        group: Optional[dist.ProcessGroup] = None,
    ):
        assert dist.is_initialized(), "torch.distributed must be initialized before LocalMesh"
        self.mesh_id = mesh_id
        self.rank = rank
        self.world = world
        self.device = torch.device(device) if device is not None else None
        self.group = group  # None -> default PG. Prefer a dedicated PG so mesh
        # p2p never shares a communicator with collectives from other threads.
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)
        self._send_tickets: Dict[int, int] = collections.defaultdict(int)  # dst -> next ticket
        self._recv_next: Dict[int, int] = collections.defaultdict(int)  # src -> next expected ticket
        self._recv_pending: Dict[int, Dict[int, MeshRecvHandle]] = collections.defaultdict(dict)
        self._queue: collections.deque = collections.deque()
        self._broken: Optional[str] = None
        self._shutdown = False
        self._stream = None
        self._thread = threading.Thread(target=self._run, name=f"LocalMesh-{mesh_id}", daemon=True)
        self._thread.start()

    # ------------------------------------------------------------- control

    @property
    def is_usable(self) -> bool:
        return self._broken is None and not self._shutdown and self._thread.is_alive()

    def mark_broken(self, reason: str) -> None:
        with self._cv:
            if self._broken is None:
                self._broken = reason
                logger.warning("mesh %s marked broken: %s", self.mesh_id, reason)
            self._cv.notify_all()
        self._fail_all_pending(MeshError(f"mesh broken: {reason}"))

    def shutdown(self) -> None:
        with self._cv:
            self._shutdown = True
            self._cv.notify_all()
        self._thread.join(timeout=5)
        self._fail_all_pending(MeshError("mesh shut down"))

    def _fail_all_pending(self, exc: Exception) -> None:
        with self._cv:
            pending = [h for d in self._recv_pending.values() for h in d.values()]
            self._recv_pending.clear()
            items = list(self._queue)
            self._queue.clear()
        for h in pending:
            if not h.future.done():
                h.future.set_exception(exc)
        for item in items:  # ("send", dst, _SendEntry)
            entry = item[-1]
            if isinstance(entry, _SendEntry) and not entry.fut.done():
                entry.fut.set_exception(exc)

    def warmup_ring(self, timeout: float = 120.0) -> None:
        """Pass one dummy tensor around the ring (r -> r+1). MUST be called on
        every rank before serving when the group is NCCL/RCCL: communicator
        init is collective and lazy, so without this the first decode chain
        deadlocks (rank r's first p2p op only happens once the token reaches
        it, which requires init to have completed on all ranks). Also primes
        exactly the pairs the pipeline uses and advances every ticket counter
        identically on all ranks."""
        if self.world < 2:
            return
        dst = (self.rank + 1) % self.world
        src = (self.rank - 1) % self.world
        dev = self.device if self.device is not None else "cpu"
        fut, ticket = self.send(torch.zeros(1, device=dev), dst=dst)
        assert ticket == 0, "warmup_ring must run before any other transfer"
        handle = self.post_recv(src, 0, (1,), "f32")
        handle.result(timeout)
        fut.result(timeout)

    # ------------------------------------------------------------ transfers

    def send(self, tensor: torch.Tensor, dst: int, ready_event=None) -> Tuple[Future, int]:
        """Enqueue a send; the per-dst ticket is assigned atomically at enqueue
        time (so ticket order == issue order, with no holes). Returns
        (future resolved when the transfer is on the wire, ticket)."""
        ticket, commit, _abort = self.send_deferred(dst)
        fut = commit(tensor, ready_event)
        return fut, ticket

    def send_deferred(self, dst: int):
        """Reserve the next ticket to `dst` NOW and supply the tensor later
        ("pre-announce"): the caller can ship the ticket to the receiver over
        TCP before the producing compute has even started, so the control
        plane runs ahead of the data plane. The comm thread holds back all
        later sends to the same dst until this entry is committed (ticket
        order is preserved).

        Returns (ticket, commit(tensor, ready_event=None) -> Future,
        abort(reason)). An aborted entry breaks the mesh (the receiver is
        already expecting this ticket; silently skipping it would desync the
        pair) — the swarm falls back to TCP."""
        entry = _SendEntry()
        with self._cv:
            if self._broken:
                raise MeshError(f"mesh broken: {self._broken}")
            ticket = self._send_tickets[dst]
            self._send_tickets[dst] = ticket + 1
            entry.ticket = ticket
            self._queue.append(("send", dst, entry))
            self._cv.notify_all()

        def commit(tensor: torch.Tensor, ready_event=None) -> Future:
            with self._cv:
                entry.tensor = tensor
                entry.ready_event = ready_event
                entry.committed = True
                self._cv.notify_all()
            return entry.fut

        def abort(reason: str) -> None:
            self.mark_broken(f"deferred send to {dst} (ticket {entry.ticket}) aborted: {reason}")

        return ticket, commit, abort

    def post_recv(self, src: int, ticket: int, shape, dtype_str: str) -> MeshRecvHandle:
        """Register an expected inbound transfer (called when the TCP meta for
        a mesh transfer arrives). Recvs are issued strictly in ticket order per
        src; out-of-order posts are held back."""
        from petals_amd.utils.serialization import _STR_TO_DTYPE

        dtype = _STR_TO_DTYPE[dtype_str] if isinstance(dtype_str, str) else dtype_str
        handle = MeshRecvHandle(shape, dtype, src)
        with self._cv:
            if self._broken:
                raise MeshError(f"mesh broken: {self._broken}")
            self._recv_pending[src][ticket] = handle
            self._cv.notify_all()
        return handle

    # ----------------------------------------------------------- comm thread

    def _ready_recvs(self) -> List[Tuple[int, int, MeshRecvHandle]]:
        """Under lock: pop recvs whose ticket is next-expected for their src."""
        ready = []
        for src, pending in self._recv_pending.items():
            while self._recv_next[src] in pending:
                t = self._recv_next[src]
                ready.append((src, t, pending.pop(t)))
                self._recv_next[src] = t + 1
        return ready

    def _run(self):
        """Issue ops non-blockingly and poll completions: the comm thread must
        NEVER host-block on one transfer while another (possibly the one the
        peer is waiting for) still needs issuing — bidirectional pushes between
        a pair would deadlock otherwise."""
        use_cuda = self.device is not None and self.device.type == "cuda"
        if use_cuda:
            torch.cuda.set_device(self.device)
            self._stream = torch.cuda.Stream(device=self.device)
        else:
            # gloo p2p ops progress only inside Work.wait() (is_completed()
            # never flips on its own) — hand each work to a waiter thread
            from concurrent.futures import ThreadPoolExecutor

            self._waiters = ThreadPoolExecutor(max_workers=32, thread_name_prefix=f"mesh-wait-{self.mesh_id}")
        outstanding: List[tuple] = []  # (kind, work, future, buf_or_None)
        send_q: Dict[int, collections.deque] = collections.defaultdict(collections.deque)
        while True:
            with self._cv:
                if (
                    not self._queue
                    and not self._any_ready_locked()
                    and not outstanding
                    and not self._any_committed_locked(send_q)
                    and not self._shutdown
                ):
                    self._cv.wait(timeout=0.5)
                if self._shutdown:
                    return
                while self._queue:  # entries arrive in ticket order per dst
                    _, dst, entry = self._queue.popleft()
                    send_q[dst].append(entry)
                sends = []
                for dst, q in send_q.items():
                    # strict ticket order per pair: stop at the first
                    # uncommitted (pre-announced, compute in flight) entry
                    while q and q[0].committed:
                        sends.append((dst, q.popleft()))
                recvs = self._ready_recvs()
            try:
                use_stream = self._stream is not None
                ctx = torch.cuda.stream(self._stream) if use_stream else _nullcontext()
                with ctx:
                    for dst, entry in sends:
                        if entry.ready_event is not None and use_stream:
                            self._stream.wait_event(entry.ready_event)
                        t = entry.tensor if entry.tensor.is_contiguous() else entry.tensor.contiguous()
                        w = dist.isend(t, dst=dst, group=self.group)
                        if use_cuda:
                            outstanding.append(("send", w, entry.fut, t))  # keep t alive until done
                        else:
                            self._waiters.submit(_wait_and_resolve, w, entry.fut, True, (t,))
                    for src, _ticket, handle in recvs:
                        buf = torch.empty(
                            handle.shape,
                            dtype=handle.dtype,
                            device=self.device if self.device is not None else "cpu",
                        )
                        w = dist.irecv(buf, src=src, group=self.group)
                        if use_cuda:
                            outstanding.append(("recv", w, handle.future, buf))
                        else:
                            self._waiters.submit(_wait_and_resolve, w, handle.future, buf, ())
                still = []
                for kind, w, fut, buf in outstanding:
                    # NCCL/RCCL: is_completed() queries the op's end-of-op GPU
                    # event, so a True result means the data is globally
                    # visible and consumable on any stream
                    if w.is_completed():
                        if not fut.done():
                            fut.set_result(buf if kind == "recv" else True)
                    else:
                        still.append((kind, w, fut, buf))
                outstanding = still
                if outstanding:
                    time.sleep(20e-6)
            except Exception as e:  # noqa: BLE001
                logger.exception("mesh comm thread failed")
                for _dst, entry in sends:
                    if not entry.fut.done():
                        entry.fut.set_exception(e)
                for _, _, h in recvs:
                    if not h.future.done():
                        h.future.set_exception(e)
                for _, _w, fut, _buf in outstanding:
                    if not fut.done():
                        fut.set_exception(e)
                self.mark_broken(repr(e))
                return

    def _any_ready_locked(self) -> bool:
        return any(self._recv_next[src] in pending for src, pending in self._recv_pending.items())

    @staticmethod
    def _any_committed_locked(send_q) -> bool:
        return any(q and q[0].committed for q in send_q.values())


# Process-local mesh registry: the thin client (and bench harnesses) in a
# server process use the same mesh for final-output delivery.
_LOCAL_MESH: Dict[str, LocalMesh] = {}


def register_local_mesh(mesh: LocalMesh) -> None:
    _LOCAL_MESH[mesh.mesh_id] = mesh


def get_local_mesh(mesh_id: Optional[str] = None) -> Optional[LocalMesh]:
    if mesh_id is not None:
        m = _LOCAL_MESH.get(mesh_id)
        return m if m is not None and m.is_usable else None
    for m in _LOCAL_MESH.values():
        if m.is_usable:
            return m
    return None


def _wait_and_resolve(work, fut: Future, result, keepalive=()):
    try:
        work.wait()
        if not fut.done():
            fut.set_result(result)
    except Exception as e:  # noqa: BLE001
        if not fut.done():
            fut.set_exception(e)
    finally:
        del keepalive


class _nullcontext:
    def __enter__(self):
        return None

    def __exit__(self, *a):
        return False
