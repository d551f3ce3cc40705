"""Client-side multi-server autoregressive inference session.

Parity with reference ``client/inference_session.py`` (:97-414): one
`_ServerSession` per span, fault recovery by replaying input history into a
replacement server's fresh KV cache, `position` setter for speculative-decoding
rollback (start_from_position), and server-to-server push (`next_servers`
metadata) so intermediate activations skip the client round-trip.

Replay protocol (cleaner than the reference's but same capability): every
server session tracks `position` (tokens in that server's KV cache) and keeps
`history` (all inputs its span has consumed). A session whose position lags
the client position receives `history[position:] + current step` in a single
multi-token step with start_from_position, rebuilding its cache exactly.
"""

from __future__ import annotations

import asyncio
import itertools
import logging
import time
import uuid
from typing import List, Optional

import torch

from petals_amd.client.routing.sequence_manager import RemoteSequenceManager
from petals_amd.data_structures import CHAIN_DELIMITER, RemoteSpanInfo
from petals_amd.p2p.transport import RpcMessage, RpcStream
from petals_amd.utils.misc import DUMMY, DUMMY_INT64, is_dummy

logger = logging.getLogger(__name__)


class _NeedFullReplay(Exception):
    """Raised when a session's input history cannot cover a replay; the chain
    must be rebuilt from block 0."""


class _ServerSession:
    """An open rpc_inference stream to one span server."""

    def __init__(self, manager: RemoteSequenceManager, span: RemoteSpanInfo, max_length: int, batch_size: int):
        self.manager = manager
        self.span = span
        self.max_length = max_length
        self.batch_size = batch_size
        self.session_id = str(uuid.uuid4())
        self.stream: Optional[RpcStream] = None
        self.position = 0
        self.history: Optional[torch.Tensor] = None  # inputs this span consumed
        self.stepped = False

    async def _aopen(self):
        uids = self.manager.block_uids[self.span.start : self.span.end]
        meta = {
            "uids": CHAIN_DELIMITER.join(uids),
            "max_length": self.max_length,
            "batch_size": self.batch_size,
            "session_id": self.session_id,
            "active_adapter": self.manager.config.active_adapter,
        }
        addr = self.manager.address_of(self.span.peer_id)
        self.stream = await self.manager.p2p.open_stream(
            addr, "petals.rpc_inference", RpcMessage(meta=meta), timeout=self.manager.config.connect_timeout
        )
        ack = await self.stream.receive(timeout=self.manager.config.request_timeout)
        assert ack.meta.get("session_open"), f"bad session ack: {ack.meta}"

    def open(self):
        self.manager.run_coroutine(self._aopen(), timeout=self.manager.config.request_timeout + 10)

    async def _asend_step(self, meta: dict, tensors: List[torch.Tensor]):
        await self.stream.send(RpcMessage(meta=meta, tensors=tensors))

    async def _arecv_step(self, timeout: float) -> RpcMessage:
        return await self.stream.receive(timeout=timeout)

    def close(self):
        if self.stream is None:
            return
        try:
            self.manager.run_coroutine(self.stream.close(), timeout=5)
        except Exception:  # noqa: BLE001
            pass
        self.stream = None


class InferenceSession:
    """Synchronous user-facing session over the full block range."""

    def __init__(self, sequence_manager: RemoteSequenceManager, max_length: int, batch_size: int = 1):
        self._manager = sequence_manager
        self._max_length = max_length
        self._batch_size = batch_size
        self._position = 0
        self._sessions: List[_ServerSession] = []
        self._closed = False
        self.output_ids: Optional[torch.Tensor] = None  # used by RemoteGenerationMixin resume

    @property
    def num_blocks(self) -> int:
        return self._manager.num_blocks

    @property
    def position(self) -> int:
        return self._position

    @position.setter
    def position(self, new_position: int) -> None:
        """Rollback for speculative decoding: next step sends
        start_from_position so servers rewind their KV caches."""
        assert new_position <= self._position, "position can only be rolled back"
        self._position = new_position
        for s in self._sessions:
            s.position = min(s.position, new_position)
            if s.history is not None and s.history.shape[1] > new_position:
                s.history = s.history[:, :new_position] if new_position > 0 else None

    @property
    def max_length(self) -> int:
        return self._max_length

    def __enter__(self) -> "InferenceSession":
        assert not self._closed
        return self

    # ------------------------------------------------------------ plumbing

    def _open_sessions(self, start: int, end: int) -> List[_ServerSession]:
        spans = self._manager.make_sequence(
            start, end, mode="min_latency", cache_tokens_needed=self._batch_size * self._max_length
        )
        spans[-1].end = min(spans[-1].end, end)
        sessions = []
        for span in spans:
            s = _ServerSession(self._manager, span, self._max_length, self._batch_size)
            s.open()
            sessions.append(s)
        return sessions

    def _update_sequence(self, server_idx: int, block_idx: int) -> None:
        """Replace the failed session (and everything it covered) with fresh
        servers; history replay happens lazily on their first step."""
        old = self._sessions[server_idx : server_idx + 1]
        update_end = old[0].span.end if old else self.num_blocks
        for s in old:
            s.close()
        new_sessions = self._open_sessions(block_idx, update_end)
        if old and old[0].history is not None:
            new_sessions[0].history = old[0].history
        self._sessions[server_idx : server_idx + 1] = new_sessions

    # ---------------------------------------------------------------- step

    def step(
        self,
        inputs: torch.Tensor,
        prompts: Optional[torch.Tensor] = None,
        hypo_ids: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        assert not self._closed
        if prompts is None or is_dummy(prompts):
            prompts = DUMMY
        else:
            assert prompts.ndim == 4 and prompts.shape[0] == self.num_blocks
        if hypo_ids is None or is_dummy(hypo_ids):
            hypo_ids = DUMMY_INT64
        else:
            assert hypo_ids.dtype == torch.int64

        inputs_device, inputs_dtype = inputs.device, inputs.dtype
        # inputs stay on their device: the in-proc/mesh paths move them without
        # ever visiting the host, and socket serialization does .cpu() itself
        inputs = inputs.detach()
        if inputs.dtype not in (torch.bfloat16, torch.float16, torch.float32):
            inputs = inputs.float()
        prompts = prompts.detach() if not is_dummy(prompts) else prompts
        n = inputs.shape[1]
        if self._position + n > self._max_length:
            raise ValueError(f"max_length exceeded: {self._position} + {n} > {self._max_length}")
        step_id = str(uuid.uuid4())

        if not self._sessions:
            self._sessions = self._open_sessions(0, self.num_blocks)

        # fast path: server-to-server push when every session is in sync
        use_push = (
            self._manager.config.use_server_to_server
            and len(self._sessions) > 1
            and is_dummy(prompts)
            and all(s.position == self._position and s.stepped for s in self._sessions)
        )
        if use_push:
            try:
                out = self._step_pushed(inputs, hypo_ids, step_id)
                self._position += n
                return out.to(device=inputs_device, dtype=inputs_dtype)
            except Exception as e:  # noqa: BLE001
                logger.warning("server-to-server push failed (%r); falling back to relayed steps", e)

        out = self._step_sequential(inputs, prompts, hypo_ids, step_id)
        self._position += n
        return out[:, -n:].to(device=inputs_device, dtype=inputs_dtype)

    def _step_sequential(self, inputs, prompts, hypo_ids, step_id) -> torch.Tensor:
        n = inputs.shape[1]
        server_idx = 0
        block_idx = 0
        flowing = inputs  # activations flowing between spans (may include replay tokens)
        while block_idx < self.num_blocks:
            for attempt_no in itertools.count():
                session = None
                try:
                    if server_idx >= len(self._sessions) or attempt_no >= 1:
                        self._update_sequence(server_idx, block_idx)
                    session = self._sessions[server_idx]
                    flowing_out = self._session_step(session, flowing, prompts, hypo_ids, step_id, n)
                    server_idx += 1
                    block_idx = session.span.end
                    flowing = flowing_out
                    self._manager.on_request_success(session.span.peer_id)
                    break
                except _NeedFullReplay:
                    # a mid-chain session lost its input history (this happens
                    # after server-to-server pushed steps, where intermediate
                    # activations never pass through the client). Rebuild the
                    # whole chain from block 0 using the first span's complete
                    # input history.
                    logger.warning("replaying the full chain from block 0 to rebuild lost caches")
                    first_history = self._sessions[0].history
                    assert first_history is not None and first_history.shape[1] >= self._position
                    for s in self._sessions:
                        s.position = 0
                        if s.span.start != 0:
                            s.history = None
                    server_idx, block_idx = 0, 0
                    flowing = torch.cat([first_history[:, : self._position], inputs], dim=1)
                    self._sessions[0].history = None
                    break
                except Exception as e:  # noqa: BLE001
                    peer = session.span.peer_id if session is not None else None
                    self._manager.on_request_failure(peer)
                    max_retries = self._manager.config.max_retries
                    if max_retries is not None and attempt_no + 1 > max_retries:
                        raise
                    delay = self._manager.get_retry_delay(attempt_no)
                    logger.warning(
                        "inference step via %s failed (%r); retry in %.1f s",
                        peer[:8] if peer else "?", e, delay,
                    )
                    time.sleep(delay)
        return flowing

    def _session_step(
        self,
        session: _ServerSession,
        flowing: torch.Tensor,
        prompts: torch.Tensor,
        hypo_ids: torch.Tensor,
        step_id: str,
        n: int,
    ) -> torch.Tensor:
        """One span's share of a step, including lazy history replay."""
        expected_len = self._position + n - session.position
        if flowing.shape[1] >= expected_len:
            span_inputs = flowing[:, -expected_len:]
        else:
            # flowing carries only the current step; prepend replay from history
            assert flowing.shape[1] == n, (flowing.shape, n)
            if session.history is None or session.history.shape[1] < self._position:
                raise _NeedFullReplay(session.span)
            span_inputs = torch.cat([session.history[:, session.position : self._position], flowing], dim=1)

        span_prompts = DUMMY if is_dummy(prompts) else prompts[session.span.start : session.span.end]
        meta = {
            "step_id": step_id,
            "start_from_position": session.position,
        }
        tensors = [span_inputs, span_prompts, hypo_ids]

        import os as _os
        import time as _time

        _trace = _os.environ.get("PETALS_AMD_STEP_TRACE")
        _t0 = _time.perf_counter()

        async def roundtrip():
            _ts = _time.perf_counter()
            await session._asend_step(meta, tensors)
            _tsent = _time.perf_counter()
            while True:
                msg = await session._arecv_step(self._manager.config.request_timeout)
                if msg.meta.get("step_id") == step_id and msg.tensors:
                    if _trace:
                        print(f"[cli] in-loop send {( _tsent-_ts)*1e3:.2f} recv {(_time.perf_counter()-_tsent)*1e3:.2f} ms", flush=True)
                    return msg.tensors[0]

        out = self._manager.run_coroutine(roundtrip(), timeout=self._manager.config.request_timeout + 10)
        if _trace:
            print(f"[cli] step total {(_time.perf_counter()-_t0)*1e3:.2f} ms (incl thread hop)", flush=True)
        assert out.shape == span_inputs.shape, f"{out.shape} vs {span_inputs.shape}"
        # bookkeeping
        base = session.history[:, : session.position] if session.history is not None else span_inputs[:, :0]
        session.history = torch.cat([base, span_inputs], dim=1)
        session.position = self._position + n
        session.stepped = True
        return out

    def _step_pushed(self, inputs: torch.Tensor, hypo_ids: torch.Tensor, step_id: str) -> torch.Tensor:
        """Send to the first span; servers hand activations to each other over
        rpc_push (RCCL for co-located spans); we await the last span's output.
        When the last span's server shares a LocalMesh with THIS process, the
        final activation also comes back over RCCL and the client stream
        carries only metadata."""
        from petals_amd.parallel.mesh import get_local_mesh

        first, last = self._sessions[0], self._sessions[-1]
        next_servers = []
        for s in self._sessions[1:]:
            addr = self._manager.address_of(s.span.peer_id)
            si = s.span.server_info
            next_servers.append(
                {
                    "addr": list(addr),
                    "session_id": s.session_id,
                    "start": s.span.start,
                    "end": s.span.end,
                    "mesh_id": getattr(si, "mesh_id", None),
                    "mesh_rank": getattr(si, "mesh_rank", None),
                }
            )
        meta = {
            "step_id": step_id,
            "start_from_position": self._position,
            "next_servers": next_servers,
        }
        last_info = last.span.server_info
        local_mesh = get_local_mesh(getattr(last_info, "mesh_id", None))
        if (
            local_mesh is not None
            and getattr(last_info, "mesh_rank", None) is not None
            and last_info.mesh_rank != local_mesh.rank
        ):
            meta["output_via_mesh"] = {"mesh_id": local_mesh.mesh_id, "rank": local_mesh.rank}
        tensors = [inputs, DUMMY, hypo_ids]

        async def roundtrip():
            await first._asend_step(meta, tensors)
            while True:
                msg = await last._arecv_step(self._manager.config.request_timeout)
                if msg.meta.get("step_id") != step_id:
                    continue
                tvm = msg.meta.get("tensors_via_mesh")
                if tvm is not None:
                    handle = local_mesh.post_recv(
                        int(tvm["src_rank"]), int(tvm["ticket"]), tvm["shape"], tvm["dtype"]
                    )
                    return await asyncio.wait_for(
                        asyncio.wrap_future(handle.future), self._manager.config.request_timeout
                    )
                if msg.tensors:
                    return msg.tensors[0]

        out = self._manager.run_coroutine(roundtrip(), timeout=self._manager.config.request_timeout + 10)
        n = inputs.shape[1]
        for s in self._sessions:
            base = s.history if s.history is not None else None
            if s.span.start == 0:  # only the first span's exact inputs are known client-side
                s.history = torch.cat([base, inputs], dim=1) if base is not None else inputs
            s.position = self._position + n
        return out

    def close(self):
        if not self._closed:
            for s in self._sessions:
                s.close()
            self._sessions.clear()
            self._closed = True

    def __exit__(self, *exc):
        self.close()

    def __del__(self):
        try:
            self.close()
        except Exception:  # noqa: BLE001
            pass
