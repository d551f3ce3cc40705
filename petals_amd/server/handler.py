"""The server's RPC surface — 7 RPCs matching the reference's wire API
(`server/handler.py:132-592`): rpc_inference (bidi stream), rpc_forward,
rpc_backward, rpc_forward_stream, rpc_backward_stream, rpc_push, rpc_info.

Single-process MI355X design: handlers are asyncio coroutines on the server
loop; all GPU work is submitted to the PriorityRuntime thread (inference
priority beats training). Server-to-server activation push (`rpc_push`)
delivers a step's inputs straight into the *next* server's open inference
session, skipping the client round-trip.
"""

from __future__ import annotations

import asyncio
import contextlib
import logging
import os
import time
from typing import Any, Dict, List, Optional, Sequence, Tuple

import torch

from petals_amd.data_structures import CHAIN_DELIMITER, InferenceMetadata, ModuleUID
from petals_amd.p2p.streaming import receive_tensors_streamed, send_tensors_streamed
from petals_amd.p2p.transport import P2PNode, RpcError, RpcMessage, RpcStream
from petals_amd.server.backend import TransformerBackend
from petals_amd.server.memory_cache import MemoryCache, TensorDescriptor
from petals_amd.server.scheduler import PriorityRuntime, TaskPrioritizer
from petals_amd.utils.misc import DUMMY, is_dummy

logger = logging.getLogger(__name__)


class _Session:
    """Server-side state of one rpc_inference session."""

    def __init__(self, session_id: str, uids: List[ModuleUID], max_length: int):
        self.session_id = session_id
        self.uids = uids
        self.max_length = max_length
        self.prefix_length = 0
        self.pushed_inputs: asyncio.Queue = asyncio.Queue()
        self.span_graph = None  # _SpanDecodeGraph once captured (GPU decode)
        self.tp_slot = 0  # shadow-rank KV slot (tensor parallelism)


class _SpanDecodeGraph:
    """Per-session hipGraph of one whole-span decode step (all blocks): the
    session's KV tensors have stable addresses for its lifetime, so a single
    capture replays for every generated token with only the device-resident
    position advancing (ops/fused_decode.DecodeContext). This gives the real
    serving path the same launch-overhead-free decode as bench.py."""

    def __init__(self, blocks, cache_pairs, batch: int, hidden_size: int, device, dtype, adapter_name,
                 initial_position: int):
        from petals_amd.ops.fused_decode import DecodeContext
        from petals_amd.utils.graphs import GraphedCallable
        from petals_amd.utils.peft import using_adapter

        self.ctx = DecodeContext(device)
        self.h_in = torch.empty(batch, 1, hidden_size, device=device, dtype=dtype)
        # warmup/capture writes garbage K/V at this position — it MUST be the
        # position the first real replay will overwrite (never the prefilled
        # prefix)
        self.ctx.set_position(initial_position)

        def span_fn():
            h = self.h_in
            self.ctx.norm_parts = None  # folded-norm hand-off starts fresh each pass
            with using_adapter(adapter_name):
                for block, (k, v) in zip(blocks, cache_pairs):
                    h = block(h, kv_cache=(k, v), ctx=self.ctx)
            return h

        self.graph = GraphedCallable(span_fn, [])

    def step(self, hidden_states: torch.Tensor, prefix_length: int) -> torch.Tensor:
        self.ctx.set_position(prefix_length)
        self.h_in.copy_(hidden_states.view(self.h_in.shape))
        return self.graph.replay()


def _dtype_str(dtype) -> str:
    from petals_amd.utils.serialization import _DTYPE_TO_STR

    return _DTYPE_TO_STR[dtype]


def _commit_if_needed(commit_cb, out: torch.Tensor) -> None:
    """Runs IN the runtime thread: hand the produced activation to a
    pre-announced mesh transfer. The CUDA event lets the comm thread isend as
    soon as the producing stream reaches this point — before the runtime's
    end-of-task synchronize."""
    if commit_cb is None:
        return
    evt = None
    if out.is_cuda:
        evt = torch.cuda.Event()
        evt.record()
    commit_cb(out, evt)


class TransformerConnectionHandler:
    def __init__(
        self,
        *,
        backends: Dict[ModuleUID, TransformerBackend],
        memory_cache: MemoryCache,
        runtime: PriorityRuntime,
        prioritizer: Optional[TaskPrioritizer] = None,
        inference_max_length: int = 8192,
        request_timeout: float = 180.0,
        session_timeout: float = 30 * 60,
        step_timeout: float = 5 * 60,
        server_info_extra=None,
        p2p: Optional[P2PNode] = None,
        adapters: Sequence[str] = (),
        mesh=None,
        default_compression: str = "none",
        tp_coord=None,
    ):
        self.backends = backends
        self.memory_cache = memory_cache
        self.runtime = runtime
        self.prioritizer = prioritizer or TaskPrioritizer()
        self.inference_max_length = inference_max_length
        self.request_timeout = request_timeout
        self.session_timeout = session_timeout
        self.step_timeout = step_timeout
        self.server_info_extra = server_info_extra or (lambda: {})
        self.adapters = tuple(adapters)
        self.p2p = p2p
        self.mesh = mesh  # LocalMesh: RCCL/xGMI hand-off tier for co-located spans
        self.default_compression = default_compression
        self.tp_coord = tp_coord  # TPCoordinator: rank-0 side of intra-server TP
        self._tp_slots = iter(range(1, 1 << 30))
        self._sessions: Dict[str, _Session] = {}

    def register(self, p2p: P2PNode) -> None:
        self.p2p = p2p
        for name, fn in (
            ("petals.rpc_info", self.rpc_info),
            ("petals.rpc_forward", self.rpc_forward),
            ("petals.rpc_forward_stream", self.rpc_forward_stream),
            ("petals.rpc_backward", self.rpc_backward),
            ("petals.rpc_backward_stream", self.rpc_backward_stream),
            ("petals.rpc_inference", self.rpc_inference),
            ("petals.rpc_push", self.rpc_push),
        ):
            p2p.add_handler(name, fn)

    # ------------------------------------------------------------- helpers

    def _parse_uids(self, meta: Dict[str, Any]) -> List[ModuleUID]:
        uids_field = meta.get("uids")
        if isinstance(uids_field, str):
            uids = uids_field.split(CHAIN_DELIMITER)
        else:
            uids = list(uids_field or ())
        if not uids:
            raise RpcError("request must name at least one module uid")
        for uid in uids:
            if uid not in self.backends:
                raise RpcError(f"uid {uid!r} is not served here")
        return uids

    def _split_prompts(self, prompts: Optional[torch.Tensor], n: int, dtype, device) -> List[Optional[torch.Tensor]]:
        if prompts is None or is_dummy(prompts):
            return [None] * n
        assert prompts.ndim == 4 and prompts.shape[0] == n, "prompts must be [num_blocks, batch, pre_seq, hidden]"
        return [p.to(device=device, dtype=dtype) for p in prompts]

    # ----------------------------------------------------------- rpc_info

    async def rpc_info(self, request: RpcMessage, stream: RpcStream) -> None:
        first = next(iter(self.backends.values()), None)
        cache_bytes_left = self.memory_cache.bytes_left
        bytes_per_token = first.cache_bytes_per_token() if first is not None else 1
        info = {
            "version": "petals_amd-0.1",
            "dht_client_mode": False,
            "cache_tokens_left": int(cache_bytes_left // max(1, bytes_per_token)),
            "adapters": list(self.adapters),
            **self.server_info_extra(),
        }
        await stream.close(RpcMessage(meta=info))

    # -------------------------------------------------------- rpc_forward

    def _forward_chain(
        self,
        uids: List[ModuleUID],
        hidden_states: torch.Tensor,
        prompts: Optional[torch.Tensor],
        active_adapter: Optional[str] = None,
    ) -> torch.Tensor:
        """Runs IN the runtime thread: chain of stateless block forwards."""
        from petals_amd.utils.peft import using_adapter

        backend0 = self.backends[uids[0]]
        device, dtype = backend0.device, backend0.dtype
        hidden_states = hidden_states.to(device=device, dtype=dtype)
        if self.tp_coord is not None:
            p = prompts if (prompts is not None and not is_dummy(prompts)) else None
            self.tp_coord.forward(hidden_states, p)
        prompt_list = self._split_prompts(prompts, len(uids), dtype, device)
        with using_adapter(active_adapter):
            for uid, prompt in zip(uids, prompt_list):
                if prompt is not None:
                    hidden_states = hidden_states.clone()
                    hidden_states[:, : prompt.shape[1]] += prompt
                hidden_states = self.backends[uid].forward(hidden_states)
        return hidden_states.cpu()

    async def _handle_forward(self, meta: Dict[str, Any], tensors: List[torch.Tensor]) -> List[torch.Tensor]:
        uids = self._parse_uids(meta)
        hidden_states = tensors[0]
        prompts = tensors[1] if len(tensors) > 1 else None
        assert hidden_states.ndim == 3
        priority = self.prioritizer.prioritize(hidden_states, points=meta.get("points", 0), type="forward")
        out = await self.runtime.submit(
            priority, self._forward_chain, uids, hidden_states, prompts, meta.get("active_adapter")
        )
        return [out]

    def _out_compressions(self, meta, n):
        c = meta.get("output_compression") or self.default_compression
        return [c] * n if c and c != "none" else None

    async def rpc_forward(self, request: RpcMessage, stream: RpcStream) -> None:
        outs = await asyncio.wait_for(self._handle_forward(request.meta, request.tensors), self.request_timeout)
        await stream.send(RpcMessage(tensors=outs), kind="end", compressions=self._out_compressions(request.meta, len(outs)))

    async def rpc_forward_stream(self, request: RpcMessage, stream: RpcStream) -> None:
        meta, tensors = await receive_tensors_streamed(stream, timeout=self.request_timeout)
        meta = {**request.meta, **meta}
        outs = await asyncio.wait_for(self._handle_forward(meta, tensors), self.request_timeout)
        await send_tensors_streamed(stream, outs, close=True)

    # ------------------------------------------------------- rpc_backward

    def _backward_chain(
        self,
        uids: List[ModuleUID],
        inputs: torch.Tensor,
        grad_outputs: torch.Tensor,
        prompts: Optional[torch.Tensor],
        active_adapter: Optional[str] = None,
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Runs IN the runtime thread. Re-runs forward to recover intermediate
        activations, then backward in reverse (parity: block_functions.py:84-141)."""
        from petals_amd.utils.peft import using_adapter

        backend0 = self.backends[uids[0]]
        device, dtype = backend0.device, backend0.dtype
        inputs = inputs.to(device=device, dtype=dtype)
        grad_outputs = grad_outputs.to(device=device, dtype=dtype)
        if self.tp_coord is not None:
            p = prompts if (prompts is not None and not is_dummy(prompts)) else None
            self.tp_coord.backward(inputs, grad_outputs, p)
        prompt_list = self._split_prompts(prompts, len(uids), dtype, device)
        # re-run forward (plain no_grad, NOT inference_mode: these activations
        # feed autograd below) to recover intermediate inputs
        inter_inputs: List[torch.Tensor] = []
        hidden = inputs
        adapter_scope = using_adapter(active_adapter)  # exception-safe via try/finally below
        adapter_scope.__enter__()
        try:
            return self._backward_chain_inner(uids, inputs, grad_outputs, prompt_list, inter_inputs, hidden)
        finally:
            adapter_scope.__exit__(None, None, None)

    def _backward_chain_inner(self, uids, inputs, grad_outputs, prompt_list, inter_inputs, hidden):
        with torch.no_grad():
            for uid, prompt in zip(uids[:-1], prompt_list[:-1]):
                if prompt is not None:
                    hidden = hidden.clone()
                    hidden[:, : prompt.shape[1]] += prompt
                inter_inputs.append(hidden)
                hidden = self.backends[uid].block(hidden)
            if prompt_list[-1] is not None:
                hidden = hidden.clone()
                hidden[:, : prompt_list[-1].shape[1]] += prompt_list[-1]
            inter_inputs.append(hidden)

        grad_prompts: List[Optional[torch.Tensor]] = []
        grad = grad_outputs
        for uid, hidden, prompt in zip(reversed(uids), reversed(inter_inputs), reversed(prompt_list)):
            # prompt was already added into `hidden`; grad wrt prompt is the
            # slice of grad wrt hidden over the prompt positions
            grad, _ = self.backends[uid].backward(hidden, grad, prompt=None)
            if prompt is not None:
                grad_prompts.append(grad[:, : prompt.shape[1]].clone())
            else:
                grad_prompts.append(None)
        grad_prompts.reverse()
        if any(gp is not None for gp in grad_prompts):
            ref = next(gp for gp in grad_prompts if gp is not None)
            grad_prompts_t = torch.stack([gp if gp is not None else torch.zeros_like(ref) for gp in grad_prompts])
        else:
            grad_prompts_t = DUMMY
        return grad.cpu(), grad_prompts_t.cpu() if not is_dummy(grad_prompts_t) else DUMMY

    async def _handle_backward(self, meta: Dict[str, Any], tensors: List[torch.Tensor]) -> List[torch.Tensor]:
        uids = self._parse_uids(meta)
        inputs, grad_outputs = tensors[0], tensors[1]
        prompts = tensors[2] if len(tensors) > 2 else None
        priority = self.prioritizer.prioritize(inputs, points=meta.get("points", 0), type="backward")
        grad_inputs, grad_prompts = await self.runtime.submit(
            priority, self._backward_chain, uids, inputs, grad_outputs, prompts, meta.get("active_adapter")
        )
        return [grad_inputs, grad_prompts]

    async def rpc_backward(self, request: RpcMessage, stream: RpcStream) -> None:
        outs = await asyncio.wait_for(self._handle_backward(request.meta, request.tensors), self.request_timeout)
        await stream.send(RpcMessage(tensors=outs), kind="end", compressions=self._out_compressions(request.meta, len(outs)))

    async def rpc_backward_stream(self, request: RpcMessage, stream: RpcStream) -> None:
        meta, tensors = await receive_tensors_streamed(stream, timeout=self.request_timeout)
        meta = {**request.meta, **meta}
        outs = await asyncio.wait_for(self._handle_backward(meta, tensors), self.request_timeout)
        await send_tensors_streamed(stream, outs, close=True)

    # ------------------------------------------------------ rpc_inference

    def _inference_step_chain(
        self,
        uids: List[ModuleUID],
        hidden_states: torch.Tensor,
        hypo_ids: torch.Tensor,
        prompts: Optional[torch.Tensor],
        handles: List[Tuple[int, int]],
        prefix_length: int,
        active_adapter: Optional[str],
        session: Optional[_Session] = None,
        keep_on_device: bool = False,
        commit_cb=None,
        reply_cb=None,
    ) -> torch.Tensor:
        """Runs IN the runtime thread: one inference step through the whole span
        (the single-process analog of reference _MergedInferenceStep)."""
        backend0 = self.backends[uids[0]]
        device, dtype = backend0.device, backend0.dtype
        if not torch.is_tensor(hidden_states):
            # MeshRecvHandle: the step was submitted BEFORE its input landed;
            # block here (the actual data dependency) instead of in the
            # asyncio loop so launch work overlaps the upstream hop's compute
            hidden_states = hidden_states.result(self.step_timeout)
        hidden_states = hidden_states.to(device=device, dtype=dtype)
        if self.tp_coord is not None and session is not None:
            hy = hypo_ids if (hypo_ids is not None and not is_dummy(hypo_ids)) else None
            p = prompts if (prompts is not None and not is_dummy(prompts)) else None
            h3d = hidden_states.view(hidden_states.shape[0], -1, hidden_states.shape[-1])
            max_chunk = backend0._estimate_max_chunk_length(h3d, prefix_length)
            self.tp_coord.step(session.tp_slot, h3d, prefix_length, hy, max_chunk, p)
        has_hypo = hypo_ids is not None and not is_dummy(hypo_ids)
        has_prompts = prompts is not None and not is_dummy(prompts)

        # fast serving path: whole-span hipGraph decode (1 new token, GPU,
        # every block on the fused path). Beam reorder runs eagerly before the
        # replay; prompts force the eager path (they change per step).
        all_fast = all(
            getattr(self.backends[uid].block, "_fast", None) is not None
            and getattr(self.backends[uid].block._fast, "graph_safe", False)
            for uid in uids
        )
        max_graph_batch = 1
        if all_fast:
            quants = {self.backends[uid].block._fast.quant for uid in uids}
            # NF4 batch 3-8 runs the native BATCH kernels since the scratch
            # fix; PETALS_AMD_GRAPH_MAX_BATCH caps span-graph capture for A/B
            default = "8"
            max_graph_batch = int(os.environ.get("PETALS_AMD_GRAPH_MAX_BATCH", default))
        if (
            session is not None
            and device.type == "cuda"
            and all_fast
            and hidden_states.shape[1] == 1
            and hidden_states.shape[0] <= max_graph_batch
            and not has_prompts
            and not os.environ.get("PETALS_AMD_NO_GRAPHS")
        ):
            if has_hypo:
                hypo = hypo_ids.to(device)
                with self.memory_cache.use_cache(*(h for pair in handles for h in pair)) as tensors:
                    for t in tensors:
                        t[...] = t[hypo]
            if session.span_graph is None:
                cache_pairs = []
                with self.memory_cache.use_cache(*(h for pair in handles for h in pair)) as tensors:
                    for i in range(len(uids)):
                        cache_pairs.append((tensors[2 * i], tensors[2 * i + 1]))
                blocks = [self.backends[uid].block for uid in uids]
                session.span_graph = _SpanDecodeGraph(
                    blocks, cache_pairs, hidden_states.shape[0], hidden_states.shape[-1],
                    device, dtype, active_adapter, initial_position=prefix_length,
                )
            if os.environ.get("PETALS_AMD_STEP_TRACE"):
                _g0 = time.perf_counter()
                out = session.span_graph.step(hidden_states, prefix_length)
                torch.cuda.synchronize()
                _g1 = time.perf_counter()
                # clone: the graph's static output is overwritten by the next
                # replay; anything leaving the runtime must own its storage
                res = out.clone() if keep_on_device else out.cpu()
                _commit_if_needed(commit_cb, res)
                if reply_cb is not None:
                    if res.is_cuda:
                        torch.cuda.current_stream().synchronize()
                    reply_cb(res)
                print(f"[rt] graph {( _g1-_g0)*1e3:.2f} out {(time.perf_counter()-_g1)*1e3:.2f} ms", flush=True)
                return res
            out = session.span_graph.step(hidden_states, prefix_length)
            out = out.clone() if keep_on_device else out.cpu()
            _commit_if_needed(commit_cb, out)
            if reply_cb is not None:
                # the client reads this tensor on ITS stream: finish ours first
                # (the scheduler syncs right after fn returns anyway, so this
                # only moves the sync before the delivery, not adds one)
                if out.is_cuda:
                    torch.cuda.current_stream().synchronize()
                reply_cb(out)
            return out

        if has_hypo:
            hypo_ids = hypo_ids.to(device)
        from petals_amd.utils.peft import using_adapter

        prompt_list = self._split_prompts(prompts, len(uids), dtype, device)
        with using_adapter(active_adapter):
            for uid, prompt, handle_pair in zip(uids, prompt_list, handles):
                if prompt is not None:
                    hidden_states = hidden_states.clone()
                    hidden_states[:, : prompt.shape[1]] += prompt
                info = InferenceMetadata(uid, prefix_length, tuple(handle_pair), active_adapter)
                (hidden_states,) = self.backends[uid].inference_step(hidden_states, hypo_ids, info)
        hidden_states = hidden_states if keep_on_device else hidden_states.cpu()
        _commit_if_needed(commit_cb, hidden_states)
        if reply_cb is not None:
            if hidden_states.is_cuda:
                torch.cuda.current_stream().synchronize()
            reply_cb(hidden_states)
        return hidden_states

    async def rpc_inference(self, request: RpcMessage, stream: RpcStream) -> None:
        meta = request.meta
        uids = self._parse_uids(meta)
        max_length = int(meta.get("max_length", 0))
        if not 0 < max_length <= self.inference_max_length:
            raise RpcError(f"max_length must be in (0, {self.inference_max_length}], got {max_length}")
        batch_size = int(meta.get("batch_size", 1))
        active_adapter = meta.get("active_adapter") or None
        session_id = meta.get("session_id") or f"anon-{id(stream)}"
        alloc_timeout = float(meta.get("alloc_timeout", self.memory_cache.alloc_timeout))

        descriptors: List[TensorDescriptor] = []
        for uid in uids:
            descriptors.extend(self.backends[uid].get_inference_cache_descriptors(batch_size, max_length))

        session = _Session(session_id, uids, max_length)
        self._sessions[session_id] = session
        t_start = time.monotonic()
        try:
            async with self.memory_cache.allocate_cache(*descriptors, timeout=alloc_timeout) as flat_handles:
                handles = [tuple(flat_handles[2 * i : 2 * i + 2]) for i in range(len(uids))]
                if self.tp_coord is not None:
                    session.tp_slot = next(self._tp_slots)
                    self.tp_coord.open_session(session.tp_slot, batch_size, max_length)
                # confirm session is open (client waits for this before step 1)
                await stream.send(RpcMessage(meta={"session_open": True, "session_id": session_id}))
                async for step_meta, tensors in self._iterate_inference_steps(stream, session):
                    if time.monotonic() - t_start > self.session_timeout:
                        raise RpcError("session timed out")
                    if "start_from_position" in step_meta:
                        pos = int(step_meta["start_from_position"])
                        if pos > session.prefix_length:
                            raise RpcError("start_from_position is ahead of the cache")
                        session.prefix_length = pos
                    hidden_states = tensors[0]
                    prompts = tensors[1] if len(tensors) > 1 else None
                    hypo_ids = tensors[2] if len(tensors) > 2 else None
                    if torch.is_tensor(hidden_states):
                        length_increment = hidden_states.shape[1] if hidden_states.numel() > 0 else 0
                    else:  # MeshRecvHandle: inputs arrive over RCCL; the shape
                        # is known from the pre-announce, so the step can be
                        # queued before the data lands (the runtime thread
                        # blocks on the actual dependency)
                        length_increment = hidden_states.shape[1]
                    if session.prefix_length + length_increment > max_length:
                        raise RpcError(
                            f"max_length exceeded: prefix {session.prefix_length} + {length_increment} > {max_length}"
                        )
                    step_start_position = session.prefix_length
                    _trace = os.environ.get("PETALS_AMD_STEP_TRACE")
                    _t0 = time.perf_counter()

                    has_prompts = prompts is not None and not is_dummy(prompts)
                    next_servers = step_meta.get("next_servers")
                    will_push = bool(next_servers) and not has_prompts and length_increment > 0
                    output_via_mesh = step_meta.get("output_via_mesh")
                    # keep the output on the GPU when it leaves over RCCL (mesh
                    # push / mesh output) or stays in-process (co-located client)
                    keep_on_device = (
                        (will_push and self._mesh_dst(next_servers[0]) is not None)
                        or (not will_push and self._mesh_out_dst(output_via_mesh) is not None)
                        or getattr(stream, "is_inproc", False)
                    )

                    # --- pre-announce: for mesh hand-offs, reserve the ticket
                    # and ship the (tiny) TCP metadata BEFORE computing, so the
                    # control plane of hop r+1 runs while hop r's GPU works;
                    # the activation is committed to the wire by the runtime
                    # thread the moment it exists (ticket order keeps pairs in
                    # sync even though metas may arrive out of order)
                    commit_cb = abort_cb = None
                    pre_kind = None
                    ack_task = None
                    if length_increment > 0:
                        out_shape = list(hidden_states.shape)  # blocks preserve shape
                        out_dtype = self.backends[uids[0]].dtype
                        if will_push:
                            mdst = self._mesh_dst(next_servers[0])
                            if mdst is not None:
                                try:
                                    ticket, commit_cb, abort_cb = self.mesh.send_deferred(mdst)
                                    push_meta = {
                                        "session_id": next_servers[0]["session_id"],
                                        "step_id": step_meta.get("step_id"),
                                        "next_servers": next_servers[1:],
                                        "start_from_position": step_start_position,
                                        "output_via_mesh": output_via_mesh,
                                        "tensors_via_mesh": {
                                            "mesh_id": self.mesh.mesh_id,
                                            "src_rank": self.mesh.rank,
                                            "ticket": ticket,
                                            "shape": out_shape,
                                            "dtype": _dtype_str(out_dtype),
                                        },
                                    }
                                    ack_task = asyncio.ensure_future(
                                        asyncio.wait_for(
                                            self.p2p.call_unary(
                                                tuple(next_servers[0]["addr"]), "petals.rpc_push",
                                                RpcMessage(meta=push_meta),
                                            ),
                                            timeout=10.0,
                                        )
                                    )
                                    pre_kind = "push"
                                except Exception as e:  # noqa: BLE001
                                    logger.warning("mesh pre-announce failed (%r); TCP push", e)
                                    commit_cb = abort_cb = None
                        elif not has_prompts:
                            modst = self._mesh_out_dst(output_via_mesh)
                            if modst is not None:
                                try:
                                    ticket, commit_cb, abort_cb = self.mesh.send_deferred(modst)
                                    await stream.send(RpcMessage(meta={
                                        "step_id": step_meta.get("step_id"),
                                        "tensors_via_mesh": self._mesh_desc_shaped(out_shape, out_dtype, ticket),
                                    }))
                                    pre_kind = "out"
                                except Exception as e:  # noqa: BLE001
                                    logger.warning("mesh out pre-announce failed (%r); stream", e)
                                    commit_cb = abort_cb = None

                    # direct reply from the runtime thread: a co-located
                    # (in-process) client gets its output the moment compute
                    # finishes, without waking the server loop first
                    reply_cb = None
                    replied_directly = False
                    if (
                        length_increment > 0
                        and not will_push
                        and self._mesh_out_dst(output_via_mesh) is None
                        and getattr(stream, "is_inproc", False)
                    ):
                        _sid = step_meta.get("step_id")
                        _peer = stream.peer

                        def reply_cb(out, _sid=_sid, _peer=_peer):
                            _peer._deliver(RpcMessage(meta={"step_id": _sid}, tensors=[out]))

                        replied_directly = True
                    if length_increment > 0:
                        priority = self.prioritizer.prioritize(hidden_states, type="inference")
                        try:
                            output = await self.runtime.submit(
                                priority,
                                self._inference_step_chain,
                                uids,
                                hidden_states,
                                hypo_ids,
                                prompts,
                                handles,
                                session.prefix_length,
                                active_adapter,
                                session,
                                keep_on_device,
                                commit_cb,
                                reply_cb,
                            )
                        except BaseException as e:
                            if abort_cb is not None:
                                # the peer already expects this ticket: a
                                # silent skip would desync the pair
                                abort_cb(f"compute failed: {e!r}")
                            if ack_task is not None:
                                ack_task.cancel()
                                with contextlib.suppress(Exception):
                                    await ack_task
                            raise
                    else:
                        output = hidden_states
                    if _trace:
                        print(f"[srv] compute {(time.perf_counter()-_t0)*1e3:.2f} ms", flush=True)
                    session.prefix_length += length_increment

                    if replied_directly:
                        continue  # the runtime thread already delivered the reply
                    if pre_kind == "out":
                        continue  # meta already on the stream; data on the mesh

                    pushed = False
                    if pre_kind == "push":
                        try:
                            await ack_task
                            pushed = True
                        except Exception as e:  # noqa: BLE001
                            logger.warning("pre-announced push not acknowledged (%r)", e)
                            if self.mesh is not None:
                                # data already committed to the wire but the peer
                                # never learned the ticket: the pair is desynced
                                self.mesh.mark_broken(f"push ack failed: {e!r}")
                    elif will_push:
                        pushed = await self._push_outputs(output, step_meta, next_servers, step_start_position)
                    if not pushed:
                        # the last server of a push chain (or any server when push
                        # is off/failed) returns outputs on its client stream
                        out_meta = {"step_id": step_meta.get("step_id")}
                        if torch.is_tensor(output) and output.device.type != "cpu" and not getattr(stream, "is_inproc", False):
                            output = output.cpu()
                        await stream.send(RpcMessage(meta=out_meta, tensors=[output]))
        finally:
            if self.tp_coord is not None and session.tp_slot:
                self.tp_coord.close_session(session.tp_slot)
            self._sessions.pop(session_id, None)

    async def _iterate_inference_steps(self, stream: RpcStream, session: _Session):
        """Multiplex client-stream steps with server-pushed steps (parity:
        handler.py:247-308)."""
        while True:
            client_task = asyncio.ensure_future(stream.receive(timeout=self.step_timeout))
            push_task = asyncio.ensure_future(session.pushed_inputs.get())
            done, pending = await asyncio.wait(
                {client_task, push_task}, return_when=asyncio.FIRST_COMPLETED, timeout=self.step_timeout
            )
            for task in pending:
                task.cancel()
                with contextlib.suppress(asyncio.CancelledError):
                    await task
            if not done:
                raise RpcError("inference step timed out")
            # both tasks may complete in the same tick; each completed task has
            # already consumed its message, so process EVERY one or a step is lost
            for task in done:
                try:
                    item = task.result()
                except RpcError as e:
                    if "closed" in str(e):
                        return  # client closed the stream: session over
                    raise
                if isinstance(item, RpcMessage):
                    if item.kind == "end" and not item.tensors and not item.meta.get("step_id"):
                        return  # graceful close
                    yield item.meta, item.tensors
                else:  # pushed step: (meta, tensors)
                    yield item

    # ------------------------------------------------- mesh (RCCL) hand-off

    def _mesh_dst(self, next_server_entry) -> Optional[int]:
        """Next-server mesh rank iff it shares a usable mesh with this server.
        Entries: {"addr", "session_id", "start", "end", "mesh_id", "mesh_rank"}."""
        if self.mesh is None or not self.mesh.is_usable or self.mesh.device is None:
            return None
        mesh_id = next_server_entry.get("mesh_id")
        mesh_rank = next_server_entry.get("mesh_rank")
        if mesh_id != self.mesh.mesh_id or mesh_rank is None or mesh_rank == self.mesh.rank:
            return None
        return int(mesh_rank)

    def _mesh_out_dst(self, output_via_mesh) -> Optional[int]:
        """Client-requested final-output mesh delivery ({mesh_id, rank})."""
        if not output_via_mesh or self.mesh is None or not self.mesh.is_usable or self.mesh.device is None:
            return None
        if output_via_mesh.get("mesh_id") != self.mesh.mesh_id:
            return None
        rank = output_via_mesh.get("rank")
        if rank is None or rank == self.mesh.rank:
            return None
        return int(rank)

    def _mesh_desc(self, tensor: torch.Tensor, ticket: int) -> Dict[str, Any]:
        return self._mesh_desc_shaped(list(tensor.shape), tensor.dtype, ticket)

    def _mesh_desc_shaped(self, shape, dtype, ticket: int) -> Dict[str, Any]:
        return {
            "mesh_id": self.mesh.mesh_id,
            "src_rank": self.mesh.rank,
            "ticket": ticket,
            "shape": list(shape),
            "dtype": _dtype_str(dtype),
        }

    async def rpc_push(self, request: RpcMessage, stream: RpcStream) -> None:
        session_id = request.meta.get("session_id")
        session = self._sessions.get(session_id)
        if session is None:
            raise RpcError(f"no active inference session {session_id!r}")
        tensors: List[Any] = list(request.tensors)
        tvm = request.meta.get("tensors_via_mesh")
        if tvm is not None:
            if self.mesh is None or not self.mesh.is_usable or tvm.get("mesh_id") != self.mesh.mesh_id:
                raise RpcError("mesh transfer addressed to a server without that mesh")
            handle = self.mesh.post_recv(
                int(tvm["src_rank"]), int(tvm["ticket"]), tvm["shape"], tvm["dtype"]
            )
            tensors = [handle]
        session.pushed_inputs.put_nowait((dict(request.meta), tensors))
        await stream.close(RpcMessage(meta={"ok": True}))

    async def _push_outputs(
        self, output: torch.Tensor, step_meta: Dict[str, Any], next_servers, step_start_position: int
    ) -> bool:
        """Push this step's output into the next server's session. next_servers
        entries: {"addr": [host, port(, "relay", peer)], "session_id", "start",
        "end", "mesh_id", "mesh_rank"}; we contact the first entry.
        start_from_position propagates down the chain so rollbacks
        (speculative decoding) rewind every span's cache. Co-located next
        servers get the activation over RCCL/xGMI (the TCP message then
        carries only metadata)."""
        try:
            entry = next_servers[0]
            meta = {
                "session_id": entry["session_id"],
                "step_id": step_meta.get("step_id"),
                "next_servers": next_servers[1:],
                "start_from_position": step_start_position,
                "output_via_mesh": step_meta.get("output_via_mesh"),
            }
            tensors = [output]
            mesh_dst = self._mesh_dst(entry)
            if mesh_dst is not None and output.device.type == self.mesh.device.type:
                _fut, ticket = self.mesh.send(output, dst=mesh_dst)
                meta["tensors_via_mesh"] = self._mesh_desc(output, ticket)
                tensors = []
            elif output.device.type != "cpu":
                output = output.cpu()
                tensors = [output]
            await asyncio.wait_for(
                self.p2p.call_unary(tuple(entry["addr"]), "petals.rpc_push", RpcMessage(meta=meta, tensors=tensors)),
                timeout=10.0,
            )
            return True
        except Exception as e:  # noqa: BLE001
            logger.warning("rpc_push to next server failed: %r", e)
            return False
