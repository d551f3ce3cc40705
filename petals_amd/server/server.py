"""Server process: owns the DHT handle, decides which blocks to serve, loads
them, serves the 7 RPCs, announces itself, and rebalances when the swarm is
uneven.

Parity with reference ``server/server.py`` (Server :46-429, ModuleContainer
:431-672, ModuleAnnouncerThread :674-768) in a single-process MI355X design:
one asyncio loop thread runs DHT + RPC handlers; one PriorityRuntime thread
owns the GPU; no handler subprocesses (the reference needed them because of
Python-level serialization costs per libp2p stream; asyncio + zero-copy
msgpack framing covers swarm-scale traffic here).
"""

from __future__ import annotations

import asyncio
import logging
import os
import random
import threading
import time
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from petals_amd.data_structures import ServerInfo, ServerState, get_dht_time, make_uid
from petals_amd.dht.node import DHTNode
from petals_amd.models.config_base import load_model_config
from petals_amd.p2p.transport import P2PNode
from petals_amd.server import block_selection
from petals_amd.server.backend import TransformerBackend
from petals_amd.server.from_pretrained import load_pretrained_block
from petals_amd.server.handler import TransformerConnectionHandler
from petals_amd.server.memory_cache import MemoryCache
from petals_amd.server.scheduler import PriorityRuntime
from petals_amd.server.throughput import get_server_throughput
from petals_amd.utils.misc import get_size_in_bytes

logger = logging.getLogger(__name__)

from petals_amd.constants import DTYPE_MAP  # noqa: E402  (parity: reference constants.py:18)


class Server:
    """One swarm server hosting a contiguous span of transformer blocks."""

    def __init__(
        self,
        model_name_or_dir: str,
        *,
        initial_peers: Sequence[Tuple[str, int]] = (),
        host: str = "127.0.0.1",
        port: int = 0,
        device: Optional[str] = None,
        torch_dtype: str = "auto",
        num_blocks: Optional[int] = None,
        block_indices: Optional[str] = None,
        dht_prefix: Optional[str] = None,
        attn_cache_tokens: int = 16384,
        max_batch_size: int = 8,
        inference_max_length: Optional[int] = None,
        throughput: str | float = "auto",
        update_period: float = 60.0,
        expiration: Optional[float] = None,
        balance_quality: float = 0.75,
        mean_balance_check_period: float = 120.0,
        quant_type: str = "none",
        adapters: Sequence[str] = (),
        public_name: Optional[str] = None,
        announce_host: Optional[str] = None,
        skip_reachability_check: bool = False,
        mesh=None,
        secure: Optional[bool] = None,
        identity_path: Optional[str] = None,
        use_relay: bool = True,
        force_relay: bool = False,
        max_chunk_size_bytes: int = 256 * 1024 * 1024,
        max_alloc_timeout: float = 600.0,
        request_timeout: float = 3 * 60.0,
        session_timeout: float = 30 * 60.0,
        step_timeout: float = 5 * 60.0,
        compression: str = "none",
        stats_report_interval: Optional[float] = None,
        cache_dir: Optional[str] = None,
        max_disk_space: Optional[int] = None,
        tensor_parallel_ranks: int = 1,
        tp_group=None,
    ):
        self.config = load_model_config(model_name_or_dir)
        self.model_name_or_dir = model_name_or_dir
        if dht_prefix:
            self.config.dht_prefix = dht_prefix
        self.initial_peers = [tuple(p) for p in initial_peers]
        self.host, self.port = host, port
        self.announce_host = announce_host
        self.skip_reachability_check = skip_reachability_check

        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        dtype = DTYPE_MAP[torch_dtype]
        if dtype is None:
            dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.torch_dtype = dtype
        self.quant_type = quant_type

        self.block_indices: Optional[List[int]] = None
        if block_indices is not None:
            start, end = block_indices.split(":")
            self.block_indices = list(range(int(start), int(end)))
            num_blocks = len(self.block_indices)
        self.num_blocks = num_blocks or self._choose_num_blocks(attn_cache_tokens)

        self.attn_cache_tokens = attn_cache_tokens
        self.max_batch_size = max_batch_size
        self.inference_max_length = inference_max_length or attn_cache_tokens
        self.update_period = update_period
        self.expiration = expiration or max(3 * update_period, 60.0)
        self.balance_quality = balance_quality
        self.mean_balance_check_period = mean_balance_check_period
        self.public_name = public_name
        self.adapters = tuple(adapters)
        self._throughput_setting = throughput
        # LocalMesh for the co-located RCCL/xGMI activation hand-off tier
        self.mesh = mesh
        if mesh is not None:
            from petals_amd.parallel.mesh import register_local_mesh

            register_local_mesh(mesh)

        self.module_uids = [make_uid(self.config.dht_prefix, i) for i in range(self.config.num_blocks)]

        # runtime state
        self.p2p: Optional[P2PNode] = None
        self.dht_node: Optional[DHTNode] = None
        self.runtime: Optional[PriorityRuntime] = None
        self.memory_cache: Optional[MemoryCache] = None
        self.backends: Dict[str, TransformerBackend] = {}
        self.handler: Optional[TransformerConnectionHandler] = None
        self.loop: Optional[asyncio.AbstractEventLoop] = None
        self._thread: Optional[threading.Thread] = None
        self._ready = threading.Event()
        self._stop = threading.Event()
        self._restart_requested = False
        self.server_info: Optional[ServerInfo] = None
        self.listen_addr: Optional[Tuple[str, int]] = None
        self._ping_agg = None  # PingAggregator for next-server RTT gossip
        self.secure = secure
        self.identity_path = identity_path
        self.use_relay = use_relay
        self.force_relay = force_relay
        self._relayed = False
        self._relay_addr: Optional[Tuple[str, int]] = None
        self.max_chunk_size_bytes = max_chunk_size_bytes
        self.max_alloc_timeout = max_alloc_timeout
        self.request_timeout = request_timeout
        self.session_timeout = session_timeout
        self.step_timeout = step_timeout
        self.compression = compression
        self.stats_report_interval = stats_report_interval
        self.cache_dir = cache_dir
        self.max_disk_space = max_disk_space
        # intra-server tensor parallelism (parallel/tp.py): this process is one
        # of `tensor_parallel_ranks` ranks sharding every served block
        self.tensor_parallel_ranks = int(tensor_parallel_ranks)
        self.tp_group = tp_group

    # -------------------------------------------------------------- sizing

    def _block_param_bytes(self) -> int:
        cfg = self.config
        dtype_bytes = get_size_in_bytes(self.torch_dtype)
        if self.quant_type == "nf4":
            dtype_bytes = 0.53125  # 4.25 bits/param (parity: block_utils.py:46)
        elif self.quant_type == "int8":
            dtype_bytes = 1.002  # 8 bits/param + per-column bf16 scales
        h, inter = cfg.hidden_size, cfg.intermediate_size
        kv = cfg.n_kv_heads * cfg.head_dim
        attn = h * h + 2 * h * kv + h * h
        n_experts = getattr(cfg, "num_local_experts", 1)
        mlp = 3 * h * inter * n_experts
        return int((attn + mlp) * dtype_bytes)

    def _choose_num_blocks(self, attn_cache_tokens: int = 16384) -> int:
        """Fit blocks to device memory: block bytes + attn-cache + autograd
        reserve (parity: server.py:275-326)."""
        if self.device.type == "cuda":
            total = torch.cuda.get_device_properties(self.device).total_memory
        else:
            total = 16 << 30
        block_bytes = self._block_param_bytes()
        cache_bytes_per_block = (
            2 * self.config.n_kv_heads * self.config.head_dim * attn_cache_tokens * get_size_in_bytes(self.torch_dtype)
        )
        autograd_reserve = 2 << 30
        usable = total * 0.92 - autograd_reserve
        n = max(1, int(usable // (block_bytes + cache_bytes_per_block)))
        return min(n, self.config.num_blocks)

    # ------------------------------------------------------------ lifecycle

    def start(self):
        self._thread = threading.Thread(target=self._run_thread, name="PetalsServer", daemon=True)
        self._thread.start()
        self._ready.wait(timeout=600)
        if not self._ready.is_set() or self.listen_addr is None:
            raise RuntimeError("server failed to start")
        return self

    def _run_thread(self):
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        self.loop = loop
        try:
            loop.run_until_complete(self._amain())
        except Exception:  # noqa: BLE001
            logger.exception("server main loop crashed")
        finally:
            self._ready.set()  # unblock start() even on failure
            try:
                loop.run_until_complete(self._ashutdown())
            except Exception:  # noqa: BLE001
                pass
            # drain leftovers (e.g. rpc_inference generators awaiting a pushed
            # step when a client session dies with the server) so closing the
            # loop doesn't finalize their coroutines on a dead loop
            try:
                pending = [t for t in asyncio.all_tasks(loop) if not t.done()]
                for t in pending:
                    t.cancel()
                if pending:
                    loop.run_until_complete(asyncio.gather(*pending, return_exceptions=True))
                loop.run_until_complete(loop.shutdown_asyncgens())
            except Exception:  # noqa: BLE001
                pass
            loop.close()

    async def _amain(self):
        identity = None
        if self.identity_path or self.secure:
            from petals_amd.p2p.identity import NodeIdentity

            identity = NodeIdentity(self.identity_path)
        self.p2p = P2PNode(identity=identity, secure=self.secure)
        await self.p2p.listen(host=self.host, port=self.port)
        # the address other peers should dial (never announce 0.0.0.0)
        host = self.announce_host or self.host
        if host in ("0.0.0.0", "::"):
            import socket

            host = socket.gethostbyname(socket.gethostname())
        self.listen_addr = (host, self.p2p.listen_addr[1])
        self.dht_node = await DHTNode.create(initial_peers=self.initial_peers, p2p=self.p2p)
        from petals_amd.p2p.relay import RelayClient, RelayHub
        from petals_amd.server.reachability import ReachabilityProtocol, check_direct_reachability

        ReachabilityProtocol(self.p2p)
        RelayHub(self.p2p)  # any reachable server can relay for NAT'd peers
        unreachable = self.force_relay
        if not unreachable and not self.skip_reachability_check and self.initial_peers:
            ok = await check_direct_reachability(self.p2p, self.listen_addr, self.initial_peers)
            unreachable = ok is False
        if unreachable:
            if not (self.use_relay and self.initial_peers):
                raise RuntimeError(
                    f"server at {self.listen_addr} is not reachable by the swarm and relays are disabled"
                )
            self._relay_addr = tuple(self.initial_peers[0])
            await RelayClient(self.p2p, self._relay_addr).start()
            self._relayed = True
            logger.warning(
                "server is not directly reachable; serving via relay %s (throughput penalized x0.2)",
                self._relay_addr,
            )

        if self.tensor_parallel_ranks > 1:
            import torch.distributed as dist

            from petals_amd.parallel.tp import TPCoordinator

            assert dist.is_initialized(), "TP serving needs torch.distributed (launch under torchrun)"
            assert dist.get_rank(self.tp_group) == 0, (
                "only TP rank 0 runs the Server; other ranks run TPShadowWorker (cli/run_server handles this)"
            )
            self._tp_coord = TPCoordinator(
                self.tp_group, self.device, self.config.hidden_size,
                self.torch_dtype if self.device.type == "cuda" else torch.float32,
            )
        else:
            self._tp_coord = None

        self.runtime = PriorityRuntime(self.device).start()
        while not self._stop.is_set():
            await self._serve_once()
            if not self._restart_requested:
                break
            self._restart_requested = False

    async def _serve_once(self):
        """Load a span of blocks and serve until shutdown or rebalance."""
        # --- choose blocks
        if self.block_indices is not None:
            block_indices = self.block_indices
        else:
            infos, _ = await self._get_infos()
            block_indices = block_selection.choose_best_blocks(self.num_blocks, infos)
        logger.info("serving blocks %s..%s", block_indices[0], block_indices[-1])

        # --- throughput estimate (cheap on CPU; measured once per device/model)
        if self._throughput_setting == "auto":
            tp = get_server_throughput(
                self.config,
                device=self.device,
                dtype=self.torch_dtype,
                num_blocks=len(block_indices),
                quant_type=self.quant_type,
            )
            throughput = tp["throughput"]
            inference_rps, forward_rps, network_rps = tp["inference_rps"], tp["forward_rps"], tp["network_rps"]
        else:
            throughput = float(self._throughput_setting)
            inference_rps = forward_rps = network_rps = throughput
        if self._relayed:
            # relayed traffic crosses the relay's uplink twice (parity:
            # reference server/throughput.py:96-107)
            throughput *= 0.2

        cache_bytes = int(
            2
            * self.config.n_kv_heads
            * self.config.head_dim
            * self.attn_cache_tokens
            * get_size_in_bytes(self.torch_dtype)
            * len(block_indices)
        )
        self.memory_cache = MemoryCache(cache_bytes, self.device, alloc_timeout=self.max_alloc_timeout)

        self.server_info = ServerInfo(
            state=ServerState.JOINING,
            throughput=throughput,
            start_block=block_indices[0],
            end_block=block_indices[-1] + 1,
            public_name=self.public_name,
            version="0.1",
            network_rps=network_rps,
            forward_rps=forward_rps,
            inference_rps=inference_rps,
            torch_dtype=str(self.torch_dtype).replace("torch.", ""),
            quant_type=self.quant_type,
            adapters=tuple(os.path.basename(os.path.normpath(a)) for a in self.adapters),
            mesh_id=self.mesh.mesh_id if self.mesh is not None else None,
            mesh_rank=self.mesh.rank if self.mesh is not None else None,
            using_relay=self._relayed,
        )
        await self._announce()

        # --- load blocks (in a worker thread: file IO + H2D copies)
        served_uids = [make_uid(self.config.dht_prefix, i) for i in block_indices]
        if self._tp_coord is not None:
            # tell the shadow ranks which span to load (they build the same
            # shard blocks and mirror every collective from here on)
            self._tp_coord.span(block_indices[0], block_indices[-1] + 1)
        self.backends = {}
        for i, uid in zip(block_indices, served_uids):
            block = await asyncio.get_event_loop().run_in_executor(
                None,
                lambda idx=i: load_pretrained_block(
                    self.model_name_or_dir, self.config, idx, torch_dtype=self.torch_dtype,
                    device=self.device, quant_type=self.quant_type,
                    tp_rank=0, tp_world=self.tensor_parallel_ranks, tp_group=self.tp_group,
                ),
            )
            for adapter_dir in self.adapters:
                from petals_amd.utils.peft import add_adapter_to_block, load_block_adapter

                ad = load_block_adapter(adapter_dir, i, self.config.block_prefix)
                if ad is not None:
                    add_adapter_to_block(block, ad)
            self.backends[uid] = TransformerBackend(
                uid, block, config=self.config, memory_cache=self.memory_cache, dtype=self.torch_dtype,
                max_chunk_size_bytes=self.max_chunk_size_bytes,
            )

        self.handler = TransformerConnectionHandler(
            backends=self.backends,
            memory_cache=self.memory_cache,
            runtime=self.runtime,
            inference_max_length=self.inference_max_length,
            p2p=self.p2p,
            adapters=tuple(os.path.basename(os.path.normpath(a)) for a in self.adapters),
            mesh=self.mesh,
            request_timeout=self.request_timeout,
            session_timeout=self.session_timeout,
            step_timeout=self.step_timeout,
            default_compression=self.compression,
            tp_coord=self._tp_coord,
        )
        self.handler.register(self.p2p)

        self.server_info.state = ServerState.ONLINE
        await self._announce()
        declare_model_info = {
            "model_type": self.config.model_type,
            "num_blocks": self.config.num_blocks,
            "dht_prefix": self.config.dht_prefix,
            "config": self.config.to_dict(),
        }
        await self.dht_node.store_many(
            [("_petals_amd.models", self.config.dht_prefix, declare_model_info, get_dht_time() + self.expiration)]
        )
        self._ready.set()

        # --- announce + health + rebalance loop
        next_balance_check = time.monotonic() + random.random() * 2 * self.mean_balance_check_period
        last_announce = time.monotonic()
        last_stats = time.monotonic()
        try:
            while not self._stop.is_set():
                if (
                    self.stats_report_interval
                    and time.monotonic() - last_stats >= self.stats_report_interval
                ):
                    st = self.runtime.stats
                    logger.info(
                        "runtime: %d tasks, %.1f s busy; cache: %.0f/%.0f MiB free",
                        st["tasks"], st["busy_time"],
                        self.memory_cache.bytes_left / (1 << 20),
                        self.memory_cache.max_size_bytes / (1 << 20),
                    )
                    last_stats = time.monotonic()
                if time.monotonic() - last_announce >= self.update_period:
                    try:
                        await self._ping_next_servers()
                    except Exception as e:  # noqa: BLE001
                        logger.debug("next-server pings failed: %r", e)
                    await self._announce()
                    last_announce = time.monotonic()
                if not self.runtime.alive:
                    raise RuntimeError("runtime thread died")
                await asyncio.sleep(min(self.update_period, 0.5))
                if self.block_indices is None and time.monotonic() > next_balance_check:
                    next_balance_check = time.monotonic() + random.random() * 2 * self.mean_balance_check_period
                    infos, _ = await self._get_infos()
                    if block_selection.should_choose_other_blocks(self.p2p.peer_id, infos, self.balance_quality):
                        logger.info("swarm is imbalanced: restarting with a new span")
                        self._restart_requested = True
                        break
        finally:
            self.server_info.state = ServerState.OFFLINE
            await self._announce()
            for backend in self.backends.values():
                backend.shutdown()
            self.backends = {}

    async def _get_infos(self):
        from petals_amd.data_structures import RemoteModuleInfo, ServerInfo as SI

        found = await self.dht_node.get_many(self.module_uids)
        infos, addrs = [], {}
        for uid in self.module_uids:
            servers = {}
            for peer_id, (value, _exp) in (found.get(uid) or {}).items():
                try:
                    info = SI.from_dict(value["info"])
                    addrs[peer_id] = tuple(value["addr"])
                except (KeyError, TypeError):
                    continue
                servers[peer_id] = info
            infos.append(RemoteModuleInfo(uid=uid, servers=servers) if servers else None)
        return infos, addrs

    async def _ping_next_servers(self, max_pinged: int = 5):
        """Ping servers most likely to be NEXT in an inference chain (those
        hosting the block right after our span) and gossip the RTTs via
        ServerInfo.next_pings — the client's min-latency router prices chained
        server->server hops with them (parity: reference server/server.py:760,
        client/routing/sequence_manager.py:253-268)."""
        if self.server_info is None or self.server_info.end_block is None:
            return
        next_block = self.server_info.end_block
        if next_block >= self.config.num_blocks:
            self.server_info.next_pings = {}
            return
        infos, addrs = await self._get_infos()
        info = infos[next_block]
        if info is None:
            return
        candidates = {
            pid: addrs[pid]
            for pid in info.servers
            if pid != self.p2p.peer_id and pid in addrs
        }
        if len(candidates) > max_pinged:
            import random as _random

            keys = _random.sample(sorted(candidates), max_pinged)
            candidates = {k: candidates[k] for k in keys}
        if not candidates:
            self.server_info.next_pings = {}
            return
        if self._ping_agg is None:
            from petals_amd.utils.ping import PingAggregator

            self._ping_agg = PingAggregator(self.p2p)
        await self._ping_agg.ping(candidates)
        import math as _math

        self.server_info.next_pings = {
            p: rtt for p, rtt in self._ping_agg.to_dict().items() if _math.isfinite(rtt)
        }

    async def _announce(self):
        if self.server_info is None:
            return
        first = next(iter(self.backends.values()), None)
        if first is not None and self.memory_cache is not None:
            bytes_per_token = first.cache_bytes_per_token() * max(1, len(self.backends))
            self.server_info.cache_tokens_left = int(self.memory_cache.bytes_left // max(1, bytes_per_token))
        uids = [
            make_uid(self.config.dht_prefix, i)
            for i in range(self.server_info.start_block or 0, self.server_info.end_block or 0)
        ]
        if not uids:
            return
        if self._relayed:
            announced_addr = [self._relay_addr[0], self._relay_addr[1], "relay", self.p2p.peer_id]
        else:
            announced_addr = list(self.listen_addr)
        value = {"info": self.server_info.to_dict(), "addr": announced_addr}
        entries = [(uid, self.p2p.peer_id, value, get_dht_time() + self.expiration) for uid in uids]
        try:
            await self.dht_node.store_many(entries)
        except Exception as e:  # noqa: BLE001
            logger.warning("announce failed: %r", e)

    async def _ashutdown(self):
        if getattr(self, "_tp_coord", None) is not None:
            self._tp_coord.shutdown()
        if self.runtime is not None:
            self.runtime.shutdown()
        if self.dht_node is not None:
            await self.dht_node.shutdown()

    @property
    def peer_id(self) -> str:
        return self.p2p.peer_id if self.p2p else ""

    def shutdown(self, timeout: float = 15.0):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=timeout)

    def is_healthy(self) -> bool:
        return (
            self._thread is not None
            and self._thread.is_alive()
            and self.runtime is not None
            and self.runtime.alive
        )
