"""Run a petals_amd server: `python -m petals_amd.cli.run_server <model> [...]`.

Parity: reference cli/run_server.py (the ~45-flag surface, trimmed to the
options meaningful in this native build; YAML config via --config).
"""

from __future__ import annotations

import argparse
import logging
import signal
import time


def parse_addr(s: str):
    host, port = s.rsplit(":", 1)
    return (host, int(port))


# reference flags that have NO meaning in this native build: each maps to a
# clear error so users migrating from the reference get told what to do
# instead of silent acceptance (reference cli/run_server.py:25-165)
_REJECTED_FLAGS = {
    "--token": "no Hugging Face Hub access in this build — point `model` at a local checkpoint dir",
    "--use_auth_token": "no Hugging Face Hub access in this build — point `model` at a local checkpoint dir",
    "--revision": "no Hugging Face Hub access in this build — check out the revision locally",
    "--host_maddrs": "no libp2p multiaddrs here — use --host/--port (TCP with optional --secure TLS)",
    "--announce_maddrs": "no libp2p multiaddrs here — use --public_ip/--announce_host",
    "--daemon_startup_timeout": "there is no p2p daemon subprocess in this build",
    "--num_handlers": "handlers are asyncio coroutines in one process, not subprocesses",
    "--prefetch_batches": "no hivemind task pools — the PriorityRuntime schedules whole requests",
    "--sender_threads": "no hivemind task pools — outputs are sent from the asyncio loop",
    "--min_batch_size": "no hivemind batching — requests are scheduled whole",
    "--custom_module_path": "register custom families via petals_amd.models (import your module before run_server)",
}


def _parse_size(s):
    if s is None:
        return None
    s = str(s).strip().upper()
    mult = 1
    for suffix, m in (("GB", 1 << 30), ("MB", 1 << 20), ("KB", 1 << 10), ("B", 1)):
        if s.endswith(suffix):
            return int(float(s[: -len(suffix)]) * m)
    return int(s) * mult


def main(argv=None):
    parser = argparse.ArgumentParser(description="petals_amd server")
    parser.add_argument("model", nargs="?", default=None, help="model preset name or local checkpoint dir")
    parser.add_argument("--converted_model_name_or_path", default=None,
                        help="deprecated alias for the positional `model` argument")
    parser.add_argument("--config", help="YAML file with defaults for any flag")
    parser.add_argument("--host", default="0.0.0.0")
    parser.add_argument("--port", type=int, default=0)
    parser.add_argument("--initial_peers", nargs="*", default=[], help="host:port of bootstrap DHT nodes")
    parser.add_argument("--new_swarm", action="store_true",
                        help="start a fresh swarm (suppresses the empty --initial_peers warning)")
    parser.add_argument("--num_blocks", type=int, default=None)
    parser.add_argument("--block_indices", type=str, default=None, help="e.g. 0:16")
    parser.add_argument("--dht_prefix", type=str, default=None)
    parser.add_argument("--device", type=str, default=None)
    parser.add_argument("--torch_dtype", type=str, default="auto")
    parser.add_argument("--quant_type", type=str, default="none", choices=["none", "nf4", "int8"])
    parser.add_argument("--attn_cache_tokens", type=int, default=16384)
    parser.add_argument("--max_batch_size", type=int, default=8)
    parser.add_argument("--max_chunk_size_bytes", type=int, default=256 * 1024 * 1024,
                        help="prefill chunking bound inside a span (attention score memory)")
    parser.add_argument("--max_alloc_timeout", type=float, default=600.0,
                        help="longest a session may wait for KV-cache memory")
    parser.add_argument("--inference_max_length", type=int, default=None)
    parser.add_argument("--request_timeout", type=float, default=180.0,
                        help="timeout for forward/backward requests")
    parser.add_argument("--session_timeout", type=float, default=30 * 60.0,
                        help="max lifetime of an inference session")
    parser.add_argument("--step_timeout", type=float, default=5 * 60.0,
                        help="max wait for the next inference step of a session")
    parser.add_argument("--compression", type=str, default="none",
                        choices=["none", "float16", "bfloat16", "blockwise_8bit"],
                        help="default wire compression for outputs (clients can override per request)")
    parser.add_argument("--throughput", default="auto")
    parser.add_argument("--update_period", type=float, default=60.0)
    parser.add_argument("--expiration", type=float, default=None)
    parser.add_argument("--balance_quality", type=float, default=0.75)
    parser.add_argument("--mean_balance_check_period", type=float, default=120.0)
    parser.add_argument("--public_name", type=str, default=None)
    parser.add_argument("--adapters", nargs="*", default=[], help="local PEFT adapter dirs to serve")
    parser.add_argument("--announce_host", type=str, default=None)
    parser.add_argument("--public_ip", type=str, default=None, help="alias for --announce_host")
    parser.add_argument("--skip_reachability_check", action="store_true")
    parser.add_argument("--force_relay", action="store_true",
                        help="serve through a circuit relay even if directly reachable (NAT-path testing)")
    parser.add_argument("--no_auto_relay", action="store_true",
                        help="refuse to serve via a circuit relay when unreachable (fail instead)")
    parser.add_argument("--secure", action="store_true",
                        help="STARTTLS transport: peer ids become certificate fingerprints")
    parser.add_argument("--identity_path", type=str, default=None,
                        help="directory holding (or receiving) this node's keypair + certificate")
    parser.add_argument("--cache_dir", type=str, default=None,
                        help="directory for throughput/disk caches (default ~/.cache/petals_amd)")
    parser.add_argument("--max_disk_space", type=str, default=None,
                        help="disk cache budget, e.g. 50GB")
    parser.add_argument("--stats_report_interval", type=float, default=None,
                        help="log runtime/cache statistics every N seconds")
    parser.add_argument("--increase_file_limit", action="store_true",
                        help="raise RLIMIT_NOFILE to the hard limit (many concurrent sessions)")
    parser.add_argument("--tensor_parallel_ranks", type=int, default=1,
                        help="shard each block across N co-located GPU ranks (RCCL all-reduce); "
                             "run one process per rank under torchrun")
    for flag, why in _REJECTED_FLAGS.items():
        parser.add_argument(flag, default=None, help=f"NOT SUPPORTED: {why}")
    args = parser.parse_args(argv)

    for flag, why in _REJECTED_FLAGS.items():
        if getattr(args, flag.lstrip("-"), None) is not None:
            parser.error(f"{flag} is not supported in petals_amd: {why}")
    model = args.model or args.converted_model_name_or_path
    if not model:
        parser.error("provide a model (preset name or local checkpoint dir)")
    if args.new_swarm and args.initial_peers:
        parser.error("--new_swarm and --initial_peers are mutually exclusive")

    if args.config:
        import yaml

        with open(args.config) as f:
            defaults = yaml.safe_load(f) or {}
        for k, v in defaults.items():
            if getattr(args, k, None) in (None, parser.get_default(k)):
                setattr(args, k, v)

    if args.increase_file_limit:
        import resource

        soft, hard = resource.getrlimit(resource.RLIMIT_NOFILE)
        resource.setrlimit(resource.RLIMIT_NOFILE, (hard, hard))

    logging.basicConfig(level=logging.INFO, format="%(asctime)s %(levelname).1s %(name)s: %(message)s")
    if args.cache_dir:
        import os as _os

        _os.environ["PETALS_AMD_CACHE"] = args.cache_dir

    if args.tensor_parallel_ranks > 1:
        # launched under torchrun, one process per GPU: rank 0 is the swarm
        # Server; ranks 1.. execute their block shards in lockstep
        import datetime
        import os as _os

        import torch
        import torch.distributed as dist

        if not dist.is_initialized():
            backend = "nccl" if torch.cuda.is_available() else "gloo"
            dist.init_process_group(backend, timeout=datetime.timedelta(seconds=600))
        rank, world = dist.get_rank(), dist.get_world_size()
        assert world == args.tensor_parallel_ranks, (
            f"torchrun world size {world} != --tensor_parallel_ranks {args.tensor_parallel_ranks}"
        )
        if torch.cuda.is_available():
            torch.cuda.set_device(int(_os.environ.get("LOCAL_RANK", rank)))
        if rank != 0:
            from petals_amd.constants import DTYPE_MAP
            from petals_amd.models.config_base import load_model_config
            from petals_amd.parallel.tp import TPShadowWorker

            config = load_model_config(model)
            device = torch.device("cuda", torch.cuda.current_device()) if torch.cuda.is_available() else torch.device("cpu")
            dtype = DTYPE_MAP[args.torch_dtype]
            if dtype is None:
                dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
            TPShadowWorker(
                model, config, device=device, torch_dtype=dtype, quant_type=args.quant_type,
                group=None, rank=rank, world=world,
            ).serve_forever()
            return

    from petals_amd.server.server import Server

    server = Server(
        model,
        initial_peers=[parse_addr(p) for p in args.initial_peers],
        host=args.host,
        port=args.port,
        device=args.device,
        torch_dtype=args.torch_dtype,
        num_blocks=args.num_blocks,
        block_indices=args.block_indices,
        dht_prefix=args.dht_prefix,
        attn_cache_tokens=args.attn_cache_tokens,
        max_batch_size=args.max_batch_size,
        inference_max_length=args.inference_max_length,
        throughput=args.throughput,
        update_period=args.update_period,
        expiration=args.expiration,
        balance_quality=args.balance_quality,
        mean_balance_check_period=args.mean_balance_check_period,
        quant_type=args.quant_type,
        public_name=args.public_name,
        adapters=args.adapters,
        announce_host=args.public_ip or args.announce_host,
        skip_reachability_check=args.skip_reachability_check,
        secure=True if args.secure else None,
        identity_path=args.identity_path,
        use_relay=not args.no_auto_relay,
        force_relay=args.force_relay,
        max_chunk_size_bytes=args.max_chunk_size_bytes,
        max_alloc_timeout=args.max_alloc_timeout,
        request_timeout=args.request_timeout,
        session_timeout=args.session_timeout,
        step_timeout=args.step_timeout,
        compression=args.compression,
        stats_report_interval=args.stats_report_interval,
        cache_dir=args.cache_dir,
        max_disk_space=_parse_size(args.max_disk_space),
        tensor_parallel_ranks=args.tensor_parallel_ranks,
    )
    server.start()
    print(f"petals_amd server listening on {server.listen_addr} peer_id={server.peer_id}", flush=True)

    stop = []
    signal.signal(signal.SIGINT, lambda *a: stop.append(1))
    signal.signal(signal.SIGTERM, lambda *a: stop.append(1))
    try:
        while not stop and server.is_healthy():
            time.sleep(1)
    finally:
        server.shutdown()


if __name__ == "__main__":
    main()
