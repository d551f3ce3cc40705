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


def main(argv=None):
    parser = argparse.ArgumentParser(description="petals_amd server")
    parser.add_argument("model", help="model preset name or local checkpoint dir")
    parser.add_argument("--config", help="YAML file with defaults for any flag")
    parser.add_argument("--host", default="0.0.0.0")
    parser.add_argument("--port", type=int, default=0)
    parser.add_argument("--initial_peers", nargs="*", default=[], help="host:port of bootstrap DHT nodes")
    parser.add_argument("--num_blocks", type=int, default=None)
    parser.add_argument("--block_indices", type=str, default=None, help="e.g. 0:16")
    parser.add_argument("--dht_prefix", type=str, default=None)
    parser.add_argument("--device", type=str, default=None)
    parser.add_argument("--torch_dtype", type=str, default="auto")
    parser.add_argument("--quant_type", type=str, default="none", choices=["none", "nf4", "int8"])
    parser.add_argument("--attn_cache_tokens", type=int, default=16384)
    parser.add_argument("--max_batch_size", type=int, default=8)
    parser.add_argument("--inference_max_length", type=int, default=None)
    parser.add_argument("--throughput", default="auto")
    parser.add_argument("--update_period", type=float, default=60.0)
    parser.add_argument("--expiration", type=float, default=None)
    parser.add_argument("--balance_quality", type=float, default=0.75)
    parser.add_argument("--mean_balance_check_period", type=float, default=120.0)
    parser.add_argument("--public_name", type=str, default=None)
    parser.add_argument("--adapters", nargs="*", default=[], help="local PEFT adapter dirs to serve")
    parser.add_argument("--announce_host", type=str, default=None)
    parser.add_argument("--skip_reachability_check", action="store_true")
    args = parser.parse_args(argv)

    if args.config:
        import yaml

        with open(args.config) as f:
            defaults = yaml.safe_load(f) or {}
        for k, v in defaults.items():
            if getattr(args, k, None) in (None, parser.get_default(k)):
                setattr(args, k, v)

    logging.basicConfig(level=logging.INFO, format="%(asctime)s %(levelname).1s %(name)s: %(message)s")
    from petals_amd.server.server import Server

    server = Server(
        args.model,
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
        announce_host=args.announce_host,
        skip_reachability_check=args.skip_reachability_check,
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
