"""Run a standalone DHT bootstrap node (parity: reference cli/run_dht.py)."""

from __future__ import annotations

import argparse
import logging
import signal
import time


def main(argv=None):
    parser = argparse.ArgumentParser(description="petals_amd DHT bootstrap node")
    parser.add_argument("--host", default="0.0.0.0")
    parser.add_argument("--port", type=int, default=31337)
    parser.add_argument("--initial_peers", nargs="*", default=[])
    args = parser.parse_args(argv)

    logging.basicConfig(level=logging.INFO)
    from petals_amd.dht.node import DHT

    peers = []
    for p in args.initial_peers:
        host, port = p.rsplit(":", 1)
        peers.append((host, int(port)))
    dht = DHT(initial_peers=peers, host=args.host, port=args.port)
    print(f"DHT bootstrap node at {dht.listen_addr} peer_id={dht.peer_id}", flush=True)

    stop = []
    signal.signal(signal.SIGINT, lambda *a: stop.append(1))
    signal.signal(signal.SIGTERM, lambda *a: stop.append(1))
    while not stop:
        time.sleep(1)
    dht.shutdown()


if __name__ == "__main__":
    main()
