#!/usr/bin/env python3
"""Parallel forward-pass throughput (parity: reference benchmarks/benchmark_forward.py)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch


def parse_addr(s):
    host, port = s.rsplit(":", 1)
    return (host, int(port))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", default="test-llama")
    parser.add_argument("--initial_peers", nargs="+", required=True)
    parser.add_argument("--dht_prefix", default=None)
    parser.add_argument("--batch_size", type=int, default=4)
    parser.add_argument("--seq_len", type=int, default=128)
    parser.add_argument("--n_steps", type=int, default=10)
    parser.add_argument("--warmup_steps", type=int, default=1)
    parser.add_argument("--device", default="cpu")
    args = parser.parse_args()

    from petals_amd.utils.auto_config import AutoDistributedModel

    model = AutoDistributedModel.from_pretrained(
        args.model, initial_peers=[parse_addr(p) for p in args.initial_peers],
        dht_prefix=args.dht_prefix, show_route=False,
    )
    h = torch.randn(args.batch_size, args.seq_len, model.config.hidden_size, device=args.device)
    t0 = None
    for i in range(args.n_steps):
        if i == args.warmup_steps:
            t0 = time.perf_counter()
        with torch.no_grad():
            model.h(h)
        print(f"step {i} done", flush=True)
    elapsed = time.perf_counter() - t0
    tokens = (args.n_steps - args.warmup_steps) * args.batch_size * args.seq_len
    print(f"forward throughput: {tokens / elapsed:.1f} tokens/sec")


if __name__ == "__main__":
    main()
