#!/usr/bin/env python3
"""Prompt-tuning fwd+bwd throughput (parity: reference benchmarks/benchmark_training.py)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch
import torch.nn.functional as F


def parse_addr(s):
    host, port = s.rsplit(":", 1)
    return (host, int(port))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", default="test-llama")
    parser.add_argument("--initial_peers", nargs="+", required=True)
    parser.add_argument("--dht_prefix", default=None)
    parser.add_argument("--device", default="cpu")
    parser.add_argument("--task", default="cls", choices=["cls", "causal_lm"])
    parser.add_argument("--pre_seq_len", type=int, default=4)
    parser.add_argument("--tuning_mode", default="deep_ptune", choices=["ptune", "deep_ptune"])
    parser.add_argument("--batch_size", type=int, default=2)
    parser.add_argument("--seq_len", type=int, default=32)
    parser.add_argument("--n_steps", type=int, default=6)
    parser.add_argument("--warmup_steps", type=int, default=1)
    args = parser.parse_args()

    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    model = AutoDistributedModelForCausalLM.from_pretrained(
        args.model, initial_peers=[parse_addr(p) for p in args.initial_peers],
        dht_prefix=args.dht_prefix, show_route=False,
        pre_seq_len=args.pre_seq_len, tuning_mode=args.tuning_mode,
    )
    opt = torch.optim.Adam([p for p in model.parameters() if p.requires_grad], lr=1e-3)
    fwd_times, bwd_times = [], []
    for i in range(args.n_steps):
        ids = torch.randint(0, model.config.vocab_size, (args.batch_size, args.seq_len))
        t0 = time.perf_counter()
        logits = model(input_ids=ids).logits
        fwd_times.append(time.perf_counter() - t0)
        loss = F.cross_entropy(logits[:, :-1].flatten(0, 1), ids[:, 1:].flatten())
        t0 = time.perf_counter()
        opt.zero_grad()
        loss.backward()
        opt.step()
        bwd_times.append(time.perf_counter() - t0)
    n = args.n_steps - args.warmup_steps
    tokens = args.batch_size * args.seq_len
    print(f"forward: {tokens * n / sum(fwd_times[args.warmup_steps:]):.1f} tok/s, "
          f"backward+step: {tokens * n / sum(bwd_times[args.warmup_steps:]):.1f} tok/s")


if __name__ == "__main__":
    main()
