#!/usr/bin/env python3
"""Client-side inference benchmark (parity: reference
benchmarks/benchmark_inference.py — tokens/sec of session-based generate,
multi-process clients supported via --n_processes)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import multiprocessing as mp
import time

import torch


def parse_addr(s):
    host, port = s.rsplit(":", 1)
    return (host, int(port))


def benchmark_inference(process_idx, args, results):
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    model = AutoDistributedModelForCausalLM.from_pretrained(
        args.model, initial_peers=[parse_addr(p) for p in args.initial_peers],
        dht_prefix=args.dht_prefix, show_route=False,
        torch_dtype=getattr(torch, args.torch_dtype),
    )
    if args.device != "cpu":
        model = model.to(args.device)  # embeddings/norm/head; blocks stay remote
    ids = torch.randint(0, model.config.vocab_size, (1, 8), device=args.device)
    with model.transformer.h.inference_session(max_length=args.seq_len) as session:
        with model.transformer.h.use_session(session):
            model(input_ids=ids)  # prefill
            step = torch.randint(0, model.config.vocab_size, (1, 1), device=args.device)
            t0 = None
            for i in range(args.seq_len - ids.shape[1] - 1):
                if i == args.warmup_steps:
                    t0 = time.perf_counter()
                model(input_ids=step)
            n_timed = args.seq_len - ids.shape[1] - 1 - args.warmup_steps
            speed = n_timed / (time.perf_counter() - t0)
    results[process_idx] = speed
    print(f"process {process_idx}: {speed:.2f} tokens/sec")


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model", default="test-llama")
    parser.add_argument("--initial_peers", nargs="+", required=True, help="host:port")
    parser.add_argument("--dht_prefix", default=None)
    parser.add_argument("--torch_dtype", default="float32")
    parser.add_argument("--device", default="cpu")
    parser.add_argument("--n_processes", default="1")
    parser.add_argument("--seq_len", type=int, default=128)
    parser.add_argument("--warmup_steps", type=int, default=4)
    args = parser.parse_args()

    n_processes = mp.cpu_count() if args.n_processes == "n_gpus" else int(args.n_processes)
    manager = mp.Manager()
    results = manager.dict()
    processes = [mp.Process(target=benchmark_inference, args=(i, args, results)) for i in range(n_processes)]
    for p in processes:
        p.start()
    for p in processes:
        p.join()
    total = sum(results.values())
    print(f"TOTAL: {total:.2f} tokens/sec across {n_processes} client process(es)")


if __name__ == "__main__":
    main()
