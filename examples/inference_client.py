#!/usr/bin/env python3
"""Minimal inference client (parity: reference examples — notebook form).

Start a swarm first:
    python -m petals_amd.cli.run_dht --host 127.0.0.1 --port 31337
    python -m petals_amd.cli.run_server llama-2-70b --initial_peers 127.0.0.1:31337 \
        --torch_dtype bfloat16 --quant_type nf4
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-2-70b")
    p.add_argument("--initial_peers", nargs="+", default=["127.0.0.1:31337"])
    p.add_argument("--device", default="cpu", help="client-side embeddings/LM head device")
    p.add_argument("--max_new_tokens", type=int, default=32)
    args = p.parse_args()

    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    peers = [tuple([h, int(pt)]) for h, pt in (a.rsplit(":", 1) for a in args.initial_peers)]
    model = AutoDistributedModelForCausalLM.from_pretrained(args.model, initial_peers=peers)
    if args.device != "cpu":
        model = model.to(args.device)

    ids = torch.randint(0, model.config.vocab_size, (1, 16), device=args.device)
    # one session reused across calls: the KV cache persists server-side
    with model.transformer.h.inference_session(max_length=16 + args.max_new_tokens) as sess:
        with model.transformer.h.use_session(sess):
            out = model.generate(ids, max_new_tokens=args.max_new_tokens, do_sample=True, top_p=0.9)
    print("generated ids:", out[0].tolist())


if __name__ == "__main__":
    main()
