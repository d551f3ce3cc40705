#!/usr/bin/env python3
"""Deep prompt tuning over a swarm (parity: reference
examples/prompt-tuning-*.ipynb, script form; synthetic data — this
environment has no dataset downloads).

The client holds the ONLY trainable parameters (per-layer prompts); servers
stay frozen and stateless across steps. Works against any running swarm:
    python examples/prompt_tuning.py --initial_peers 127.0.0.1:31337
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-2-70b")
    p.add_argument("--initial_peers", nargs="+", default=["127.0.0.1:31337"])
    p.add_argument("--pre_seq_len", type=int, default=8)
    p.add_argument("--tuning_mode", default="deep_ptune", choices=["ptune", "deep_ptune"])
    p.add_argument("--batch_size", type=int, default=4)
    p.add_argument("--seq_len", type=int, default=32)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--lr", type=float, default=1e-2)
    args = p.parse_args()

    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    peers = [tuple([h, int(pt)]) for h, pt in (a.rsplit(":", 1) for a in args.initial_peers)]
    model = AutoDistributedModelForCausalLM.from_pretrained(
        args.model, initial_peers=peers,
        pre_seq_len=args.pre_seq_len, tuning_mode=args.tuning_mode,
    )
    trainable = [p for p in model.parameters() if p.requires_grad]
    n = sum(p.numel() for p in trainable)
    print(f"trainable parameters (client-side prompts): {n}")
    opt = torch.optim.AdamW(trainable, lr=args.lr)

    for step in range(args.steps):
        ids = torch.randint(0, model.config.vocab_size, (args.batch_size, args.seq_len))
        logits = model(input_ids=ids).logits
        loss = F.cross_entropy(
            logits[:, :-1].reshape(-1, logits.shape[-1]), ids[:, 1:].reshape(-1)
        )
        opt.zero_grad()
        loss.backward()
        opt.step()
        print(f"step {step}: loss {loss.item():.4f}")


if __name__ == "__main__":
    main()
