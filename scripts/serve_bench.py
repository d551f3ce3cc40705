"""Full-stack serving measurement: a REAL server (DHT + TCP RPC + scheduler +
per-session span hipGraph) + a thin client on the same box — the end-to-end
number users of the framework get, including wire serialization.

Usage: python scripts/serve_bench.py [--model llama-2-7b] [--new-tokens 64]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-2-7b")
    p.add_argument("--new-tokens", type=int, default=64)
    p.add_argument("--prompt-len", type=int, default=32)
    p.add_argument("--quant", default="none")
    args = p.parse_args()

    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    boot = DHT(host="127.0.0.1")
    t0 = time.time()
    server = Server(
        args.model, initial_peers=[boot.listen_addr], host="127.0.0.1",
        device="cuda", torch_dtype="bfloat16", dht_prefix="serve-bench",
        throughput=1000.0, quant_type=args.quant,
    ).start()
    print(f"server up in {time.time()-t0:.1f}s, blocks={server.num_blocks}", flush=True)

    model = AutoDistributedModelForCausalLM.from_pretrained(
        args.model, initial_peers=[boot.listen_addr], dht_prefix="serve-bench",
        show_route=False, max_retries=2,
    )
    # thin client computes embeddings + LM head on the GPU too
    model = model.to(device="cuda", dtype=torch.bfloat16)
    torch.manual_seed(0)
    ids = torch.randint(0, model.config.vocab_size, (1, args.prompt_len))

    try:
        _run(model, ids, args)
    finally:
        model.transformer.h.sequence_manager.shutdown()
        server.shutdown()
        boot.shutdown()
        os._exit(0)  # skip interpreter-teardown races between HIP and daemon threads


def _run(model, ids, args):
    # --- isolate raw session step latency (client + wire + server compute),
    # no embeddings/head/generate logic
    H = model.config.hidden_size
    dev = model.transformer.embed_tokens.weight.device
    h = torch.randn(1, 1, H, device=dev, dtype=model.transformer.embed_tokens.weight.dtype) * 0.02
    with model.transformer.h.inference_session(max_length=64) as sess:
        sess.step(h)  # session open + first step (graph capture server-side)
        import time as _t

        t0 = _t.perf_counter()
        for _ in range(20):
            sess.step(h)
        dt = (_t.perf_counter() - t0) / 20
        print(f"raw session step: {dt*1000:.2f} ms/token (client+wire+server)", flush=True)

    # warmup generation (includes session setup + graph capture)
    out = model.generate(ids, max_new_tokens=8, do_sample=False)
    assert out.shape[1] == args.prompt_len + 8

    t0 = time.perf_counter()
    out = model.generate(ids, max_new_tokens=args.new_tokens, do_sample=False)
    elapsed = time.perf_counter() - t0
    tps = args.new_tokens / elapsed
    print(f"FULL-STACK serving: {tps:.2f} tokens/s ({elapsed/args.new_tokens*1000:.1f} ms/token) "
          f"model={args.model} quant={args.quant} [includes TCP wire + client embeds/head on GPU]", flush=True)


if __name__ == "__main__":
    main()
