"""Full-stack serving measurement: a REAL server process (DHT + TCP RPC +
scheduler + per-session span hipGraph) + a thin client process — the
end-to-end number users of the framework get, including wire serialization.

Usage: python scripts/serve_bench.py [--model llama-2-7b] [--new-tokens 64]
"""

import argparse
import os
import subprocess
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
    p.add_argument("--subprocess", action="store_true",
                   help="server in a separate process (NB: two processes sharing one GPU "
                        "context-switch on every hop; use for isolation tests only)")
    args = p.parse_args()

    if os.environ.get("PETALS_AMD_GC_TUNE"):
        import gc

        gc.collect()
        gc.freeze()
        gc.set_threshold(100000, 50, 50)

    from petals_amd.dht.node import DHT
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    boot = DHT(host="127.0.0.1")
    t0 = time.time()
    server = proc = None
    if not args.subprocess:
        from petals_amd.server.server import Server

        server = Server(
            args.model, initial_peers=[boot.listen_addr], host="127.0.0.1",
            device="cuda", torch_dtype="bfloat16", dht_prefix="serve-bench",
            throughput=1000.0, quant_type=args.quant,
        ).start()
    else:
        proc = subprocess.Popen(
            [sys.executable, "-m", "petals_amd.cli.run_server", args.model,
             "--host", "127.0.0.1", "--initial_peers", f"127.0.0.1:{boot.listen_addr[1]}",
             "--torch_dtype", "bfloat16", "--dht_prefix", "serve-bench",
             "--throughput", "1000", "--quant_type", args.quant],
            cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
            env={**os.environ},
        )
        while True:
            line = proc.stdout.readline()
            if "listening on" in line:
                break
            if proc.poll() is not None:
                raise RuntimeError("server died during startup")
        # drain further server output in a thread: a full pipe buffer would
        # BLOCK the server on its own log writes
        import threading

        def _drain():
            with open("gpurun_out/server_sub.log", "w") as f:
                for line in proc.stdout:
                    f.write(line)

        os.makedirs("gpurun_out", exist_ok=True)
        threading.Thread(target=_drain, daemon=True).start()
    print(f"server up in {time.time()-t0:.1f}s", flush=True)

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
        if server is not None:
            server.shutdown()
        if proc is not None:
            proc.terminate()
        boot.shutdown()
        os._exit(0)  # skip interpreter-teardown races between HIP and daemon threads


def _run(model, ids, args):
    # --- isolate raw session step latency (client + wire + server compute)
    H = model.config.hidden_size
    dev = model.transformer.embed_tokens.weight.device
    h = torch.randn(1, 1, H, device=dev, dtype=model.transformer.embed_tokens.weight.dtype) * 0.02
    with model.transformer.h.inference_session(max_length=64) as sess:
        sess.step(h)  # session open + first step (graph capture server-side)
        t0 = time.perf_counter()
        for _ in range(20):
            sess.step(h)
        dt = (time.perf_counter() - t0) / 20
        print(f"raw session step: {dt*1000:.2f} ms/token (client+wire+server)", flush=True)

    # warmup generation (includes session setup + graph capture)
    out = model.generate(ids, max_new_tokens=8, do_sample=False)
    assert out.shape[1] == args.prompt_len + 8

    t0 = time.perf_counter()
    out = model.generate(ids, max_new_tokens=args.new_tokens, do_sample=False)
    elapsed = time.perf_counter() - t0
    tps = args.new_tokens / elapsed
    print(f"FULL-STACK serving: {tps:.2f} tokens/s ({elapsed/args.new_tokens*1000:.1f} ms/token) "
          f"model={args.model} quant={args.quant} "
          f"[{'separate server process' if args.subprocess else 'co-located (one GPU context)'}]", flush=True)


if __name__ == "__main__":
    main()
