"""Characterize the serving-cadence stall seen in round 1: replay a llama-2-70b
whole-span decode graph in a loop with a controlled host-side gap between
steps (serving syncs every token; bench.py's back-to-back loop does not) and
report per-step time distribution per gap size.

Round-1 evidence (PARITY.md "serving cadence"): with the ~1 ms gaps of real
serving, every ~3rd step ran ~4.8x slower while sclk cycled through a 95 MHz
idle level; back-to-back replays never stalled.

Usage: python scripts/stall_probe.py [--blocks 80] [--steps 60]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-2-70b")
    p.add_argument("--blocks", type=int, default=0, help="0 = all")
    p.add_argument("--steps", type=int, default=60)
    p.add_argument("--quant", default="nf4")
    p.add_argument("--gaps-ms", default="0,0.3,0.7,1.5,3,6")
    args = p.parse_args()

    from petals_amd.models.config_base import load_model_config
    from petals_amd.ops.fused_decode import DecodeContext
    from petals_amd.server.from_pretrained import build_empty_block, init_random_block_
    from petals_amd.utils.graphs import GraphedCallable

    config = load_model_config(args.model)
    n_blocks = args.blocks or config.num_blocks
    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)
    dtype = torch.bfloat16
    H = config.hidden_size
    max_len = 512

    t0 = time.time()
    blocks, caches = [], []
    for i in range(n_blocks):
        blk = build_empty_block(config, i, device, dtype)
        init_random_block_(blk, config, i)
        blk = blk.eval()
        blk.optimize_for_inference(quant=args.quant)
        blocks.append(blk)
        ks, vs = blk.kv_cache_shape(1, max_len)
        caches.append((torch.zeros(ks, device=device, dtype=dtype), torch.zeros(vs, device=device, dtype=dtype)))
    print(f"built {n_blocks} blocks in {time.time()-t0:.1f}s", flush=True)

    ctx = DecodeContext(device)
    ctx.set_position(128)
    h_in = torch.randn(1, 1, H, device=device, dtype=dtype) * 0.02

    def span_fn():
        h = h_in
        for blk, (k, v) in zip(blocks, caches):
            h = blk(h, kv_cache=(k, v), ctx=ctx)
        return h

    g = GraphedCallable(span_fn, [])
    pos = 130

    for gap_ms in [float(x) for x in args.gaps_ms.split(",")]:
        # warm
        for _ in range(5):
            ctx.set_position(pos); pos += 1
            g.replay()
            torch.cuda.synchronize()
        times = []
        for _ in range(args.steps):
            if gap_ms > 0:
                time.sleep(gap_ms / 1000)
            t1 = time.perf_counter()
            ctx.set_position(pos); pos += 1
            g.replay()
            out = g.static_outputs[0].cpu()  # same per-step sync as serving
            times.append((time.perf_counter() - t1) * 1000)
        times_sorted = sorted(times)
        mean = sum(times) / len(times)
        p50 = times_sorted[len(times) // 2]
        p95 = times_sorted[int(len(times) * 0.95)]
        mx = times_sorted[-1]
        n_slow = sum(1 for t in times if t > 1.5 * p50)
        print(f"gap {gap_ms:5.1f} ms: mean {mean:6.2f}  p50 {p50:6.2f}  p95 {p95:6.2f}  "
              f"max {mx:6.2f}  slow-steps {n_slow}/{len(times)}", flush=True)


if __name__ == "__main__":
    main()
