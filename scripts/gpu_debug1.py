"""Debug: attn_decode NaN at (gq=4, kv_len=1) + kernel-level microbenches."""

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import math
import time

import torch

from petals_amd import ops

hip = ops._load_hip_ops()
assert hip is not None


def attn_debug():
    b, kv_heads, hd, lmax, kv_len = 2, 8, 128, 640, 1
    kc = torch.zeros(b, kv_heads, lmax, hd, device="cuda", dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    kc[:, :, :kv_len] = 0.5
    vc[:, :, :kv_len] = 0.25
    empty = torch.empty(0, device="cuda")
    kv_len_t = torch.tensor([kv_len], dtype=torch.int32, device="cuda")
    for gq in (1, 2, 4, 8):
        q = torch.randn(b, kv_heads * gq * hd, device="cuda")
        for splits in (1, 2, 3, 0):
            out = hip.attn_decode_fused(q, kc, vc, kv_len_t, gq, splits, empty, empty, 1 / math.sqrt(hd))
            print(f"gq={gq} splits={splits}: nan={torch.isnan(out).any().item()} "
                  f"out[0,:3]={out[0,:3].tolist()} expect~0.25", flush=True)


def bench_kernel(fn, n=50):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def gemv_bench():
    ws = torch.empty(64 * 57344, dtype=torch.float32, device="cuda")
    for in_dim, out_dim, name in ((8192, 10240, "qkv"), (8192, 8192, "o"), (8192, 57344, "gateup"), (28672, 8192, "down"), (8192, 32000, "head")):
        wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.02).to(torch.bfloat16)
        x = torch.randn(1, in_dim, device="cuda")
        t = bench_kernel(lambda: hip.gemv_bf16(wt, x, ws, None, 0))
        gb = in_dim * out_dim * 2 / 1e9
        print(f"gemv {name} [{in_dim},{out_dim}]: {t*1e6:.1f} us, {gb/t:.0f} GB/s", flush=True)


def rmsnorm_bench():
    x = torch.randn(1, 8192, device="cuda").to(torch.bfloat16)
    w = torch.ones(8192, device="cuda", dtype=torch.bfloat16)
    t = bench_kernel(lambda: hip.rms_norm_f32out(x, w, 1e-5))
    print(f"rms_norm_f32out [1,8192]: {t*1e6:.1f} us", flush=True)


def attn_bench():
    b, kv_heads, gq, hd, lmax = 1, 8, 8, 128, 4096
    kc = (torch.randn(b, kv_heads, lmax, hd, device="cuda") * 0.3).to(torch.bfloat16)
    vc = kc.clone()
    q = torch.randn(b, kv_heads * gq * hd, device="cuda")
    empty = torch.empty(0, device="cuda")
    for kv_len in (128, 1024, 4000):
        kv_len_t = torch.tensor([kv_len], dtype=torch.int32, device="cuda")
        t = bench_kernel(lambda: hip.attn_decode_fused(q, kc, vc, kv_len_t, gq, 0, empty, empty, 1 / math.sqrt(hd)))
        gb = 2 * kv_heads * kv_len * hd * 2 / 1e9
        print(f"attn_decode kv_len={kv_len}: {t*1e6:.1f} us, {gb/t:.0f} GB/s", flush=True)


def layer_decode_bench():
    """One full fused 70B layer decode step, timed."""
    from petals_amd.models.config_base import load_model_config
    from petals_amd.server.from_pretrained import build_empty_block, init_random_block_

    config = load_model_config("llama-2-70b")
    blk = build_empty_block(config, 0, "cuda", torch.bfloat16)
    init_random_block_(blk, config, 0)
    blk = blk.eval().optimize_for_inference()
    ks, vs = blk.kv_cache_shape(1, 256)
    k = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
    v = torch.zeros(vs, device="cuda", dtype=torch.bfloat16)
    h = torch.randn(1, 1, config.hidden_size, device="cuda", dtype=torch.bfloat16)
    with torch.inference_mode():
        t = bench_kernel(lambda: blk(h, kv_cache=(k, v), prefix_length=100), n=100)
    gb = 1.69
    print(f"70B layer decode: {t*1e6:.1f} us ({gb/t:.0f} GB/s vs 1.69 GB weights; ideal ~280us)", flush=True)

    # python-overhead check: measure without sync inside loop but calling python chain
    with torch.inference_mode():
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(100):
            blk(h, kv_cache=(k, v), prefix_length=100)
        wall_nosync = (time.perf_counter() - t0) / 100
        torch.cuda.synchronize()
    print(f"70B layer decode (no per-call sync): {wall_nosync*1e6:.1f} us", flush=True)




def nf4_bench():
    ws = torch.empty(64 * 57344, dtype=torch.float32, device="cuda")
    for in_dim, out_dim, name in ((8192, 10240, "qkv"), (8192, 8192, "o"), (8192, 57344, "gateup"), (28672, 8192, "down")):
        wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.02).to(torch.bfloat16)
        packed, absmax = hip.nf4_quantize(wt)
        del wt
        x = torch.randn(1, in_dim, device="cuda")
        t = bench_kernel(lambda: hip.gemv_nf4(packed, absmax, x, ws, None, 0))
        gb = (packed.numel() + absmax.numel() * 2) / 1e9
        print(f"gemv_nf4 {name} [{in_dim},{out_dim}]: {t*1e6:.1f} us, {gb/t:.0f} GB/s packed ({gb*1000:.0f}MB)", flush=True)
        torch.cuda.empty_cache()




def int8_bench():
    for name, in_dim, out_dim in (("qkv", 8192, 10240), ("gateup", 8192, 57344), ("down", 28672, 8192)):
        wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.02).float()
        scale = (wt.abs().amax(dim=0).clamp_min(1e-8) / 127.0)
        q = torch.round(wt / scale).clamp(-127, 127).to(torch.int8).contiguous()
        sc = scale.to(torch.bfloat16).contiguous()
        del wt
        x = torch.randn(1, in_dim, device="cuda")
        ws = torch.empty(0, device="cuda")
        t = bench_kernel(lambda: hip.gemv_int8(q, sc, x, ws, None, 0))
        gb = q.numel() / 1e9
        print(f"gemv_int8 {name} [{in_dim},{out_dim}]: {t*1e6:.1f} us, {gb/t:.0f} GB/s ({gb*1000:.0f}MB)", flush=True)
        torch.cuda.empty_cache()


def prefill_attn_bench():
    import math
    from petals_amd.ops import reference

    for s in (512, 2048, 4096):
        b, qh, kvh, hd = 1, 64, 8, 128  # llama-2-70b shape
        q = (torch.randn(b, qh, s, hd, device="cuda") * 0.3).to(torch.bfloat16)
        k = (torch.randn(b, kvh, s, hd, device="cuda") * 0.3).to(torch.bfloat16)
        v = k.clone()
        sc = 1.0 / math.sqrt(hd)
        t_mfma = bench_kernel(lambda: hip.attn_prefill_fused(q, k, v, s, 0, sc, True), n=20)
        t_ref = bench_kernel(lambda: reference.attention(q, k, v, causal=True), n=5)
        flops = 4 * b * qh * s * s * hd / 2  # causal half
        print(f"prefill attn S={s}: mfma {t_mfma*1e3:.2f} ms ({flops/t_mfma/1e12:.0f} TF) "
              f"vs matmul+softmax {t_ref*1e3:.2f} ms", flush=True)


if __name__ == "__main__":
    _all = {
        "attn_debug": attn_debug,
        "rmsnorm": rmsnorm_bench,
        "gemv": gemv_bench,
        "attn": attn_bench,
        "layer": layer_decode_bench,
        "nf4": nf4_bench,
        "int8": int8_bench,
        "prefill": prefill_attn_bench,
    }
    picked = [a for a in sys.argv[1:] if a in _all]
    for name, fn in _all.items():
        if not picked or name in picked:
            fn()
