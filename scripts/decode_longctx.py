"""Decode-attention bandwidth sweep over kv_len (llama-2-70b shape)."""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from petals_amd import ops

hip = ops._load_hip_ops()
B, KV, GQ, HD = 1, 8, 8, 128  # llama-2-70b
empty = torch.empty(0, dtype=torch.float32, device="cuda")

for kv_len in (4096, 16384, 32768, 65536, 131072):
    lmax = kv_len
    k_cache = (torch.randn(B, KV, lmax, HD, device="cuda") * 0.3).to(torch.bfloat16)
    v_cache = k_cache.clone()
    q = torch.randn(B, KV * GQ * HD, device="cuda")
    kvl = torch.tensor([kv_len], dtype=torch.int32, device="cuda")
    best = (None, float("inf"))
    for splits in (0, 32, 64, 128, 256, 512):
        try:
            for _ in range(3):
                hip.attn_decode_fused(q, k_cache, v_cache, kvl, GQ, splits, empty, empty, 1 / math.sqrt(HD))
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(20):
                hip.attn_decode_fused(q, k_cache, v_cache, kvl, GQ, splits, empty, empty, 1 / math.sqrt(HD))
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 20
        except RuntimeError as e:
            print(f"kv={kv_len} splits={splits}: {e}", flush=True)
            continue
        gb = 2 * B * KV * kv_len * HD * 2 / 1e9  # K+V bytes read
        print(f"kv={kv_len:6d} splits={splits:3d}: {dt*1e6:7.1f} us  {gb/dt:6.0f} GB/s", flush=True)
        if dt < best[1]:
            best = (splits, dt)
    print(f"  -> best splits={best[0]}  {best[1]*1e6:.1f} us", flush=True)
    del k_cache, v_cache
    torch.cuda.empty_cache()
