"""Time the MFMA prefill attention kernel (llama-2-70b head shape) and report
achieved TFLOP/s. Usage: python scripts/prefill_time.py [S ...]"""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from petals_amd import ops
from petals_amd.ops import reference

hip = ops._load_hip_ops()
lens = [int(x) for x in sys.argv[1:]] or [2048, 4096, 8192]
b, qh, kvh, hd = 1, 64, 8, 128
for s in lens:
    torch.manual_seed(0)
    q = (torch.randn(b, qh, s, hd, device="cuda") * 0.3).to(torch.bfloat16)
    k = (torch.randn(b, kvh, s, hd, device="cuda") * 0.3).to(torch.bfloat16)
    v = (torch.randn(b, kvh, s, hd, device="cuda") * 0.3).to(torch.bfloat16)
    sc = 1.0 / math.sqrt(hd)
    out = hip.attn_prefill_fused(q, k, v, s, 0, sc, True)
    if s <= 2048:  # numerics check vs fp32 reference
        ref = reference.attention(q.float().cpu(), k.float().cpu(), v.float().cpu(), causal=True)
        err = (out.float().cpu() - ref).abs().max().item()
        assert err < 3e-2, f"S={s} max err {err}"
    torch.cuda.synchronize()
    n = 10
    t0 = time.perf_counter()
    for _ in range(n):
        hip.attn_prefill_fused(q, k, v, s, 0, sc, True)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / n
    flops = 2 * 2 * b * qh * hd * (s * (s + 1) / 2)  # causal QK^T + PV
    print(f"S={s:6d}  {dt*1e3:7.3f} ms  {flops/dt/1e12:7.1f} TF", flush=True)
