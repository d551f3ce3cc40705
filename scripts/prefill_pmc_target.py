"""PMC target: repeated prefill-attention launches at the llama-2-70b shape.

Run under `rocprofv3 --pmc <counters> -- python scripts/prefill_pmc_target.py [S]`
(counters in their own run; no trace domains).
"""
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from petals_amd import ops

hip = ops._load_hip_ops()
s = int(sys.argv[1]) if len(sys.argv) > 1 else 4096
b, qh, kvh, hd = 1, 64, 8, 128
q = (torch.randn(b, qh, s, hd, device="cuda") * 0.3).to(torch.bfloat16)
k = (torch.randn(b, kvh, s, hd, device="cuda") * 0.3).to(torch.bfloat16)
v = k.clone()
sc = 1.0 / math.sqrt(hd)
for _ in range(20):
    hip.attn_prefill_fused(q, k, v, s, 0, sc, True)
torch.cuda.synchronize()
print("done")
