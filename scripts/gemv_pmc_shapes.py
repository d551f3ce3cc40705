"""PMC target: run each 70B NF4 gemv shape separately (30 reps each) so
per-shape counters can be compared (why do qkv/o sit at ~2.2-2.5 TB/s packed
while gateup reaches 4.8?)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from petals_amd import ops
hip = ops._load_hip_ops()
ws = torch.empty(512 * 57344, dtype=torch.float32, device="cuda")
shapes = [(8192, 10240, 160), (8192, 8192, 160), (8192, 57344, 64), (28672, 8192, 256)]
for in_dim, out_dim, splits in shapes:
    wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.02).to(torch.bfloat16)
    packed, absmax = hip.nf4_quantize(wt)
    amt = absmax.t().contiguous()
    del wt
    x = torch.randn(1, in_dim, device="cuda")
    for _ in range(30):
        hip.gemv_nf4(packed, absmax, x, ws, None, 0, splits, None, amt)
    torch.cuda.synchronize()
    del packed, absmax, amt
    torch.cuda.empty_cache()
print("done")
