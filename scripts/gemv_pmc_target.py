import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from petals_amd import ops
hip = ops._load_hip_ops()
ws = torch.empty(64 * 57344, dtype=torch.float32, device="cuda")
wt = (torch.randn(8192, 10240, device="cuda") * 0.02).to(torch.bfloat16)
packed, absmax = hip.nf4_quantize(wt)
x = torch.randn(1, 8192, device="cuda")
for _ in range(30):
    hip.gemv_bf16(wt, x, ws, None, 0)
    hip.gemv_nf4(packed, absmax, x, ws, None, 0)
torch.cuda.synchronize()
print("done")
