"""Localize prefill-attention numerics failure: per-head/per-row error map."""
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from petals_amd import ops

hip = ops._load_hip_ops()


def ref_attn(q, k, v, off, causal):
    # fp32 reference on GPU, GQA by head repeat
    b, qh, s, hd = q.shape
    kvh = k.shape[1]
    kf = k.float().repeat_interleave(qh // kvh, dim=1)
    vf = v.float().repeat_interleave(qh // kvh, dim=1)
    scores = q.float() @ kf.transpose(-1, -2) / math.sqrt(hd)
    if causal:
        kv_len = kf.shape[2]
        qpos = torch.arange(s, device=q.device)[:, None] + off
        kpos = torch.arange(kv_len, device=q.device)[None, :]
        scores = scores.masked_fill(kpos > qpos, float("-inf"))
    return torch.softmax(scores, dim=-1) @ vf


def run_case(name, b, qh, kvh, s, hd, off, causal):
    torch.manual_seed(9)
    kv_len = off + s
    lmax = kv_len + 16
    q = (torch.randn(b, qh, s, hd, device="cuda") * 0.5).to(torch.bfloat16)
    k = torch.zeros(b, kvh, lmax, hd, device="cuda", dtype=torch.bfloat16)
    v = torch.zeros_like(k)
    k[:, :, :kv_len] = (torch.randn(b, kvh, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    v[:, :, :kv_len] = (torch.randn(b, kvh, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    out = hip.attn_prefill_fused(q, k, v, kv_len, off, 1.0 / math.sqrt(hd), causal)
    ref = ref_attn(q, k[:, :, :kv_len], v[:, :, :kv_len], off, causal)
    err = (out.float() - ref).abs()  # [b, qh, s, hd]
    print(f"{name}: max={err.max().item():.4f}")
    if err.max().item() > 0.05:
        perhead = err.amax(dim=(2, 3))
        print("  per-head max:\n", perhead.cpu().numpy().round(3))
        bb, hh = divmod(perhead.argmax().item(), qh)
        rows = err[bb, hh].amax(dim=1).cpu().numpy().round(3)
        bad = [i for i, e in enumerate(rows) if e > 0.05]
        print(f"  worst head ({bb},{hh}): bad rows {bad[:40]}{'...' if len(bad) > 40 else ''} of {s}")


kvt = os.environ.get("PETALS_PREFILL_KVT", "64(default)")
print(f"KVT={kvt}")
run_case("case0 b2 qh8 kvh2 s67 hd128 causal", 2, 8, 2, 67, 128, 0, True)
run_case("case1 b1 qh4 kvh4 s200 hd128 causal", 1, 4, 4, 200, 128, 0, True)
run_case("case2 b1 qh8 kvh8 s33 hd64 off50", 1, 8, 8, 33, 64, 50, True)
run_case("case3 b2 qh4 kvh1 s64 hd128 noncausal", 2, 4, 1, 64, 128, 0, False)
run_case("uniform s64", 1, 4, 4, 64, 128, 0, True)
run_case("uniform s128", 1, 4, 4, 128, 128, 0, True)
run_case("tiny s16", 1, 2, 2, 16, 128, 0, True)
