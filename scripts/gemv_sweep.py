"""Sweep gemv split counts per decode shape to find the bandwidth knee."""

import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from petals_amd import ops

hip = ops._load_hip_ops()


def bench(fn, n=100):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n


def main():
    ws = torch.empty(512 * 57344, dtype=torch.float32, device="cuda")
    shapes = [(8192, 10240, "qkv"), (8192, 8192, "o"), (8192, 57344, "gateup"),
              (28672, 8192, "down"), (8192, 32000, "head"), (16384, 106496, "405b-gateup")]
    for in_dim, out_dim, name in shapes:
        try:
            wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.02).to(torch.bfloat16)
        except RuntimeError:
            print(f"{name}: OOM, skip"); continue
        x = torch.randn(1, in_dim, device="cuda")
        gb = in_dim * out_dim * 2 / 1e9
        best = (0, 0)
        row = []
        for splits in (0, 8, 16, 32, 64, 128, 256, 448):
            if splits * 64 > in_dim and splits != 0:
                continue
            t = bench(lambda: hip.gemv_bf16(wt, x, ws, None, 0, splits))
            bw = gb / t
            row.append(f"s{splits}:{bw:.0f}")
            if bw > best[0]:
                best = (bw, splits)
        print(f"gemv {name} [{in_dim}x{out_dim}] {gb*1000:.0f}MB: {' '.join(row)}  BEST s{best[1]} {best[0]:.0f} GB/s", flush=True)
        del wt
        torch.cuda.empty_cache()


def nf4_sweep():
    ws = torch.empty(512 * 57344, dtype=torch.float32, device="cuda")
    shapes = [(8192, 10240, "qkv"), (8192, 8192, "o"), (8192, 57344, "gateup"), (28672, 8192, "down")]
    for in_dim, out_dim, name in shapes:
        wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.02).to(torch.bfloat16)
        packed, absmax = hip.nf4_quantize(wt)
        del wt
        x = torch.randn(1, in_dim, device="cuda")
        gb = (packed.numel() + absmax.numel() * 2) / 1e9
        amt = absmax.t().contiguous()
        for label, amt_arg in (("am_t", amt),):
            row, best = [], (0, 0)
            for splits in (0, 16, 32, 64, 96, 128, 160, 192, 256, 320, 448):
                if splits * 16 > in_dim and splits:
                    continue
                t = bench(lambda: hip.gemv_nf4(packed, absmax, x, ws, None, 0, splits, None, amt_arg))
                bw = gb / t
                row.append(f"s{splits}:{bw:.0f}")
                if bw > best[0]:
                    best = (bw, splits)
            print(f"gemv_nf4[{label}] {name} [{in_dim}x{out_dim}] {gb*1000:.0f}MB: {' '.join(row)}  "
                  f"BEST s{best[1]} {best[0]:.0f} GB/s", flush=True)
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
    nf4_sweep()
