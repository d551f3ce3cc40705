"""CPU checks for the fused-path weight re-permutations: the [q|k|v] column
orders used by BloomFastPath/FalconFastPath must agree with the blocks' own
checkpoint-layout QKV splits (models/bloom/block.py, models/falcon/block.py)."""

import torch

from petals_amd.ops.fused_decode import _bloom_qkv_perm, _falcon_qkv_perm


def test_bloom_qkv_perm_matches_block_split():
    qh, hd = 4, 16
    H = qh * hd
    torch.manual_seed(0)
    w = torch.randn(3 * H, H)  # HF layout: rows = fused out features
    x = torch.randn(2, 5, H)

    fused = (x @ w.t()).view(2, 5, qh, 3, hd)  # the block's split (block.py)
    q_ref = fused[..., 0, :].reshape(2, 5, qh * hd)
    k_ref = fused[..., 1, :].reshape(2, 5, qh * hd)
    v_ref = fused[..., 2, :].reshape(2, 5, qh * hd)

    perm = _bloom_qkv_perm(qh, hd)
    wt = w.t()[:, perm]  # [H, 3H] in [q|k|v] order (fast-path layout)
    qkv = x @ wt
    assert torch.allclose(qkv[..., : qh * hd], q_ref, atol=1e-5)
    assert torch.allclose(qkv[..., qh * hd : 2 * qh * hd], k_ref, atol=1e-5)
    assert torch.allclose(qkv[..., 2 * qh * hd :], v_ref, atol=1e-5)
    # bias permutes the same way
    b = torch.randn(3 * H)
    assert torch.allclose((x @ w.t() + b).view(2, 5, qh, 3, hd)[..., 0, :].reshape(2, 5, -1),
                          (qkv + b[perm])[..., : qh * hd], atol=1e-5)


def test_falcon_qkv_perm_matches_block_split():
    qh, kh, hd = 8, 2, 16
    gq = qh // kh
    H = qh * hd
    out = (qh + 2 * kh) * hd
    torch.manual_seed(1)
    w = torch.randn(out, H)
    x = torch.randn(2, 3, H)

    fused = (x @ w.t()).view(2, 3, kh, gq + 2, hd)  # new-decoder split (block.py)
    q_ref = fused[..., :-2, :].reshape(2, 3, qh * hd)
    k_ref = fused[..., -2, :].reshape(2, 3, kh * hd)
    v_ref = fused[..., -1, :].reshape(2, 3, kh * hd)

    perm = _falcon_qkv_perm(qh, kh, hd)
    wt = w.t()[:, perm]
    qkv = x @ wt
    assert torch.allclose(qkv[..., : qh * hd], q_ref, atol=1e-5)
    assert torch.allclose(qkv[..., qh * hd : (qh + kh) * hd], k_ref, atol=1e-5)
    assert torch.allclose(qkv[..., (qh + kh) * hd :], v_ref, atol=1e-5)
