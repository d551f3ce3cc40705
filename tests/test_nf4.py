"""NF4 quantization: CPU reference roundtrip + GPU kernel parity."""

import math

import pytest
import torch

from petals_amd.ops import nf4


def test_cpu_roundtrip_error_bound():
    torch.manual_seed(0)
    w = torch.randn(64, 256) * 0.02
    packed, absmax = nf4.quantize(w)
    assert packed.shape == (64, 128) and packed.dtype == torch.uint8
    assert absmax.shape == (64, 4) and absmax.dtype == torch.bfloat16
    deq = nf4.dequantize(packed, absmax).float()
    err = (deq - w).abs()
    # NF4 with block 64: worst-case relative error per block is bounded by the
    # largest inter-level gap (~0.15 of absmax)
    blocks = w.reshape(64, 4, 64)
    bound = blocks.abs().amax(-1, keepdim=True).expand_as(blocks).reshape(64, 256) * 0.16 + 1e-6
    assert (err <= bound).all()
    # and typical error is much smaller
    assert err.mean() < 0.02 * w.abs().mean() * 5


def test_exact_levels_roundtrip():
    absmax_scale = 0.37
    levels = nf4.NF4_LEVELS * absmax_scale
    w = levels.repeat(4).reshape(1, 64)
    packed, absmax = nf4.quantize(w)
    deq = nf4.dequantize(packed, absmax).float()
    assert torch.allclose(deq, w.to(torch.bfloat16).float(), atol=3e-3)


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
class TestNF4GPU:
    @pytest.fixture(scope="class")
    def hip(self):
        from petals_amd import ops

        mod = ops._load_hip_ops()
        assert mod is not None
        return mod

    def test_quantize_matches_cpu(self, hip):
        torch.manual_seed(1)
        w = (torch.randn(128, 512) * 0.05).to(torch.bfloat16)
        packed_g, absmax_g = hip.nf4_quantize(w.cuda())
        packed_c, absmax_c = nf4.quantize(w.float())
        assert torch.equal(absmax_g.cpu(), absmax_c)
        # levels can tie at midpoints; compare dequantized values instead
        deq_g = hip.nf4_dequantize(packed_g, absmax_g).float().cpu()
        deq_c = nf4.dequantize(packed_c, absmax_c).float()
        assert (deq_g - deq_c).abs().max() < 2e-3

    def test_gemv_nf4(self, hip):
        torch.manual_seed(2)
        in_dim, out_dim = 1024, 2048
        wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.05).to(torch.bfloat16)
        packed, absmax = hip.nf4_quantize(wt)
        deq = hip.nf4_dequantize(packed, absmax)
        x = torch.randn(1, in_dim, device="cuda")
        ws = torch.empty(0, device="cuda")
        y = hip.gemv_nf4(packed, absmax, x, ws, None, 0)
        ref = x @ deq.float()  # same quantized weights -> near-exact match
        assert torch.allclose(y, ref, atol=2e-2, rtol=2e-2), (y - ref).abs().max()

    def test_fast_path_nf4_close_to_bf16(self, hip):
        from petals_amd.models import get_model_block
        from petals_amd.models.config_base import load_model_config
        from petals_amd.server.from_pretrained import init_random_block_

        cfg = load_model_config("test-llama")
        cfg.hidden_size, cfg.num_attention_heads, cfg.num_key_value_heads, cfg.intermediate_size = 512, 4, 2, 1024
        blk = get_model_block(cfg, 0)
        init_random_block_(blk, cfg, 0)
        sd = blk.state_dict()

        outs = {}
        for quant in ("none", "nf4"):
            b = get_model_block(cfg, 0)
            b.load_state_dict(sd)
            b = b.to("cuda", torch.bfloat16).eval().optimize_for_inference(quant=quant)
            ks, vs = b.kv_cache_shape(1, 16)
            k = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
            v = torch.zeros(vs, device="cuda", dtype=torch.bfloat16)
            torch.manual_seed(3)
            x = torch.randn(1, 5, 512, device="cuda", dtype=torch.bfloat16) * 0.3
            with torch.inference_mode():
                h = b(x, kv_cache=(k, v), prefix_length=0)
                h = b(h[:, -1:], kv_cache=(k, v), prefix_length=5)
            outs[quant] = h.float().cpu()
        diff = (outs["none"] - outs["nf4"]).abs().max()
        assert diff < 0.2, diff  # quantization error, not a wiring bug


def test_int8_weight_roundtrip_error():
    """Per-out-column symmetric int8 (the _FastWeight int8 scheme) keeps
    relative weight error within ~1% on gaussian weights (CPU math)."""
    import torch

    torch.manual_seed(2)
    w = torch.randn(512, 768) * 0.05
    scale = w.abs().amax(dim=0).clamp_min(1e-8) / 127.0
    q = torch.round(w / scale).clamp(-127, 127).to(torch.int8)
    deq = q.float() * scale
    rel = (deq - w).norm() / w.norm()
    assert rel < 0.012, rel.item()
    assert (deq - w).abs().max() <= scale.max() * 0.5 + 1e-6
