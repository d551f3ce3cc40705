"""Exact-match tests of the native LlamaBlock vs HuggingFace transformers
(golden-reference pattern, reference tests/test_block_exact_match.py)."""

import pytest
import torch

from petals_amd.models.llama.block import LlamaBlock
from petals_amd.models.llama.config import LlamaConfig

HF_CFG = dict(
    hidden_size=64,
    num_hidden_layers=2,
    num_attention_heads=4,
    num_key_value_heads=2,
    intermediate_size=128,
    vocab_size=100,
    max_position_embeddings=128,
)


@pytest.fixture(scope="module")
def hf_model():
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(0)
    cfg = transformers.LlamaConfig(**HF_CFG)
    return transformers.LlamaForCausalLM(cfg).eval()


def make_native_block(hf_model, layer_idx=0) -> LlamaBlock:
    cfg = LlamaConfig(
        hidden_size=64,
        num_hidden_layers=2,
        num_attention_heads=4,
        num_key_value_heads=2,
        intermediate_size=128,
        vocab_size=100,
        max_position_embeddings=128,
        layer_norm_eps=hf_model.config.rms_norm_eps,
    )
    block = LlamaBlock(cfg, layer_idx=layer_idx).eval()
    prefix = f"model.layers.{layer_idx}."
    sd = {k[len(prefix):]: v for k, v in hf_model.state_dict().items() if k.startswith(prefix)}
    block.load_state_dict(sd)
    return block


@torch.no_grad()
def test_block_forward_matches_hf(hf_model):
    block = make_native_block(hf_model)
    hf_layer = hf_model.model.layers[0]
    rotary = hf_model.model.rotary_emb

    torch.manual_seed(1)
    x = torch.randn(2, 9, 64)
    position_ids = torch.arange(9).unsqueeze(0).expand(2, -1)
    pos_emb = rotary(x, position_ids)
    causal = torch.full((9, 9), float("-inf")).triu(1).view(1, 1, 9, 9).expand(2, -1, -1, -1)
    ref = hf_layer(x, attention_mask=causal, position_ids=position_ids, position_embeddings=pos_emb)
    if isinstance(ref, tuple):
        ref = ref[0]
    out = block(x)
    assert torch.allclose(out, ref, atol=1e-5, rtol=1e-4), (out - ref).abs().max()


@torch.no_grad()
def test_block_incremental_inference_matches_forward(hf_model):
    """Token-by-token decoding with the KV cache == one-shot forward."""
    block = make_native_block(hf_model)
    torch.manual_seed(2)
    x = torch.randn(1, 12, 64)
    full = block(x)

    k_shape, v_shape = block.kv_cache_shape(batch_size=1, max_length=16)
    k_cache = torch.zeros(k_shape)
    v_cache = torch.zeros(v_shape)
    outs = []
    # prefill 5 tokens, then decode one at a time
    outs.append(block(x[:, :5], kv_cache=(k_cache, v_cache), prefix_length=0))
    for t in range(5, 12):
        outs.append(block(x[:, t : t + 1], kv_cache=(k_cache, v_cache), prefix_length=t))
    step = torch.cat(outs, dim=1)
    assert torch.allclose(step, full, atol=1e-5, rtol=1e-4), (step - full).abs().max()


@torch.no_grad()
def test_full_logits_match_hf(hf_model):
    """Embeddings + both blocks + norm + lm_head vs HF full model."""
    blocks = [make_native_block(hf_model, i) for i in range(2)]
    torch.manual_seed(3)
    ids = torch.randint(0, 100, (2, 7))
    h = hf_model.model.embed_tokens(ids)
    for blk in blocks:
        h = blk(h)
    from petals_amd.ops import rms_norm

    h = rms_norm(h, hf_model.model.norm.weight, hf_model.config.rms_norm_eps)
    logits = h @ hf_model.lm_head.weight.T
    ref = hf_model(ids).logits
    assert torch.allclose(logits, ref, atol=1e-4, rtol=1e-3), (logits - ref).abs().max()


def test_block_backward_runs(hf_model):
    block = make_native_block(hf_model)
    x = torch.randn(1, 6, 64, requires_grad=True)
    out = block(x)
    out.square().mean().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
