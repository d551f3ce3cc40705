"""Exact-match vs HuggingFace for BLOOM, Falcon and Mixtral (golden-reference
pattern): save an HF checkpoint, load blocks through the server-side loader,
run the local chain (embeddings -> blocks -> norm -> head), compare logits and
incremental decoding."""

import pytest
import torch

from petals_amd.models.config_base import load_model_config
from petals_amd.server.from_pretrained import load_pretrained_block

transformers = pytest.importorskip("transformers")

FAMILIES = {
    "bloom": dict(
        hf_config=lambda t: t.BloomConfig(hidden_size=64, n_head=4, n_layer=3, vocab_size=100),
        embed=lambda m: lambda ids: m.transformer.word_embeddings_layernorm(m.transformer.word_embeddings(ids)),
        norm=lambda m: m.transformer.ln_f,
        head=lambda m: m.lm_head,
    ),
    "falcon-new": dict(
        hf_config=lambda t: t.FalconConfig(
            hidden_size=64, num_attention_heads=4, num_hidden_layers=3, vocab_size=100,
            new_decoder_architecture=True, num_kv_heads=2, bias=False, parallel_attn=True,
        ),
        embed=lambda m: m.transformer.word_embeddings,
        norm=lambda m: m.transformer.ln_f,
        head=lambda m: m.lm_head,
    ),
    "falcon-7b-style": dict(
        hf_config=lambda t: t.FalconConfig(
            hidden_size=64, num_attention_heads=4, num_hidden_layers=3, vocab_size=100,
            new_decoder_architecture=False, multi_query=True, bias=False, parallel_attn=True,
        ),
        embed=lambda m: m.transformer.word_embeddings,
        norm=lambda m: m.transformer.ln_f,
        head=lambda m: m.lm_head,
    ),
    "mixtral": dict(
        hf_config=lambda t: t.MixtralConfig(
            hidden_size=64, num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
            intermediate_size=128, vocab_size=100, num_local_experts=4, num_experts_per_tok=2,
        ),
        embed=lambda m: m.model.embed_tokens,
        norm=lambda m: m.model.norm,
        head=lambda m: m.lm_head,
    ),
}

HF_CLASSES = {
    "bloom": "BloomForCausalLM",
    "falcon-new": "FalconForCausalLM",
    "falcon-7b-style": "FalconForCausalLM",
    "mixtral": "MixtralForCausalLM",
}


@pytest.mark.parametrize("family", list(FAMILIES))
def test_family_exact_match(family, tmp_path):
    spec = FAMILIES[family]
    torch.manual_seed(0)
    hf_cfg = spec["hf_config"](transformers)
    hf_model = getattr(transformers, HF_CLASSES[family])(hf_cfg).eval()
    ckpt = tmp_path / family
    hf_model.save_pretrained(ckpt, safe_serialization=True)

    config = load_model_config(str(ckpt))
    blocks = [
        load_pretrained_block(str(ckpt), config, i, torch_dtype=torch.float32)
        for i in range(config.num_blocks)
    ]

    torch.manual_seed(1)
    ids = torch.randint(0, 100, (2, 9))
    with torch.no_grad():
        ref = hf_model(ids).logits
        h = spec["embed"](hf_model)(ids)
        for blk in blocks:
            h = blk(h)
        out = spec["head"](hf_model)(spec["norm"](hf_model)(h))
    assert torch.allclose(out, ref, atol=2e-4, rtol=1e-3), (out - ref).abs().max()

    # incremental decode == one-shot
    with torch.no_grad():
        h_full = spec["embed"](hf_model)(ids)
        full_out = h_full
        caches = []
        for blk in blocks:
            ks, vs = blk.kv_cache_shape(2, 12)
            caches.append((torch.zeros(ks), torch.zeros(vs)))
        parts = []
        x = h_full[:, :5]
        for t_start, t_end in ((0, 5), (5, 6), (6, 9)):
            x = h_full[:, t_start:t_end]
            for blk, kv in zip(blocks, caches):
                x = blk(x, kv_cache=kv, prefix_length=t_start)
            parts.append(x)
        inc = torch.cat(parts, dim=1)
        ref_hidden = h_full
        for blk in blocks:
            ref_hidden = blk(ref_hidden)
    assert torch.allclose(inc, ref_hidden, atol=1e-4, rtol=1e-3), (inc - ref_hidden).abs().max()
