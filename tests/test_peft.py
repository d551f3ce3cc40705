"""LoRA adapter serving (reference tests/test_peft.py pattern): adapter output
must equal a merged-weight model (W' = W + scale * B @ A)."""

import json
import os

import pytest
import torch

from petals_amd.models import get_model_block
from petals_amd.models.config_base import load_model_config
from petals_amd.server.from_pretrained import init_random_block_
from petals_amd.utils.peft import BlockAdapter, add_adapter_to_block, load_block_adapter, using_adapter

RANK, ALPHA = 4, 8

# per-family adapter targets: (HF checkpoint path prefix, projections trained)
_FAMILY = {
    "test-llama": ("model.layers", ("self_attn.q_proj", "mlp.down_proj")),
    "test-falcon": ("transformer.h", ("self_attention.query_key_value", "mlp.dense_4h_to_h")),
    "test-bloom": ("h", ("self_attention.query_key_value", "mlp.dense_4h_to_h")),
}


def _module_by_path(block, path):
    mod = block
    for part in path.split("."):
        mod = getattr(mod, part)
    return mod


def make_adapter_dir(tmp_path, config, n_blocks, seed=11, model="test-llama"):
    from safetensors.torch import save_file

    from petals_amd.models import get_model_block

    torch.manual_seed(seed)
    prefix, projs = _FAMILY[model]
    probe = get_model_block(config, 0)
    tensors = {}
    for i in range(n_blocks):
        for proj in projs:
            lin = _module_by_path(probe, proj)
            in_d, out_d = lin.in_features, lin.out_features
            tensors[f"base_model.model.{prefix}.{i}.{proj}.lora_A.weight"] = torch.randn(RANK, in_d) * 0.05
            tensors[f"base_model.model.{prefix}.{i}.{proj}.lora_B.weight"] = torch.randn(out_d, RANK) * 0.05
    d = tmp_path / "test-adapter"
    d.mkdir()
    with open(d / "adapter_config.json", "w") as f:
        json.dump({"r": RANK, "lora_alpha": ALPHA, "peft_type": "LORA"}, f)
    save_file(tensors, str(d / "adapter_model.safetensors"))
    return str(d)


_KEY_TO_PATH = {
    "q": "self_attn.q_proj", "k": "self_attn.k_proj", "v": "self_attn.v_proj",
    "o": "self_attn.o_proj", "gate": "mlp.gate_proj", "up": "mlp.up_proj",
    "down": "mlp.down_proj",
    "qkv": "self_attention.query_key_value", "dense": "self_attention.dense",
    "h4h": "mlp.dense_h_to_4h", "4hh": "mlp.dense_4h_to_h",
}


def merged_block(block, adapter: BlockAdapter):
    """Clone the block with LoRA merged into the dense weights."""
    import copy

    m = copy.deepcopy(block)
    with torch.no_grad():
        for key, (a, b, scale) in adapter.projections.items():
            _module_by_path(m, _KEY_TO_PATH[key]).weight += (b.float() @ a.float()) * scale
    return m


@pytest.mark.parametrize("model", ["test-llama", "test-falcon", "test-bloom"])
def test_block_adapter_matches_merged(tmp_path, model):
    cfg = load_model_config(model)
    block = get_model_block(cfg, 0)
    init_random_block_(block, cfg, 0)
    adapter_dir = make_adapter_dir(tmp_path, cfg, 1, model=model)
    ad = load_block_adapter(adapter_dir, 0, _FAMILY[model][0])
    expected = {"q", "down"} if model == "test-llama" else {"qkv", "4hh"}
    assert ad is not None and set(ad.projections) == expected
    add_adapter_to_block(block, ad)

    torch.manual_seed(1)
    x = torch.randn(2, 7, cfg.hidden_size)
    base = block(x)
    with using_adapter("test-adapter"):
        with_lora = block(x)
    ref = merged_block(block, ad)(x)
    assert not torch.allclose(with_lora, base, atol=1e-5)
    assert torch.allclose(with_lora, ref, atol=1e-5), (with_lora - ref).abs().max()


@pytest.mark.parametrize("model", ["test-llama", "test-falcon", "test-bloom"])
def test_block_adapter_with_cache(tmp_path, model):
    cfg = load_model_config(model)
    block = get_model_block(cfg, 0)
    init_random_block_(block, cfg, 0)
    adapter_dir = make_adapter_dir(tmp_path, cfg, 1, model=model)
    ad = load_block_adapter(adapter_dir, 0, _FAMILY[model][0])
    add_adapter_to_block(block, ad)

    torch.manual_seed(2)
    x = torch.randn(1, 6, cfg.hidden_size)
    with using_adapter("test-adapter"):
        full = block(x)
        ks, vs = block.kv_cache_shape(1, 8)
        k, v = torch.zeros(ks), torch.zeros(vs)
        parts = [block(x[:, :3], kv_cache=(k, v), prefix_length=0)]
        for t in range(3, 6):
            parts.append(block(x[:, t : t + 1], kv_cache=(k, v), prefix_length=t))
    assert torch.allclose(torch.cat(parts, 1), full, atol=1e-5)


def test_unknown_adapter_raises():
    cfg = load_model_config("test-llama")
    block = get_model_block(cfg, 0)
    init_random_block_(block, cfg, 0)
    x = torch.randn(1, 3, cfg.hidden_size)
    with using_adapter("missing"), pytest.raises(KeyError):
        block(x)


def test_swarm_serves_adapter(tmp_path):
    """End-to-end: client selects the adapter by name; output == merged model."""
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    cfg = load_model_config("test-llama")
    adapter_dir = make_adapter_dir(tmp_path, cfg, cfg.num_blocks)

    boot = DHT(host="127.0.0.1")
    server = Server(
        "test-llama", initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
        torch_dtype="float32", block_indices=f"0:{cfg.num_blocks}", dht_prefix="peft-llama",
        throughput=1.0, adapters=[adapter_dir],
    ).start()
    try:
        base = AutoDistributedModelForCausalLM.from_pretrained(
            "test-llama", initial_peers=[boot.listen_addr], dht_prefix="peft-llama",
            show_route=False, max_retries=1,
        )
        torch.manual_seed(3)
        ids = torch.randint(0, 128, (1, 5))
        with torch.no_grad():
            out_base = base(input_ids=ids).logits
        base.transformer.h.sequence_manager.config.active_adapter = "test-adapter"
        with torch.no_grad():
            out_lora = base(input_ids=ids).logits
        assert not torch.allclose(out_base, out_lora, atol=1e-5)

        # golden: local chain with merged weights
        blocks = []
        for i in range(cfg.num_blocks):
            blk = get_model_block(cfg, i)
            init_random_block_(blk, cfg, i)
            ad = load_block_adapter(adapter_dir, i, "model.layers")
            blocks.append(merged_block(blk, ad))
        h = base.transformer.embed_tokens(ids)
        with torch.no_grad():
            for blk in blocks:
                h = blk(h)
            h = base.transformer.norm(h)
            ref = base.lm_head(h)
        assert torch.allclose(out_lora, ref, atol=1e-4), (out_lora - ref).abs().max()
        base.transformer.h.sequence_manager.shutdown()
    finally:
        server.shutdown()
        boot.shutdown()
