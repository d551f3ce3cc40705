"""Speculative decoding: greedy output must be bit-identical to plain
generate() for any draft model (perfect or adversarial), with server-side
KV rollback via start_from_position (reference tests/test_speculative_generation.py)."""

import pytest
import torch

HF_CFG = dict(
    hidden_size=64, num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
    intermediate_size=128, vocab_size=128, max_position_embeddings=256, tie_word_embeddings=False,
)


@pytest.fixture(scope="module")
def swarm(tmp_path_factory):
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(0)
    cfg = transformers.LlamaConfig(**HF_CFG)
    hf_model = transformers.LlamaForCausalLM(cfg).eval()
    path = tmp_path_factory.mktemp("spec_ckpt")
    hf_model.save_pretrained(path, safe_serialization=True)

    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server

    boot = DHT(host="127.0.0.1")
    servers = [
        Server(str(path), initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
               torch_dtype="float32", block_indices=spec, dht_prefix="spec-llama",
               throughput=1.0).start()
        for spec in ("0:2", "2:4")
    ]
    yield boot, str(path), hf_model
    for s in servers:
        s.shutdown()
    boot.shutdown()


@pytest.fixture(scope="module")
def client_model(swarm):
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    boot, path, _ = swarm
    model = AutoDistributedModelForCausalLM.from_pretrained(
        path, initial_peers=[boot.listen_addr], dht_prefix="spec-llama",
        show_route=False, max_retries=1, min_backoff=0.2,
    )
    yield model
    model.transformer.h.sequence_manager.shutdown()


class _BadDraft:
    """Adversarial draft: always proposes token 7."""

    def __call__(self, input_ids=None, **kw):
        class Out:
            pass

        o = Out()
        logits = torch.full((input_ids.shape[0], input_ids.shape[1], 128), -10.0)
        logits[..., 7] = 10.0
        o.logits = logits
        return o


@pytest.mark.parametrize("draft_kind", ["perfect", "bad"])
def test_speculative_equals_greedy(client_model, swarm, draft_kind):
    from petals_amd.models.llama.speculative_model import DistributedLlamaForSpeculativeGeneration

    _, _, hf_model = swarm
    torch.manual_seed(7)
    ids = torch.randint(0, 128, (1, 6))
    ref = client_model.generate(ids, max_new_tokens=10, do_sample=False)

    draft = hf_model if draft_kind == "perfect" else _BadDraft()
    spec = DistributedLlamaForSpeculativeGeneration(client_model, draft)
    out = spec.generate(ids, max_new_tokens=10, speculative_tokens=3)
    assert torch.equal(out, ref), (out.tolist(), ref.tolist())
