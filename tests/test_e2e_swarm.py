"""End-to-end mini-swarm on CPU (BASELINE config #1 pattern): two fp32 servers
hosting disjoint spans + a thin client; generate() and training fwd/bwd are
exact-matched against the local HF model (golden-reference pattern, reference
tests/test_full_model.py)."""

import os
import tempfile

import pytest
import torch

HF_CFG = dict(
    hidden_size=64,
    num_hidden_layers=4,
    num_attention_heads=4,
    num_key_value_heads=2,
    intermediate_size=128,
    vocab_size=128,
    max_position_embeddings=256,
    tie_word_embeddings=False,
)


@pytest.fixture(scope="module")
def hf_checkpoint(tmp_path_factory):
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(0)
    cfg = transformers.LlamaConfig(**HF_CFG)
    model = transformers.LlamaForCausalLM(cfg).eval()
    path = tmp_path_factory.mktemp("llama_ckpt")
    model.save_pretrained(path, safe_serialization=True)
    return str(path), model


@pytest.fixture(scope="module")
def swarm(hf_checkpoint):
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server

    path, _ = hf_checkpoint
    bootstrap = DHT(host="127.0.0.1")
    servers = []
    # server1: blocks 0..2, server2: blocks 2..4, server3 overlaps 1..4
    for spec in ("0:2", "2:4", "1:4"):
        s = Server(
            path,
            initial_peers=[bootstrap.listen_addr],
            host="127.0.0.1",
            device="cpu",
            torch_dtype="float32",
            block_indices=spec,
            dht_prefix="test-llama-e2e",
            throughput=1.0,
            update_period=2.0,
        )
        s.start()
        servers.append(s)
    yield bootstrap, servers, path
    for s in servers:
        s.shutdown()
    bootstrap.shutdown()


@pytest.fixture(scope="module")
def client_model(swarm, hf_checkpoint):
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    bootstrap, _, path = swarm
    model = AutoDistributedModelForCausalLM.from_pretrained(
        path,
        initial_peers=[bootstrap.listen_addr],
        dht_prefix="test-llama-e2e",
        show_route=False,
        max_retries=2,
        min_backoff=0.2,
        request_timeout=30.0,
    )
    yield model
    model.transformer.h.sequence_manager.shutdown()


def test_forward_exact_match(client_model, hf_checkpoint):
    _, hf_model = hf_checkpoint
    torch.manual_seed(1)
    ids = torch.randint(0, 128, (2, 10))
    with torch.no_grad():
        ref = hf_model(ids).logits
        out = client_model(input_ids=ids).logits
    assert out.shape == ref.shape
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-3), (out - ref).abs().max()


def test_greedy_generate_matches_hf(client_model, hf_checkpoint):
    _, hf_model = hf_checkpoint
    torch.manual_seed(2)
    ids = torch.randint(0, 128, (1, 5))
    ref = hf_model.generate(ids, max_new_tokens=6, do_sample=False)
    out = client_model.generate(ids, max_new_tokens=6, do_sample=False)
    assert torch.equal(out, ref), (out, ref)


def test_batched_generate(client_model):
    torch.manual_seed(3)
    ids = torch.randint(0, 128, (3, 4))
    out = client_model.generate(ids, max_new_tokens=5, do_sample=False)
    assert out.shape == (3, 9)


def test_sampling_generate_deterministic(client_model):
    ids = torch.randint(0, 128, (1, 4))
    torch.manual_seed(0)
    out1 = client_model.generate(ids, max_new_tokens=5, do_sample=True, top_k=10)
    torch.manual_seed(0)
    out2 = client_model.generate(ids, max_new_tokens=5, do_sample=True, top_k=10)
    assert torch.equal(out1, out2)


def test_hf_generate_logits_processor_and_stopping(client_model, hf_checkpoint):
    """transformers.GenerationMixin delegation: logits processors and stopping
    criteria work against the swarm and match the local HF model exactly."""
    import transformers
    from transformers import LogitsProcessorList, StoppingCriteriaList
    from transformers.generation.logits_process import NoRepeatNGramLogitsProcessor
    from transformers.generation.stopping_criteria import MaxLengthCriteria

    _, hf_model = hf_checkpoint
    torch.manual_seed(11)
    ids = torch.randint(0, 128, (1, 5))
    processors = LogitsProcessorList([NoRepeatNGramLogitsProcessor(2)])
    criteria = StoppingCriteriaList([MaxLengthCriteria(max_length=11)])
    ref = hf_model.generate(
        ids, max_new_tokens=8, do_sample=False,
        logits_processor=LogitsProcessorList([NoRepeatNGramLogitsProcessor(2)]),
        stopping_criteria=StoppingCriteriaList([MaxLengthCriteria(max_length=11)]),
    )
    out = client_model.generate(
        ids, max_new_tokens=8, do_sample=False,
        logits_processor=processors, stopping_criteria=criteria,
    )
    assert torch.equal(out, ref), (out, ref)


def test_hf_generate_beam_search_matches_local(client_model, hf_checkpoint):
    """HF beam search via delegation: server-side KV reorder through hypo_ids."""
    _, hf_model = hf_checkpoint
    torch.manual_seed(12)
    ids = torch.randint(0, 128, (1, 5))
    ref = hf_model.generate(ids, max_new_tokens=6, num_beams=3, do_sample=False,
                            early_stopping=True)
    out = client_model.generate(ids, max_new_tokens=6, num_beams=3, do_sample=False,
                                early_stopping=True)
    assert torch.equal(out, ref), (out, ref)


def test_inference_matches_forward(client_model):
    """Token-by-token session logits == one-shot forward logits."""
    torch.manual_seed(4)
    ids = torch.randint(0, 128, (1, 8))
    with torch.no_grad():
        ref = client_model(input_ids=ids).logits
    outs = []
    with client_model.transformer.h.inference_session(max_length=16) as sess:
        with client_model.transformer.h.use_session(sess):
            with torch.no_grad():
                outs.append(client_model(input_ids=ids[:, :3]).logits)
                for t in range(3, 8):
                    outs.append(client_model(input_ids=ids[:, t : t + 1]).logits)
    step_logits = torch.cat(outs, dim=1)
    assert torch.allclose(step_logits, ref, atol=1e-4, rtol=1e-3), (step_logits - ref).abs().max()


def test_training_forward_backward(client_model, hf_checkpoint):
    """Remote fwd+bwd grads match local HF grads (prompt-tuning style: grads
    wrt inputs_embeds)."""
    _, hf_model = hf_checkpoint
    torch.manual_seed(5)
    ids = torch.randint(0, 128, (2, 6))

    embeds_ref = hf_model.get_input_embeddings()(ids).detach().requires_grad_(True)
    ref_out = hf_model(inputs_embeds=embeds_ref).logits
    ref_loss = ref_out.square().mean()
    ref_loss.backward()

    embeds = hf_model.get_input_embeddings()(ids).detach().requires_grad_(True)
    out = client_model(inputs_embeds=embeds).logits
    loss = out.square().mean()
    loss.backward()

    assert torch.allclose(out, ref_out, atol=1e-4, rtol=1e-3)
    assert torch.allclose(embeds.grad, embeds_ref.grad, atol=1e-4, rtol=1e-3), (
        (embeds.grad - embeds_ref.grad).abs().max()
    )


def test_auto_span_selection(hf_checkpoint):
    """Servers without pinned blocks pick the least-covered span (block_selection)."""
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server

    path, _ = hf_checkpoint
    boot = DHT(host="127.0.0.1")
    s1 = Server(path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
                torch_dtype="float32", num_blocks=2, dht_prefix="auto-span", throughput=1.0).start()
    import time as _t

    _t.sleep(0.5)
    s2 = Server(path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
                torch_dtype="float32", num_blocks=2, dht_prefix="auto-span", throughput=1.0).start()
    try:
        covered = sorted(
            set(range(s1.server_info.start_block, s1.server_info.end_block))
            | set(range(s2.server_info.start_block, s2.server_info.end_block))
        )
        assert covered == [0, 1, 2, 3], (s1.server_info, s2.server_info)
    finally:
        s1.shutdown()
        s2.shutdown()
        boot.shutdown()


def test_inference_failover_mid_session(hf_checkpoint):
    """Kill the server an open session is using; the session must recover by
    replaying its input history into a redundant server, with logits matching
    the uninterrupted reference (the petals failover semantics,
    reference client/inference_session.py history replay)."""
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    path, hf_model = hf_checkpoint
    boot = DHT(host="127.0.0.1")
    servers = []
    for _ in range(2):  # two full-coverage servers
        s = Server(
            path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
            torch_dtype="float32", block_indices="0:4", dht_prefix="test-failover",
            throughput=1.0, update_period=1.0,
        )
        servers.append(s.start())
    model = AutoDistributedModelForCausalLM.from_pretrained(
        path, initial_peers=[boot.listen_addr], dht_prefix="test-failover",
        show_route=False, max_retries=3, min_backoff=0.2, request_timeout=20.0,
    )
    try:
        torch.manual_seed(7)
        ids = torch.randint(0, 128, (1, 8))
        with torch.no_grad():
            ref = model(input_ids=ids).logits  # uninterrupted reference

        outs = []
        with model.transformer.h.inference_session(max_length=16) as sess:
            with model.transformer.h.use_session(sess):
                with torch.no_grad():
                    outs.append(model(input_ids=ids[:, :4]).logits)
                    # kill the server this session is attached to
                    used_peer = sess._sessions[0].span.peer_id
                    victim = next(s for s in servers if s.peer_id == used_peer)
                    victim.shutdown()
                    for t in range(4, 8):
                        outs.append(model(input_ids=ids[:, t : t + 1]).logits)
                survivor = next(s for s in servers if s.peer_id != used_peer)
                assert sess._sessions[0].span.peer_id == survivor.peer_id, "session must have failed over"
        step_logits = torch.cat(outs, dim=1)
        assert torch.allclose(step_logits, ref, atol=1e-4, rtol=1e-3), (step_logits - ref).abs().max()
    finally:
        model.transformer.h.sequence_manager.shutdown()
        for s in servers:
            s.shutdown()
        boot.shutdown()


def test_server_to_server_push_used(client_model, monkeypatch):
    """Steady-state multi-span decode must take the rpc_push fast path (server
    hands activations to the next server; the client only talks to the first
    and last spans) — and still match the one-shot forward logits."""
    from petals_amd.client import inference_session as isess

    calls = {"push": 0}
    orig = isess.InferenceSession._step_pushed

    def spy(self, *a, **k):
        calls["push"] += 1
        return orig(self, *a, **k)

    monkeypatch.setattr(isess.InferenceSession, "_step_pushed", spy)
    torch.manual_seed(9)
    ids = torch.randint(0, 128, (1, 6))
    with torch.no_grad():
        ref = client_model(input_ids=ids).logits
    outs = []
    with client_model.transformer.h.inference_session(max_length=12) as sess:
        with client_model.transformer.h.use_session(sess):
            with torch.no_grad():
                outs.append(client_model(input_ids=ids[:, :3]).logits)
                for t in range(3, 6):
                    outs.append(client_model(input_ids=ids[:, t : t + 1]).logits)
    assert calls["push"] >= 1, "server-to-server push was never used"
    step_logits = torch.cat(outs, dim=1)
    assert torch.allclose(step_logits, ref, atol=1e-4, rtol=1e-3), (step_logits - ref).abs().max()


def test_beam_search_matches_local_mirror(client_model, hf_checkpoint):
    """Remote beam search (beams = server-side batch rows, KV caches reordered
    via hypo_ids each step) vs a local mirror of the same algorithm that
    recomputes from scratch every step: identical sequences prove the
    server-side cache reorder is correct."""
    import torch.nn.functional as F

    _, hf_model = hf_checkpoint
    torch.manual_seed(3)
    ids = torch.randint(0, 128, (1, 5))
    num_beams, new_tokens = 2, 5
    out = client_model.generate(ids, max_new_tokens=new_tokens, num_beams=num_beams)

    expanded = ids.expand(num_beams, -1)
    with torch.no_grad():
        logits = hf_model(expanded).logits[:, -1, :].float()
    logprobs = F.log_softmax(logits[0:1], dim=-1)
    scores, next_tokens = logprobs.topk(num_beams, dim=-1)
    beam_scores = scores[0]
    sequences = torch.cat([expanded, next_tokens[0][:, None]], dim=1)
    for _ in range(new_tokens - 1):
        with torch.no_grad():
            logits = hf_model(sequences).logits[:, -1, :].float()  # full recompute, no KV cache
        logprobs = F.log_softmax(logits, dim=-1)
        total = beam_scores[:, None] + logprobs
        vocab = total.shape[-1]
        beam_scores, flat_idx = total.reshape(-1).topk(num_beams)
        beam_idx, token_idx = flat_idx // vocab, flat_idx % vocab
        sequences = torch.cat([sequences[beam_idx], token_idx[:, None]], dim=1)
    ref = sequences[beam_scores.argmax()][None]
    assert torch.equal(out, ref), (out, ref)


@pytest.mark.parametrize("family", ["bloom", "mixtral"])
def test_training_grads_via_inputs_embeds_nonllama(tmp_path, family):
    """Training through caller-provided inputs_embeds (the reference's prompt
    -tuning client mode) must match local HF for families whose model wrapper
    transforms the embedding stream — regression for the BLOOM embedding
    LayerNorm being skipped on the inputs_embeds path (applied only on the
    input_ids path: generate() matched while training grads were ~3e-2 off)."""
    transformers = pytest.importorskip("transformers")
    import os as _os

    torch.manual_seed(0)
    if family == "bloom":
        cfg = transformers.BloomConfig(hidden_size=64, n_head=4, n_layer=3, vocab_size=100)
        hf = transformers.BloomForCausalLM(cfg).eval()
    else:
        cfg = transformers.MixtralConfig(
            hidden_size=64, num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
            intermediate_size=128, vocab_size=100, num_local_experts=4, num_experts_per_tok=2,
        )
        hf = transformers.MixtralForCausalLM(cfg).eval()
    path = _os.path.join(str(tmp_path), "ckpt")
    hf.save_pretrained(path, safe_serialization=True)

    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    boot = DHT(host="127.0.0.1")
    server = Server(
        path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
        torch_dtype="float32", block_indices="0:3", dht_prefix=f"{family}-egrad",
        throughput=1.0,
    ).start()
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            path, initial_peers=[boot.listen_addr], dht_prefix=f"{family}-egrad",
            show_route=False, max_retries=1,
        )
        ids = torch.randint(0, 100, (1, 5))
        er = hf.get_input_embeddings()(ids).detach().requires_grad_(True)
        hf(inputs_embeds=er).logits.square().mean().backward()
        e = hf.get_input_embeddings()(ids).detach().requires_grad_(True)
        model(inputs_embeds=e).logits.square().mean().backward()
        assert torch.allclose(e.grad, er.grad, atol=1e-4, rtol=1e-3), (
            (e.grad - er.grad).abs().max()
        )
        model.transformer.h.sequence_manager.shutdown()
    finally:
        server.shutdown()
        boot.shutdown()


def test_chunked_prefill_matches_forward(tmp_path, hf_checkpoint):
    """A prefill longer than max_chunk_size_bytes allows is split into
    sequential chunks server-side (backend.inference_step); logits must
    equal the unchunked forward."""
    path, hf_model = hf_checkpoint
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    boot = DHT(host="127.0.0.1")
    # 512 B budget => 2-3 token chunks for hidden 64 / 4 heads / 14 tokens
    budget = 512
    server = Server(
        path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
        torch_dtype="float32", block_indices="0:4", dht_prefix="chunked-e2e",
        throughput=1.0, max_chunk_size_bytes=budget,
    ).start()
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            path, initial_peers=[boot.listen_addr], dht_prefix="chunked-e2e",
            show_route=False, max_retries=1,
        )
        torch.manual_seed(9)
        ids = torch.randint(0, 128, (1, 14))
        with torch.no_grad():
            ref = hf_model(input_ids=ids).logits
        # sanity: the budget really forces chunking for this shape
        # (backend formula: n_heads * batch * 4 * worst_case_length)
        assert budget // (4 * 1 * 4 * ids.shape[1]) < ids.shape[1]

        with model.transformer.h.inference_session(max_length=20) as sess:
            with model.transformer.h.use_session(sess):
                with torch.no_grad():
                    out = model(input_ids=ids).logits
        assert torch.allclose(out, ref, atol=1e-4, rtol=1e-3), (out - ref).abs().max()
        model.transformer.h.sequence_manager.shutdown()
    finally:
        server.shutdown()
        boot.shutdown()


def test_worker_thread_mode(swarm, monkeypatch):
    """PETALS_AMD_WORKER_THREAD=1 restores the background-thread client loop
    (default is the inline pump); generate must still work. Uses a FRESH
    client: the shared client_model may hold a resumable session from an
    earlier test with a different batch size."""
    from petals_amd.client.remote_worker import reset_worker
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    bootstrap, _, path = swarm
    monkeypatch.setenv("PETALS_AMD_WORKER_THREAD", "1")
    reset_worker()
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            path, initial_peers=[bootstrap.listen_addr], dht_prefix="test-llama-e2e",
            show_route=False, max_retries=2, min_backoff=0.2,
        )
        ids = torch.randint(0, 128, (1, 4))
        out = model.generate(ids, max_new_tokens=4, do_sample=False)
        assert out.shape == (1, 8)
        model.transformer.h.sequence_manager.shutdown()
    finally:
        monkeypatch.delenv("PETALS_AMD_WORKER_THREAD")
        reset_worker()


@pytest.mark.parametrize("family", ["llama", "bloom"])
def test_sequence_classification_e2e(tmp_path, family):
    """Distributed*ForSequenceClassification: forward through the swarm +
    backward into the classifier head (reference model.py parity)."""
    transformers = pytest.importorskip("transformers")
    import os as _os
    import time as _time

    torch.manual_seed(0)
    if family == "llama":
        cfg = transformers.LlamaConfig(
            hidden_size=64, num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
            intermediate_size=128, vocab_size=128, max_position_embeddings=256,
            tie_word_embeddings=False,
        )
        hf = transformers.LlamaForCausalLM(cfg)
    else:
        cfg = transformers.BloomConfig(hidden_size=64, n_head=4, n_layer=2, vocab_size=128)
        hf = transformers.BloomForCausalLM(cfg)
    path = _os.path.join(str(tmp_path), "ckpt")
    hf.eval().save_pretrained(path, safe_serialization=True)

    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedConfig

    boot = DHT(host="127.0.0.1")
    server = Server(
        path, initial_peers=[boot.listen_addr], host="127.0.0.1", device="cpu",
        torch_dtype="float32", block_indices="0:2", dht_prefix=f"cls-{family}",
        throughput=1.0,
    ).start()
    try:
        _time.sleep(1)
        if family == "llama":
            from petals_amd.models.llama.model import DistributedLlamaForSequenceClassification as CLS
        else:
            from petals_amd.models.bloom.model import DistributedBloomForSequenceClassification as CLS
        m = CLS.from_pretrained(
            path, initial_peers=[boot.listen_addr], dht_prefix=f"cls-{family}", num_labels=3,
            show_route=False, max_retries=3, min_backoff=0.3,
        )
        ids = torch.randint(0, 128, (2, 7))
        out = m(input_ids=ids)
        assert out.logits.shape == (2, 3)
        loss = torch.nn.functional.cross_entropy(out.logits, torch.tensor([0, 2]))
        loss.backward()
        grads = [p.grad for p in m.score.parameters() if p.grad is not None]
        assert grads and all(torch.isfinite(g).all() for g in grads)
        m.transformer.h.sequence_manager.shutdown()
    finally:
        server.shutdown()
        boot.shutdown()
