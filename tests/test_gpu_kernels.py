"""Numerics tests for the CDNA4 HIP kernel suite vs plain-PyTorch fp32
references (run on a real MI355X via `pytest -m gpu`)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


@pytest.fixture(scope="module")
def hip():
    from petals_amd import ops

    mod = ops._load_hip_ops()
    assert mod is not None, f"HIP extension must load on a GPU box: {ops._hip_import_error!r}"
    return mod


@requires_gpu
def test_rms_norm(hip):
    from petals_amd.ops import reference

    torch.manual_seed(0)
    x = torch.randn(5, 4096, device="cuda").to(torch.bfloat16)
    w = torch.randn(4096, device="cuda").to(torch.bfloat16)
    out = hip.rms_norm(x, w, 1e-5)
    ref = reference.rms_norm(x.float().cpu(), w.float().cpu(), 1e-5)
    assert torch.allclose(out.float().cpu(), ref, atol=2e-2, rtol=2e-2)

    out32 = hip.rms_norm_f32out(x, w, 1e-5)
    assert out32.dtype == torch.float32
    assert torch.allclose(out32.cpu(), ref, atol=2e-2, rtol=2e-2)


@requires_gpu
def test_swiglu(hip):
    torch.manual_seed(0)
    g = torch.randn(3, 1000, device="cuda").to(torch.bfloat16)
    u = torch.randn(3, 1000, device="cuda").to(torch.bfloat16)
    out = hip.swiglu(g, u)
    ref = torch.nn.functional.silu(g.float()) * u.float()
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2)


@requires_gpu
def test_apply_rope(hip):
    from petals_amd.ops import reference

    torch.manual_seed(0)
    b, qh, kh, s, hd = 2, 8, 2, 7, 128
    q = torch.randn(b, qh, s, hd, device="cuda").to(torch.bfloat16)
    k = torch.randn(b, kh, s, hd, device="cuda").to(torch.bfloat16)
    cos, sin = reference.build_rope_cache(hd, 64)
    pos = torch.arange(3, 3 + s).unsqueeze(0).expand(b, s).contiguous().cuda()
    q2, k2 = hip.apply_rope(q, k, cos.cuda(), sin.cuda(), pos)
    qr, kr = reference.apply_rope(q.float().cpu(), k.float().cpu(), cos, sin, pos.cpu())
    assert torch.allclose(q2.float().cpu(), qr, atol=2e-2, rtol=2e-2)
    assert torch.allclose(k2.float().cpu(), kr, atol=2e-2, rtol=2e-2)


@requires_gpu
@pytest.mark.parametrize("batch", [1, 2, 4])
@pytest.mark.parametrize("shape", [(512, 1024), (4096, 4096), (1000, 1536)])
def test_gemv_plain(hip, batch, shape):
    torch.manual_seed(1)
    in_dim, out_dim = shape
    wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.05).to(torch.bfloat16)
    x = torch.randn(batch, in_dim, device="cuda")
    ws = torch.empty(0, device="cuda")
    y = hip.gemv_bf16(wt, x, ws, None, 0)  # plain f32
    ref = x @ wt.float()
    assert torch.allclose(y, ref, atol=1e-2, rtol=1e-2), (y - ref).abs().max()


@requires_gpu
def test_gemv_residual(hip):
    torch.manual_seed(2)
    wt = (torch.randn(2048, 1024, device="cuda") * 0.05).to(torch.bfloat16)
    x = torch.randn(2, 2048, device="cuda")
    res = torch.randn(2, 1024, device="cuda").to(torch.bfloat16)
    ws = torch.empty(0, device="cuda")
    y = hip.gemv_bf16(wt, x, ws, res, 2)  # residual bf16
    ref = res.float() + x @ wt.float()
    assert torch.allclose(y.float(), ref, atol=5e-2, rtol=5e-2)


@requires_gpu
def test_gemv_swiglu(hip):
    torch.manual_seed(3)
    wt = (torch.randn(1024, 2 * 768, device="cuda") * 0.05).to(torch.bfloat16)
    x = torch.randn(1, 1024, device="cuda")
    ws = torch.empty(0, device="cuda")
    y = hip.gemv_bf16(wt, x, ws, None, 3)  # swiglu f32
    full = x @ wt.float()
    g, u = full[:, :768], full[:, 768:]
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(y, ref, atol=2e-2, rtol=2e-2)


@requires_gpu
@pytest.mark.parametrize("gq,kv_heads,kv_len", [(8, 8, 500), (1, 32, 77), (4, 8, 1)])
def test_attn_decode(hip, gq, kv_heads, kv_len):
    from petals_amd.ops import reference

    torch.manual_seed(4)
    b, hd, lmax = 2, 128, 640
    q = torch.randn(b, kv_heads * gq * hd, device="cuda")
    k_cache = torch.zeros(b, kv_heads, lmax, hd, device="cuda", dtype=torch.bfloat16)
    v_cache = torch.zeros_like(k_cache)
    k_cache[:, :, :kv_len] = (torch.randn(b, kv_heads, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    v_cache[:, :, :kv_len] = (torch.randn(b, kv_heads, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    kv_len_t = torch.tensor([kv_len], dtype=torch.int32, device="cuda")
    empty = torch.empty(0, device="cuda")
    out = hip.attn_decode_fused(q, k_cache, v_cache, kv_len_t, gq, 0, empty, empty, 1.0 / math.sqrt(hd))

    q_ref = q.view(b, kv_heads * gq, 1, hd).float().cpu()
    ref = reference.attention(
        q_ref,
        k_cache[:, :, :kv_len].float().cpu(),
        v_cache[:, :, :kv_len].float().cpu(),
        causal=False,
    )
    ref = ref.view(b, -1)
    assert torch.allclose(out.cpu(), ref, atol=2e-2, rtol=2e-2), (out.cpu() - ref).abs().max()


@requires_gpu
def test_llama_block_fast_decode_matches_cpu(hip):
    """Full fused decode path vs the fp32 CPU block with the same weights."""
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.server.from_pretrained import init_random_block_

    cfg = load_model_config("test-llama")
    cfg.hidden_size, cfg.num_attention_heads, cfg.num_key_value_heads, cfg.intermediate_size = 512, 4, 2, 1024
    blk_cpu = get_model_block(cfg, 0)
    init_random_block_(blk_cpu, cfg, 0)
    blk_cpu = blk_cpu.float().eval()

    blk_gpu = get_model_block(cfg, 0)
    blk_gpu.load_state_dict(blk_cpu.state_dict())
    blk_gpu = blk_gpu.to("cuda", torch.bfloat16).eval().optimize_for_inference()

    torch.manual_seed(5)
    B, S = 2, 9
    x = torch.randn(B, S, 512) * 0.5
    ks, vs = blk_cpu.kv_cache_shape(B, 32)
    kc, vc = torch.zeros(ks), torch.zeros(vs)
    kg = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
    vg = torch.zeros(vs, device="cuda", dtype=torch.bfloat16)

    # prefill 6 tokens (fast prefill path), then decode 3 (fused kernels)
    y_cpu, y_gpu = [], []
    y_cpu.append(blk_cpu(x[:, :6], kv_cache=(kc, vc), prefix_length=0))
    y_gpu.append(blk_gpu(x[:, :6].cuda().bfloat16(), kv_cache=(kg, vg), prefix_length=0))
    for t in range(6, S):
        y_cpu.append(blk_cpu(x[:, t : t + 1], kv_cache=(kc, vc), prefix_length=t))
        y_gpu.append(blk_gpu(x[:, t : t + 1].cuda().bfloat16(), kv_cache=(kg, vg), prefix_length=t))
    ref = torch.cat(y_cpu, 1)
    out = torch.cat([y.float().cpu() for y in y_gpu], 1)
    assert torch.allclose(out, ref, atol=0.05, rtol=0.05), (out - ref).abs().max()


@requires_gpu
def test_gpu_training_backward_grads(tmp_path):
    """Client training fwd+bwd through a GPU server (bf16 fused autograd
    path): input grads must track a local fp32 CPU swarm's within bf16
    tolerance (cosine > 0.99). An HF checkpoint on disk pins identical
    weights on both servers (the preset random init differs per device)."""
    transformers = pytest.importorskip("transformers")
    import os as _os

    torch.manual_seed(0)
    cfg = transformers.LlamaConfig(
        hidden_size=64, num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        intermediate_size=128, vocab_size=128, max_position_embeddings=256,
        tie_word_embeddings=False,
    )
    path = _os.path.join(str(tmp_path), "ckpt")
    transformers.LlamaForCausalLM(cfg).eval().save_pretrained(path, safe_serialization=True)

    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    def run(device, dtype, prefix):
        boot = DHT(host="127.0.0.1")
        server = Server(
            path, initial_peers=[boot.listen_addr], host="127.0.0.1",
            device=device, torch_dtype=dtype, block_indices="0:4",
            dht_prefix=prefix, throughput=1.0,
        ).start()
        try:
            model = AutoDistributedModelForCausalLM.from_pretrained(
                path, initial_peers=[boot.listen_addr], dht_prefix=prefix,
                show_route=False, max_retries=1,
            )
            torch.manual_seed(8)
            ids = torch.randint(0, 128, (1, 6))
            e = model.transformer.embed_tokens(ids).detach().float().requires_grad_(True)
            model(inputs_embeds=e).logits.float().square().mean().backward()
            g = e.grad.clone()
            model.transformer.h.sequence_manager.shutdown()
            return g
        finally:
            server.shutdown()
            boot.shutdown()

    g_gpu = run("cuda", "bfloat16", "gpu-train")
    g_cpu = run("cpu", "float32", "cpu-train")
    assert torch.isfinite(g_gpu).all()
    cos = torch.nn.functional.cosine_similarity(g_gpu.flatten(), g_cpu.flatten(), dim=0)
    assert cos > 0.99, cos


@requires_gpu
def test_gpu_two_span_push_chain():
    """Two GPU servers (spans 0:2 and 2:4) in one process: decode steps chain
    server-to-server over rpc_push with per-span hipGraphs; generate must
    produce the same ids as a single-server swarm."""
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    boot = DHT(host="127.0.0.1")
    servers = [
        Server(
            "test-llama", initial_peers=[boot.listen_addr], host="127.0.0.1",
            device="cuda", torch_dtype="bfloat16", block_indices=spec,
            dht_prefix="gpu-2span", throughput=1.0,
        ).start()
        for spec in ("0:2", "2:4")
    ]
    boot_single = DHT(host="127.0.0.1")
    single = Server(
        "test-llama", initial_peers=[boot_single.listen_addr], host="127.0.0.1",
        device="cuda", torch_dtype="bfloat16", block_indices="0:4",
        dht_prefix="gpu-1span", throughput=1.0,
    ).start()
    try:
        torch.manual_seed(2)
        ids = torch.randint(0, 128, (1, 5))
        outs = {}
        for prefix, peers in (("gpu-2span", boot), ("gpu-1span", boot_single)):
            model = AutoDistributedModelForCausalLM.from_pretrained(
                "test-llama", initial_peers=[peers.listen_addr], dht_prefix=prefix,
                show_route=False, max_retries=1,
            )
            outs[prefix] = model.generate(ids, max_new_tokens=8, do_sample=False)
            model.transformer.h.sequence_manager.shutdown()
        assert torch.equal(outs["gpu-2span"], outs["gpu-1span"]), outs
    finally:
        for s in servers:
            s.shutdown()
        single.shutdown()
        boot.shutdown()
        boot_single.shutdown()


@requires_gpu
def test_gpu_session_rollback_start_from_position():
    """Speculative-decoding rollback against a GPU server with span graphs:
    rewinding the session position and re-stepping must give the same logits
    as a fresh session fed the same prefix (KV overwritten in place)."""
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    boot = DHT(host="127.0.0.1")
    server = Server(
        "test-llama", initial_peers=[boot.listen_addr], host="127.0.0.1",
        device="cuda", torch_dtype="bfloat16", block_indices="0:4",
        dht_prefix="gpu-rb", throughput=1.0,
    ).start()
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            "test-llama", initial_peers=[boot.listen_addr], dht_prefix="gpu-rb",
            show_route=False, max_retries=1,
        )
        torch.manual_seed(4)
        ids = torch.randint(0, 128, (1, 6))
        extra = torch.randint(0, 128, (1, 3))
        with model.transformer.h.inference_session(max_length=16) as sess:
            with model.transformer.h.use_session(sess):
                with torch.no_grad():
                    model(input_ids=ids)           # position 6
                    model(input_ids=extra)         # position 9 (to be discarded)
                sess.position = 6                  # speculative rollback
                with torch.no_grad():
                    logits_rb = model(input_ids=extra[:, :1]).logits
        with model.transformer.h.inference_session(max_length=16) as sess2:
            with model.transformer.h.use_session(sess2):
                with torch.no_grad():
                    model(input_ids=ids)
                    logits_fresh = model(input_ids=extra[:, :1]).logits
        assert torch.allclose(logits_rb, logits_fresh, atol=2e-2, rtol=2e-2), (
            (logits_rb - logits_fresh).abs().max())
        model.transformer.h.sequence_manager.shutdown()
    finally:
        server.shutdown()
        boot.shutdown()


@requires_gpu
@pytest.mark.parametrize("preset", ["test-llama", "test-mixtral", "test-bloom-hd64", "test-falcon-hd64"])
def test_gpu_server_e2e_generate(preset):
    """Tiny swarm per family: 1 GPU server (bf16, fused path + span graphs)
    serving a real client generate()."""
    from petals_amd.dht.node import DHT
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    boot = DHT(host="127.0.0.1")
    server = Server(
        preset,
        initial_peers=[boot.listen_addr],
        host="127.0.0.1",
        device="cuda",
        torch_dtype="bfloat16",
        block_indices="0:4",
        dht_prefix=f"gpu-e2e-{preset}",
        throughput=1.0,
    ).start()
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            preset, initial_peers=[boot.listen_addr], dht_prefix=f"gpu-e2e-{preset}",
            show_route=False, max_retries=1,
        )
        ids = torch.randint(0, 128, (1, 5))
        out = model.generate(ids, max_new_tokens=8, do_sample=False)
        assert out.shape == (1, 13)
        model.transformer.h.sequence_manager.shutdown()
    finally:
        server.shutdown()
        boot.shutdown()


@requires_gpu
def test_mixtral_block_fast_decode_matches_cpu(hip):
    """Mixtral fused path (attention kernels + routed expert GEMVs) vs CPU fp32."""
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.server.from_pretrained import init_random_block_

    cfg = load_model_config("test-mixtral")
    cfg.hidden_size, cfg.num_attention_heads, cfg.num_key_value_heads, cfg.intermediate_size = 512, 4, 2, 1024
    blk_cpu = get_model_block(cfg, 0)
    init_random_block_(blk_cpu, cfg, 0)
    blk_cpu = blk_cpu.float().eval()

    blk_gpu = get_model_block(cfg, 0)
    blk_gpu.load_state_dict(blk_cpu.state_dict())
    blk_gpu = blk_gpu.to("cuda", torch.bfloat16).eval().optimize_for_inference()
    assert blk_gpu._fast is not None

    torch.manual_seed(6)
    B, S = 1, 8
    x = torch.randn(B, S, 512) * 0.5
    ks, vs = blk_cpu.kv_cache_shape(B, 16)
    kc, vc = torch.zeros(ks), torch.zeros(vs)
    kg = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
    vg = torch.zeros(vs, device="cuda", dtype=torch.bfloat16)

    y_cpu = [blk_cpu(x[:, :5], kv_cache=(kc, vc), prefix_length=0)]
    y_gpu = [blk_gpu(x[:, :5].cuda().bfloat16(), kv_cache=(kg, vg), prefix_length=0)]
    for t in range(5, S):
        y_cpu.append(blk_cpu(x[:, t : t + 1], kv_cache=(kc, vc), prefix_length=t))
        y_gpu.append(blk_gpu(x[:, t : t + 1].cuda().bfloat16(), kv_cache=(kg, vg), prefix_length=t))
    ref = torch.cat(y_cpu, 1)
    out = torch.cat([y.float().cpu() for y in y_gpu], 1)
    # bf16 routing can pick different experts on near-ties; require close overall
    assert torch.allclose(out, ref, atol=0.08, rtol=0.08), (out - ref).abs().max()


@requires_gpu
@pytest.mark.parametrize("quant", ["none", "nf4"])
def test_moe_grouped_gemm_matches_dense(hip, quant):
    """Prefill grouped MFMA GEMM (expert-sorted padded tiles, NF4 dequant
    fused into the LDS B staging) vs per-expert dense matmuls."""
    from petals_amd.ops.fused_moe import _StackedExperts, sort_pairs_by_expert

    torch.manual_seed(11)
    E, IN, OUT, T, K = 8, 512, 1152, 96, 2  # IN % 64 == 0, OUT % 128 == 0
    wts = [(torch.randn(IN, OUT, device="cuda") * 0.05).to(torch.bfloat16) for _ in range(E)]
    st = _StackedExperts(wts, hip, quant)
    assert st.gemm_ok
    x = (torch.randn(T, IN, device="cuda") * 0.5).to(torch.bfloat16)
    sel = torch.randint(0, E, (T, K), device="cuda")
    # make one expert empty and one over-full to exercise segment padding
    sel[:40, 0] = 3
    sel[sel == 7] = 1

    sorted_pairs, tile_expert = sort_pairs_by_expert(sel, E)
    out = st.moe_gemm(x, sorted_pairs, tile_expert, K, T * K)
    assert out.shape == (T * K, OUT)

    flat = sel.reshape(-1)
    for pair in range(0, T * K, 7):
        e = int(flat[pair])
        ref = (x[pair // K].float() @ st.dense(e).float()).to(torch.bfloat16)
        got = out[pair]
        assert torch.allclose(got.float(), ref.float(), atol=0.02, rtol=0.02), (
            pair, e, (got.float() - ref.float()).abs().max())


@requires_gpu
@pytest.mark.parametrize("quant", ["none", "nf4"])
def test_mixtral_decode_graph_capture(hip, quant):
    """Device-routed MoE decode is hipGraph-safe: capture one decode step,
    replay over new tokens, match the eager fused path exactly."""
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.ops.fused_decode import DecodeContext
    from petals_amd.server.from_pretrained import init_random_block_
    from petals_amd.utils.graphs import GraphedCallable

    cfg = load_model_config("test-mixtral")
    cfg.hidden_size, cfg.num_attention_heads, cfg.num_key_value_heads, cfg.intermediate_size = 512, 4, 2, 1024
    blk = get_model_block(cfg, 0)
    init_random_block_(blk, cfg, 0)
    blk = blk.to("cuda", torch.bfloat16).eval().optimize_for_inference(quant=quant)
    assert blk._fast is not None and blk._fast.graph_safe, "device-routed MoE must be graph-safe"

    blk2 = get_model_block(cfg, 0)
    init_random_block_(blk2, cfg, 0)
    blk2 = blk2.to("cuda", torch.bfloat16).eval().optimize_for_inference(quant=quant)

    torch.manual_seed(7)
    ks, vs = blk.kv_cache_shape(1, 16)
    kg = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
    vg = torch.zeros(vs, device="cuda", dtype=torch.bfloat16)
    kg2, vg2 = kg.clone(), vg.clone()
    xs = [torch.randn(1, 1, 512, device="cuda", dtype=torch.bfloat16) * 0.5 for _ in range(4)]

    ctx = DecodeContext(torch.device("cuda"))
    ctx.set_position(0)
    h_in = torch.empty_like(xs[0])

    def step():
        return blk(h_in, kv_cache=(kg, vg), ctx=ctx)

    g = GraphedCallable(step, [])
    outs_graph, outs_eager = [], []
    for t, x in enumerate(xs):
        ctx.set_position(t)
        h_in.copy_(x)
        outs_graph.append(g.replay().clone())
        outs_eager.append(blk2(x, kv_cache=(kg2, vg2), prefix_length=t))
    for og, oe in zip(outs_graph, outs_eager):
        assert torch.allclose(og.float(), oe.float(), atol=1e-3, rtol=1e-3), (og - oe).abs().max()


@requires_gpu
@pytest.mark.parametrize("case", [
    dict(b=2, qh=8, kvh=2, s=67, hd=128, off=0, causal=True),
    dict(b=1, qh=4, kvh=4, s=200, hd=128, off=0, causal=True),
    dict(b=1, qh=8, kvh=8, s=33, hd=64, off=50, causal=True),   # chunk over a cache prefix
    dict(b=2, qh=4, kvh=1, s=64, hd=128, off=0, causal=False),
])
def test_attn_prefill_mfma(hip, case):
    """MFMA flash prefill vs fp32-softmax reference."""
    from petals_amd.ops import reference

    torch.manual_seed(9)
    b, qh, kvh, s, hd, off = case["b"], case["qh"], case["kvh"], case["s"], case["hd"], case["off"]
    kv_len = off + s
    lmax = kv_len + 16
    q = (torch.randn(b, qh, s, hd, device="cuda") * 0.5).to(torch.bfloat16)
    k = torch.zeros(b, kvh, lmax, hd, device="cuda", dtype=torch.bfloat16)
    v = torch.zeros_like(k)
    k[:, :, :kv_len] = (torch.randn(b, kvh, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    v[:, :, :kv_len] = (torch.randn(b, kvh, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    import math
    out = hip.attn_prefill_fused(q, k, v, kv_len, off, 1.0 / math.sqrt(hd), case["causal"])
    ref = reference.attention(
        q.float().cpu(), k[:, :, :kv_len].float().cpu(), v[:, :, :kv_len].float().cpu(),
        causal=case["causal"], kv_offset=off,
    )
    err = (out.float().cpu() - ref).abs().max()
    assert torch.allclose(out.float().cpu(), ref, atol=3e-2, rtol=3e-2), err


@requires_gpu
def test_gpu_speculative_and_lora_e2e(tmp_path):
    """GPU server with the fused fast path + span graphs: speculative rollback
    (start_from_position) and a LoRA adapter, end to end."""
    import json

    import torch as T
    from safetensors.torch import save_file

    from petals_amd.dht.node import DHT
    from petals_amd.models.config_base import load_model_config
    from petals_amd.models.llama.speculative_model import DistributedLlamaForSpeculativeGeneration
    from petals_amd.server.server import Server
    from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

    cfg = load_model_config("test-llama-hd128")
    # synthetic LoRA adapter for every block
    ad_dir = tmp_path / "gpu-adapter"
    ad_dir.mkdir()
    T.manual_seed(11)
    tensors = {}
    for i in range(cfg.num_blocks):
        tensors[f"base_model.model.model.layers.{i}.self_attn.q_proj.lora_A.weight"] = T.randn(4, 512) * 0.03
        tensors[f"base_model.model.model.layers.{i}.self_attn.q_proj.lora_B.weight"] = T.randn(512, 4) * 0.03
    with open(ad_dir / "adapter_config.json", "w") as f:
        json.dump({"r": 4, "lora_alpha": 8}, f)
    save_file(tensors, str(ad_dir / "adapter_model.safetensors"))

    boot = DHT(host="127.0.0.1")
    server = Server(
        "test-llama-hd128", initial_peers=[boot.listen_addr], host="127.0.0.1", device="cuda",
        torch_dtype="bfloat16", block_indices="0:4", dht_prefix="gpu-spec", throughput=1.0,
        adapters=[str(ad_dir)],
    ).start()
    try:
        model = AutoDistributedModelForCausalLM.from_pretrained(
            "test-llama-hd128", initial_peers=[boot.listen_addr], dht_prefix="gpu-spec",
            show_route=False, max_retries=1,
        )
        assert server.backends and all(b.block._fast is not None for b in server.backends.values())
        ids = T.randint(0, 128, (1, 6))
        ref = model.generate(ids, max_new_tokens=10, do_sample=False)

        class BadDraft:
            def __call__(self, input_ids=None, **kw):
                class Out: ...
                o = Out()
                logits = T.full((input_ids.shape[0], input_ids.shape[1], 128), -10.0)
                logits[..., 5] = 10.0
                o.logits = logits
                return o

        spec = DistributedLlamaForSpeculativeGeneration(model, BadDraft())
        out = spec.generate(ids, max_new_tokens=10, speculative_tokens=3)
        assert T.equal(out, ref), (out.tolist(), ref.tolist())

        # adapter changes the output
        model.transformer.h.sequence_manager.config.active_adapter = "gpu-adapter"
        out_lora = model.generate(ids, max_new_tokens=10, do_sample=False)
        assert out_lora.shape == ref.shape
        model.transformer.h.sequence_manager.shutdown()
    finally:
        server.shutdown()
        boot.shutdown()


@requires_gpu
def test_layer_norm(hip):
    torch.manual_seed(3)
    x = (torch.randn(5, 1024, device="cuda") * 2 + 0.3).to(torch.bfloat16)
    w = torch.randn(1024, device="cuda").to(torch.bfloat16)
    b = torch.randn(1024, device="cuda").to(torch.bfloat16)
    ref = torch.nn.functional.layer_norm(x.float().cpu(), (1024,), w.float().cpu(), b.float().cpu(), 1e-5)
    out = hip.layer_norm(x, w, b, 1e-5)
    assert torch.allclose(out.float().cpu(), ref, atol=3e-2, rtol=3e-2), (out.float().cpu() - ref).abs().max()
    out32 = hip.layer_norm_f32out(x, w, b, 1e-5)
    assert out32.dtype == torch.float32
    assert torch.allclose(out32.cpu(), ref, atol=3e-2, rtol=3e-2)


@requires_gpu
def test_gemv_bias_and_gelu(hip):
    torch.manual_seed(4)
    in_dim, out_dim, B = 512, 768, 2
    wt = (torch.randn(in_dim, out_dim, device="cuda") * 0.05).to(torch.bfloat16)
    x = torch.randn(B, in_dim, device="cuda")
    bias = (torch.randn(out_dim, device="cuda") * 0.5).to(torch.bfloat16)
    ws = torch.empty(0, device="cuda")
    ref_lin = x.cpu() @ wt.float().cpu() + bias.float().cpu()
    # plain f32 + bias
    out = hip.gemv_bf16(wt, x, ws, None, 0, 0, bias)
    assert torch.allclose(out.cpu(), ref_lin, atol=2e-2, rtol=2e-2)
    # gelu(tanh) epilogue + bias
    out_g = hip.gemv_bf16(wt, x, ws, None, 4, 0, bias)
    ref_g = torch.nn.functional.gelu(ref_lin, approximate="tanh")
    assert torch.allclose(out_g.cpu(), ref_g, atol=2e-2, rtol=2e-2), (out_g.cpu() - ref_g).abs().max()
    # nf4 path with bias
    packed, absmax = hip.nf4_quantize(wt)
    wt_dq = hip.nf4_dequantize(packed, absmax)
    ref_nf4 = torch.nn.functional.gelu(x.cpu() @ wt_dq.float().cpu() + bias.float().cpu(), approximate="tanh")
    out_n = hip.gemv_nf4(packed, absmax, x, ws, None, 4, 0, bias)
    assert torch.allclose(out_n.cpu(), ref_nf4, atol=2e-2, rtol=2e-2)


@requires_gpu
def test_attn_decode_alibi(hip):
    """ALiBi decode attention vs the reference with an explicit bias matrix."""
    from petals_amd.ops import reference

    torch.manual_seed(6)
    B, H, hd, kv_len, lmax = 2, 8, 64, 73, 96
    q = torch.randn(B, H, 1, hd, device="cuda").float()
    k_cache = torch.zeros(B, H, lmax, hd, device="cuda", dtype=torch.bfloat16)
    v_cache = torch.zeros_like(k_cache)
    k_cache[:, :, :kv_len] = (torch.randn(B, H, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    v_cache[:, :, :kv_len] = (torch.randn(B, H, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    slopes = reference.build_alibi_slopes(H).to("cuda", torch.float32)
    kvl = torch.tensor([kv_len], dtype=torch.int32, device="cuda")
    empty = torch.empty(0, dtype=torch.float32, device="cuda")
    scale = 1.0 / math.sqrt(hd)
    out = hip.attn_decode_fused(
        q.view(B, H * hd).contiguous(), k_cache, v_cache, kvl, 1, 0, empty, empty, scale, slopes
    )
    k_pos = torch.arange(kv_len, dtype=torch.float32)
    bias = (slopes.cpu()[:, None, None] * k_pos[None, None, :]).unsqueeze(0)
    ref = reference.attention(
        q.float().cpu(), k_cache[:, :, :kv_len].float().cpu(), v_cache[:, :, :kv_len].float().cpu(),
        causal=False, attn_bias=bias,
    ).view(B, H * hd)
    assert torch.allclose(out.float().cpu(), ref, atol=3e-2, rtol=3e-2), (out.float().cpu() - ref).abs().max()


@requires_gpu
def test_kv_cache_write(hip):
    torch.manual_seed(7)
    B, qh, kh, hd, lmax, pos = 2, 4, 4, 64, 32, 11
    qkv = torch.randn(B, (qh + 2 * kh) * hd, device="cuda")
    k_cache = torch.zeros(B, kh, lmax, hd, device="cuda", dtype=torch.bfloat16)
    v_cache = torch.zeros_like(k_cache)
    p = torch.tensor([pos], dtype=torch.int32, device="cuda")
    hip.kv_cache_write(qkv, p, k_cache, v_cache, qh, kh)
    k_ref = qkv[:, qh * hd : (qh + kh) * hd].view(B, kh, hd)
    v_ref = qkv[:, (qh + kh) * hd :].view(B, kh, hd)
    assert torch.allclose(k_cache[:, :, pos].float(), k_ref, atol=1e-2, rtol=1e-2)
    assert torch.allclose(v_cache[:, :, pos].float(), v_ref, atol=1e-2, rtol=1e-2)
    assert k_cache[:, :, pos + 1].abs().max() == 0 and k_cache[:, :, pos - 1].abs().max() == 0


def _block_fused_vs_cpu(preset: str, hidden: int, adapter: bool = False):
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.server.from_pretrained import init_random_block_
    from petals_amd.utils.peft import add_adapter_to_block, using_adapter

    cfg = load_model_config(preset)
    blk_cpu = get_model_block(cfg, 0)
    init_random_block_(blk_cpu, cfg, 0)
    blk_cpu = blk_cpu.float().eval()

    blk_gpu = get_model_block(cfg, 0)
    blk_gpu.load_state_dict(blk_cpu.state_dict())
    blk_gpu = blk_gpu.to("cuda", torch.bfloat16).eval().optimize_for_inference()
    assert blk_gpu._fast is not None, f"{preset} must take the fused path"

    ctx_mgr = None
    if adapter:
        # synthetic LoRA on qkv + the MLP down projection: the fused decode
        # path must match the CPU eager adapter application
        from petals_amd.utils.peft import BlockAdapter

        torch.manual_seed(17)
        qkv_lin = (blk_cpu.self_attention.query_key_value if hasattr(blk_cpu, "self_attention")
                   else blk_cpu.self_attn.q_proj)
        down_lin = (blk_cpu.mlp.dense_4h_to_h if hasattr(blk_cpu.mlp, "dense_4h_to_h")
                    else blk_cpu.mlp.down_proj)
        qkv_key = "qkv" if hasattr(blk_cpu, "self_attention") else "q"
        down_key = "4hh" if hasattr(blk_cpu.mlp, "dense_4h_to_h") else "down"
        projections = {
            qkv_key: (torch.randn(4, qkv_lin.in_features) * 0.05,
                      torch.randn(qkv_lin.out_features, 4) * 0.05, 2.0),
            down_key: (torch.randn(4, down_lin.in_features) * 0.05,
                       torch.randn(down_lin.out_features, 4) * 0.05, 2.0),
        }
        for blk in (blk_cpu, blk_gpu):
            ad = BlockAdapter(name="t", projections={k: (a.clone(), b.clone(), s)
                                                     for k, (a, b, s) in projections.items()})
            add_adapter_to_block(blk, ad)
        ctx_mgr = using_adapter("t")

    import contextlib

    with (ctx_mgr or contextlib.nullcontext()):
        torch.manual_seed(5)
        B, S = 2, 9
        x = torch.randn(B, S, hidden) * 0.5
        ks, vs = blk_cpu.kv_cache_shape(B, 32)
        kc, vc = torch.zeros(ks), torch.zeros(vs)
        kg = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
        vg = torch.zeros(vs, device="cuda", dtype=torch.bfloat16)

        y_cpu = [blk_cpu(x[:, :6], kv_cache=(kc, vc), prefix_length=0)]
        y_gpu = [blk_gpu(x[:, :6].cuda().bfloat16(), kv_cache=(kg, vg), prefix_length=0)]
        for t in range(6, S):
            y_cpu.append(blk_cpu(x[:, t : t + 1], kv_cache=(kc, vc), prefix_length=t))
            y_gpu.append(blk_gpu(x[:, t : t + 1].cuda().bfloat16(), kv_cache=(kg, vg), prefix_length=t))
    ref = torch.cat(y_cpu, 1)
    out = torch.cat([y.float().cpu() for y in y_gpu], 1)
    assert torch.allclose(out, ref, atol=0.05, rtol=0.05), (out - ref).abs().max()


@requires_gpu
def test_bloom_block_fast_decode_matches_cpu(hip):
    """BLOOM fused path (LayerNorm + ALiBi + GELU biases) vs fp32 CPU block."""
    _block_fused_vs_cpu("test-bloom-hd64", 256)


@requires_gpu
def test_folded_norm_decode_matches_unfolded(hip, monkeypatch):
    """PETALS_AMD_FOLD_NORM: RMSNorm weights folded into the NF4 weights +
    inv_rms from producer-side sumsq partials must match the unfolded fused
    path, including the ctx (norm_parts) hand-off between chained blocks."""
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.ops.fused_decode import DecodeContext
    from petals_amd.server.from_pretrained import init_random_block_

    cfg = load_model_config("test-llama-hd128")

    def build(fold):
        monkeypatch.setenv("PETALS_AMD_FOLD_NORM", "1" if fold else "0")
        blks = []
        for i in range(2):
            blk = get_model_block(cfg, i)
            init_random_block_(blk, cfg, i)
            blk = blk.to("cuda", torch.bfloat16).eval().optimize_for_inference(quant="nf4")
            assert blk._fast.fold_norm == fold
            blks.append(blk)
        return blks

    ref_blks = build(False)
    fold_blks = build(True)
    torch.manual_seed(13)
    B, T = 2, 4
    xs = [torch.randn(B, 1, cfg.hidden_size, device="cuda", dtype=torch.bfloat16) * 0.5 for _ in range(T)]

    def run(blks, use_ctx):
        ks, vs = blks[0].kv_cache_shape(B, 16)
        caches = [(torch.zeros(ks, device="cuda", dtype=torch.bfloat16),
                   torch.zeros(vs, device="cuda", dtype=torch.bfloat16)) for _ in blks]
        ctx = DecodeContext(torch.device("cuda")) if use_ctx else None
        outs = []
        for t, x in enumerate(xs):
            if ctx is not None:
                ctx.set_position(t)
                ctx.norm_parts = None  # span loop resets each pass
            h = x
            for blk, (k, v) in zip(blks, caches):
                h = blk(h, kv_cache=(k, v), prefix_length=t, ctx=ctx)
            outs.append(h)
        return outs

    ref = run(ref_blks, use_ctx=False)
    got_plain = run(fold_blks, use_ctx=False)   # per-block sumsq_rows
    got_ctx = run(fold_blks, use_ctx=True)      # producer-side parts hand-off
    for t in range(T):
        # folded weights quantize on a slightly different NF4 grid -> small tol
        assert torch.allclose(got_plain[t].float(), ref[t].float(), atol=0.06, rtol=0.06), (
            t, (got_plain[t].float() - ref[t].float()).abs().max())
        # parts-vs-sumsq_rows summation order differs by ULPs; bf16 rounding
        # can flip on a few elements
        assert torch.allclose(got_ctx[t].float(), got_plain[t].float(), atol=0.02, rtol=0.02), (
            t, (got_ctx[t].float() - got_plain[t].float()).abs().max())

    # prefill path with folded weights (scale-only norm) must also match
    xp = torch.randn(1, 6, cfg.hidden_size, device="cuda", dtype=torch.bfloat16) * 0.5
    pr = ref_blks[0](xp)
    pf = fold_blks[0](xp)
    assert torch.allclose(pf.float(), pr.float(), atol=0.06, rtol=0.06), (
        (pf.float() - pr.float()).abs().max())


@requires_gpu
def test_nf4_block_decode_batch_beyond_kernel_cap(hip):
    """Batch 6 NF4 decode: the gemv kernel caps at BATCH=4, so the block
    splits into sub-batches (decode_step_auto) instead of falling back to
    the dense prefill path; numerics must match per-row batch-1 decode."""
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.server.from_pretrained import init_random_block_

    cfg = load_model_config("test-llama-hd128")
    blk = get_model_block(cfg, 0)
    init_random_block_(blk, cfg, 0)
    blk = blk.to("cuda", torch.bfloat16).eval().optimize_for_inference(quant="nf4")

    torch.manual_seed(21)
    B, T = 6, 3
    xs = [torch.randn(B, 1, cfg.hidden_size, device="cuda", dtype=torch.bfloat16) * 0.5 for _ in range(T)]
    ks, vs = blk.kv_cache_shape(B, 16)
    k6 = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
    v6 = torch.zeros_like(k6)
    out6 = [blk(x, kv_cache=(k6, v6), prefix_length=t) for t, x in enumerate(xs)]

    ks1, vs1 = blk.kv_cache_shape(1, 16)
    for b in range(B):
        k1 = torch.zeros(ks1, device="cuda", dtype=torch.bfloat16)
        v1 = torch.zeros_like(k1)
        for t, x in enumerate(xs):
            ref = blk(x[b : b + 1], kv_cache=(k1, v1), prefix_length=t)
            got = out6[t][b : b + 1]
            assert torch.allclose(got.float(), ref.float(), atol=0.02, rtol=0.02), (
                b, t, (got.float() - ref.float()).abs().max())


@requires_gpu
@pytest.mark.parametrize("preset,hidden", [("test-falcon-hd64", 256), ("test-bloom-hd64", 256)])
def test_block_fast_decode_with_adapter_matches_cpu(hip, preset, hidden):
    """LoRA on the falcon/bloom FUSED decode paths (qkv delta permuted into
    the fused [q|k|v] layout, kv_cache_write / rope_cache_write) vs CPU."""
    _block_fused_vs_cpu(preset, hidden, adapter=True)


@requires_gpu
def test_bloom_dual_graph_with_big_vocab_head(hip):
    """Regression for the round-1 dual-graph fault: a bloom-layout span graph
    must survive the client's LM-head graph growing the shared gemv workspace
    (bloom's 250k vocab). _get_ws retires replaced workspaces so captured raw
    pointers stay valid."""
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.ops.fused_decode import DecodeContext
    from petals_amd.server.from_pretrained import init_random_block_
    from petals_amd.utils.graphs import GraphedCallable

    cfg = load_model_config("test-bloom-hd64")
    blk = get_model_block(cfg, 0)
    init_random_block_(blk, cfg, 0)
    blk = blk.to("cuda", torch.bfloat16).eval().optimize_for_inference()
    assert blk._fast is not None and blk._fast.graph_safe

    H, vocab = cfg.hidden_size, 250880  # bloom-176b-sized head
    ks, vs = blk.kv_cache_shape(1, 16)
    kg = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
    vg = torch.zeros_like(kg)
    ctx = DecodeContext(torch.device("cuda"))
    ctx.set_position(0)
    h_in = torch.randn(1, 1, H, device="cuda", dtype=torch.bfloat16) * 0.5

    g_span = GraphedCallable(lambda: blk(h_in, kv_cache=(kg, vg), ctx=ctx), [])

    # capture the head AFTER the span graph — its gemv workspace request (64 *
    # vocab floats) is far larger than the span's and used to free the span's
    head_t = (torch.randn(H, vocab, device="cuda") * 0.02).to(torch.bfloat16)
    norm_w = torch.ones(H, device="cuda", dtype=torch.bfloat16)
    ws_empty = torch.empty(0, device="cuda")
    h_last = torch.randn(1, H, device="cuda", dtype=torch.bfloat16)

    def head_fn():
        from petals_amd.ops.fused_decode import _get_ws

        xn = hip.rms_norm_f32out(h_last, norm_w, 1e-5)
        return hip.gemv_bf16(head_t, xn, _get_ws(torch.device("cuda", 0), "gemv", 64 * vocab), None, 0)

    g_head = GraphedCallable(head_fn, [])

    # interleaved replays: span graph must still write valid outputs
    blk2 = get_model_block(cfg, 0)
    init_random_block_(blk2, cfg, 0)
    blk2 = blk2.to("cuda", torch.bfloat16).eval().optimize_for_inference()
    kg2, vg2 = kg.clone(), vg.clone()
    for t in range(4):
        ctx.set_position(t)
        x = torch.randn(1, 1, H, device="cuda", dtype=torch.bfloat16) * 0.5
        h_in.copy_(x)
        out = g_span.replay().clone()
        g_head.replay()
        ref = blk2(x, kv_cache=(kg2, vg2), prefix_length=t)
        assert torch.allclose(out.float(), ref.float(), atol=1e-3, rtol=1e-3), (
            (out - ref).abs().max()
        )


@requires_gpu
def test_falcon_block_fast_decode_matches_cpu(hip):
    """Falcon new-decoder fused path (parallel attn+MLP, rope GQA) vs CPU."""
    _block_fused_vs_cpu("test-falcon-hd64", 256)


@requires_gpu
def test_falcon_gq29_block_fast_decode_matches_cpu(hip):
    """falcon-180b head geometry (gq=29): multi-group MFMA decode vs CPU."""
    _block_fused_vs_cpu("test-falcon-gq29", 1856)


@requires_gpu
def test_falcon_mqa71_block_fast_decode_matches_cpu(hip):
    """falcon-7b geometry (old decoder, MQA gq=71, single LN) vs CPU."""
    _block_fused_vs_cpu("test-falcon-mqa71", 4544)


class _FakeTPAllReduce:
    """Simulates a 2-rank all-reduce on ONE GPU: both shard threads rendezvous
    at a barrier, partials are summed, both continue with the total — the
    exact dataflow of dist.all_reduce over RCCL, minus the wire."""

    def __init__(self, world=2):
        import threading

        self.barrier = threading.Barrier(world)
        self.world = world
        self.slots = {}
        self.lock = threading.Lock()
        self.gen = 0

    def __call__(self, t):
        with self.lock:
            self.slots[len(self.slots)] = t
        idx = self.barrier.wait()
        if idx == 0:
            total = sum(v.float() for v in self.slots.values())
            for v in self.slots.values():
                v.copy_(total.to(v.dtype))
            torch.cuda.synchronize()
        self.barrier.wait()
        if idx == 0:
            self.slots.clear()
        self.barrier.wait()


@requires_gpu
@pytest.mark.parametrize(
    "model,quant",
    [
        ("test-llama-hd128", "none"),
        ("test-llama-hd128", "nf4"),
        ("test-falcon-hd64", "none"),  # single-reduce parallel residual
        ("test-falcon-hd64", "nf4"),
        ("test-bloom-hd64", "none"),  # ALiBi slope slicing + rank-0 biases
        ("test-bloom-hd64", "nf4"),
    ],
)
def test_tp_fused_shards_match_full_block(hip, model, quant):
    """Two TP shards of a block (fused NF4/MFMA path, simulated all-reduce)
    decode EXACTLY like the unsharded fused block — llama, falcon and bloom."""
    import threading

    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.parallel.tp import build_tp_block
    from petals_amd.server.from_pretrained import init_random_block_

    cfg = load_model_config(model)
    full = get_model_block(cfg, 0)
    init_random_block_(full, cfg, 0)
    sd = {k: v.clone() for k, v in full.state_dict().items()}
    full = full.to("cuda", torch.bfloat16).eval().optimize_for_inference(quant=quant)

    reducer = _FakeTPAllReduce(2)
    shards = []
    for r in range(2):
        blk = build_tp_block(cfg, 0, rank=r, world=2)
        blk.load_from_full_state_dict(sd)
        blk = blk.to("cuda", torch.bfloat16).eval().optimize_for_inference(quant=quant)
        assert blk._fast is not None, "TP shard must take the fused path"
        blk._fast._tp_allreduce_ = reducer
        shards.append(blk)

    torch.manual_seed(9)
    T = 4
    xs = [torch.randn(1, 1, cfg.hidden_size, device="cuda", dtype=torch.bfloat16) * 0.5 for _ in range(T)]

    ks, vs = full.kv_cache_shape(1, 16)
    kf = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
    vf = torch.zeros_like(kf)
    ref = [full(x, kv_cache=(kf, vf), prefix_length=t) for t, x in enumerate(xs)]

    outs = [None, None]
    errs = []

    def run_rank(r):
        try:
            torch.cuda.set_device(0)
            blk = shards[r]
            ks_, vs_ = blk.kv_cache_shape(1, 16)
            k = torch.zeros(ks_, device="cuda", dtype=torch.bfloat16)
            v = torch.zeros_like(k)
            res = []
            for t, x in enumerate(xs):
                res.append(blk(x, kv_cache=(k, v), prefix_length=t))
            outs[r] = res
        except Exception as e:  # noqa: BLE001
            errs.append((r, repr(e)))
            try:
                reducer.barrier.abort()
            except Exception:  # noqa: BLE001
                pass

    threads = [threading.Thread(target=run_rank, args=(r,)) for r in range(2)]
    for th in threads:
        th.start()
    for th in threads:
        th.join(timeout=120)
    assert not errs, errs
    for t in range(T):
        a, b = outs[0][t].float(), outs[1][t].float()
        assert torch.allclose(a, b), "TP ranks must agree on the reduced output"
        assert torch.allclose(a, ref[t].float(), atol=0.05, rtol=0.05), (a - ref[t].float()).abs().max()


@requires_gpu
@pytest.mark.parametrize("gq,kv_heads,kv_len", [(29, 1, 333), (71, 1, 95), (32, 2, 200)])
def test_attn_decode_big_gq(hip, gq, kv_heads, kv_len):
    """gq > 16: the decode kernel loops ceil(gq/16) A-fragment head groups."""
    from petals_amd.ops import reference

    torch.manual_seed(6)
    b, hd, lmax = 2, 64, 512
    q = torch.randn(b, kv_heads * gq * hd, device="cuda")
    k_cache = torch.zeros(b, kv_heads, lmax, hd, device="cuda", dtype=torch.bfloat16)
    v_cache = torch.zeros_like(k_cache)
    k_cache[:, :, :kv_len] = (torch.randn(b, kv_heads, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    v_cache[:, :, :kv_len] = (torch.randn(b, kv_heads, kv_len, hd, device="cuda") * 0.5).to(torch.bfloat16)
    kv_len_t = torch.tensor([kv_len], dtype=torch.int32, device="cuda")
    empty = torch.empty(0, device="cuda")
    out = hip.attn_decode_fused(q, k_cache, v_cache, kv_len_t, gq, 0, empty, empty, 1.0 / math.sqrt(hd))

    q_ref = q.view(b, kv_heads * gq, 1, hd).float().cpu()
    kr = k_cache[:, :, :kv_len].float().cpu()
    vr = v_cache[:, :, :kv_len].float().cpu()
    ref = reference.attention(q_ref, kr, vr, causal=False).view(b, -1)
    assert torch.allclose(out.cpu(), ref, atol=2e-2, rtol=2e-2), (out.cpu() - ref).abs().max()


@requires_gpu
def test_gemv_int8(hip):
    """Weight-only int8 gemv (per-column scales) vs the same math in torch."""
    torch.manual_seed(11)
    in_dim, out_dim, B = 1024, 1536, 2
    w = (torch.randn(in_dim, out_dim, device="cuda") * 0.04).float()
    scale = w.abs().amax(dim=0).clamp_min(1e-8) / 127.0
    q = torch.round(w / scale).clamp(-127, 127).to(torch.int8).contiguous()
    sc_bf16 = scale.to(torch.bfloat16).contiguous()
    x = torch.randn(B, in_dim, device="cuda")
    ws = torch.empty(0, device="cuda")
    ref = x.cpu() @ (q.float().cpu() * sc_bf16.float().cpu())
    out = hip.gemv_int8(q, sc_bf16, x, ws, None, 0)
    assert torch.allclose(out.cpu(), ref, atol=2e-2, rtol=2e-2), (out.cpu() - ref).abs().max()
    # residual epilogue
    res = torch.randn(B, out_dim, device="cuda").to(torch.bfloat16)
    out_r = hip.gemv_int8(q, sc_bf16, x, ws, res, 2)
    assert torch.allclose(out_r.float().cpu(), ref + res.float().cpu(), atol=4e-2, rtol=4e-2)


@requires_gpu
def test_llama_block_int8_decode_matches_cpu(hip):
    """Full fused decode on weight-only int8 vs the fp32 CPU block."""
    from petals_amd.models import get_model_block
    from petals_amd.models.config_base import load_model_config
    from petals_amd.server.from_pretrained import init_random_block_

    cfg = load_model_config("test-llama-hd128")
    blk_cpu = get_model_block(cfg, 0)
    init_random_block_(blk_cpu, cfg, 0)
    blk_cpu = blk_cpu.float().eval()

    blk_gpu = get_model_block(cfg, 0)
    blk_gpu.load_state_dict(blk_cpu.state_dict())
    blk_gpu = blk_gpu.to("cuda", torch.bfloat16).eval().optimize_for_inference(quant="int8")
    assert blk_gpu._fast is not None and blk_gpu._fast.quant == "int8"

    torch.manual_seed(5)
    B, S, H = 2, 9, cfg.hidden_size
    x = torch.randn(B, S, H) * 0.5
    ks, vs = blk_cpu.kv_cache_shape(B, 32)
    kc, vc = torch.zeros(ks), torch.zeros(vs)
    kg = torch.zeros(ks, device="cuda", dtype=torch.bfloat16)
    vg = torch.zeros(vs, device="cuda", dtype=torch.bfloat16)

    y_cpu = [blk_cpu(x[:, :6], kv_cache=(kc, vc), prefix_length=0)]
    y_gpu = [blk_gpu(x[:, :6].cuda().bfloat16(), kv_cache=(kg, vg), prefix_length=0)]
    for t in range(6, S):
        y_cpu.append(blk_cpu(x[:, t : t + 1], kv_cache=(kc, vc), prefix_length=t))
        y_gpu.append(blk_gpu(x[:, t : t + 1].cuda().bfloat16(), kv_cache=(kg, vg), prefix_length=t))
    ref = torch.cat(y_cpu, 1)
    out = torch.cat([y.float().cpu() for y in y_gpu], 1)
    # int8 weights: looser bound than bf16/nf4 block tests
    assert torch.allclose(out, ref, atol=0.08, rtol=0.08), (out - ref).abs().max()
