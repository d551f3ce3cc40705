"""Intra-server tensor parallelism through the REAL serving stack (gloo
world 2 on CPU): rank 0 runs the full Server over its block SHARDS, rank 1
runs the TPShadowWorker lockstep loop; a client's generate() and training
fwd+bwd must exact-match the local HF model. On an MI355X node the identical
code paths run rank-per-GPU over RCCL/xGMI."""

import os
import socket

import pytest
import torch
import torch.multiprocessing as mp

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


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank, world, dist_port, ckpt_dir, fail_q):
    try:
        import torch.distributed as dist
        import transformers

        dist.init_process_group(
            backend="gloo", init_method=f"tcp://127.0.0.1:{dist_port}", rank=rank, world_size=world
        )
        from petals_amd.models.config_base import load_model_config

        path = os.path.join(ckpt_dir, "ckpt")

        if rank != 0:
            from petals_amd.parallel.tp import TPShadowWorker

            config = load_model_config(path)
            TPShadowWorker(
                path, config, device=torch.device("cpu"), torch_dtype=torch.float32,
                quant_type="none", group=None, rank=rank, world=world,
            ).serve_forever()
            dist.destroy_process_group()
            return

        from petals_amd.dht.node import DHT
        from petals_amd.server.server import Server
        from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

        hf_model = transformers.AutoModelForCausalLM.from_pretrained(path).eval()
        boot = DHT(host="127.0.0.1")
        server = Server(
            path,
            initial_peers=[boot.listen_addr],
            host="127.0.0.1",
            device="cpu",
            torch_dtype="float32",
            block_indices="0:4",
            dht_prefix="tp-serve",
            throughput=1.0,
            tensor_parallel_ranks=world,
        ).start()
        model = AutoDistributedModelForCausalLM.from_pretrained(
            path, initial_peers=[boot.listen_addr], dht_prefix="tp-serve",
            show_route=False, max_retries=1, min_backoff=0.2,
        )
        torch.manual_seed(31)
        ids = torch.randint(0, 128, (1, 5))
        ref = hf_model.generate(ids, max_new_tokens=6, do_sample=False)
        out = model.generate(ids, max_new_tokens=6, do_sample=False)
        assert torch.equal(out, ref), (out, ref)

        # training fwd+bwd across TP shards: grads match local HF
        embeds_ref = hf_model.get_input_embeddings()(ids).detach().requires_grad_(True)
        hf_model(inputs_embeds=embeds_ref).logits.square().mean().backward()
        embeds = hf_model.get_input_embeddings()(ids).detach().requires_grad_(True)
        model(inputs_embeds=embeds).logits.square().mean().backward()
        assert torch.allclose(embeds.grad, embeds_ref.grad, atol=1e-4, rtol=1e-3), (
            (embeds.grad - embeds_ref.grad).abs().max()
        )

        model.transformer.h.sequence_manager.shutdown()
        server.shutdown()  # broadcasts OP_SHUTDOWN to the shadow
        boot.shutdown()
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback

        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}\n{traceback.format_exc()}")
        raise


@pytest.mark.parametrize("family", ["llama", "falcon", "bloom"])
def test_tp_serving_exact_match_world2(tmp_path, family):
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(0)
    if family == "llama":
        cfg = transformers.LlamaConfig(**HF_CFG)
        hf_model = transformers.LlamaForCausalLM(cfg)
    elif family == "falcon":
        cfg = transformers.FalconConfig(
            hidden_size=64, num_attention_heads=4, num_hidden_layers=4, vocab_size=128,
            new_decoder_architecture=True, num_kv_heads=2, bias=False, parallel_attn=True,
        )
        hf_model = transformers.FalconForCausalLM(cfg)
    else:
        cfg = transformers.BloomConfig(hidden_size=64, n_head=4, n_layer=4, vocab_size=128)
        hf_model = transformers.BloomForCausalLM(cfg)
    hf_model.eval().save_pretrained(
        os.path.join(str(tmp_path), "ckpt"), safe_serialization=True
    )
    port = _free_port()
    ctx = mp.get_context("spawn")
    fail_q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, str(tmp_path), fail_q)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    for p in procs:
        if p.is_alive():
            p.terminate()
    errors = []
    while not fail_q.empty():
        errors.append(fail_q.get())
    assert not errors, errors[0]
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
