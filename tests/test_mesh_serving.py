"""Full serving stack over the LocalMesh tier (gloo world 2 on CPU): two real
Server processes share a mesh; pushed decode steps hand activations rank->rank
through it and the final activation returns to the client's process the same
way. Generate output is exact-matched against local HF. On an MI355X node the
identical code paths run over RCCL/xGMI (backend "nccl")."""

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
        from petals_amd.dht.node import DHT
        from petals_amd.parallel.mesh import LocalMesh
        from petals_amd.server.server import Server

        path = os.path.join(ckpt_dir, "ckpt")
        hf_model = transformers.LlamaForCausalLM.from_pretrained(path).eval() if rank == 0 else None

        mesh = LocalMesh("test-serve-mesh", rank, world, device=torch.device("cpu"))

        boot_addr = [None]
        boot = None
        if rank == 0:
            boot = DHT(host="127.0.0.1")
            boot_addr = [list(boot.listen_addr)]
        dist.broadcast_object_list(boot_addr, src=0)
        initial_peers = [tuple(boot_addr[0])]

        spans = {0: "0:2", 1: "2:4"}
        server = Server(
            path,
            initial_peers=initial_peers,
            host="127.0.0.1",
            device="cpu",
            torch_dtype="float32",
            block_indices=spans[rank],
            dht_prefix="mesh-serve",
            throughput=1.0,
            update_period=2.0,
            mesh=mesh,
        ).start()
        dist.barrier()

        if rank == 0:
            from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

            model = AutoDistributedModelForCausalLM.from_pretrained(
                path, initial_peers=initial_peers, dht_prefix="mesh-serve",
                show_route=False, max_retries=1, min_backoff=0.2,
            )
            torch.manual_seed(3)
            ids = torch.randint(0, 128, (1, 5))
            ref = hf_model.generate(ids, max_new_tokens=8, do_sample=False)
            out = model.generate(ids, max_new_tokens=8, do_sample=False)
            assert torch.equal(out, ref), (out, ref)
            # the mesh must actually have carried activations: rank0's server
            # pushed to rank1 over it...
            assert mesh._send_tickets.get(1, 0) > 0, "no mesh sends from rank 0 to rank 1"
            # ...and rank1 sent the final activations back to this process
            assert mesh._recv_next.get(1, 0) > 0, "no mesh deliveries from rank 1"
        dist.barrier()
        # --- failure path: a broken mesh must fall back to TCP push with the
        # exact same outputs (mark_broken on both ranks, as a real RCCL error
        # would do on both ends of the pair)
        mesh.mark_broken("test-induced failure")
        dist.barrier()
        if rank == 0:
            sent_before = dict(mesh._send_tickets)
            torch.manual_seed(7)
            ids2 = torch.randint(0, 128, (1, 4))
            ref2 = hf_model.generate(ids2, max_new_tokens=6, do_sample=False)
            out2 = model.generate(ids2, max_new_tokens=6, do_sample=False)
            assert torch.equal(out2, ref2), (out2, ref2)
            assert dict(mesh._send_tickets) == sent_before, "broken mesh must not be used"
            model.transformer.h.sequence_manager.shutdown()
        dist.barrier()
        server.shutdown()
        if boot is not None:
            boot.shutdown()
        mesh.shutdown()
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback

        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}\n{traceback.format_exc()}")
        raise


def test_mesh_serving_exact_match_world2(tmp_path):
    transformers = pytest.importorskip("transformers")
    torch.manual_seed(0)
    cfg = transformers.LlamaConfig(**HF_CFG)
    transformers.LlamaForCausalLM(cfg).eval().save_pretrained(
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
    errors = []
    while not fail_q.empty():
        errors.append(fail_q.get())
    assert not errors, errors[0]
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
