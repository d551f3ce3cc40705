#!/usr/bin/env python3
"""Flagship benchmark: single-batch autoregressive generate, Llama-2-70B,
blocks pipelined across N MI355X GPUs over RCCL/xGMI (BASELINE.json metric).

  python bench.py --gpus N --steps K --warmup W [--model llama-2-70b]

For N>1 the driver launches this under torch.distributed.run with one rank per
GPU; rank r hosts a contiguous span of the 80 blocks (the petals_amd span
architecture), activations hop rank->rank via RCCL send/recv (the framework's
intra-node hand-off path, petals_amd/parallel/pipeline.py), rank 0 holds
embeddings + final norm + LM head (the thin petals client role). Weights are
random-init bf16 (no network for checkpoints), data is a synthetic prompt.

A "step" = one generated token. Weak scaling: each GPU keeps the whole model's
per-token read spread over N, i.e. per-GPU work shrinks but single-batch
decode is sequential across spans (petals semantics: more servers host bigger
models; they do not parallelize one batch).
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, flush=True)


def _emit_result(args, world, use_cuda, elapsed, B, stack):
    ms_per_step = elapsed / args.steps * 1000
    tokens_per_s = args.steps * B / elapsed
    baseline = 6.0  # reference headline: Llama-2-70B, 6 tok/s on the public swarm
    par = ("swarm-pp%d" % world if world > 1 else "swarm-single") if stack == "serve" else (
        f"pp{world}" if world > 1 else "single")
    result = {
        "metric": "single-batch generate tokens/sec, Llama-2-70B across 1/2/4/8 MI355X servers",
        "value": tokens_per_s,
        "unit": "tokens/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": ms_per_step,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": tokens_per_s / baseline if args.model == "llama-2-70b" else None,
        "dtype": ("bf16" if args.quant == "none" else f"{args.quant}-weights/bf16-compute") if use_cuda else "fp32",
        "data": "synthetic prompt, random-init weights (no network)",
        "config": {
            "model": args.model,
            "global_batch": B,
            "seq_len": args.prompt_len + args.warmup + args.steps,
            "prompt_len": args.prompt_len,
            "parallelism": par,
            "quant": args.quant if use_cuda else "none",
            "stack": stack,
        },
    }
    print(json.dumps(result), flush=True)


def run_serve_stack(args, rank, world, use_cuda, config) -> bool:
    """The REAL serving stack, rank-per-GPU: every rank runs a full Server
    (DHT announce, handler RPCs, PriorityRuntime, MemoryCache, per-session
    span hipGraphs) on its span; rank 0 additionally runs the thin client
    (embeddings + LM head + sampling). Adjacent spans hand activations over
    the RCCL/xGMI mesh (parallel/mesh.py); the co-located client<->rank0 hop
    is in-process (p2p/transport.py InProcStream). This is what a user of
    `petals_amd.cli.run_server` + AutoDistributedModelForCausalLM gets.

    Returns True if the measurement was emitted; False after a COORDINATED
    failure (all ranks agree via all-reduce), so main() can fall back to the
    bare pipeline measurement instead of hanging the job."""
    import torch.distributed as dist

    B = args.batch
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0))) if use_cuda else torch.device("cpu")
    quant = args.quant if use_cuda else "none"

    def healthy(ok: bool, what: str) -> bool:
        """Coordinated go/no-go: min over ranks."""
        if world == 1:
            return ok
        t = torch.tensor([1.0 if ok else 0.0], device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        if t.item() < 0.5:
            log(f"[bench] serve stack aborted at stage: {what}")
            return False
        return True

    mesh = boot = server = None
    ok = True
    try:
        if world > 1:
            from petals_amd.parallel.mesh import LocalMesh

            # dedicated process group: mesh p2p must never share a communicator
            # with the benchmark's own barriers/all-reduces
            pg = dist.new_group(list(range(world)))
            mesh = LocalMesh("bench-mesh", rank, world, device=device, group=pg)

        from petals_amd.dht.node import DHT
        from petals_amd.parallel.pipeline import split_blocks
        from petals_amd.server.server import Server

        boot_addr = [None]
        if rank == 0:
            boot = DHT(host="127.0.0.1")
            boot_addr = [list(boot.listen_addr)]
        if world > 1:
            dist.broadcast_object_list(boot_addr, src=0)
        initial_peers = [tuple(boot_addr[0])]

        spans = split_blocks(config.num_blocks, world)
        my_span = spans[rank]
        t0 = time.time()
        server = Server(
            args.model,
            initial_peers=initial_peers,
            host="127.0.0.1",
            device=str(device),
            torch_dtype="bfloat16" if use_cuda else "float32",
            block_indices=f"{my_span.start}:{my_span.stop}",
            dht_prefix="bench-serve",
            throughput=1000.0,
            quant_type=quant,
            mesh=mesh,
            update_period=30.0,
        ).start()
        log(f"[bench] serve stack: rank {rank} serving blocks {my_span.start}:{my_span.stop} "
            f"in {time.time()-t0:.1f}s")
    except Exception as e:  # noqa: BLE001
        print(f"[bench] rank {rank}: serve stack startup failed: {e!r}", flush=True)
        ok = False

    if not healthy(ok, "server startup"):
        _serve_cleanup(server, mesh, boot, use_cuda)
        return False

    try:
        if mesh is not None:
            # NCCL communicator init is lazy AND collective: prime the ring
            # eagerly or the first decode chain deadlocks
            mesh.warmup_ring()
    except Exception as e:  # noqa: BLE001
        print(f"[bench] rank {rank}: mesh warmup failed: {e!r}", flush=True)
        ok = False
    if not healthy(ok, "mesh warmup"):
        _serve_cleanup(server, mesh, boot, use_cuda)
        return False

    def barrier():
        if world > 1:
            dist.barrier()

    elapsed = 0.0
    client_ok = True
    if rank == 0:
        model = None
        try:
            from petals_amd.utils.auto_config import AutoDistributedModelForCausalLM

            model = AutoDistributedModelForCausalLM.from_pretrained(
                args.model, initial_peers=initial_peers, dht_prefix="bench-serve",
                show_route=False, max_retries=2, request_timeout=60.0,
            )
            if use_cuda:
                model = model.to(device=device, dtype=torch.bfloat16)
            gen = torch.Generator().manual_seed(1234)
            prompt = torch.randint(0, config.vocab_size, (B, args.prompt_len), generator=gen)
            if use_cuda:
                prompt = prompt.to(device)
            max_len = args.prompt_len + args.warmup + args.steps + 8
            with model.transformer.h.inference_session(max_length=max_len, batch_size=B) as sess, \
                    model.transformer.h.use_session(sess):
                out = model.generate(prompt, max_new_tokens=max(args.warmup, 1), do_sample=False,
                                     eos_token_id=-1)
                barrier()
                if use_cuda:
                    torch.cuda.synchronize(device)
                t_start = time.perf_counter()
                out = model.generate(out[:, -1:], max_new_tokens=args.steps, do_sample=False,
                                     eos_token_id=-1)
                if use_cuda:
                    torch.cuda.synchronize(device)
                elapsed = time.perf_counter() - t_start
                barrier()
                assert out.shape[1] == args.steps + 1
        except Exception as e:  # noqa: BLE001
            print(f"[bench] client failed: {e!r}", flush=True)
            client_ok = False
            barrier()  # release peers from the timed-region barriers
            barrier()
        finally:
            if model is not None:
                model.transformer.h.sequence_manager.shutdown()
    else:
        barrier()  # start of timed region
        barrier()  # end of timed region

    if not healthy(client_ok, "client generate"):
        _serve_cleanup(server, mesh, boot, use_cuda)
        return False

    # max over ranks (only rank 0 measured, but keep the collective contract)
    if world > 1:
        e = torch.tensor([elapsed], device=device if use_cuda else "cpu")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    if rank == 0:
        _emit_result(args, world, use_cuda, elapsed, B, "serve")

    _serve_cleanup(server, mesh, boot, use_cuda)
    if world > 1:
        dist.barrier()
    return True


def _serve_cleanup(server, mesh, boot, use_cuda):
    try:
        if server is not None:
            server.shutdown()
        if mesh is not None:
            mesh.shutdown()
        if boot is not None:
            boot.shutdown()
    except Exception:  # noqa: BLE001
        pass
    if use_cuda:
        import gc

        gc.collect()
        torch.cuda.empty_cache()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=64)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--model", default="llama-2-70b")
    p.add_argument("--prompt-len", type=int, default=128)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--device", default="cuda")
    p.add_argument("--quant", default="nf4", choices=["none", "nf4", "int8"], help="BASELINE config #3 names NF4 for the 70B pipeline; --quant none measures pure bf16")
    p.add_argument("--stack", default="serve", choices=["pipeline", "serve"],
                   help="serve (default): the REAL serving stack (DHT + Server + handler + client "
                        "sessions) with the RCCL mesh hand-off — what a user gets; pipeline: bare "
                        "rank-per-GPU RCCL pipeline (kernel-harness upper bound)")
    args = p.parse_args()

    import torch.distributed as dist

    from petals_amd.models.config_base import load_model_config
    from petals_amd.parallel.pipeline import PipelineStage, init_process_group_from_env, split_blocks
    from petals_amd.server.from_pretrained import build_empty_block, init_random_block_

    rank, world = init_process_group_from_env()
    assert world == args.gpus or args.gpus == 1, f"WORLD_SIZE={world} but --gpus={args.gpus}"
    world = max(world, 1)

    use_cuda = args.device == "cuda" and torch.cuda.is_available()
    if not use_cuda:
        args.quant = "none"
    if use_cuda:
        device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
        torch.cuda.set_device(device)
        dtype = torch.bfloat16
    else:
        device = torch.device("cpu")
        dtype = torch.float32

    config = load_model_config(args.model)
    if os.environ.get("PETALS_AMD_BENCH_BLOCKS"):  # debug: cap the span size
        config.num_hidden_layers = int(os.environ["PETALS_AMD_BENCH_BLOCKS"])
    H = config.hidden_size
    B = args.batch
    max_len = args.prompt_len + args.warmup + args.steps + 8

    if args.stack == "serve":
        if run_serve_stack(args, rank, world, use_cuda, config):
            if world > 1:
                dist.destroy_process_group()
            os._exit(0)  # skip interpreter-teardown races between HIP and daemon threads
        log("[bench] serve stack failed; measuring the bare pipeline instead")
        args.stack = "pipeline"

    spans = split_blocks(config.num_blocks, world)
    my_span = spans[rank]
    log(f"[bench] model={args.model} blocks={config.num_blocks} world={world} "
        f"span[0]={list(spans[0])[:1]}..{list(spans[0])[-1:]} dtype={dtype}")

    # --- build this rank's span (deterministic random init, bf16, fused path)
    t0 = time.time()
    blocks = []
    for i in my_span:
        blk = build_empty_block(config, i, device, dtype)
        init_random_block_(blk, config, i)
        blk = blk.eval()
        if use_cuda and hasattr(blk, "optimize_for_inference"):
            blk.optimize_for_inference(quant=args.quant)
        blocks.append(blk)
    kv_caches = []
    for blk in blocks:
        ks, vs = blk.kv_cache_shape(B, max_len)
        kv_caches.append((
            torch.zeros(ks, device=device, dtype=dtype),
            torch.zeros(vs, device=device, dtype=dtype),
        ))
    stage = PipelineStage(blocks, rank, world, device, H, dtype)
    log(f"[bench] rank {rank}: {len(blocks)} blocks built in {time.time()-t0:.1f}s")

    # --- rank 0 client-side parts
    if rank == 0:
        gen = torch.Generator().manual_seed(1234)
        embed = (torch.randn(config.vocab_size, H, generator=gen) * 0.02).to(device=device, dtype=dtype)
        norm_w = torch.ones(H, device=device, dtype=dtype)
        head_t = embed.t().contiguous()  # tied head, [H, vocab] for gemv/matmul
        prompt = torch.randint(0, config.vocab_size, (B, args.prompt_len), generator=gen).to(device)

    from petals_amd import ops

    hip = ops._load_hip_ops() if use_cuda else None
    if use_cuda and hip is None:
        raise RuntimeError(f"HIP extension required on GPU: {ops._hip_import_error!r}")

    def head_logits(h_last: torch.Tensor) -> torch.Tensor:
        """h_last [B, H] -> logits [B, vocab] (final norm + tied head)."""
        if hip is not None:
            xn = hip.rms_norm_f32out(h_last.to(dtype), norm_w, config.layer_norm_eps)
            ws = torch.empty(0, device=device)
            return hip.gemv_bf16(head_t, xn, ws, None, 0)
        xn = ops.reference.rms_norm(h_last, norm_w, config.layer_norm_eps)
        return xn @ head_t.float() if head_t.dtype != xn.dtype else xn @ head_t

    def run_pipeline(h: torch.Tensor, prefix: int, seq: int) -> torch.Tensor:
        """Push [B, seq, H] through all spans; returns last hidden on rank 0."""
        if world == 1:
            return stage.forward_span(h, kv_caches, prefix)
        if rank == 0:
            h = stage.forward_span(h, kv_caches, prefix)
            stage.send(h)
            return stage.recv(B, seq)  # final output comes back from last rank
        else:
            h = stage.recv(B, seq)
            h = stage.forward_span(h, kv_caches, prefix)
            stage.send(h)
            return h  # unused on non-zero ranks

    def one_token(token_ids, prefix: int):
        """Generate the next token given current ids [B,1] (eager path)."""
        if rank == 0:
            h = embed[token_ids.view(-1)].view(B, 1, H)
        else:
            h = torch.empty(B, 1, H, device=device, dtype=dtype)
        out = run_pipeline(h, prefix, 1)
        if rank == 0:
            logits = head_logits(out[:, -1, :])
            return logits.argmax(dim=-1, keepdim=True)
        return token_ids

    # ---- hipGraph decode: capture each rank's whole span step once; replay
    # per token with only the device position advancing (utils/graphs.py)
    graph_state = {}

    def build_graphs(cur_id):
        from petals_amd.ops.fused_decode import DecodeContext
        from petals_amd.utils.graphs import GraphedCallable

        ctx = DecodeContext(device)
        ctx.set_position(args.prompt_len)
        gs = {"ctx": ctx}
        if rank == 0:
            gs["cur_id"] = cur_id.clone()

            def span_fn():
                h = embed.index_select(0, gs["cur_id"].view(-1)).view(B, 1, H)
                ctx.norm_parts = None  # folded-norm hand-off starts fresh each pass
                for blk, (k, v) in zip(blocks, kv_caches):
                    h = blk(h, kv_cache=(k, v), ctx=ctx)
                return h

            gs["g_span"] = GraphedCallable(span_fn, [])
            if world == 1:
                def head_fn():
                    logits = head_logits(gs["g_span"].static_outputs[0][:, -1, :])
                    gs["cur_id"].copy_(logits.argmax(dim=-1, keepdim=True))
                    return logits
                if not os.environ.get("PETALS_AMD_NO_HEAD_GRAPH"):  # debug bisect
                    gs["g_head"] = GraphedCallable(head_fn, [])
                else:
                    gs["head_eager"] = head_fn
            else:
                gs["h_back"] = torch.empty(B, 1, H, device=device, dtype=dtype)

                def head_fn():
                    logits = head_logits(gs["h_back"][:, -1, :])
                    gs["cur_id"].copy_(logits.argmax(dim=-1, keepdim=True))
                    return logits
                if not os.environ.get("PETALS_AMD_NO_HEAD_GRAPH"):  # debug bisect
                    gs["g_head"] = GraphedCallable(head_fn, [])
                else:
                    gs["head_eager"] = head_fn
        else:
            gs["h_in"] = torch.empty(B, 1, H, device=device, dtype=dtype)

            def span_fn():
                h = gs["h_in"]
                ctx.norm_parts = None  # folded-norm hand-off starts fresh each pass
                for blk, (k, v) in zip(blocks, kv_caches):
                    h = blk(h, kv_cache=(k, v), ctx=ctx)
                return h

            gs["g_span"] = GraphedCallable(span_fn, [])
        return gs

    def one_token_graphed(prefix: int):
        gs = graph_state
        gs["ctx"].set_position(prefix)
        if rank == 0:
            h = gs["g_span"].replay()
            if world > 1:
                stage.send(h)
                dist.recv(gs["h_back"], src=stage.prev_rank)
            if "g_head" in gs:
                gs["g_head"].replay()
            else:
                gs["head_eager"]()
        else:
            dist.recv(gs["h_in"], src=stage.prev_rank)
            gs["g_span"].replay()
            stage.send(gs["g_span"].static_outputs[0])

    # --- prefill
    prefix = 0
    if rank == 0:
        h = embed[prompt.view(-1)].view(B, args.prompt_len, H)
    else:
        h = torch.empty(B, args.prompt_len, H, device=device, dtype=dtype)
    out = run_pipeline(h, 0, args.prompt_len)
    prefix = args.prompt_len
    if rank == 0:
        cur = head_logits(out[:, -1, :]).argmax(dim=-1, keepdim=True)
    else:
        cur = torch.zeros(B, 1, dtype=torch.long, device=device)

    # --- capture decode graphs (GPU) or stay eager (CPU / MoE / capture failure)
    graph_safe = all(getattr(getattr(blk, "_fast", None), "graph_safe", False) for blk in blocks)
    use_graphs = use_cuda and graph_safe and not os.environ.get("PETALS_AMD_NO_GRAPHS")
    if use_graphs:
        try:
            graph_state.update(
                build_graphs(cur if rank == 0 else torch.zeros(B, 1, dtype=torch.long, device=device))
            )
        except Exception as e:  # noqa: BLE001
            log(f"[bench] graph capture failed ({e!r}); falling back to eager decode")
            use_graphs = False
    if use_graphs:
        step_fn = one_token_graphed
    else:
        def step_fn(p):
            nonlocal cur
            cur = one_token(cur, p)

    # --- warmup decode
    for _ in range(args.warmup):
        step_fn(prefix)
        prefix += 1

    # --- timed region
    if world > 1:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize(device)
    t_start = time.perf_counter()
    for _ in range(args.steps):
        step_fn(prefix)
        prefix += 1
    if use_cuda:
        torch.cuda.synchronize(device)
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t_start

    # max over ranks
    if world > 1:
        e = torch.tensor([elapsed], device=device)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    if rank == 0:
        _emit_result(args, world, use_cuda, elapsed, B, "pipeline")

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
