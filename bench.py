#!/usr/bin/env python3
"""Flagship serving benchmark: pipeline-parallel decode throughput.

Measures the reference's headline metric (BASELINE.json): output
tokens/sec for DeepSeek-Coder-V2-Lite, pipeline-parallel over N MI355X
GPUs, synthetic data + random-init weights, bf16 compute.  One rank per
GPU over RCCL when launched via torch.distributed.run; single-process
at N=1.

A "step" = one decode iteration producing one token for every sequence
in the global batch (micro-batched through the pipeline stages).
Prefill is untimed warm-up state; W warm-up decode steps run untimed;
exactly K steps are timed between barrier+synchronize fences; the
elapsed time is MAX over ranks.
"""

from __future__ import annotations

import argparse
import json
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=192)  # ~2 s timed
    #  region at the headline config: long enough for the
    #  driver's gpu-busy sampler, still
    #  well under a minute end to end
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--model", type=str, default="deepseek-v2-lite")
    p.add_argument("--batch", type=int, default=64, help="global batch (sequences)")
    p.add_argument("--prefill", type=int, default=512, help="synthetic prompt length")
    p.add_argument("--quant", dest="quant", action="store_true", default=None,
                   help="4-bit (w4a16) weights — the reference's headline "
                        "precision (DeepSeek-Coder-V2-Lite 4-bit, "
                        "/root/reference/README.md:26-35).  Default: ON for "
                        "deepseek models (the headline config), off for the "
                        "bf16 llama configs")
    p.add_argument("--bf16", dest="quant", action="store_false",
                   help="bf16 weights instead of the headline 4-bit")
    p.add_argument("--bits", type=int, default=4, choices=[4, 8],
                   help="quant width when --quant is active (w4a16/w8a16)")
    p.add_argument("--micro", type=int, default=0,
                   help="micro-batches (0 = auto: max(world, 1), capped by batch)")
    p.add_argument("--no-graph", action="store_true",
                   help="disable hipGraph-captured decode (debugging)")
    p.add_argument("--graph", action="store_true",
                   help="force hipGraph decode even at large batch")
    return p.parse_args()


def main():
    import os
    import sys

    args = parse_args()
    if args.quant is None:
        args.quant = "deepseek" in args.model

    # --gpus N without a torchrun rendezvous: self-launch N ranks so a
    # plain `python bench.py --gpus 8` runs 8 REAL pipeline stages (and
    # can never mislabel a 1-process run as pp8)
    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        import socket
        import subprocess
        with socket.socket() as sock:  # a free rendezvous port
            sock.bind(("127.0.0.1", 0))
            port = sock.getsockname()[1]
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}",
               "--master-addr", "127.0.0.1", "--master-port", str(port),
               os.path.abspath(__file__), *sys.argv[1:]]
        sys.exit(subprocess.call(cmd))

    from mlx_sharding_amd.parallel.rccl import (PipelineWorker,
                                                build_stage_model,
                                                init_distributed)
    from mlx_sharding_amd.utils.presets import get_preset

    rank, world, device = init_distributed()
    if world != args.gpus:
        raise SystemExit(
            f"bench.py: launched with WORLD_SIZE={world} but --gpus "
            f"{args.gpus}; the labels would lie — relaunch with matching "
            f"values (torchrun --nproc-per-node N bench.py --gpus N)")
    n_gpus = world
    use_gpu = torch.cuda.is_available()
    if not use_gpu and args.model in ("deepseek-v2-lite", "llama-3-8b",
                                      "llama-3-70b", "gemma-2-9b"):
        # CPU smoke: shrink to the debug model so a no-GPU run finishes
        args.model = "debug-deepseek" if "deepseek" in args.model else "debug-llama"
        args.batch = min(args.batch, 4)
        args.prefill = min(args.prefill, 32)

    config = get_preset(args.model, quant=args.quant, bits=args.bits)
    quant_for = None
    if args.quant:
        qc = config.quantization
        quant_for = lambda prefix: qc  # noqa: E731  (quantize every linear)
    model = build_stage_model(config, rank, world, device, quant_for=quant_for)
    dtype = torch.bfloat16
    worker = PipelineWorker(model, rank, world, device, dtype)

    n_micro = args.micro or max(world, 1)
    n_micro = max(1, min(n_micro, args.batch))
    while args.batch % n_micro:
        n_micro -= 1
    micro = args.batch // n_micro

    torch.manual_seed(1234 + rank)
    ids = [torch.randint(0, config.vocab_size, (micro, args.prefill))
           for _ in range(n_micro)]

    def sync():
        if use_gpu:
            torch.cuda.synchronize()
        if world > 1:
            torch.distributed.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    # ---- prefill (untimed state setup; measured separately for TTFT) ----
    # one throwaway prefill first: hipBLASLt GEMM heuristics/code-objects
    # warm up on first use of each shape (~1.4 s one-time, not TTFT)
    worker.prefill(ids, micro, n_micro, args.prefill)
    sync()
    t_pf = time.perf_counter()
    tokens = worker.prefill(ids, micro, n_micro, args.prefill)
    sync()
    ttft_s = time.perf_counter() - t_pf

    if worker.is_first and tokens is None:
        raise RuntimeError("stage 0 did not receive tokens")

    # ---- warmup ----
    for _ in range(args.warmup):
        tokens = worker.decode_step(tokens, micro, n_micro)

    # hipGraph decode wins in the launch-bound regime (small batch:
    # ~100+ kernels/step at ~5 us launch each); at batch >= 32 the
    # per-step time is kernel-execution-bound and graph replay measures
    # SLOWER (hipBLASLt kernels run ~3x slower under replay on this
    # stack — see docs/PERFORMANCE.md), so default off there.  Multi-
    # rank graphs are ON in the same small-µbatch regime: the graphs
    # are comm-free (RCCL hops run eager between replays, writing the
    # graphs' static buffers) and capture is rank-local, so a capture
    # failure degrades to eager decode instead of hanging the job.
    use_graph = use_gpu and not args.no_graph and \
        (args.graph or micro <= 16)
    if use_graph:
        capacity = args.prefill + args.warmup + args.steps + 16
        worker.enable_graph_decode(tokens, micro, n_micro, capacity)
        # two replays to settle (the capture warmup rewound the position)
        for _ in range(2):
            tokens = worker.decode_step(tokens, micro, n_micro)

    # ---- timed region ----
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        tokens = worker.decode_step(tokens, micro, n_micro)
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        total_tokens = args.batch * args.steps
        value = total_tokens / elapsed
        result = {
            "metric": f"output tokens/sec (decode), {args.model} PP",
            "value": round(value, 3),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            # global batch is FIXED as N grows (total work fixed, the
            # model just spreads over more stages) = strong scaling
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic (random ids, random-init weights)",
            "config": {
                "model": args.model,
                "global_batch": args.batch,
                "seq_len": args.prefill,
                "parallelism": f"pp{n_gpus}",
                "micro_batches": n_micro,
                "weights": (f"int{args.bits}-w{args.bits}a16"
                            if args.quant else "bf16"),
                "p50_ttft_ms": round(ttft_s * 1000, 3),
            },
        }
        print(json.dumps(result), flush=True)

    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
