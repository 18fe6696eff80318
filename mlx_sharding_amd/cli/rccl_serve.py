"""`mlx-sharding-rccl-serve`: OpenAI API over an in-node RCCL pipeline.

Launch with one rank per GPU:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 -m mlx_sharding_amd.cli.rccl_serve \
        --model /path/to/checkpoint --port 8080
"""

from __future__ import annotations

import argparse


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--model", type=str, required=True)
    p.add_argument("--host", type=str, default="127.0.0.1")
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--quantize", type=int, choices=[4, 8], default=None,
                   help="quantize a dense checkpoint to w4a16/w8a16 at load "
                        "(the fast path on MI355X)")
    p.add_argument("--quantize-group-size", type=int, default=64)
    args = p.parse_args(argv)

    from transformers import AutoTokenizer

    from ..config import ModelConfig
    from ..parallel.rccl import PipelineWorker, init_distributed, split_layers
    from ..parallel.rccl_serve import (RcclModelProvider, RcclPipeline)
    from ..server.openai_api import run
    from ..utils.loading import load_model

    rank, world, device = init_distributed()
    cfg = ModelConfig.load(args.model)
    s, e = split_layers(cfg.num_hidden_layers, world)[rank]
    q = (args.quantize, args.quantize_group_size) if args.quantize else None
    model, _ = load_model(args.model, s, e, device=str(device), quantize=q)
    worker = PipelineWorker(model, rank, world, device)
    pipeline = RcclPipeline(worker)

    if rank == 0:
        tokenizer = AutoTokenizer.from_pretrained(args.model)
        provider = RcclModelProvider(args, pipeline, tokenizer)
        server = run(args.host, args.port, provider)
        print(f"RCCL pipeline API on {args.host}:{server.server_address[1]} "
              f"(pp{world})", flush=True)
        try:
            server.serve_forever()
        finally:
            pipeline.shutdown()
    else:
        pipeline.worker_loop()


if __name__ == "__main__":
    main()
