"""Greedy generation CLI — parity with /root/reference/generate.py.

Loads the first shard locally, chains remote shard servers, streams
text, and prints the reference's prompt/generation tokens-per-sec
metrics (generate.py:115-122).
"""

from __future__ import annotations

import argparse

import torch


def main(argv=None):
    p = argparse.ArgumentParser(description="LLM pipeline inference")
    # defaults mirror the reference CLI (/root/reference/generate.py:14-17)
    p.add_argument("--model", type=str, default="shard_0")
    p.add_argument("--prompt", type=str,
                   default="how to write quicksort in python")
    p.add_argument("--max_tokens", "--max-tokens", type=int, default=512)
    p.add_argument("--server_address", "--server-address", type=str, default="",
                   help="comma-separated list of remote shard addresses")
    p.add_argument("--start_layer", "--start-layer", type=int, default=None)
    p.add_argument("--end_layer", "--end-layer", type=int, default=None)
    p.add_argument("--temp", type=float, default=0.0,
                   help="sampling temperature (0 = greedy, the reference CLI default)")
    p.add_argument("--device", type=str,
                   default="cuda" if torch.cuda.is_available() else "cpu")
    p.add_argument("--quantize", type=int, choices=[4, 8], default=None,
                   help="quantize a dense checkpoint to w4a16/w8a16 at load")
    p.add_argument("--quantize-group-size", type=int, default=64)
    args = p.parse_args(argv)

    from transformers import AutoTokenizer

    from ..parallel.engine import (GenerationStats, SamplingParams,
                                   stream_generate)
    from ..parallel.grpc_transport import make_clients
    from ..utils.detokenizer import StreamingDetokenizer
    from ..utils.loading import get_model_path, load_model

    # local directory or HF repo id (reference utils.py:33-39)
    path = get_model_path(args.model)
    tokenizer = AutoTokenizer.from_pretrained(str(path))
    q = (args.quantize, args.quantize_group_size) if args.quantize else None
    model, config = load_model(path, args.start_layer, args.end_layer,
                               device=args.device, quantize=q)
    remotes = make_clients(args.server_address.split(",")) if args.server_address else []

    messages = [{"role": "user", "content": args.prompt}]
    try:
        text = tokenizer.apply_chat_template(messages, tokenize=False,
                                             add_generation_prompt=True)
    except Exception:  # tokenizer without a chat template
        text = args.prompt
    ids = torch.tensor([tokenizer.encode(text)], device=args.device)

    detok = StreamingDetokenizer(tokenizer)
    stats = GenerationStats()
    eos = [tokenizer.eos_token_id] if tokenizer.eos_token_id is not None else []
    print("=" * 10)
    for tid, _ in stream_generate(ids, model, remotes,
                                  max_tokens=args.max_tokens,
                                  params=SamplingParams(temperature=args.temp),
                                  eos_token_ids=eos, stats=stats):
        print(detok.add_token(tid), end="", flush=True)
    print(detok.finalize())
    print("=" * 10)
    print(f"Prompt: {stats.prompt_tps:.3f} tokens-per-sec")
    print(f"Generation: {stats.generation_tps:.3f} tokens-per-sec")


if __name__ == "__main__":
    main()
