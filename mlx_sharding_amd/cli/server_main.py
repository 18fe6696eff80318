"""`mlx-sharding-server` CLI — parity with /root/reference/shard/main.py:4-14."""

from __future__ import annotations

import argparse

import torch


def main():
    p = argparse.ArgumentParser(description="Start a pipeline shard server")
    p.add_argument("--model", type=str, required=True,
                   help="model checkpoint directory (full or pre-sharded)")
    p.add_argument("-s", "--start-layer", type=int, default=None)
    p.add_argument("-e", "--end-layer", type=int, default=None)
    p.add_argument("--port", type=int, default=0,
                   help="listen port (0 = auto-assign and print)")
    p.add_argument("--device", type=str,
                   default="cuda" if torch.cuda.is_available() else "cpu")
    p.add_argument("--quantize", type=int, choices=[4, 8], default=None,
                   help="quantize a dense checkpoint to w4a16/w8a16 at load")
    p.add_argument("--quantize-group-size", type=int, default=64)
    args = p.parse_args()

    from ..server.shard_server import serve
    q = (args.quantize, args.quantize_group_size) if args.quantize else None
    serve(args.model, args.start_layer, args.end_layer, port=args.port,
          device=args.device, quantize=q)


if __name__ == "__main__":
    main()
