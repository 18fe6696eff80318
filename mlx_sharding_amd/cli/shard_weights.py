"""Offline per-stage weight splitter CLI — parity with
/root/reference/sharding_weight.py:74-92 (same on-disk layout)."""

from __future__ import annotations

import argparse


def main(argv=None):
    p = argparse.ArgumentParser(description="Split a checkpoint into one stage's weights")
    p.add_argument("--model_path", "--model-path", type=str, required=True)
    p.add_argument("--output_dir", "--output-dir", type=str, required=True)
    p.add_argument("--start_layer", "--start-layer", type=int, required=True)
    p.add_argument("--end_layer", "--end-layer", type=int, required=True)
    args = p.parse_args(argv)

    from ..utils.loading import save_sharded_weights
    out = save_sharded_weights(args.model_path, args.output_dir,
                               args.start_layer, args.end_layer)
    print(f"wrote {out}")


if __name__ == "__main__":
    main()
