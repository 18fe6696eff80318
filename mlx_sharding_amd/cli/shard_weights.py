"""Offline per-stage weight splitter CLI — parity with
/root/reference/sharding_weight.py:74-92 (same on-disk layout)."""

from __future__ import annotations

import argparse


def main(argv=None):
    p = argparse.ArgumentParser(description="Split a checkpoint into one stage's weights")
    # the reference names this flag --model (sharding_weight.py:77);
    # --model_path kept as an alias
    p.add_argument("--model", "--model_path", "--model-path",
                   dest="model_path", type=str, required=True)
    p.add_argument("--output_dir", "--output-dir", type=str, required=True)
    p.add_argument("--start_layer", "--start-layer", type=int, required=True)
    p.add_argument("--end_layer", "--end-layer", type=int, required=True)
    # accepted for command-line compatibility (the reference requires it,
    # sharding_weight.py:85; we read the layer count from config.json)
    p.add_argument("--total_layers", "--total-layers", type=int, default=None)
    args = p.parse_args(argv)

    from ..utils.loading import save_sharded_weights
    out = save_sharded_weights(args.model_path, args.output_dir,
                               args.start_layer, args.end_layer)
    print(f"wrote {out}")


if __name__ == "__main__":
    main()
