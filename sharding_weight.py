#!/usr/bin/env python3
"""Root-level weight splitter CLI (parity with /root/reference/sharding_weight.py)."""
from mlx_sharding_amd.cli.shard_weights import main

if __name__ == "__main__":
    main()
