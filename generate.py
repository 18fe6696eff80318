#!/usr/bin/env python3
"""Root-level greedy generation CLI (parity with /root/reference/generate.py)."""
from mlx_sharding_amd.cli.generate import main

if __name__ == "__main__":
    main()
