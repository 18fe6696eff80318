"""Per-layer KV cache: preallocated contiguous, grow-by-chunk.

Layout [B, n_kv_heads, S, D] with independent K and V head dims (MLA
stores K at qk_nope+qk_rope=192 and V at 128 — the tuple head_dim the
reference exposes at /root/reference/shard/server/model/deepseek_v2.py:120-125).

The reference's cache lifecycle: one fresh cache per
generation request, exactly one multi-token prefill then T=1 decode
steps.  Sessions own their caches here (no global mutable CACHE).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch


class KVCache:
    CHUNK = 1024

    def __init__(self, n_kv_heads: int, k_head_dim: int, v_head_dim: int,
                 dtype: torch.dtype = torch.bfloat16,
                 device: torch.device | str = "cpu",
                 batch_size: int = 1):
        self.n_kv_heads = n_kv_heads
        self.k_head_dim = k_head_dim
        self.v_head_dim = v_head_dim
        # an integer cache dtype silently truncates K/V on copy (torch
        # casts without complaint) — fail loudly instead
        if not dtype.is_floating_point:
            raise TypeError(f"KV cache dtype must be floating, got {dtype}")
        self.dtype = dtype
        self.device = torch.device(device)
        self.batch_size = batch_size
        self.offset = 0
        self._k: Optional[torch.Tensor] = None
        self._v: Optional[torch.Tensor] = None
        # hipGraph-captured decode: position lives on device (int32 [1]);
        # update() appends by index_copy_ and python offset is not advanced.
        self.graph_pos: Optional[torch.Tensor] = None

    def _ensure(self, batch: int, needed: int):
        cap = 0 if self._k is None else self._k.shape[2]
        if self._k is not None and self._k.shape[0] != batch:
            raise ValueError(f"KV cache batch mismatch: {self._k.shape[0]} vs {batch}")
        if needed <= cap and self._k is not None:
            return
        new_cap = ((needed + self.CHUNK - 1) // self.CHUNK) * self.CHUNK
        nk = torch.empty(batch, self.n_kv_heads, new_cap, self.k_head_dim,
                         dtype=self.dtype, device=self.device)
        nv = torch.empty(batch, self.n_kv_heads, new_cap, self.v_head_dim,
                         dtype=self.dtype, device=self.device)
        if self._k is not None and self.offset > 0:
            nk[:, :, : self.offset] = self._k[:, :, : self.offset]
            nv[:, :, : self.offset] = self._v[:, :, : self.offset]
        self._k, self._v = nk, nv

    def update(self, k: torch.Tensor, v: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        """Append k/v [B, n_kv, T, D]; return full views [B, n_kv, offset+T, D].

        In graph mode (graph_pos set) the append is an index_copy_ at the
        device position and the FULL buffers are returned — the caller
        passes graph_pos to the attention op, which reads the length on
        device (capture-safe: no python-int shapes)."""
        if self.graph_pos is not None:
            idx = self.graph_pos.to(torch.long)
            self._k.index_copy_(2, idx, k)
            self._v.index_copy_(2, idx, v)
            return self._k, self._v
        B, H, T, _ = k.shape
        self._ensure(B, self.offset + T)
        self._k[:, :, self.offset: self.offset + T] = k
        self._v[:, :, self.offset: self.offset + T] = v
        self.offset += T
        return (self._k[:, :, : self.offset], self._v[:, :, : self.offset])

    def append_mla(self, kvh: torch.Tensor, kpe: torch.Tensor):
        """Fused MLA append: kvh [B, T, nh, nope+vd] (kv_b output) and
        roped kpe [B, T, rope] scatter straight into the caches — no
        cat/head-expand/index_copy round trips.  GPU-only (the HIP
        extension provides the kernel); returns (k, v) like update()."""
        from .. import ops as _ops
        ext = _ops.hip_ext()
        B, T = kvh.shape[0], kvh.shape[1]
        if self.graph_pos is not None:
            ext.mla_append_kv(kvh, kpe, self._k, self._v,
                              pos=self.graph_pos)
            return self._k, self._v
        self._ensure(B, self.offset + T)
        ext.mla_append_kv(kvh, kpe, self._k, self._v, pos0=self.offset)
        self.offset += T
        return (self._k[:, :, : self.offset], self._v[:, :, : self.offset])

    def append_rope_kv(self, k: torch.Tensor, v: torch.Tensor,
                       cos: torch.Tensor, sin: torch.Tensor,
                       interleaved: bool = False):
        """Fused GQA append: RoPE the new k rows [B, T, Hkv, D] and
        scatter k/v straight into the caches (replaces a rope launch +
        two index_copy launches).  GPU-only; returns (k, v) like
        update()."""
        from .. import ops as _ops
        ext = _ops.hip_ext()
        B, T = k.shape[0], k.shape[1]
        if self.graph_pos is not None:
            ext.rope_append_kv(k, v, cos, sin, self._k, self._v,
                               pos=self.graph_pos, interleaved=interleaved)
            return self._k, self._v
        self._ensure(B, self.offset + T)
        ext.rope_append_kv(k, v, cos, sin, self._k, self._v,
                           pos0=self.offset, interleaved=interleaved)
        self.offset += T
        return (self._k[:, :, : self.offset], self._v[:, :, : self.offset])

    @property
    def k(self) -> Optional[torch.Tensor]:
        return None if self._k is None else self._k[:, :, : self.offset]

    @property
    def v(self) -> Optional[torch.Tensor]:
        return None if self._v is None else self._v[:, :, : self.offset]

    def keys_buffer(self) -> Optional[torch.Tensor]:
        """Full backing buffer (for the HIP decode kernel, which indexes by offset)."""
        return self._k

    def values_buffer(self) -> Optional[torch.Tensor]:
        return self._v

    @property
    def capacity(self) -> int:
        return 0 if self._k is None else self._k.shape[2]

    def ensure_capacity(self, cap: int, batch: Optional[int] = None):
        b = batch if batch is not None else (
            self._k.shape[0] if self._k is not None else self.batch_size)
        self._ensure(b, cap)

    def trim(self, n: int):
        """Roll back to the first ``n`` cached positions (prefix-cache
        reuse: a follow-up request whose prompt shares an n-token prefix
        with the previous generation re-uses those K/V rows and only
        prefills the tail)."""
        if n < 0 or n > self.offset:
            raise ValueError(f"trim({n}) outside [0, {self.offset}]")
        self.offset = n
        self.graph_pos = None

    def reset(self):
        self.offset = 0
        self.graph_pos = None


def make_cache(layer_specs: List[Tuple[int, int, int]], dtype: torch.dtype,
               device, batch_size: int = 1) -> List[KVCache]:
    """Build one KVCache per (n_kv_heads, k_dim, v_dim) layer spec."""
    return [KVCache(h, kd, vd, dtype=dtype, device=device, batch_size=batch_size)
            for (h, kd, vd) in layer_specs]
