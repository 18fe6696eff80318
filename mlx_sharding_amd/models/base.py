"""Shared building blocks for sharded stage models.

The sharding pattern mirrors the reference (e.g.
/root/reference/shard/server/model/llama.py:26-36): a stage owns a
contiguous [start, end) layer range; the first stage owns embed_tokens,
the last owns the final norm + lm_head.  Unlike the reference we do NOT
materialize IdentityBlock placeholders — layer modules are keyed by
their *global* index in a ModuleDict so weight keys stay globally
consistent without num_hidden_layers dummy entries.
"""

from __future__ import annotations

from typing import Callable, Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..config import ModelConfig, QuantConfig, ShardSpec
from ..ops.kvcache import KVCache

QuantPredicate = Callable[[str], bool]


class Linear(nn.Module):
    """Dense or MLX-affine-quantized linear, chosen per checkpoint key.

    Dense path → F.linear (hipBLASLt on ROCm).  Quantized path → the
    w4a16/w8a16 dequant-GEMM op.  Parameter names (weight/scales/biases)
    match the MLX checkpoint layout so pre-quantized checkpoints load
    directly (/root/reference/shard/utils.py:54-65 predicate: a
    `.scales` key exists for the module).
    """

    def __init__(self, in_features: int, out_features: int,
                 quant: Optional[QuantConfig] = None, bias: bool = False,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.quant = quant
        if quant is None:
            self.weight = nn.Parameter(
                torch.empty(out_features, in_features, dtype=dtype), requires_grad=False)
        else:
            per_word = 32 // quant.bits
            wdtype = torch.uint32 if hasattr(torch, "uint32") else torch.int32
            self.weight = nn.Parameter(
                torch.empty(out_features, in_features // per_word, dtype=wdtype),
                requires_grad=False)
            self.scales = nn.Parameter(
                torch.empty(out_features, in_features // quant.group_size, dtype=dtype),
                requires_grad=False)
            self.biases = nn.Parameter(
                torch.empty(out_features, in_features // quant.group_size, dtype=dtype),
                requires_grad=False)
        if bias:
            self.bias = nn.Parameter(torch.empty(out_features, dtype=dtype),
                                     requires_grad=False)
        else:
            self.bias = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.quant is None:
            return ops.linear(x, self.weight, self.bias)
        y = ops.quantized_linear(x, self.weight, self.scales, self.biases,
                                 self.quant.group_size, self.quant.bits)
        if self.bias is not None:
            y = y + self.bias
        return y


class RMSNorm(nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5, weight_offset: float = 0.0,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        self.eps = eps
        self.weight_offset = weight_offset
        self.weight = nn.Parameter(torch.ones(dim, dtype=dtype), requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rms_norm(x, self.weight, self.eps, self.weight_offset)


class StageModel(nn.Module):
    """Base class for one pipeline stage of a sharded model.

    Subclasses set up, inside ``self.model`` (an inner module named so
    state-dict keys match the checkpoint's ``model.*`` prefix):
      - ``embed_tokens`` when the shard owns it
      - ``layers``: ModuleDict {str(global_idx): DecoderLayer}
      - ``norm`` when last
    and ``self.lm_head`` when last (or tie to embeddings).
    """

    rope_attn_scale: float = 1.0

    def __init__(self, config: ModelConfig, shard: ShardSpec):
        super().__init__()
        self.config = config
        self.shard = shard
        self._rt_len = 0
        self._rt_dev = None

    # -- RoPE tables -------------------------------------------------------
    def _rope_tables(self, device, min_len: int):
        """Host-precomputed cos/sin tables [N, D/2] fp32 (guide Appendix B:
        no per-element trig on device); grown lazily, device-resident."""
        if self._rt_len < min_len or self._rt_dev != device:
            n = max(1024, 1 << max(int(min_len) - 1, 1).bit_length())
            pos = torch.arange(n, device=device)
            inv = self.rope_inv_freq.to(device)
            from .. import ops as _ops
            cos, sin = _ops.rope_cos_sin(pos, inv, self.rope_attn_scale)
            self._rt_cos = cos.contiguous()
            self._rt_sin = sin.contiguous()
            self._rt_len, self._rt_dev = n, device
        return self._rt_cos, self._rt_sin

    def rope_for(self, cache0, T: int, device):
        """(cos, sin, offset) for this forward; handles graph-pos mode
        (device position tensor) and the eager python-offset mode."""
        gp = cache0.graph_pos if cache0 is not None else None
        if gp is not None:
            cos_t, sin_t = self._rope_tables(device, cache0.capacity)
            idx = gp.to(torch.long)
            return cos_t.index_select(0, idx), sin_t.index_select(0, idx), 0
        offset = cache0.offset if cache0 is not None else 0
        cos_t, sin_t = self._rope_tables(device, offset + T)
        return cos_t[offset: offset + T], sin_t[offset: offset + T], offset

    @property
    def fp_dtype(self) -> torch.dtype:
        """Activation dtype = first floating parameter's dtype (a packed
        uint32 quant weight can come first on stages without embeddings
        — casting activations to THAT truncates them to garbage)."""
        return next((q.dtype for q in self.parameters()
                     if q.is_floating_point()), torch.bfloat16)

    # -- cache ------------------------------------------------------------
    def cache_specs(self) -> List[Tuple[int, int, int]]:
        """(n_kv_heads, k_head_dim, v_head_dim) for each *owned* layer."""
        raise NotImplementedError

    def make_cache(self, dtype: Optional[torch.dtype] = None, batch_size: int = 1,
                   device=None) -> List[KVCache]:
        # dtype from the first FLOATING parameter: on a quantized stage
        # without embeddings the first parameter is a packed uint32
        # weight, and a uint32 KV cache silently truncates every K/V
        # value to zero (caught as PP-parity divergence)
        p = next(self.parameters())
        fp = next((q for q in self.parameters() if q.is_floating_point()), p)
        from ..ops.kvcache import make_cache
        return make_cache(self.cache_specs(),
                          dtype=dtype or fp.dtype,
                          device=device if device is not None else p.device,
                          batch_size=batch_size)

    # -- weights ----------------------------------------------------------
    @classmethod
    def owns_key(cls, key: str, shard: ShardSpec) -> bool:
        """Key-routing rule — identical to the reference splitter
        (/root/reference/sharding_weight.py:17-24): layer keys by index
        range, embeddings on the first shard, final norm + lm_head on
        the last."""
        if key.startswith("model.layers."):
            idx = int(key.split(".")[2])
            return shard.owns(idx)
        if key.startswith("model.embed_tokens"):
            return shard.is_first
        if key.startswith("model.norm") or key.startswith("lm_head"):
            return shard.is_last
        return False

    @classmethod
    def sanitize(cls, weights: Dict[str, torch.Tensor], shard: ShardSpec) -> Dict[str, torch.Tensor]:
        """Filter a merged weight dict down to this shard's keys and drop
        buffers that are recomputed (rotary inv_freq), matching
        /root/reference/shard/server/model/llama.py:92-107."""
        return {k: v for k, v in weights.items()
                if cls.owns_key(k, shard) and "rotary_emb.inv_freq" not in k}

    def load_weights(self, weights: Dict[str, torch.Tensor], strict: bool = True):
        weights = self.sanitize(weights, self.shard)
        missing, unexpected = self.load_state_dict(weights, strict=False)
        # tied lm_head shows up as missing when the checkpoint ties embeddings
        missing = [m for m in missing if not self._key_optional(m)]
        if strict and (missing or unexpected):
            raise RuntimeError(
                f"weight mismatch: missing={missing[:8]} unexpected={unexpected[:8]}")
        return missing, unexpected

    def _key_optional(self, key: str) -> bool:
        return False

    # -- forward ----------------------------------------------------------
    def forward(self, x: torch.Tensor, cache: Optional[List[KVCache]] = None) -> torch.Tensor:
        """x: [B, T] int64 token ids on the first shard, else [B, T, H]
        hidden states.  Returns hidden states, or logits on the last shard."""
        raise NotImplementedError


def owned_layer_indices(shard: ShardSpec) -> List[int]:
    return list(range(shard.start_layer, shard.end_layer))
