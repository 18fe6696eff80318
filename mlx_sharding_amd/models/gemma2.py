"""Gemma-2 stage model.

Parity quirks preserved from /root/reference/shard/server/model/gemma2.py:
input scaling h *= sqrt(hidden_size) (:43), final-logit softcapping (:82-84),
tied output head = embedding so the *last* shard also instantiates
embed_tokens (:23-24).  Plus gemma2 architecture specifics: (1+w) RMSNorm,
pre/post sandwich norms, attn-logit softcapping, sliding-window attention
on even layers, gelu-tanh MLP.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..config import ModelConfig, ShardSpec
from ..ops.kvcache import KVCache
from .base import Linear, RMSNorm, StageModel, owned_layer_indices
from .llama import _Inner


class Gemma2Attention(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str, layer_idx: int):
        super().__init__()
        H = cfg.hidden_size
        self.n_heads = cfg["num_attention_heads"]
        self.n_kv_heads = cfg.get("num_key_value_heads", self.n_heads)
        self.head_dim = cfg.get("head_dim") or H // self.n_heads
        self.scale = float(cfg.get("query_pre_attn_scalar", self.head_dim)) ** -0.5
        self.softcap = float(cfg.get("attn_logit_softcapping") or 0.0)
        # sliding window on even layers (HF Gemma2 convention)
        self.sliding_window = int(cfg.get("sliding_window", 4096)) if layer_idx % 2 == 0 else 0
        q = lambda name: quant_for(f"{prefix}.{name}")
        self.q_proj = Linear(H, self.n_heads * self.head_dim, q("q_proj"))
        self.k_proj = Linear(H, self.n_kv_heads * self.head_dim, q("k_proj"))
        self.v_proj = Linear(H, self.n_kv_heads * self.head_dim, q("v_proj"))
        self.o_proj = Linear(self.n_heads * self.head_dim, H, q("o_proj"))
        self._fused_qkv = None  # set by fuse_model()

    def forward(self, x, cos, sin, cache: Optional[KVCache]):
        B, T, _ = x.shape
        if self._fused_qkv is not None:
            q, k, v = self._fused_qkv(x)
        else:
            q, k, v = self.q_proj(x), self.k_proj(x), self.v_proj(x)
        q = q.view(B, T, self.n_heads, self.head_dim)
        k = k.view(B, T, self.n_kv_heads, self.head_dim)
        v = v.view(B, T, self.n_kv_heads, self.head_dim)
        q = ops.apply_rope(q, cos, sin).transpose(1, 2)
        offset = 0
        gp = None
        if cache is not None and ops.use_native(x):
            # fused rope-k + cache scatter (see llama.py)
            gp = cache.graph_pos
            if gp is None:
                offset = cache.offset
            k, v = cache.append_rope_kv(k, v, cos, sin)
        else:
            k = ops.apply_rope(k, cos, sin).transpose(1, 2)
            v = v.transpose(1, 2)
            if cache is not None:
                gp = cache.graph_pos
                if gp is None:
                    offset = cache.offset
                k, v = cache.update(k, v)
        out = ops.attention(q, k, v, self.scale, causal_offset=offset,
                            softcap=self.softcap,
                            sliding_window=self.sliding_window, pos_dev=gp)
        return self.o_proj(out.transpose(1, 2).reshape(B, T, -1))


class Gemma2MLP(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str):
        super().__init__()
        H = cfg.hidden_size
        I = cfg["intermediate_size"]
        q = lambda name: quant_for(f"{prefix}.{name}")
        self.gate_proj = Linear(H, I, q("gate_proj"))
        self.up_proj = Linear(H, I, q("up_proj"))
        self.down_proj = Linear(I, H, q("down_proj"))
        self._fused_gu = None  # set by fuse_model()

    def forward(self, x):
        if self._fused_gu is not None:
            g, u = self._fused_gu(x)
        else:
            g, u = self.gate_proj(x), self.up_proj(x)
        return self.down_proj(ops.geglu(g, u))


class Gemma2DecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str, layer_idx: int):
        super().__init__()
        H = cfg.hidden_size
        eps = cfg.get("rms_norm_eps", 1e-6)
        self.self_attn = Gemma2Attention(cfg, quant_for, f"{prefix}.self_attn", layer_idx)
        self.mlp = Gemma2MLP(cfg, quant_for, f"{prefix}.mlp")
        self.input_layernorm = RMSNorm(H, eps, weight_offset=1.0)
        self.post_attention_layernorm = RMSNorm(H, eps, weight_offset=1.0)
        self.pre_feedforward_layernorm = RMSNorm(H, eps, weight_offset=1.0)
        self.post_feedforward_layernorm = RMSNorm(H, eps, weight_offset=1.0)

    def forward(self, x, cos, sin, cache):
        h = x + self.post_attention_layernorm(
            self.self_attn(self.input_layernorm(x), cos, sin, cache))
        return h + self.post_feedforward_layernorm(
            self.mlp(self.pre_feedforward_layernorm(h)))


class Gemma2StageModel(StageModel):
    model_type = "gemma2"

    def __init__(self, config: ModelConfig, shard: ShardSpec, quant_for=None):
        super().__init__(config, shard)
        quant_for = quant_for or (lambda prefix: None)
        H = config.hidden_size
        self.model = _Inner()
        # gemma2 quirk: last shard needs embeds too (tied output head,
        # /root/reference/shard/server/model/gemma2.py:23-24)
        if shard.is_first or shard.is_last:
            self.model.embed_tokens = nn.Embedding(config.vocab_size, H,
                                                   dtype=torch.bfloat16)
        layers = nn.ModuleDict()
        for i in owned_layer_indices(shard):
            layers[str(i)] = Gemma2DecoderLayer(config, quant_for, f"model.layers.{i}", i)
        self.model.layers = layers
        if shard.is_last:
            self.model.norm = RMSNorm(H, config.get("rms_norm_eps", 1e-6),
                                      weight_offset=1.0)
        self.final_softcap = float(config.get("final_logit_softcapping") or 0.0)
        head_dim = config.get("head_dim") or H // config["num_attention_heads"]
        inv = ops.rope_freqs(head_dim, float(config.get("rope_theta", 10000.0)))
        self.register_buffer("rope_inv_freq", inv, persistent=False)
        # bf16-rounded sqrt(H) normalizer as a buffer: creating it per
        # forward would be an uncapturable H2D copy under hipGraph
        self.register_buffer(
            "embed_scale",
            torch.tensor(config.hidden_size ** 0.5, dtype=torch.bfloat16),
            persistent=False)

    @classmethod
    def owns_key(cls, key: str, shard: ShardSpec) -> bool:
        if key.startswith("model.embed_tokens"):
            return shard.is_first or shard.is_last
        return StageModel.owns_key.__func__(cls, key, shard)

    def cache_specs(self) -> List[Tuple[int, int, int]]:
        cfg = self.config
        hd = cfg.get("head_dim") or cfg.hidden_size // cfg["num_attention_heads"]
        nkv = cfg.get("num_key_value_heads", cfg["num_attention_heads"])
        return [(nkv, hd, hd) for _ in range(self.shard.n_layers)]

    def forward(self, x: torch.Tensor, cache: Optional[List[KVCache]] = None) -> torch.Tensor:
        if self.shard.is_first:
            h = self.model.embed_tokens(x)
            h = h * self.embed_scale.to(h.dtype)
        else:
            h = x
        T = h.shape[1]
        cos, sin, _ = self.rope_for(cache[0] if cache else None, T, h.device)
        for j, i in enumerate(owned_layer_indices(self.shard)):
            c = cache[j] if cache is not None else None
            h = self.model.layers[str(i)](h, cos, sin, c)
        if self.shard.is_last:
            h = self.model.norm(h)
            h = h @ self.model.embed_tokens.weight.t()
            if self.final_softcap > 0:
                h = ops.softcap(h, self.final_softcap)
        return h
