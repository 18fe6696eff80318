"""Llama-family stage model (covers Mistral via the model_type remap).

Capability parity with /root/reference/shard/server/model/llama.py:
same shard semantics (embed on first, norm+lm_head on last, tied
embeddings via lm_head fallback), same key routing; compute runs on
our op layer (HIP kernels on GPU, torch reference on CPU).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..config import ModelConfig, ShardSpec
from ..ops.kvcache import KVCache
from .base import Linear, RMSNorm, StageModel, owned_layer_indices


class LlamaAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str):
        super().__init__()
        H = cfg.hidden_size
        self.n_heads = cfg["num_attention_heads"]
        self.n_kv_heads = cfg.get("num_key_value_heads", self.n_heads)
        self.head_dim = cfg.get("head_dim") or H // self.n_heads
        self.scale = self.head_dim ** -0.5
        # qwen2 (served through this class via MODEL_REMAPPING) hardcodes
        # QKV bias on / o_proj bias off in its HF reference; llama-family
        # configs carry an explicit attention_bias flag for all four
        qkv_bias = bool(cfg.get("attention_bias", cfg.model_type == "qwen2"))
        o_bias = bool(cfg.get("attention_bias", False))
        q = lambda name: quant_for(f"{prefix}.{name}")
        self.q_proj = Linear(H, self.n_heads * self.head_dim, q("q_proj"), qkv_bias)
        self.k_proj = Linear(H, self.n_kv_heads * self.head_dim, q("k_proj"), qkv_bias)
        self.v_proj = Linear(H, self.n_kv_heads * self.head_dim, q("v_proj"), qkv_bias)
        self.o_proj = Linear(self.n_heads * self.head_dim, H, q("o_proj"), o_bias)
        self._fused_qkv = None  # set by fuse_model()

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                cache: Optional[KVCache]) -> torch.Tensor:
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
            # fused rope-k + scatter into the caches (one kernel instead
            # of rope + two index_copy launches)
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
                            pos_dev=gp)
        out = out.transpose(1, 2).reshape(B, T, -1)
        return self.o_proj(out)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str):
        super().__init__()
        H = cfg.hidden_size
        I = cfg["intermediate_size"]
        q = lambda name: quant_for(f"{prefix}.{name}")
        self.gate_proj = Linear(H, I, q("gate_proj"))
        self.up_proj = Linear(H, I, q("up_proj"))
        self.down_proj = Linear(I, H, q("down_proj"))
        self._fused_gu = None  # set by fuse_model()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._fused_gu is not None:
            g, u = self._fused_gu(x)
        else:
            g, u = self.gate_proj(x), self.up_proj(x)
        return self.down_proj(ops.swiglu(g, u))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str):
        super().__init__()
        eps = cfg.get("rms_norm_eps", 1e-5)
        self.self_attn = LlamaAttention(cfg, quant_for, f"{prefix}.self_attn")
        self.mlp = LlamaMLP(cfg, quant_for, f"{prefix}.mlp")
        self.input_layernorm = RMSNorm(cfg.hidden_size, eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, eps)

    def forward(self, x, cos, sin, cache):
        attn = self.self_attn(self.input_layernorm(x), cos, sin, cache)
        # fused: h = x + attn; y = rms_norm(h) in one kernel
        pn = self.post_attention_layernorm
        y, h = ops.rms_norm_residual(attn, x, pn.weight, pn.eps,
                                     pn.weight_offset)
        return h + self.mlp(y)


class _Inner(nn.Module):
    pass


class LlamaStageModel(StageModel):
    model_type = "llama"

    def __init__(self, config: ModelConfig, shard: ShardSpec, quant_for=None):
        super().__init__(config, shard)
        quant_for = quant_for or (lambda prefix: None)
        H = config.hidden_size
        self.model = _Inner()
        if shard.is_first:
            self.model.embed_tokens = nn.Embedding(config.vocab_size, H,
                                                   dtype=torch.bfloat16)
        layers = nn.ModuleDict()
        for i in owned_layer_indices(shard):
            layers[str(i)] = LlamaDecoderLayer(config, quant_for, f"model.layers.{i}")
        self.model.layers = layers
        self.tie_word_embeddings = bool(config.get("tie_word_embeddings", False))
        if shard.is_last:
            self.model.norm = RMSNorm(H, config.get("rms_norm_eps", 1e-5))
            if not self.tie_word_embeddings:
                self.lm_head = Linear(H, config.vocab_size, quant_for("lm_head"))
            elif not shard.is_first:
                # tied head needs the embedding table on the last shard too
                self.model.embed_tokens = nn.Embedding(config.vocab_size, H,
                                                       dtype=torch.bfloat16)
        head_dim = config.get("head_dim") or H // config["num_attention_heads"]
        inv = ops.rope_freqs(head_dim, float(config.get("rope_theta", 10000.0)),
                             config.get("rope_scaling"))
        self.register_buffer("rope_inv_freq", inv, persistent=False)

    def _key_optional(self, key: str) -> bool:
        # a tied checkpoint may or may not carry lm_head.weight
        return key.startswith("lm_head") and self.tie_word_embeddings

    def cache_specs(self) -> List[Tuple[int, int, int]]:
        cfg = self.config
        hd = cfg.get("head_dim") or cfg.hidden_size // cfg["num_attention_heads"]
        nkv = cfg.get("num_key_value_heads", cfg["num_attention_heads"])
        return [(nkv, hd, hd) for _ in range(self.shard.n_layers)]

    def forward(self, x: torch.Tensor, cache: Optional[List[KVCache]] = None) -> torch.Tensor:
        if self.shard.is_first:
            h = self.model.embed_tokens(x)
        else:
            h = x
        T = h.shape[1]
        cos, sin, _ = self.rope_for(cache[0] if cache else None, T, h.device)
        for j, i in enumerate(owned_layer_indices(self.shard)):
            c = cache[j] if cache is not None else None
            h = self.model.layers[str(i)](h, cos, sin, c)
        if self.shard.is_last:
            h = self.model.norm(h)
            if self.tie_word_embeddings:
                h = h @ self.model.embed_tokens.weight.t()
            else:
                h = self.lm_head(h)
        return h
