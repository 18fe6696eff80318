"""DeepSeek-V2 stage model: MLA attention + MoE with stacked expert weights.

Parity with /root/reference/shard/server/model/deepseek_v2.py: same shard
semantics, same stacked `mlp.switch_mlp.*` expert-weight layout produced
by its sanitize (:101-112) so pre-stacked checkpoints interoperate, tuple
(K, V) head dims exposed for the cache (:120-125), YaRN-scaled partial
RoPE on the qk_rope slice with mscale-adjusted softmax scale.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from .. import ops
from ..config import ModelConfig, QuantConfig, ShardSpec
from ..ops.kvcache import KVCache
from .base import Linear, RMSNorm, StageModel, owned_layer_indices
from .llama import LlamaMLP, _Inner


class SwitchMLP(nn.Module):
    """Stacked routed experts: gate/up [E, I, H], down [E, H, I].

    Parameter names match the reference's stacked layout
    (`mlp.switch_mlp.{gate,up,down}_proj.weight[/scales/biases]`).
    """

    def __init__(self, n_experts: int, hidden: int, inter: int,
                 quant: Optional[QuantConfig] = None,
                 dtype: torch.dtype = torch.bfloat16):
        super().__init__()
        self.n_experts = n_experts
        self.quant = quant
        self.gate_proj = _StackedLinear(n_experts, hidden, inter, quant, dtype)
        self.up_proj = _StackedLinear(n_experts, hidden, inter, quant, dtype)
        self.down_proj = _StackedLinear(n_experts, inter, hidden, quant, dtype)

    def resident_dense(self):
        """Budget-capped resident bf16 expert copies for quant decode
        (ops.expert_dequant_resident; None when memory-tight)."""
        if self.quant is None:
            return None
        g, u, d = self.gate_proj, self.up_proj, self.down_proj
        return ops.expert_dequant_resident(
            (g.weight, g.scales, g.biases),
            (u.weight, u.scales, u.biases),
            (d.weight, d.scales, d.biases),
            self.quant.group_size, self.quant.bits)

    def forward_subs(self, x_flat: torch.Tensor, subs,
                     max_tok: int = 4) -> torch.Tensor:
        """Run from prebuilt fused-gating sub-range arrays (GPU decode)."""
        if self.quant is not None:
            # the fp16-dequant packed MFMA kernels (moe_w4f16.hip) beat
            # streaming resident bf16 copies (4x less weight traffic,
            # measured 5334 vs 5005 tok/s end to end) — packed is the
            # decode default; MLXS_AMD_MOE_RESIDENT=1 restores the old
            # resident-bf16 path for A/B
            import os
            if os.environ.get("MLXS_AMD_MOE_RESIDENT"):
                dense = self.resident_dense()
                if dense is not None:
                    return ops.grouped_expert_mlp_subs(
                        x_flat, dense[0], dense[1], dense[2], subs, max_tok)
            g, u, d = self.gate_proj, self.up_proj, self.down_proj
            return ops.grouped_expert_mlp_quant_subs(
                x_flat,
                (g.weight, g.scales, g.biases),
                (u.weight, u.scales, u.biases),
                (d.weight, d.scales, d.biases),
                subs, self.quant.group_size, self.quant.bits)
        return ops.grouped_expert_mlp_subs(
            x_flat, self.gate_proj.weight, self.up_proj.weight,
            self.down_proj.weight, subs, max_tok)

    def forward(self, x_flat: torch.Tensor, weights: torch.Tensor,
                indices: torch.Tensor) -> torch.Tensor:
        if self.quant is not None:
            g, u, d = self.gate_proj, self.up_proj, self.down_proj
            return ops.grouped_expert_mlp_quant(
                x_flat,
                (g.weight, g.scales, g.biases),
                (u.weight, u.scales, u.biases),
                (d.weight, d.scales, d.biases),
                weights, indices, self.quant.group_size, self.quant.bits)
        return ops.grouped_expert_mlp(
            x_flat, self.gate_proj.weight, self.up_proj.weight,
            self.down_proj.weight, weights, indices)


class _StackedLinear(nn.Module):
    """Holds stacked per-expert weights [E, out, in] (+ quant triplet)."""

    def __init__(self, E: int, in_features: int, out_features: int,
                 quant: Optional[QuantConfig], dtype: torch.dtype):
        super().__init__()
        self.quantc = quant
        if quant is None:
            self.weight = nn.Parameter(
                torch.empty(E, out_features, in_features, dtype=dtype),
                requires_grad=False)
        else:
            per_word = 32 // quant.bits
            wdtype = torch.uint32 if hasattr(torch, "uint32") else torch.int32
            self.weight = nn.Parameter(
                torch.empty(E, out_features, in_features // per_word, dtype=wdtype),
                requires_grad=False)
            self.scales = nn.Parameter(
                torch.empty(E, out_features, in_features // quant.group_size, dtype=dtype),
                requires_grad=False)
            self.biases = nn.Parameter(
                torch.empty(E, out_features, in_features // quant.group_size, dtype=dtype),
                requires_grad=False)

    def dense(self) -> torch.Tensor:
        if self.quantc is None:
            return self.weight
        outs = [ops.dequantize(self.weight[e], self.scales[e], self.biases[e],
                               self.quantc.group_size, self.quantc.bits)
                for e in range(self.weight.shape[0])]
        return torch.stack(outs)


class MLAAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str):
        super().__init__()
        H = cfg.hidden_size
        self.n_heads = cfg["num_attention_heads"]
        self.qk_nope = int(cfg.get("qk_nope_head_dim", 128))
        self.qk_rope = int(cfg.get("qk_rope_head_dim", 64))
        self.v_head_dim = int(cfg.get("v_head_dim", 128))
        self.kv_lora_rank = int(cfg.get("kv_lora_rank", 512))
        self.q_lora_rank = cfg.get("q_lora_rank")
        self.qk_head_dim = self.qk_nope + self.qk_rope
        q = lambda name: quant_for(f"{prefix}.{name}")
        eps = cfg.get("rms_norm_eps", 1e-6)
        if self.q_lora_rank:
            self.q_a_proj = Linear(H, self.q_lora_rank, q("q_a_proj"))
            self.q_a_layernorm = RMSNorm(self.q_lora_rank, eps)
            self.q_b_proj = Linear(self.q_lora_rank, self.n_heads * self.qk_head_dim,
                                   q("q_b_proj"))
        else:
            self.q_proj = Linear(H, self.n_heads * self.qk_head_dim, q("q_proj"))
        self._fused_qkv = None  # set by fuse_model()
        self.kv_a_proj_with_mqa = Linear(H, self.kv_lora_rank + self.qk_rope,
                                         q("kv_a_proj_with_mqa"))
        self.kv_a_layernorm = RMSNorm(self.kv_lora_rank, eps)
        self.kv_b_proj = Linear(self.kv_lora_rank,
                                self.n_heads * (self.qk_nope + self.v_head_dim),
                                q("kv_b_proj"))
        self.o_proj = Linear(self.n_heads * self.v_head_dim, H, q("o_proj"))

        rs = cfg.get("rope_scaling") or {}
        mscale_all = float(rs.get("mscale_all_dim", 0.0) or 0.0)
        factor = float(rs.get("factor", 1.0))
        scale = self.qk_head_dim ** -0.5
        if mscale_all and factor > 1.0:
            scale = scale * ops.yarn_mscale(factor, mscale_all) ** 2
        self.scale = scale
        # weight-absorbed decode (lazily built from kv_b_proj): W_UK
        # [nh, nope, rank] folds K up-projection into q; W_UV
        # [nh, rank, vd] folds V up-projection out of the attention
        # output, so decode attends over the COMPRESSED cache
        # (kv_lora_rank + qk_rope per token instead of
        # nh*(nope+rope) + nh*vd — 8.9x less KV traffic for V2-Lite).
        self._w_uk = None
        self._w_uv = None

    def absorbed_supported(self) -> bool:
        """True when the HIP decode kernel handles the absorbed shape
        (G = n_heads MQA over Dk = rank+rope, Dv = rank)."""
        ext = ops.hip_ext()
        if ext is None:
            return False
        dk = self.kv_lora_rank + self.qk_rope
        return (dk % 8 == 0 and self.kv_lora_rank % 8 == 0
                and ext.attn_decode_shape_ok(self.n_heads, self.kv_lora_rank))

    def _ensure_absorb(self, device, dtype):
        if self._w_uk is not None:
            return
        w = self.kv_b_proj.weight
        if self.kv_b_proj.quant is not None:
            qc = self.kv_b_proj.quant
            w = ops.dequantize(w, self.kv_b_proj.scales, self.kv_b_proj.biases,
                               qc.group_size, qc.bits)
        w = w.to(device=device, dtype=dtype)
        w = w.view(self.n_heads, self.qk_nope + self.v_head_dim,
                   self.kv_lora_rank)
        self._w_uk = w[:, : self.qk_nope, :].contiguous()
        self._w_uv = w[:, self.qk_nope:, :].transpose(1, 2).contiguous()

    def forward(self, x, cos, sin, cache: Optional[KVCache]):
        if cache is not None and cache.n_kv_heads == 1 and self.n_heads > 1:
            return self._forward_absorbed(x, cos, sin, cache)
        return self._forward_naive(x, cos, sin, cache)

    def _forward_absorbed(self, x, cos, sin, cache: KVCache):
        """Compressed-cache path: cache rows are [c_kv | k_pe] (rank +
        rope).  Decode = MQA over the compressed rows with q/out
        absorbed through W_UK/W_UV; prefill expands K/V from the
        compressed prefix with kv_b_proj and runs the MFMA prefill
        kernel (the standard MLA split: expanded MHA at prefill,
        absorbed MQA at decode)."""
        B, T, _ = x.shape
        nh, rank, rope = self.n_heads, self.kv_lora_rank, self.qk_rope
        ckv = None
        if self._fused_qkv is not None:
            qh, ckv = self._fused_qkv(x)
        elif self.q_lora_rank:
            qh = self.q_b_proj(self.q_a_layernorm(self.q_a_proj(x)))
        else:
            qh = self.q_proj(x)
        qh = qh.view(B, T, nh, self.qk_head_dim)
        q_nope = qh[..., : self.qk_nope]
        q_pe = ops.apply_rope(qh[..., self.qk_nope:], cos, sin,
                              interleaved=True)

        if ckv is None:
            ckv = self.kv_a_proj_with_mqa(x)
        c_kv, k_pe = ckv.split([rank, rope], dim=-1)
        c_kv = self.kv_a_layernorm(c_kv)
        k_pe = ops.apply_rope(k_pe.view(B, T, 1, rope), cos, sin,
                              interleaved=True).view(B, T, rope)

        gp = cache.graph_pos
        offset = 0 if gp is not None else cache.offset
        if ops.use_native(x):
            # fused scatter into the compressed cache (the mla_append
            # kernel with nh=1, vd=0: rows are [c_kv | k_pe]) — replaces
            # a cat + index_copy pair per layer
            k_all, _ = cache.append_mla(c_kv.reshape(B, T, 1, rank), k_pe)
        else:
            row = torch.cat([c_kv, k_pe], dim=-1).unsqueeze(1)
            k_all, _ = cache.update(row, row[..., :0])

        if T == 1:
            self._ensure_absorb(x.device, x.dtype)
            # q absorb: [nh, B, nope] @ [nh, nope, rank] -> [nh, B, rank]
            qn = q_nope.permute(2, 0, 1, 3).reshape(nh, B * T, self.qk_nope)
            q_abs = torch.bmm(qn, self._w_uk)           # [nh, B, rank]
            qf = torch.cat([q_abs.view(nh, B, T, rank).permute(1, 0, 2, 3),
                            q_pe.transpose(1, 2)], dim=-1)  # [B,nh,1,rank+rope]
            o = ops.attention(qf, k_all, k_all[..., :rank], self.scale,
                              causal_offset=offset, pos_dev=gp)  # [B,nh,1,rank]
            # out absorb: [nh, B, rank] @ [nh, rank, vd] -> [nh, B, vd]
            ov = torch.bmm(o.permute(1, 0, 2, 3).reshape(nh, B * T, rank),
                           self._w_uv)
            out = ov.view(nh, B, T, self.v_head_dim).permute(1, 2, 0, 3)
            return self.o_proj(out.reshape(B, T, -1))

        # prefill: expand the WHOLE compressed prefix (covers chunked
        # prefill, where earlier chunks live only in the cache)
        S = k_all.shape[2] if gp is not None else cache.offset
        ckv_all = k_all[:, 0, :S, :rank]
        kpe_all = k_all[:, 0, :S, rank:]
        kvh = self.kv_b_proj(ckv_all).view(B, S, nh,
                                           self.qk_nope + self.v_head_dim)
        ext = ops.hip_ext() if ops.use_native(x) else None
        if ext is not None:
            # fused expansion straight into cache-layout K/V temps: the
            # cat + head-expand + transpose + .contiguous() chain cost
            # ~30 ms/prefill of pure copies (torch-profiler attribution)
            k = kvh.new_empty(B, nh, S, self.qk_head_dim)
            v = kvh.new_empty(B, nh, S, self.v_head_dim)
            ext.mla_append_kv(kvh, kpe_all, k, v, pos0=0)
        else:
            k = torch.cat([kvh[..., : self.qk_nope],
                           kpe_all.reshape(B, S, 1, rope)
                           .expand(B, S, nh, rope)],
                          dim=-1).transpose(1, 2).contiguous()
            v = kvh[..., self.qk_nope:].transpose(1, 2).contiguous()
        qf = torch.cat([q_nope, q_pe], dim=-1).transpose(1, 2)
        out = ops.attention(qf, k, v, self.scale, causal_offset=offset)
        return self.o_proj(out.transpose(1, 2).reshape(B, T, -1))

    def _forward_naive(self, x, cos, sin, cache: Optional[KVCache]):
        B, T, _ = x.shape
        ckv = None
        if self._fused_qkv is not None:
            qh, ckv = self._fused_qkv(x)
        elif self.q_lora_rank:
            qh = self.q_b_proj(self.q_a_layernorm(self.q_a_proj(x)))
        else:
            qh = self.q_proj(x)
        qh = qh.view(B, T, self.n_heads, self.qk_head_dim)
        q_nope = qh[..., : self.qk_nope]
        q_pe = qh[..., self.qk_nope:]

        if ckv is None:
            ckv = self.kv_a_proj_with_mqa(x)
        c_kv, k_pe = ckv.split([self.kv_lora_rank, self.qk_rope], dim=-1)
        c_kv = self.kv_a_layernorm(c_kv)
        kvh = self.kv_b_proj(c_kv).view(B, T, self.n_heads, self.qk_nope + self.v_head_dim)

        # partial RoPE (interleaved / MLX-traditional convention)
        q_pe = ops.apply_rope(q_pe, cos, sin, interleaved=True)
        k_pe = ops.apply_rope(k_pe.view(B, T, 1, self.qk_rope), cos, sin,
                              interleaved=True)
        qf = torch.cat([q_nope, q_pe], dim=-1).transpose(1, 2)

        offset = 0
        gp = None
        if cache is not None and ops.use_native(x):
            # fused append: kv_b output + roped k_pe scatter straight
            # into the caches (no cat/expand/index_copy)
            gp = cache.graph_pos
            if gp is None:
                offset = cache.offset
            k, v = cache.append_mla(kvh, k_pe.reshape(B, T, self.qk_rope))
            if gp is None:
                pass  # views already sized to offset+T
        else:
            k_nope = kvh[..., : self.qk_nope]
            v = kvh[..., self.qk_nope:].transpose(1, 2)
            k = torch.cat([k_nope,
                           k_pe.expand(B, T, self.n_heads, self.qk_rope)],
                          dim=-1).transpose(1, 2)
            if cache is not None:
                gp = cache.graph_pos
                if gp is None:
                    offset = cache.offset
                k, v = cache.update(k, v)
        out = ops.attention(qf, k, v, self.scale, causal_offset=offset,
                            pos_dev=gp)
        return self.o_proj(out.transpose(1, 2).reshape(B, T, -1))


class DeepseekV2MoE(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str):
        super().__init__()
        H = cfg.hidden_size
        I_moe = int(cfg.get("moe_intermediate_size", 1408))
        self.n_experts = int(cfg.get("n_routed_experts", 64))
        self.top_k = int(cfg.get("num_experts_per_tok", 6))
        self.n_group = int(cfg.get("n_group", 1) or 1)
        self.topk_group = int(cfg.get("topk_group", 1) or 1)
        self.topk_method = cfg.get("topk_method", "greedy")
        self.routed_scaling_factor = float(cfg.get("routed_scaling_factor", 1.0))
        self.norm_topk_prob = bool(cfg.get("norm_topk_prob", False))
        self.gate = Linear(H, self.n_experts, None, dtype=torch.bfloat16)
        squant = quant_for(f"{prefix}.switch_mlp.gate_proj")
        self.switch_mlp = SwitchMLP(self.n_experts, H, I_moe, squant)
        n_shared = int(cfg.get("n_shared_experts", 0) or 0)
        if n_shared:
            shared_cfg = ModelConfig(cfg.model_type, dict(cfg.raw))
            shared_cfg.raw["intermediate_size"] = I_moe * n_shared
            self.shared_experts = LlamaMLP(shared_cfg, quant_for,
                                           f"{prefix}.shared_experts")
        else:
            self.shared_experts = None

    def _fused_gate_ok(self, n_tokens: int) -> bool:
        return (self.topk_method != "group_limited_greedy"
                and n_tokens <= 128 and self.n_experts <= 64
                and self.top_k <= 8)

    def forward(self, x):
        B, T, H = x.shape
        flat = x.reshape(-1, H)
        if (ops.use_native(flat)
                and self._fused_gate_ok(flat.shape[0])):
            # fused gating: one kernel for softmax+topk+sort+subranges
            # (32-token sub-ranges for the MFMA w4 kernels, 4 for bf16)
            logits = self.gate(flat.to(self.gate.weight.dtype))
            # 16-token sub-ranges for BOTH quant sub-paths: the bf16
            # MFMA kernels over resident copies, and the fp16-dequant
            # packed kernels (moe_w4f16.hip) in memory-tight mode
            mt = 16
            subs = ops.moe_gate_subranges(logits, self.top_k,
                                          self.routed_scaling_factor,
                                          self.norm_topk_prob, max_tok=mt)
            y = self.switch_mlp.forward_subs(flat, subs, mt)
        else:
            logits = self.gate(flat.to(self.gate.weight.dtype)).float()
            n_group = self.n_group if self.topk_method == "group_limited_greedy" else 1
            w, idx = ops.moe_gate(logits, self.top_k, n_group, self.topk_group,
                                  self.routed_scaling_factor, self.norm_topk_prob)
            w = w.to(x.dtype)
            y = self.switch_mlp(flat, w, idx)
        if self.shared_experts is not None:
            y = y + self.shared_experts(flat)
        return y.reshape(B, T, H)


class DeepseekV2DecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, quant_for, prefix: str, layer_idx: int):
        super().__init__()
        eps = cfg.get("rms_norm_eps", 1e-6)
        self.self_attn = MLAAttention(cfg, quant_for, f"{prefix}.self_attn")
        first_dense = int(cfg.get("first_k_dense_replace", 0))
        freq = int(cfg.get("moe_layer_freq", 1))
        is_moe = (cfg.get("n_routed_experts") is not None
                  and layer_idx >= first_dense and layer_idx % freq == 0)
        if is_moe:
            self.mlp = DeepseekV2MoE(cfg, quant_for, f"{prefix}.mlp")
        else:
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


class DeepseekV2StageModel(StageModel):
    model_type = "deepseek_v2"

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
            layers[str(i)] = DeepseekV2DecoderLayer(config, quant_for,
                                                    f"model.layers.{i}", i)
        self.model.layers = layers
        if shard.is_last:
            self.model.norm = RMSNorm(H, config.get("rms_norm_eps", 1e-6))
            self.lm_head = Linear(H, config.vocab_size, quant_for("lm_head"))
        inv = ops.rope_freqs(int(config.get("qk_rope_head_dim", 64)),
                             float(config.get("rope_theta", 10000.0)),
                             config.get("rope_scaling"))
        rs = config.get("rope_scaling") or {}
        factor = float(rs.get("factor", 1.0))
        mscale = float(rs.get("mscale", 1.0) or 1.0)
        mscale_all = float(rs.get("mscale_all_dim", 0.0) or 0.0)
        # HF/mlx_lm yarn scales cos/sin by the RATIO
        # yarn_mscale(factor, mscale) / yarn_mscale(factor, mscale_all_dim)
        # (DeepseekV2YarnRotaryEmbedding._mscale).  DeepSeek-V2 configs
        # have mscale == mscale_all_dim, making the ratio exactly 1.0;
        # applying the numerator alone scaled roped q·k components ~1.59x.
        if rs:
            num = ops.yarn_mscale(factor, mscale)
            den = ops.yarn_mscale(factor, mscale_all) if mscale_all else 1.0
            self.rope_attn_scale = num / den
        else:
            self.rope_attn_scale = 1.0
        self.register_buffer("rope_inv_freq", inv, persistent=False)

    @classmethod
    def sanitize(cls, weights: Dict[str, torch.Tensor], shard: ShardSpec) -> Dict[str, torch.Tensor]:
        """Stack per-expert `mlp.experts.{e}.*` into `mlp.switch_mlp.*`
        (the reference's layout, deepseek_v2.py:101-112), then filter."""
        out = dict(weights)
        # find expert keys: model.layers.N.mlp.experts.E.{gate,up,down}_proj.{weight,scales,biases}
        expert_groups: Dict[str, Dict[int, str]] = {}
        for k in list(out.keys()):
            parts = k.split(".")
            if "experts" in parts and "shared_experts" not in k:
                ei = parts.index("experts")
                e = int(parts[ei + 1])
                stacked_key = ".".join(parts[:ei]) + ".switch_mlp." + ".".join(parts[ei + 2:])
                expert_groups.setdefault(stacked_key, {})[e] = k
        for stacked_key, members in expert_groups.items():
            n = max(members) + 1
            out[stacked_key] = torch.stack([out.pop(members[e]) for e in range(n)])
        return StageModel.sanitize.__func__(cls, out, shard)

    def cache_specs(self) -> List[Tuple[int, int, int]]:
        cfg = self.config
        if self._use_absorbed():
            # compressed MLA cache: one MQA "head" of [c_kv | k_pe] rows
            rank = int(cfg.get("kv_lora_rank", 512))
            rope = int(cfg.get("qk_rope_head_dim", 64))
            return [(1, rank + rope, 0) for _ in range(self.shard.n_layers)]
        qk = int(cfg.get("qk_nope_head_dim", 128)) + int(cfg.get("qk_rope_head_dim", 64))
        vd = int(cfg.get("v_head_dim", 128))
        nh = cfg["num_attention_heads"]
        return [(nh, qk, vd) for _ in range(self.shard.n_layers)]

    def _use_absorbed(self) -> bool:
        try:
            p = next(self.parameters())
        except StopIteration:
            return False
        if not p.is_cuda:
            return False
        for layer in self.model.layers.values():
            return layer.self_attn.absorbed_supported()
        return False

    def forward(self, x: torch.Tensor, cache: Optional[List[KVCache]] = None) -> torch.Tensor:
        h = self.model.embed_tokens(x) if self.shard.is_first else x
        T = h.shape[1]
        cos, sin, _ = self.rope_for(cache[0] if cache else None, T, h.device)
        for j, i in enumerate(owned_layer_indices(self.shard)):
            c = cache[j] if cache is not None else None
            h = self.model.layers[str(i)](h, cos, sin, c)
        if self.shard.is_last:
            h = self.model.norm(h)
            h = self.lm_head(h)
        return h
