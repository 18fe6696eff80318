"""Pure-PyTorch reference implementations of every compute op.

These are the numerics ground truth the HIP/CDNA4 kernels are tested
against (fp32 accumulation throughout), and the execution path on CPU
(the reference's "plumbing" config: TinyLlama 2-stage over localhost
gRPC on CPU needs zero GPU code).

Op inventory mirrors SURVEY.md §2.4 (the ops the reference runs inside
MLX/mlx_lm; e.g. RMSNorm+RoPE via the blocks built at
/root/reference/shard/server/model/llama.py:31).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# Normalization
# ---------------------------------------------------------------------------

def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5,
             weight_offset: float = 0.0) -> torch.Tensor:
    """RMSNorm with fp32 accumulation.

    ``weight_offset=1.0`` gives gemma2's ``(1 + w)`` weighting.
    """
    dt = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    xn = xf * torch.rsqrt(var + eps)
    w = weight.float() + weight_offset
    return (xn * w).to(dt)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------

def rope_freqs(head_dim: int, base: float = 10000.0,
               scaling: Optional[dict] = None) -> torch.Tensor:
    """Inverse frequencies for the rotated half-dim, with optional scaling.

    ``scaling`` follows HF ``rope_scaling`` dicts: supports llama3-style
    and YaRN (DeepSeek-V2) factors.  Returns fp32 [head_dim/2].
    """
    inv_freq = 1.0 / (base ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim))
    if not scaling:
        return inv_freq
    rtype = scaling.get("rope_type", scaling.get("type", ""))
    if rtype == "llama3":
        factor = scaling["factor"]
        lo = scaling.get("low_freq_factor", 1.0)
        hi = scaling.get("high_freq_factor", 4.0)
        orig = scaling.get("original_max_position_embeddings", 8192)
        wavelen = 2 * math.pi / inv_freq
        low_len = orig / lo
        high_len = orig / hi
        new = torch.where(wavelen > low_len, inv_freq / factor, inv_freq)
        smooth = (orig / wavelen - lo) / (hi - lo)
        smoothed = (1 - smooth) / factor * inv_freq + smooth * inv_freq
        mid = (wavelen <= low_len) & (wavelen >= high_len)
        return torch.where(mid, smoothed, new)
    if rtype == "yarn":
        # DeepSeek-V2 YaRN: ramp between low/high correction dims.
        factor = scaling["factor"]
        orig = scaling.get("original_max_position_embeddings", 4096)
        beta_fast = scaling.get("beta_fast", 32.0)
        beta_slow = scaling.get("beta_slow", 1.0)

        def corr_dim(num_rot):
            return (head_dim * math.log(orig / (num_rot * 2 * math.pi))) / (2 * math.log(base))

        low = max(math.floor(corr_dim(beta_fast)), 0)
        high = min(math.ceil(corr_dim(beta_slow)), head_dim - 1)
        rng = torch.arange(head_dim // 2, dtype=torch.float32)
        ramp = torch.clamp((rng - low) / max(high - low, 1e-3), 0.0, 1.0)
        mask = 1.0 - ramp  # 1 → extrapolate (keep), 0 → interpolate (divide by factor)
        return inv_freq * mask + (inv_freq / factor) * (1.0 - mask)
    return inv_freq


def yarn_mscale(factor: float, mscale: float = 1.0) -> float:
    """YaRN attention-scale adjustment (DeepSeek-V2 `mscale`)."""
    if factor <= 1.0:
        return 1.0
    return 0.1 * mscale * math.log(factor) + 1.0


def rope_cos_sin(positions: torch.Tensor, inv_freq: torch.Tensor,
                 attn_scale: float = 1.0) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables [T, head_dim/2] (fp32) for given absolute positions."""
    ang = positions.float()[:, None] * inv_freq[None, :].to(positions.device)
    return torch.cos(ang) * attn_scale, torch.sin(ang) * attn_scale


def apply_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               interleaved: bool = False) -> torch.Tensor:
    """Rotate ``x`` [..., T, n_heads, D] by per-position cos/sin [T, D/2].

    ``interleaved=False`` is the HF-llama half-split convention
    (x1 = x[..., :D/2], x2 = x[..., D/2:]); ``interleaved=True`` pairs
    (x[2i], x[2i+1]) — the MLX `traditional` convention used by
    DeepSeek-V2's rope slice.
    """
    dt = x.dtype
    xf = x.float()
    D = x.shape[-1]
    # reshape cos/sin for broadcast over head dim: [T, 1, D/2]
    c = cos[..., :, None, :]
    s = sin[..., :, None, :]
    if interleaved:
        x1 = xf[..., 0::2]
        x2 = xf[..., 1::2]
        o1 = x1 * c - x2 * s
        o2 = x2 * c + x1 * s
        out = torch.stack([o1, o2], dim=-1).reshape(xf.shape)
    else:
        x1 = xf[..., : D // 2]
        x2 = xf[..., D // 2:]
        o1 = x1 * c - x2 * s
        o2 = x2 * c + x1 * s
        out = torch.cat([o1, o2], dim=-1)
    return out.to(dt)


# ---------------------------------------------------------------------------
# Attention
# ---------------------------------------------------------------------------

def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              scale: float, causal_offset: int = 0,
              softcap: float = 0.0,
              sliding_window: int = 0) -> torch.Tensor:
    """Masked SDPA with fp32 accumulation.

    q: [B, n_heads, Tq, Dqk]; k: [B, n_kv, Tk, Dqk]; v: [B, n_kv, Tk, Dv].
    Causal: query i (at absolute position causal_offset + i) attends keys
    [0, causal_offset + i].  GQA via head-group broadcast.  ``softcap``
    applies gemma2's tanh attn-logit softcapping; ``sliding_window`` > 0
    restricts keys to the trailing window.
    """
    B, H, Tq, Dq = q.shape
    Hkv = k.shape[1]
    if Hkv != H:
        rep = H // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    qf, kf, vf = q.float(), k.float(), v.float()
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if softcap and softcap > 0:
        scores = torch.tanh(scores / softcap) * softcap
    Tk = k.shape[2]
    qpos = torch.arange(Tq, device=q.device)[:, None] + causal_offset
    kpos = torch.arange(Tk, device=q.device)[None, :]
    mask = kpos > qpos
    if sliding_window and sliding_window > 0:
        mask = mask | (kpos <= qpos - sliding_window)
    scores = scores.masked_fill(mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = torch.matmul(probs, vf)
    return out.to(q.dtype)


# ---------------------------------------------------------------------------
# MLPs / activations
# ---------------------------------------------------------------------------

def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up with fp32 math."""
    dt = gate.dtype
    return (F.silu(gate.float()) * up.float()).to(dt)


def geglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """gelu_tanh(gate) * up — gemma2's MLP activation."""
    dt = gate.dtype
    return (F.gelu(gate.float(), approximate="tanh") * up.float()).to(dt)


def softcap(x: torch.Tensor, cap: float) -> torch.Tensor:
    """gemma2 logit softcapping: cap * tanh(x / cap)."""
    dt = x.dtype
    return (torch.tanh(x.float() / cap) * cap).to(dt)


# ---------------------------------------------------------------------------
# Quantized linear (MLX affine w4a16 / w8a16)
# ---------------------------------------------------------------------------

def dequantize(w_q: torch.Tensor, scales: torch.Tensor, biases: torch.Tensor,
               group_size: int, bits: int) -> torch.Tensor:
    """Unpack MLX affine-quantized weights to the scales' dtype.

    w_q: [out, in*bits/32] uint32, little-endian packed (lowest bits =
    first element).  scales/biases: [out, in/group_size].
    w[o, i] = scales[o, i // gs] * q[o, i] + biases[o, i // gs].
    """
    out_dim = w_q.shape[0]
    per_word = 32 // bits
    in_dim = w_q.shape[1] * per_word
    shifts = torch.arange(0, 32, bits, device=w_q.device, dtype=torch.int64)
    mask = (1 << bits) - 1
    q = (w_q.to(torch.int64)[:, :, None] >> shifts[None, None, :]) & mask
    q = q.reshape(out_dim, in_dim).float()
    s = scales.float().repeat_interleave(group_size, dim=1)[:, :in_dim]
    b = biases.float().repeat_interleave(group_size, dim=1)[:, :in_dim]
    return (q * s + b).to(scales.dtype)


def quantize(w: torch.Tensor, group_size: int = 64, bits: int = 4):
    """MLX-affine quantize: returns (w_q uint32, scales, biases).

    Per group: q = round((w - min) / scale), scale = (max-min)/(2^bits-1),
    packed little-endian into uint32.  Matches `dequantize` above and the
    MLX checkpoint layout (so pre-quantized MLX checkpoints interoperate).
    """
    out_dim, in_dim = w.shape
    assert in_dim % group_size == 0
    n_groups = in_dim // group_size
    wg = w.float().reshape(out_dim, n_groups, group_size)
    wmax = wg.max(-1).values
    wmin = wg.min(-1).values
    qmax = (1 << bits) - 1
    scale = (wmax - wmin) / qmax
    scale = torch.where(scale.abs() < 1e-10, torch.ones_like(scale), scale)
    q = torch.clamp(torch.round((wg - wmin[..., None]) / scale[..., None]), 0, qmax)
    q = q.reshape(out_dim, in_dim).to(torch.int64)
    per_word = 32 // bits
    qw = q.reshape(out_dim, in_dim // per_word, per_word)
    shifts = torch.arange(0, 32, bits, dtype=torch.int64)
    packed = (qw << shifts[None, None, :]).sum(-1)
    # uint32 packing via int32 view (torch has no uint32 arithmetic pre-2.3 on all backends)
    packed = (packed & 0xFFFFFFFF).to(torch.int64)
    w_q = packed.to(torch.uint32) if hasattr(torch, "uint32") else packed.to(torch.int32)
    return w_q, scale.to(w.dtype), wmin.to(w.dtype)


def quantized_linear(x: torch.Tensor, w_q: torch.Tensor, scales: torch.Tensor,
                     biases: torch.Tensor, group_size: int, bits: int) -> torch.Tensor:
    """y = x @ dequant(W)^T — reference path (materializes W)."""
    w = dequantize(w_q, scales, biases, group_size, bits)
    return F.linear(x, w.to(x.dtype))


# ---------------------------------------------------------------------------
# MoE (DeepSeek-V2 group-limited greedy top-k)
# ---------------------------------------------------------------------------

def moe_gate(router_logits: torch.Tensor, top_k: int, n_group: int = 1,
             topk_group: int = 1, routed_scaling_factor: float = 1.0,
             norm_topk_prob: bool = False):
    """Softmax router → group-limited greedy top-k.

    router_logits: [N, E].  Returns (weights [N, top_k], indices [N, top_k]).
    Matches DeepSeek-V2 MoEGate (greedy scoring_func=softmax,
    topk_method=greedy/group_limited_greedy).
    """
    scores = torch.softmax(router_logits.float(), dim=-1)
    N, E = scores.shape
    if n_group > 1:
        gscores = scores.reshape(N, n_group, E // n_group)
        group_top = gscores.max(dim=-1).values  # [N, n_group]
        top_groups = torch.topk(group_top, k=topk_group, dim=-1).indices
        gmask = torch.zeros(N, n_group, device=scores.device, dtype=torch.bool)
        gmask.scatter_(1, top_groups, True)
        mask = gmask[:, :, None].expand(N, n_group, E // n_group).reshape(N, E)
        scores = scores.masked_fill(~mask, 0.0)
    weights, indices = torch.topk(scores, k=top_k, dim=-1)
    if norm_topk_prob:
        weights = weights / (weights.sum(-1, keepdim=True) + 1e-20)
    weights = weights * routed_scaling_factor
    return weights, indices


def grouped_expert_mlp(x: torch.Tensor, gate_w: torch.Tensor, up_w: torch.Tensor,
                       down_w: torch.Tensor, weights: torch.Tensor,
                       indices: torch.Tensor) -> torch.Tensor:
    """Gather-style grouped expert SwiGLU MLP, reference implementation.

    x: [N, H]; gate_w/up_w: [E, I, H]; down_w: [E, H, I];
    weights/indices: [N, top_k].  Returns [N, H].
    Equivalent of the stacked `switch_mlp` path the reference enables at
    /root/reference/shard/server/model/deepseek_v2.py:101-112.
    """
    N, K = indices.shape
    out = torch.zeros_like(x, dtype=torch.float32)
    flat_idx = indices.reshape(-1)
    flat_w = weights.reshape(-1).float()
    xrep = x.repeat_interleave(K, dim=0).float()
    for e in torch.unique(flat_idx):
        m = flat_idx == e
        xe = xrep[m]
        h = swiglu(xe @ gate_w[e].float().t(), xe @ up_w[e].float().t())
        ye = h @ down_w[e].float().t()
        out.index_add_(0, torch.nonzero(m, as_tuple=True)[0] // K, ye * flat_w[m, None])
    return out.to(x.dtype)


# ---------------------------------------------------------------------------
# Sampling
# ---------------------------------------------------------------------------

def apply_repetition_penalty(logits: torch.Tensor, context: torch.Tensor,
                             penalty: float) -> torch.Tensor:
    """Penalize tokens in ``context`` (CTRL-style): x>0 → x/p, x<0 → x*p.

    Matches the reference's windowed implementation
    (/root/reference/shard/utils.py:152-177 via mlx_lm).
    """
    if penalty == 1.0 or context.numel() == 0:
        return logits
    out = logits.clone()
    vals = out[..., context]
    vals = torch.where(vals > 0, vals / penalty, vals * penalty)
    out[..., context] = vals
    return out


def top_p_sample(logits: torch.Tensor, top_p: float, temperature: float,
                 generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Nucleus sampling on the last-position logits [B, V] → [B] token ids."""
    probs = torch.softmax(logits.float() / temperature, dim=-1)
    sorted_probs, sorted_idx = torch.sort(probs, descending=True, dim=-1)
    cum = torch.cumsum(sorted_probs, dim=-1)
    keep = cum - sorted_probs < top_p  # keep tokens until cumulative mass reaches top_p
    keep[..., 0] = True
    filtered = torch.where(keep, sorted_probs, torch.zeros_like(sorted_probs))
    filtered = filtered / filtered.sum(-1, keepdim=True)
    pick = torch.multinomial(filtered, 1, generator=generator).squeeze(-1)
    return sorted_idx.gather(-1, pick[..., None]).squeeze(-1)


def sample(logits: torch.Tensor, temperature: float = 0.0, top_p: float = 1.0,
           generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Sample token ids from last-position logits [B, V]."""
    if temperature <= 0.0:
        return logits.argmax(dim=-1)
    if top_p < 1.0:
        return top_p_sample(logits, top_p, temperature, generator)
    probs = torch.softmax(logits.float() / temperature, dim=-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)
