"""Op dispatch: HIP/CDNA4 kernels on GPU, torch reference on CPU.

Policy (deliberate, per the MI355X-native design):
  * On CUDA (= ROCm/HIP) tensors, the hand-written gfx950 extension
    ``mlx_sharding_amd._hip_ops`` MUST be present — ops raise loudly if
    it is missing so a silent eager fallback can never masquerade as the
    native path.  Set ``MLXS_AMD_FORCE_TORCH=1`` only for debugging.
  * On CPU tensors the torch reference implementations run (the
    reference's CPU plumbing config needs no GPU code).
  * Plain dense GEMMs (qkv/o projections, dense MLP matmuls, prefill
    QK^T/PV batch matmuls) go through torch.matmul → hipBLASLt/rocBLAS,
    which is the intended library path; everything fused/nonstandard
    (norms, rope, decode attention, w4a16, MoE gather) is ours.
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from . import reference as ref

_EXT = None
_EXT_ERR: Optional[str] = None
_TRIED = False


def _load_ext():
    global _EXT, _EXT_ERR, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    try:
        from mlx_sharding_amd import _hip_ops  # built in-tree by setup.py / __graft_entry__.build()
        _EXT = _hip_ops
    except Exception as e:  # noqa: BLE001
        _EXT = None
        _EXT_ERR = repr(e)
    return _EXT


def hip_ext():
    """The loaded HIP extension module, or None (CPU-only environments)."""
    return _load_ext()


def _force_torch() -> bool:
    return os.environ.get("MLXS_AMD_FORCE_TORCH", "0") == "1"


def _require_ext(op_name: str):
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            f"mlx_sharding_amd: op '{op_name}' called on a GPU tensor but the "
            f"HIP extension mlx_sharding_amd._hip_ops is not importable "
            f"({_EXT_ERR}). Build it with `python setup.py build_ext --inplace` "
            f"(or __graft_entry__.build()). Refusing to fall back to eager torch "
            f"on GPU; set MLXS_AMD_FORCE_TORCH=1 only for debugging."
        )
    return ext


def _use_hip(t: torch.Tensor) -> bool:
    return t.is_cuda and not _force_torch()


def use_native(t: torch.Tensor) -> bool:
    """Model-level guard for choosing a fused-HIP code path: the tensor
    is on GPU, the extension is importable, and MLXS_AMD_FORCE_TORCH is
    not overriding (so a FORCE_TORCH A/B really runs eager torch)."""
    return t.is_cuda and not _force_torch() and _load_ext() is not None


# ---------------------------------------------------------------------------
# Dispatched ops
# ---------------------------------------------------------------------------

def rms_norm(x, weight, eps: float = 1e-5, weight_offset: float = 0.0):
    if _use_hip(x):
        return _require_ext("rms_norm").rms_norm(x, weight, eps, weight_offset)
    return ref.rms_norm(x, weight, eps, weight_offset)


def rms_norm_residual(x, residual, weight, eps: float = 1e-5, weight_offset: float = 0.0):
    """Fused h = x + residual; y = rms_norm(h). Returns (y, h)."""
    if _use_hip(x):
        y, h = _require_ext("rms_norm_residual").rms_norm_residual(
            x, residual, weight, eps, weight_offset)
        return y, h
    h = x + residual
    return ref.rms_norm(h, weight, eps, weight_offset), h


def apply_rope(x, cos, sin, interleaved: bool = False):
    if _use_hip(x):
        return _require_ext("apply_rope").apply_rope(
            x, cos.contiguous(), sin.contiguous(), interleaved)
    return ref.apply_rope(x, cos, sin, interleaved)


def attention(q, k, v, scale: float, causal_offset: int = 0,
              softcap: float = 0.0, sliding_window: int = 0, pos_dev=None):
    """Attention over the current K/V views.

    q: [B, Hq, Tq, Dk]; k/v: [B, Hkv, S, D*] (cache views on the GPU
    path).  Decode (Tq == 1) runs the hand-written flash-decode kernel;
    ``pos_dev`` (int32 [1] on device) switches it to graph-capture mode:
    k/v are FULL cache buffers and the kernel reads S = pos+1 on device.
    Prefill composes hipBLASLt batch GEMMs with fp32 softmax.
    """
    if _use_hip(q):
        ext = _require_ext("attention")
        cache_layout = k.stride(3) == 1 and k.stride(2) == k.shape[3]
        if q.shape[2] == 1 and cache_layout:
            return ext.attn_decode(q, k, v, scale, softcap, sliding_window,
                                   pos_dev)
        if (cache_layout
                and ext.attn_prefill_shape_ok(q.shape[3], v.shape[3])):
            return ext.attn_prefill(q, k, v, scale, softcap, sliding_window,
                                    causal_offset)
        return _prefill_attention_gpu(q, k, v, scale, causal_offset,
                                      softcap, sliding_window)
    if pos_dev is not None:  # CPU fallback for tests of the graph path
        S = int(pos_dev.item()) + 1
        return ref.attention(q, k[:, :, :S], v[:, :, :S], scale,
                             causal_offset=S - 1, softcap=softcap,
                             sliding_window=sliding_window)
    return ref.attention(q, k, v, scale, causal_offset, softcap, sliding_window)


def _prefill_attention_gpu(q, k, v, scale, causal_offset, softcap, sliding_window):
    """Prefill path: QK^T and PV on hipBLASLt (library GEMMs), mask +
    softmax in fp32.  (Flash-prefill HIP kernel is the planned upgrade.)"""
    B, Hq, Tq, Dk = q.shape
    Hkv, S = k.shape[1], k.shape[2]
    G = Hq // Hkv
    qg = q.view(B, Hkv, G, Tq, Dk)
    scores = torch.matmul(qg, k.unsqueeze(2).transpose(-1, -2)).float() * scale
    if softcap and softcap > 0:
        scores = torch.tanh(scores / softcap) * softcap
    qpos = torch.arange(Tq, device=q.device)[:, None] + causal_offset
    kpos = torch.arange(S, device=q.device)[None, :]
    mask = kpos > qpos
    if sliding_window and sliding_window > 0:
        mask = mask | (kpos <= qpos - sliding_window)
    scores = scores.masked_fill(mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1).to(q.dtype)
    out = torch.matmul(probs, v.unsqueeze(2))
    return out.reshape(B, Hq, Tq, v.shape[3])


def swiglu(gate, up):
    if _use_hip(gate):
        return _require_ext("swiglu").glu(gate, up, False)
    return ref.swiglu(gate, up)


def geglu(gate, up):
    if _use_hip(gate):
        return _require_ext("geglu").glu(gate, up, True)
    return ref.geglu(gate, up)


def softcap(x, cap: float):
    if _use_hip(x):
        return _require_ext("softcap").softcap(x, cap)
    return ref.softcap(x, cap)


_GEMV_MAX_M = 64  # above this, dequant + hipBLASLt GEMM wins


_DQ_CACHE_BYTES = 0


def _dq_cache_ok(nbytes: int) -> bool:
    """Budget for resident dequantized bf16 copies (HBM-for-throughput
    trade, docs/DECISIONS.md).  Defaults to 96 GB — generous on a
    288 GB part, and a guard against OOM on checkpoints whose
    dequantized experts would not fit (e.g. 236B-class models).
    MLXS_AMD_NO_DQ_CACHE=1 disables caching entirely;
    MLXS_AMD_DQ_CACHE_GB overrides the budget."""
    global _DQ_CACHE_BYTES
    if os.environ.get("MLXS_AMD_NO_DQ_CACHE"):
        return False
    cap = float(os.environ.get("MLXS_AMD_DQ_CACHE_GB", "96")) * (1 << 30)
    if _DQ_CACHE_BYTES + nbytes > cap:
        return False
    _DQ_CACHE_BYTES += nbytes
    return True


def linear(x, weight, bias=None):
    """Dense linear.  The MFMA decode GEMV (MLXS_AMD_DENSE_GEMV=1) was
    a NEGATIVE result at batch-64 decode shapes: hipBLASLt's small-GEMM
    kernels run ~13 us in-graph and the split-K fp32-atomic traffic
    (M*O*4*nk bytes) rivals the weight bytes on small-O projections
    (docs/PERFORMANCE.md).  Kept for shape-specific re-tuning."""
    if (os.environ.get("MLXS_AMD_DENSE_GEMV")
            and _use_hip(x) and weight.dtype == torch.bfloat16
            and x.dtype == torch.bfloat16 and weight.shape[-1] % 32 == 0
            and weight.is_contiguous()):
        lead = x.shape[:-1]
        x2 = x.reshape(-1, x.shape[-1])
        if 0 < x2.shape[0] <= 64:
            y = _require_ext("linear").dense_gemv(x2, weight)
            if bias is not None:
                y = y + bias
            return y.reshape(*lead, y.shape[-1])
    # LDS-tiled M<=64 GEMM (dense_gemm64): measured 1.0-1.5 TB/s vs
    # hipBLASLt's 3.4-5.5 on the llama-70B shapes after three
    # iterations (staging flattening, row-tile amortization) — kept as
    # opt-in infrastructure (MLXS_AMD_GEMM64=1), NOT the default
    if (os.environ.get("MLXS_AMD_GEMM64")
            and _use_hip(x) and weight.dtype == torch.bfloat16
            and x.dtype == torch.bfloat16 and weight.shape[-1] >= 16384
            and weight.shape[-1] % 8 == 0 and weight.is_contiguous()):
        lead = x.shape[:-1]
        x2 = x.reshape(-1, x.shape[-1])
        if 0 < x2.shape[0] <= 64:
            y = _require_ext("linear").dense_gemm64(x2, weight)
            if bias is not None:
                y = y + bias
            return y.reshape(*lead, y.shape[-1])
    # K-segmented coalesced-A GEMM (gemm_kseg.hip) — DEFAULT on exactly
    # the regime it measured faster than hipBLASLt (3.96 vs 3.27 TB/s,
    # tools/gemm_kseg_probe.py): M<=64 decode with K-long huge
    # projections (llama-70B down_proj class).  Everything else stays
    # on the library.  MLXS_AMD_NO_KSEG=1 opts out.
    if (not os.environ.get("MLXS_AMD_NO_KSEG")
            and _use_hip(x) and weight.dtype == torch.bfloat16
            and x.dtype == torch.bfloat16
            and weight.shape[-1] >= 16384 and weight.shape[0] >= 8192
            and weight.shape[-1] % 256 == 0 and weight.is_contiguous()):
        lead = x.shape[:-1]
        x2 = x.reshape(-1, x.shape[-1])
        if 0 < x2.shape[0] <= 64:
            ksegs = max(1, min(16, 512 // max(1, weight.shape[0] // 64)))
            y = _require_ext("linear").gemm_m64_kseg(x2, weight, ksegs)
            if bias is not None:
                y = y + bias
            return y.reshape(*lead, y.shape[-1])
    return torch.nn.functional.linear(x, weight, bias)


def quantized_linear(x, w_q, scales, biases, group_size: int, bits: int):
    if _use_hip(x):
        ext = _require_ext("quantized_linear")
        lead = x.shape[:-1]
        H = x.shape[-1]
        x2 = x.reshape(-1, H)
        M = x2.shape[0]
        # Dense projections: dequantize ONCE and keep the bf16 copy
        # resident (a few GB on a 288 GB part), then hipBLASLt — the
        # per-call w4 GEMV kernels are dequant-VALU-pipe-bound (~25 µs
        # vs ~13 µs; PMC in docs/PERFORMANCE.md) and only win when
        # memory is tight (MLXS_AMD_NO_DQ_CACHE=1 disables the cache;
        # routed EXPERT weights — the bulk of an MoE checkpoint — stay
        # packed and use the w4 MFMA kernels).
        w = getattr(w_q, "_mlxs_dqw", None)
        O = w_q.shape[0]
        if w is None and _dq_cache_ok(O * H * 2):
            w = ext.dequant(w_q, scales, biases, H, group_size, bits)
            w_q._mlxs_dqw = w
        if w is not None:
            y = linear(x2, w)
        elif (M <= _GEMV_MAX_M and group_size in (32, 64, 128)
              and not os.environ.get("MLXS_AMD_NO_W4F16")):
            # fp16-dequant MFMA GEMV (pk_fma dequant over repacked
            # words — see moe_w4f16.hip header)
            y = ext.w4f16_gemv(_to_f16_cached(x).reshape(-1, H),
                               repack_w4(w_q, bits),
                               scales, biases, group_size, bits)
        elif M <= _GEMV_MAX_M:
            y = ext.w4a16_gemv(x2, w_q, scales, biases, group_size, bits)
        else:
            w = ext.dequant(w_q, scales, biases, H, group_size, bits)
            y = torch.nn.functional.linear(x2, w)
        return y.reshape(*lead, y.shape[-1])
    return ref.quantized_linear(x, w_q, scales, biases, group_size, bits)


def moe_gate(router_logits, top_k: int, n_group: int = 1, topk_group: int = 1,
             routed_scaling_factor: float = 1.0, norm_topk_prob: bool = False):
    # Router is tiny ([N, E]); fp32 torch path is fine on both devices.
    return ref.moe_gate(router_logits, top_k, n_group, topk_group,
                        routed_scaling_factor, norm_topk_prob)


_MOE_GEMM_MIN_N = 256  # above this token count, per-expert hipBLASLt GEMMs win


def make_expert_subranges(indices: torch.Tensor, weights: torch.Tensor,
                          n_experts: int, max_tok: int = 4):
    """Sort (token, expert) pairs by expert and split each expert's run
    into sub-ranges of <= max_tok tokens (the grouped kernels' unit of
    work).  Built entirely with device ops — no host sync; padded tail
    slots carry cnt=0 and early-exit in the kernel.

    Returns (sub_expert, sub_off, sub_cnt, sorted_tok, sorted_wt, order),
    all int32/fp32 on indices.device, with len(sub_*) = E + P // max_tok.
    """
    N, K = indices.shape
    P = N * K
    dev = indices.device
    flat_e = indices.reshape(-1).long()
    order = torch.argsort(flat_e, stable=True)
    tok = torch.arange(N, device=dev, dtype=torch.int64).repeat_interleave(K)
    sorted_tok = tok[order].to(torch.int32)
    sorted_wt = weights.reshape(-1).float()[order].contiguous()
    counts = torch.bincount(flat_e, minlength=n_experts)
    start = torch.cumsum(counts, 0) - counts
    nsub = (counts + max_tok - 1) // max_tok
    sub_start = torch.cumsum(nsub, 0) - nsub
    s_upper = n_experts + P // max_tok
    # marker[slot] = expert starting at that slot; empty experts scatter -1
    # (they share sub_start with their successor, amax keeps the real one).
    # One extra slot absorbs trailing empty experts.  No torch.nonzero —
    # it would host-sync on every MoE layer.
    marker = torch.full((s_upper + 1,), -1, dtype=torch.int64, device=dev)
    eidx_all = torch.arange(n_experts, device=dev)
    src = torch.where(nsub > 0, eidx_all, torch.full_like(eidx_all, -1))
    marker.scatter_reduce_(0, sub_start, src, reduce="amax",
                           include_self=True)
    sub_expert = torch.cummax(marker[:s_upper], 0).values.clamp(min=0)
    sub_idx = torch.arange(s_upper, device=dev) - sub_start[sub_expert]
    sub_off = (start[sub_expert] + max_tok * sub_idx).to(torch.int32)
    sub_cnt = (counts[sub_expert] - max_tok * sub_idx) \
        .clamp(0, max_tok).to(torch.int32)
    return (sub_expert.to(torch.int32).contiguous(), sub_off.contiguous(),
            sub_cnt.contiguous(), sorted_tok.contiguous(), sorted_wt, order)


def _cached_t(w: torch.Tensor) -> torch.Tensor:
    if w.requires_grad or not w.is_leaf:
        return w.transpose(1, 2).contiguous()
    t = getattr(w, "_mlxs_t", None)
    if t is None:
        t = w.transpose(1, 2).contiguous()
        w._mlxs_t = t
    return t


def _cached_gu_t(gate_w: torch.Tensor, up_w: torch.Tensor) -> torch.Tensor:
    """Column-concatenated [E, H, 2I] transposed gate|up operand: ONE
    wide-N bmm instead of two measured 2563 -> 2122 us at prefill shapes
    (+21% MFU, tools/bmm_probe.py); the glu then reads the two halves
    as strided views in place."""
    t = getattr(gate_w, "_mlxs_gut", None)
    if t is not None:
        return t
    E, I, H = gate_w.shape
    t = torch.empty(E, H, 2 * I, dtype=gate_w.dtype, device=gate_w.device)
    t[..., :I] = gate_w.transpose(1, 2)
    t[..., I:] = up_w.transpose(1, 2)
    if not gate_w.requires_grad and gate_w.is_leaf:
        gate_w._mlxs_gut = t
    return t


def _moe_prefill_gemm(x, gate_w, up_w, down_w, weights, indices,
                      dequant_all=None):
    """Large-N path: expert-padded batched GEMMs (hipBLASLt bmm).

    Tokens are scattered into an [E, cap, H] padded tensor (cap = max
    tokens on one expert) so the whole MoE layer is 3 bmm launches —
    a per-expert python GEMM loop costs ~400 host-side launches per
    layer and made prefill host-bound."""
    N, K = indices.shape
    H = x.shape[-1]
    dev = x.device
    flat_e = indices.reshape(-1).long()
    order = torch.argsort(flat_e, stable=True)
    sorted_e = flat_e[order]
    tok = torch.arange(N, device=dev).repeat_interleave(K)[order]
    wts = weights.reshape(-1).float()[order]
    if dequant_all is not None:
        gate_w, up_w, down_w = dequant_all()
    E = gate_w.shape[0]
    counts = torch.zeros(E, dtype=torch.long, device=dev)
    counts.scatter_add_(0, sorted_e, torch.ones_like(sorted_e))
    start = torch.cumsum(counts, 0) - counts
    # pad cap to a 128 multiple: hipBLASLt strided-batch bmm faults on
    # odd M at these shapes (observed on ROCm 7.0 torch), and aligned M
    # picks better tiles anyway
    cap = (int(counts.max()) + 127) // 128 * 128  # one host sync per layer
    P = N * K
    slot = torch.arange(P, device=dev) - start[sorted_e]
    dst = sorted_e * cap + slot
    xp = x.new_zeros(E * cap, H)
    ext = hip_ext() if x.is_cuda else None
    if ext is not None and H % 4 == 0:
        # vectorized row scatter / per-token gather-reduce kernels:
        # torch's advanced-indexing + fp32 index_add_ run ~5x off
        # roofline at 16K-token prefill shapes (~45 ms per prefill)
        ext.moe_scatter_rows(x, xp, tok.to(torch.int32),
                             dst.to(torch.int32))
    else:
        xp[dst] = x[tok]
    xp = xp.view(E, cap, H)
    # transposed-B strided bmm memory-faults in this torch/hipBLASLt build
    # (reproduced at [64,1664,2048]x[64,2048,1408] bf16) — materialize the
    # transposed operand instead, cached on the weight tensor.  gate|up
    # run as ONE wide-N bmm (+21% MFU) with the glu reading the halves
    # as strided views.
    I = gate_w.shape[1]
    gu = torch.bmm(xp, _cached_gu_t(gate_w, up_w))
    hh = swiglu(gu[..., :I], gu[..., I:])
    d = torch.bmm(hh, _cached_t(down_w)).reshape(E * cap, H)
    if ext is not None and H % 4 == 0:
        inv = torch.empty_like(order)
        inv[order] = torch.arange(P, device=dev)
        pair_pos = dst[inv].view(N, K).to(torch.int32)
        pair_wts = weights.reshape(N, K).float()
        return ext.moe_gather_reduce(d, pair_pos, pair_wts, N)
    y = d[dst].float() * wts[:, None]
    out = torch.zeros(N, H, device=dev, dtype=torch.float32)
    out.index_add_(0, tok, y)
    return out.to(x.dtype)


def moe_gate_subranges(router_logits_bf16, top_k: int,
                       routed_scaling_factor: float = 1.0,
                       norm_topk_prob: bool = False, max_tok: int = 4):
    """Fused gating: softmax + greedy top-k + expert sort + sub-range
    build in ONE kernel (N<=128 tokens, E<=64 experts, k<=8)."""
    ext = _require_ext("moe_gate_subranges")
    N, E = router_logits_bf16.shape
    s_upper = E + (N * top_k) // max_tok
    return tuple(ext.moe_gate_subranges(router_logits_bf16, top_k, s_upper,
                                        max_tok, routed_scaling_factor,
                                        norm_topk_prob))


def grouped_expert_mlp_subs(x, gate_w, up_w, down_w, subs, max_tok: int = 4):
    """Run the grouped expert MLP from prebuilt sub-range arrays.

    max_tok selects the kernel family the sub-ranges were built for:
    16 -> MFMA 16-token tiles (weights streamed ~once per activated
    expert), <=4 -> scalar v_dot2c kernels."""
    ext = _require_ext("grouped_expert_mlp")
    sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt = subs
    P = sorted_tok.shape[0]
    h = ext.moe_gateup_grouped(x, gate_w, up_w, sub_e, sub_off, sub_cnt,
                               sorted_tok, P, max_tok)
    out = ext.moe_down_grouped(h, down_w, sub_e, sub_off, sub_cnt,
                               sorted_tok, sorted_wt, x.shape[0], max_tok)
    return out.to(x.dtype)


def _to_f16_cached(x: torch.Tensor) -> torch.Tensor:
    """bf16 -> fp16 cast (exact), memoized on the tensor object: sibling
    projections sharing one activation (o_proj + shared experts + MoE
    within a layer) pay the cast launch once.  Activations are fresh
    objects per step (never mutated in place), so the cache cannot go
    stale; under graph capture the cast is captured once and the cached
    buffer is reused by later ops in the same graph."""
    c = getattr(x, "_mlxs_f16", None)
    if c is not None:
        return c
    c = x.to(torch.float16)
    try:
        x._mlxs_f16 = c
    except Exception:  # noqa: BLE001 — non-leaf views may refuse attrs
        pass
    return c


def repack_w4(wq: torch.Tensor, bits: int) -> torch.Tensor:
    """Offline nibble/byte interleave of packed quant words for the
    fp16-dequant MFMA kernels (moe_w4f16.hip): reorders each u32 so
    ((w >> 4j) & 0x000F000F) emits element pairs in natural k-order —
    the k-permutation lives in the DATA, not in per-load VALU shuffles.
    Cached on the packed tensor (one-time, at first use)."""
    cached = getattr(wq, "_mlxs_rp", None)
    if cached is not None:
        return cached
    w = wq.view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    if bits == 4:
        even = ((w & 0xF) | (((w >> 8) & 0xF) << 4)
                | (((w >> 16) & 0xF) << 8) | (((w >> 24) & 0xF) << 12))
        odd = (((w >> 4) & 0xF) | (((w >> 12) & 0xF) << 4)
               | (((w >> 20) & 0xF) << 8) | (((w >> 28) & 0xF) << 12))
        rp = even | (odd << 16)
    elif bits == 8:
        rp = ((w & 0xFF) | (((w >> 16) & 0xFF) << 8)
              | (((w >> 8) & 0xFF) << 16) | (((w >> 24) & 0xFF) << 24))
    else:
        raise ValueError(f"unsupported bits {bits}")
    # wrap to int32 bit pattern (values >= 2^31 are valid u32 words)
    rp = torch.where(rp >= 2 ** 31, rp - 2 ** 32, rp).to(torch.int32)
    rp = rp.view(wq.dtype).contiguous()
    if not wq.requires_grad and wq.is_leaf:
        wq._mlxs_rp = rp
    return rp


def grouped_expert_mlp_quant_subs(x, gate, up, down, subs,
                                  group_size: int, bits: int):
    """Quantized grouped experts over prebuilt 16-token sub-ranges via
    the fp16-dequant MFMA kernels (moe_w4f16.hip): fused gate+up+SiLU
    in one pass, pk_fma dequant (~16 VALU per 8 weights vs ~36 for the
    old per-element cvt+fma kernel), activations cast to fp16 (exact)."""
    ext = _require_ext("grouped_expert_mlp_quant")
    sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt = subs
    P = sorted_tok.shape[0]
    if group_size % 32 != 0 or x.shape[1] % 32 != 0:
        raise ValueError(
            "quantized experts require group_size % 32 == 0 and hidden % 32 "
            f"== 0 (got gs={group_size}, H={x.shape[1]})")
    x16 = _to_f16_cached(x)
    gq = repack_w4(gate[0], bits)
    uq = repack_w4(up[0], bits)
    dq = repack_w4(down[0], bits)
    hh = ext.moe_w4f16_gateup(x16, gq, uq, gate[1], gate[2], up[1], up[2],
                              sub_e, sub_off, sub_cnt, sorted_tok, P,
                              group_size, bits)
    out = ext.moe_w4f16_down(hh, dq, down[1], down[2], sub_e, sub_off,
                             sub_cnt, sorted_tok, sorted_wt, x.shape[0],
                             group_size, bits)
    return out.to(x.dtype)


def grouped_expert_mlp(x, gate_w, up_w, down_w, weights, indices):
    if _use_hip(x):
        _require_ext("grouped_expert_mlp")
        if x.shape[0] * indices.shape[1] >= _MOE_GEMM_MIN_N:
            return _moe_prefill_gemm(x, gate_w, up_w, down_w, weights, indices)
        E = gate_w.shape[0]
        sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt, _ = \
            make_expert_subranges(indices, weights, E, max_tok=4)
        return grouped_expert_mlp_subs(
            x, gate_w, up_w, down_w,
            (sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt))
    return ref.grouped_expert_mlp(x, gate_w, up_w, down_w, weights, indices)


def expert_dequant(gate, up, down, group_size: int, bits: int):
    """Dequantize stacked expert triplets to bf16, CACHED on the packed
    gate tensor under the dq-cache budget.  Rationale: re-dequantizing
    churns ~1 GB of fresh allocations per layer per prefill, the expert
    bmms measured 3.2x slower against just-written operands, and the
    decode-regime w4 MFMA kernels are dequant-VALU-pipe-bound at ~6
    tokens/expert (PMC, docs/PERFORMANCE.md) — MI355X has 288 GB of
    HBM3E, so spending the budget on resident bf16 expert copies is the
    right trade.  Returns (gate_w, up_w, down_w) dense [E, *, *]."""
    cached = getattr(gate[0], "_mlxs_dq", None)
    if cached is not None:
        return cached
    ext = _require_ext("expert_dequant")
    E = gate[0].shape[0]
    pw = 32 // bits
    H = gate[0].shape[-1] * pw
    I = down[0].shape[-1] * pw
    gw = ext.dequant(gate[0].reshape(-1, H // pw), gate[1].reshape(-1, H // group_size),
                     gate[2].reshape(-1, H // group_size), H, group_size, bits)
    uw = ext.dequant(up[0].reshape(-1, H // pw), up[1].reshape(-1, H // group_size),
                     up[2].reshape(-1, H // group_size), H, group_size, bits)
    dw = ext.dequant(down[0].reshape(-1, I // pw), down[1].reshape(-1, I // group_size),
                     down[2].reshape(-1, I // group_size), I, group_size, bits)
    res = (gw.view(E, -1, H), uw.view(E, -1, H), dw.view(E, -1, I))
    if _dq_cache_ok(sum(t.numel() * 2 for t in res)):
        gate[0]._mlxs_dq = res
    return res


def expert_dequant_resident(gate, up, down, group_size: int, bits: int):
    """The cached dense triplet if the budget allows holding it (build
    on first call), else None — decode uses this to choose between the
    bf16 MFMA MoE kernels and the packed w4 kernels."""
    cached = getattr(gate[0], "_mlxs_dq", None)
    if cached is not None:
        return cached
    pw = 32 // bits
    # dense bytes = packed words * elems/word * 2 B, per matrix
    est = 2 * pw * (gate[0].numel() + up[0].numel() + down[0].numel())
    cap = float(os.environ.get("MLXS_AMD_DQ_CACHE_GB", "96")) * (1 << 30)
    if os.environ.get("MLXS_AMD_NO_DQ_CACHE") or _DQ_CACHE_BYTES + est > cap:
        return None
    res = expert_dequant(gate, up, down, group_size, bits)
    return res if getattr(gate[0], "_mlxs_dq", None) is not None else None


def grouped_expert_mlp_quant(x, gate, up, down, weights, indices,
                             group_size: int, bits: int):
    """Quantized stacked experts: gate/up/down are (w_q, scales, biases)
    triplets with stacked [E, ...] layout."""
    if _use_hip(x):
        ext = _require_ext("grouped_expert_mlp_quant")
        E = gate[0].shape[0]
        P = indices.numel()
        if P >= _MOE_GEMM_MIN_N:
            H = x.shape[-1]
            I = down[0].shape[-1] * (32 // bits)
            pw = 32 // bits

            return _moe_prefill_gemm(
                x, None, None, None, weights, indices,
                dequant_all=lambda: expert_dequant(gate, up, down,
                                                   group_size, bits))
        sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt, _ = \
            make_expert_subranges(indices, weights, E, max_tok=16)
        return grouped_expert_mlp_quant_subs(
            x, gate, up, down,
            (sub_e, sub_off, sub_cnt, sorted_tok, sorted_wt),
            group_size, bits)
    # CPU reference: dequantize then dense grouped MLP
    E = gate[0].shape[0]
    gw = torch.stack([ref.dequantize(gate[0][e], gate[1][e], gate[2][e],
                                     group_size, bits) for e in range(E)])
    uw = torch.stack([ref.dequantize(up[0][e], up[1][e], up[2][e],
                                     group_size, bits) for e in range(E)])
    dw = torch.stack([ref.dequantize(down[0][e], down[1][e], down[2][e],
                                     group_size, bits) for e in range(E)])
    return ref.grouped_expert_mlp(x, gw, uw, dw, weights, indices)


# Sampling runs on [B, V] once per token — torch argmax/multinomial are
# library kernels; fine on both devices.

def sample(logits, temperature: float = 0.0, top_p: float = 1.0, generator=None):
    return ref.sample(logits, temperature, top_p, generator)


apply_repetition_penalty = ref.apply_repetition_penalty
rope_freqs = ref.rope_freqs
rope_cos_sin = ref.rope_cos_sin
yarn_mscale = ref.yarn_mscale
dequantize = ref.dequantize
quantize = ref.quantize
