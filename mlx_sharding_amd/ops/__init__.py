"""Op dispatch: HIP/CDNA4 kernels on GPU, torch reference on CPU.

Policy (deliberate, per the MI355X-native design):
  * On CUDA (= ROCm/HIP) tensors, the hand-written gfx950 extension
    ``mlx_sharding_amd._hip_ops`` MUST be present — ops raise loudly if
    it is missing so a silent eager fallback can never masquerade as the
    native path.  Set ``MLXS_AMD_FORCE_TORCH=1`` only for debugging.
  * On CPU tensors the torch reference implementations run (the
    reference's CPU plumbing config needs no GPU code).
  * Plain dense GEMMs (qkv/o projections, dense MLP matmuls) go through
    torch.nn.functional.linear → hipBLASLt/rocBLAS, which is the
    intended library path; everything fused/nonstandard is ours.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

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
        return _require_ext("rms_norm_residual").rms_norm_residual(
            x, residual, weight, eps, weight_offset)
    h = x + residual
    return ref.rms_norm(h, weight, eps, weight_offset), h


def apply_rope(x, cos, sin, interleaved: bool = False):
    if _use_hip(x):
        return _require_ext("apply_rope").apply_rope(x, cos, sin, interleaved)
    return ref.apply_rope(x, cos, sin, interleaved)


def attention(q, k, v, scale: float, causal_offset: int = 0,
              softcap: float = 0.0, sliding_window: int = 0):
    """Attention over the *current* keys (prefill uses this with full K/V)."""
    if _use_hip(q):
        ext = _require_ext("attention")
        return ext.attention(q, k, v, scale, causal_offset, softcap, sliding_window)
    return ref.attention(q, k, v, scale, causal_offset, softcap, sliding_window)


def swiglu(gate, up):
    if _use_hip(gate):
        return _require_ext("swiglu").swiglu(gate, up)
    return ref.swiglu(gate, up)


def geglu(gate, up):
    if _use_hip(gate):
        return _require_ext("geglu").geglu(gate, up)
    return ref.geglu(gate, up)


def softcap(x, cap: float):
    if _use_hip(x):
        return _require_ext("softcap").softcap(x, cap)
    return ref.softcap(x, cap)


def quantized_linear(x, w_q, scales, biases, group_size: int, bits: int):
    if _use_hip(x):
        return _require_ext("quantized_linear").quantized_linear(
            x, w_q, scales, biases, group_size, bits)
    return ref.quantized_linear(x, w_q, scales, biases, group_size, bits)


def moe_gate(router_logits, top_k: int, n_group: int = 1, topk_group: int = 1,
             routed_scaling_factor: float = 1.0, norm_topk_prob: bool = False):
    # Router is tiny ([N, E]); fp32 torch path is fine on both devices.
    return ref.moe_gate(router_logits, top_k, n_group, topk_group,
                        routed_scaling_factor, norm_topk_prob)


def grouped_expert_mlp(x, gate_w, up_w, down_w, weights, indices):
    if _use_hip(x):
        return _require_ext("grouped_expert_mlp").grouped_expert_mlp(
            x, gate_w, up_w, down_w, weights, indices)
    return ref.grouped_expert_mlp(x, gate_w, up_w, down_w, weights, indices)


# Sampling runs on [B, V] once per token — dispatched for the GPU decode path.

def sample(logits, temperature: float = 0.0, top_p: float = 1.0, generator=None):
    return ref.sample(logits, temperature, top_p, generator)


apply_repetition_penalty = ref.apply_repetition_penalty
rope_freqs = ref.rope_freqs
rope_cos_sin = ref.rope_cos_sin
yarn_mscale = ref.yarn_mscale
dequantize = ref.dequantize
quantize = ref.quantize
