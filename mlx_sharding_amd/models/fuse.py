"""Inference-time projection fusion.

Column-fuses sibling linears that share an input (q/k/v, gate/up, MLA
q+kv_a) into ONE GEMM per group: at decode M is tiny, so each extra
hipBLASLt launch costs ~10 µs of latency — fusing roughly halves the
dense-GEMM count per layer.  The original parameters become VIEWS into
the fused buffer (no memory duplication, state_dict unaffected), which
is why fusion must run AFTER the model is on its final device.
"""

from __future__ import annotations

from typing import List, Sequence

import torch

from .. import ops
from .base import Linear


class FusedColumns:
    """One GEMM over the concatenated output rows of several Linears."""

    def __init__(self, linears: Sequence[Linear]):
        assert len({l.in_features for l in linears}) == 1
        # all-or-nothing bias (qwen2 QKV all carry biases; mixed groups
        # are rejected and left unfused by fuse_model's except)
        has_bias = [l.bias is not None for l in linears]
        assert all(has_bias) or not any(has_bias), "mixed-bias group"
        self.bias = None
        if all(has_bias):
            bcat = torch.cat([l.bias.data for l in linears], dim=0).contiguous()
            off = 0
            for l in linears:
                l.bias.data = bcat[off: off + l.out_features]
                off += l.out_features
            self.bias = bcat
        quants = {id(l.quant) if l.quant is None else (l.quant.group_size, l.quant.bits)
                  for l in linears}
        self.quant = linears[0].quant
        assert all((l.quant is None) == (self.quant is None) for l in linears)
        self.splits = [l.out_features for l in linears]

        w = torch.cat([l.weight.data for l in linears], dim=0).contiguous()
        off = 0
        for l in linears:
            l.weight.data = w[off: off + l.weight.shape[0]]
            off += l.weight.shape[0]
        self.weight = w
        if self.quant is not None:
            s = torch.cat([l.scales.data for l in linears], dim=0).contiguous()
            b = torch.cat([l.biases.data for l in linears], dim=0).contiguous()
            off = 0
            for l in linears:
                n = l.scales.shape[0]
                l.scales.data = s[off: off + n]
                l.biases.data = b[off: off + n]
                off += n
            self.scales, self.biases = s, b

    def __call__(self, x: torch.Tensor) -> List[torch.Tensor]:
        if self.quant is None:
            y = ops.linear(x, self.weight, self.bias)
        else:
            y = ops.quantized_linear(x, self.weight, self.scales, self.biases,
                                     self.quant.group_size, self.quant.bits)
            if self.bias is not None:
                y = y + self.bias
        return list(torch.split(y, self.splits, dim=-1))


def fuse_model(model) -> int:
    """Walk the stage model and fuse sibling projections in-place.
    Returns the number of fused groups.  Call AFTER .to(device)."""
    from .llama import LlamaAttention, LlamaMLP
    from .gemma2 import Gemma2Attention, Gemma2MLP
    from .deepseek_v2 import MLAAttention

    n = 0
    for mod in model.modules():
        try:
            if isinstance(mod, (LlamaAttention, Gemma2Attention)):
                mod._fused_qkv = FusedColumns([mod.q_proj, mod.k_proj, mod.v_proj])
                n += 1
            elif isinstance(mod, (LlamaMLP, Gemma2MLP)):
                mod._fused_gu = FusedColumns([mod.gate_proj, mod.up_proj])
                n += 1
            elif isinstance(mod, MLAAttention) and not mod.q_lora_rank:
                mod._fused_qkv = FusedColumns([mod.q_proj, mod.kv_a_proj_with_mqa])
                n += 1
        except AssertionError:
            continue  # mixed quant/bias group — leave unfused
    return n
