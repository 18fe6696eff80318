"""Model/shard configuration.

The on-disk carrier is ``config.json`` exactly as in the reference
(/root/reference/shard/utils.py:33-39 injects ``start_layer``/``end_layer``
at load time; /root/reference/sharding_weight.py:48-60 bakes them into a
pre-sharded checkpoint).  We keep the raw dict around so arch-specific
model classes can pull their own fields.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Optional


@dataclass
class ShardSpec:
    """Contiguous layer range [start, end) owned by one pipeline stage."""

    start_layer: int
    end_layer: int
    num_hidden_layers: int

    @property
    def is_first(self) -> bool:
        return self.start_layer == 0

    @property
    def is_last(self) -> bool:
        return self.end_layer == self.num_hidden_layers

    def owns(self, layer_idx: int) -> bool:
        return self.start_layer <= layer_idx < self.end_layer

    @property
    def n_layers(self) -> int:
        return self.end_layer - self.start_layer

    def __post_init__(self):
        if not (0 <= self.start_layer < self.end_layer <= self.num_hidden_layers):
            raise ValueError(
                f"invalid shard range [{self.start_layer}, {self.end_layer}) "
                f"for {self.num_hidden_layers} layers"
            )


@dataclass
class QuantConfig:
    """MLX-style affine quantization stanza from config.json.

    w = scales * q + biases, per group of `group_size` input elements;
    q packed little-endian into uint32 (8 nibbles per word at bits=4).
    Matches the checkpoint format the reference loads via
    /root/reference/shard/utils.py:54-65.
    """

    group_size: int = 64
    bits: int = 4

    @classmethod
    def from_dict(cls, d: dict) -> "QuantConfig":
        return cls(group_size=int(d.get("group_size", 64)), bits=int(d.get("bits", 4)))


@dataclass
class ModelConfig:
    model_type: str
    raw: dict = field(default_factory=dict)

    @property
    def num_hidden_layers(self) -> int:
        return int(self.raw["num_hidden_layers"])

    @property
    def hidden_size(self) -> int:
        return int(self.raw["hidden_size"])

    @property
    def vocab_size(self) -> int:
        return int(self.raw["vocab_size"])

    @property
    def quantization(self) -> Optional[QuantConfig]:
        q = self.raw.get("quantization")
        return QuantConfig.from_dict(q) if q else None

    @property
    def start_layer(self) -> int:
        return int(self.raw.get("start_layer", 0))

    @property
    def end_layer(self) -> int:
        return int(self.raw.get("end_layer", self.num_hidden_layers))

    def shard(self, start_layer: Optional[int] = None, end_layer: Optional[int] = None) -> ShardSpec:
        s = self.start_layer if start_layer is None else start_layer
        e = self.end_layer if end_layer is None else end_layer
        return ShardSpec(s, e, self.num_hidden_layers)

    def get(self, key: str, default: Any = None) -> Any:
        return self.raw.get(key, default)

    def __getitem__(self, key: str) -> Any:
        return self.raw[key]

    @classmethod
    def from_dict(cls, d: dict) -> "ModelConfig":
        return cls(model_type=d["model_type"], raw=dict(d))

    @classmethod
    def load(cls, model_path: str | Path) -> "ModelConfig":
        p = Path(model_path) / "config.json"
        with open(p) as f:
            return cls.from_dict(json.load(f))
