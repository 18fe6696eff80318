"""Model registry — arch name → stage-model class.

Mirrors the reference's dynamic-import registry
(/root/reference/shard/utils.py:14-30) including the mistral→llama remap.
"""

from __future__ import annotations

from typing import Dict, Type

from .base import StageModel
from .llama import LlamaStageModel
from .gemma2 import Gemma2StageModel
from .deepseek_v2 import DeepseekV2StageModel

MODEL_REMAPPING: Dict[str, str] = {
    "mistral": "llama",
    # beyond-parity: qwen2 is llama-shaped with QKV-only attention
    # biases (handled in LlamaAttention); the reference supports only
    # llama/mistral/gemma2/deepseek_v2 (shard/utils.py:14-17)
    "qwen2": "llama",
}

_REGISTRY: Dict[str, Type[StageModel]] = {
    "llama": LlamaStageModel,
    "gemma2": Gemma2StageModel,
    "deepseek_v2": DeepseekV2StageModel,
}


def get_model_class(model_type: str) -> Type[StageModel]:
    model_type = MODEL_REMAPPING.get(model_type, model_type)
    try:
        return _REGISTRY[model_type]
    except KeyError:
        raise ValueError(
            f"model type '{model_type}' not supported "
            f"(supported: {sorted(_REGISTRY)})") from None


def register_model(model_type: str, cls: Type[StageModel]):
    _REGISTRY[model_type] = cls
