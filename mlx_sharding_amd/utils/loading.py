"""Checkpoint loading and the offline per-stage weight splitter.

On-disk layout is the reference's declared compatibility surface
(/root/reference/sharding_weight.py): per-stage file
``model-{start:05d}-{end:05d}.safetensors`` + a rewritten
``model.safetensors.index.json`` + ``config.json`` carrying
start_layer/end_layer; dynamic mode loads a FULL checkpoint and filters
in memory via the same key-routing rule (utils.py:33-68 + sanitize).
"""

from __future__ import annotations

import glob
import json
import shutil
from pathlib import Path
from typing import Dict, Optional, Tuple

import torch
from safetensors.torch import load_file, save_file

from ..config import ModelConfig, QuantConfig
from ..models import get_model_class
from ..models.base import StageModel


def get_model_path(path_or_repo: str | Path) -> Path:
    """Resolve a local directory OR a HF hub repo id to a checkpoint
    directory, mirroring the reference's get_model_path use
    (/root/reference/shard/utils.py:33-39 via mlx_lm).

    A local path wins; otherwise the repo id is resolved through
    huggingface_hub's snapshot cache (HF_HOME), preferring the offline
    cache and only then attempting a download (no-op in air-gapped
    environments — the cached snapshot is the offline fallback)."""
    p = Path(path_or_repo)
    if p.exists():
        return p
    try:
        from huggingface_hub import snapshot_download
    except ImportError:
        raise FileNotFoundError(
            f"{path_or_repo} is not a local directory and huggingface_hub "
            f"is not installed to resolve it as a repo id") from None
    patterns = ["*.json", "*.safetensors", "*.model", "tokenizer*", "*.txt"]
    try:
        return Path(snapshot_download(repo_id=str(path_or_repo),
                                      allow_patterns=patterns,
                                      local_files_only=True))
    except Exception:
        pass  # not cached — try the network (may be unavailable)
    try:
        return Path(snapshot_download(repo_id=str(path_or_repo),
                                      allow_patterns=patterns))
    except Exception as e:  # noqa: BLE001
        raise FileNotFoundError(
            f"cannot resolve model {path_or_repo!r}: not a local directory, "
            f"not in the HF cache (HF_HOME), and download failed ({e})"
        ) from None


def load_weights(model_path: str | Path) -> Dict[str, torch.Tensor]:
    """Glob and merge all *.safetensors in the checkpoint directory
    (reference: /root/reference/shard/utils.py:40-45)."""
    model_path = Path(model_path)
    files = sorted(glob.glob(str(model_path / "*.safetensors")))
    if not files:
        raise FileNotFoundError(f"no *.safetensors found in {model_path}")
    weights: Dict[str, torch.Tensor] = {}
    for f in files:
        weights.update(load_file(f))
    return weights


def make_quant_predicate(config: ModelConfig, weights: Dict[str, torch.Tensor]):
    """quant_for(prefix) -> Optional[QuantConfig].

    A module is quantized iff the checkpoint carries a `{prefix}.scales`
    tensor — the reference's class_predicate
    (/root/reference/shard/utils.py:54-65)."""
    qc = config.quantization
    if qc is None:
        return lambda prefix: None
    scale_prefixes = {k[: -len(".scales")] for k in weights if k.endswith(".scales")}
    # stacked expert weights sanitize to switch_mlp.*; map their per-expert form too
    extra = set()
    for p in scale_prefixes:
        if ".experts." in p:
            head, _, tail = p.partition(".experts.")
            proj = tail.split(".", 1)[1] if "." in tail else tail
            extra.add(f"{head}.switch_mlp.{proj}")
    scale_prefixes |= extra

    def quant_for(prefix: str) -> Optional[QuantConfig]:
        return qc if prefix in scale_prefixes else None

    return quant_for


_NO_QUANT_KEYS = ("embed_tokens", "norm", "layernorm", "gate.weight",
                  "rotary", "inv_freq")


def quantize_weights(weights: Dict[str, torch.Tensor], config: ModelConfig,
                     group_size: int = 64, bits: int = 4) -> None:
    """Quantize a bf16/fp16 checkpoint's linear weights IN PLACE to the
    MLX affine triplet layout (weight packed uint32 + scales + biases) —
    so any dense checkpoint can run the int4/int8 path, which on this
    hardware is the FAST path (docs/PERFORMANCE.md).  Embeddings, norms
    and tiny router gates stay dense (mirrors what MLX's nn.quantize
    covers for these models).  Sets config's quantization stanza."""
    from ..ops import reference as ref
    for k in list(weights.keys()):
        w = weights[k]
        if (not k.endswith(".weight") or w.dim() not in (2, 3)
                or not w.is_floating_point()
                or w.shape[-1] % group_size != 0
                or any(tag in k for tag in _NO_QUANT_KEYS)):
            continue
        if w.dim() == 3:  # stacked [E, out, in] expert weights
            trip = [ref.quantize(w[e].to(torch.bfloat16), group_size, bits)
                    for e in range(w.shape[0])]
            wq = torch.stack([t[0] for t in trip])
            sc = torch.stack([t[1] for t in trip])
            bi = torch.stack([t[2] for t in trip])
        else:
            wq, sc, bi = ref.quantize(w.to(torch.bfloat16), group_size, bits)
        weights[k] = wq
        weights[k[:-len(".weight")] + ".scales"] = sc
        weights[k[:-len(".weight")] + ".biases"] = bi
    config.raw["quantization"] = {"group_size": group_size, "bits": bits}


def load_model(model_path: str | Path,
               start_layer: Optional[int] = None,
               end_layer: Optional[int] = None,
               device: str = "cpu",
               dtype: Optional[torch.dtype] = None,
               quantize: Optional[Tuple[int, int]] = None) -> Tuple[StageModel, ModelConfig]:
    """Load one pipeline stage from a checkpoint directory.

    Works with both pre-sharded checkpoints (config.json carries
    start/end_layer) and full checkpoints + explicit CLI range —
    equivalent by the shared key-routing rule.
    ``model_path`` may be a local directory or a HF hub repo id
    (resolved via the snapshot cache, reference utils.py:33-39).
    ``quantize=(bits, group_size)`` converts a dense checkpoint to the
    w4a16/w8a16 layout at load (quantize_weights)."""
    model_path = get_model_path(model_path)
    config = ModelConfig.load(model_path)
    shard = config.shard(start_layer, end_layer)
    weights = load_weights(model_path)
    if quantize is not None and config.quantization is None:
        bits, gs = quantize
        quantize_weights(weights, config, group_size=gs, bits=bits)
    cls = get_model_class(config.model_type)
    quant_for = make_quant_predicate(config, weights)
    model = cls(config, shard, quant_for=quant_for)
    if dtype is not None:
        for k in list(weights):
            if weights[k].is_floating_point():
                weights[k] = weights[k].to(dtype)
    model.load_weights(weights)
    model.to(device)
    model.eval()
    for p in model.parameters():
        p.requires_grad_(False)
    from ..models.fuse import fuse_model
    fuse_model(model)  # after .to(device): params become views of fused buffers
    return model, config


# ---------------------------------------------------------------------------
# Offline splitter (sharding_weight.py compatibility surface)
# ---------------------------------------------------------------------------

def _route_key(key: str, start: int, end: int, total: int,
               embed_on_last: bool = False) -> bool:
    """The reference's routing rule (/root/reference/sharding_weight.py:17-24).

    embed_on_last fixes a reference asymmetry for tied-head models
    (gemma2): its DYNAMIC path keeps embeddings on the last shard
    (shard/server/model/gemma2.py:98) but its offline splitter routes
    them only to the first (sharding_weight.py:21), so a pre-sharded
    gemma2 last stage could never tie its output head.  Emitting the
    extra key is a superset the reference's own sanitize accepts."""
    if key.startswith("model.layers."):
        idx = int(key.split(".")[2])
        return start <= idx < end
    if key.startswith("model.embed_tokens"):
        return start == 0 or (embed_on_last and end == total)
    if key.startswith("model.norm") or key.startswith("lm_head"):
        return end == total
    return False


def save_sharded_weights(model_path: str | Path, output_dir: str | Path,
                         start_layer: int, end_layer: int) -> Path:
    """Write one stage's weights + rewritten index + config.json, and copy
    tokenizer/aux files — byte-layout compatible with the reference
    splitter (/root/reference/sharding_weight.py:10-71)."""
    model_path = Path(model_path)
    output_dir = Path(output_dir)
    output_dir.mkdir(parents=True, exist_ok=True)
    config = ModelConfig.load(model_path)
    total = config.num_hidden_layers
    weights = load_weights(model_path)
    embed_on_last = config.model_type == "gemma2"  # tied head on last shard
    kept = {k: v for k, v in weights.items()
            if _route_key(k, start_layer, end_layer, total, embed_on_last)}
    shard_name = f"model-{start_layer:05d}-{end_layer:05d}.safetensors"
    save_file(kept, str(output_dir / shard_name), metadata={"format": "mlx"})

    # rewritten index, named with the shard range like the reference
    # (sharding_weight.py:42-46: model-SSSSS-EEEEE.safetensors.index.json)
    index = {
        "metadata": {"total_size": sum(v.numel() * v.element_size() for v in kept.values())},
        "weight_map": {k: shard_name for k in kept},
    }
    with open(output_dir / f"{shard_name}.index.json", "w") as f:
        json.dump(index, f, indent=2)

    cfg = dict(config.raw)
    cfg["start_layer"] = start_layer
    cfg["end_layer"] = end_layer
    with open(output_dir / "config.json", "w") as f:
        json.dump(cfg, f, indent=2)

    copy_other_files(model_path, output_dir)
    return output_dir / shard_name


def copy_other_files(model_path: Path, output_dir: Path):
    """Copy tokenizer and aux files, excluding weights/index/config;
    subdirectories are copied recursively like the reference's copytree
    (sharding_weight.py:63-71)."""
    for p in Path(model_path).iterdir():
        if p.is_dir():
            shutil.copytree(p, output_dir / p.name, symlinks=False,
                            dirs_exist_ok=True)
            continue
        if p.suffix == ".safetensors" or p.name in (
                "config.json", "model.safetensors.index.json"):
            continue
        shutil.copy2(p, output_dir / p.name)
