"""Named model-architecture presets for bench/tests (random-init weights;
there is no network for checkpoints).  Shapes follow the published HF
configs of the models BASELINE.json names."""

from __future__ import annotations

from ..config import ModelConfig

PRESETS = {
    # headline model: DeepSeek-Coder-V2-Lite-Instruct (15.7B total, 2.4B active)
    "deepseek-v2-lite": {
        "model_type": "deepseek_v2",
        "hidden_size": 2048,
        "num_hidden_layers": 27,
        "intermediate_size": 10944,
        "moe_intermediate_size": 1408,
        "num_attention_heads": 16,
        "vocab_size": 102400,
        "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0,
        "q_lora_rank": None,
        "kv_lora_rank": 512,
        "qk_nope_head_dim": 128,
        "qk_rope_head_dim": 64,
        "v_head_dim": 128,
        "n_routed_experts": 64,
        "n_shared_experts": 2,
        "num_experts_per_tok": 6,
        "n_group": 1,
        "topk_group": 1,
        "topk_method": "greedy",
        "norm_topk_prob": False,
        "routed_scaling_factor": 1.0,
        "first_k_dense_replace": 1,
        "moe_layer_freq": 1,
        "max_position_embeddings": 163840,
        "rope_scaling": {"type": "yarn", "factor": 40.0, "beta_fast": 32,
                         "beta_slow": 1, "mscale": 0.707, "mscale_all_dim": 0.707,
                         "original_max_position_embeddings": 4096},
    },
    "llama-3-8b": {
        "model_type": "llama",
        "hidden_size": 4096,
        "num_hidden_layers": 32,
        "intermediate_size": 14336,
        "num_attention_heads": 32,
        "num_key_value_heads": 8,
        "vocab_size": 128256,
        "rms_norm_eps": 1e-5,
        "rope_theta": 500000.0,
        "max_position_embeddings": 8192,
    },
    "llama-3-70b": {
        "model_type": "llama",
        "hidden_size": 8192,
        "num_hidden_layers": 80,
        "intermediate_size": 28672,
        "num_attention_heads": 64,
        "num_key_value_heads": 8,
        "vocab_size": 128256,
        "rms_norm_eps": 1e-5,
        "rope_theta": 500000.0,
        "max_position_embeddings": 8192,
    },
    "gemma-2-9b": {
        "model_type": "gemma2",
        "hidden_size": 3584,
        "num_hidden_layers": 42,
        "intermediate_size": 14336,
        "num_attention_heads": 16,
        "num_key_value_heads": 8,
        "head_dim": 256,
        "vocab_size": 256000,
        "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0,
        "query_pre_attn_scalar": 256,
        "attn_logit_softcapping": 50.0,
        "final_logit_softcapping": 30.0,
        "sliding_window": 4096,
        "max_position_embeddings": 8192,
    },
    "tinyllama-1.1b": {
        "model_type": "llama",
        "hidden_size": 2048,
        "num_hidden_layers": 22,
        "intermediate_size": 5632,
        "num_attention_heads": 32,
        "num_key_value_heads": 4,
        "vocab_size": 32000,
        "rms_norm_eps": 1e-5,
        "rope_theta": 10000.0,
    },
    # Qwen2-7B-Instruct shapes (config.json of Qwen/Qwen2-7B-Instruct);
    # llama-family remap with QKV-only attention biases
    "qwen2-7b": {
        "model_type": "qwen2",
        "hidden_size": 3584,
        "num_hidden_layers": 28,
        "intermediate_size": 18944,
        "num_attention_heads": 28,
        "num_key_value_heads": 4,
        "vocab_size": 152064,
        "rms_norm_eps": 1e-6,
        "rope_theta": 1000000.0,
        "tie_word_embeddings": False,
        "max_position_embeddings": 32768,
    },

    # small debug model (CPU-runnable; 8 layers so a CPU smoke of the
    # 8-stage pipeline gives every rank at least one layer)
    "debug-llama": {
        "model_type": "llama",
        "hidden_size": 256,
        "num_hidden_layers": 8,
        "intermediate_size": 512,
        "num_attention_heads": 8,
        "num_key_value_heads": 4,
        "vocab_size": 1024,
        "rms_norm_eps": 1e-5,
        "rope_theta": 10000.0,
    },
    "debug-deepseek": {
        "model_type": "deepseek_v2",
        "hidden_size": 128,
        "num_hidden_layers": 8,
        "intermediate_size": 256,
        "moe_intermediate_size": 64,
        "num_attention_heads": 4,
        "vocab_size": 512,
        "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0,
        "q_lora_rank": None,
        "kv_lora_rank": 64,
        "qk_nope_head_dim": 32,
        "qk_rope_head_dim": 16,
        "v_head_dim": 32,
        "n_routed_experts": 8,
        "n_shared_experts": 1,
        "num_experts_per_tok": 2,
        "first_k_dense_replace": 1,
        "moe_layer_freq": 1,
        "rope_scaling": {"type": "yarn", "factor": 4.0, "beta_fast": 32,
                         "beta_slow": 1, "mscale": 0.707, "mscale_all_dim": 0.707,
                         "original_max_position_embeddings": 4096},
    },
}


def get_preset(name: str, quant: bool = False,
               group_size: int = 64, bits: int = 4) -> ModelConfig:
    raw = dict(PRESETS[name])
    if quant:
        raw["quantization"] = {"group_size": group_size, "bits": bits}
    return ModelConfig.from_dict(raw)
