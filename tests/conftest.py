import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (run on MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tiny_llama_config():
    from mlx_sharding_amd.config import ModelConfig
    return ModelConfig.from_dict({
        "model_type": "llama", "hidden_size": 64, "num_hidden_layers": 4,
        "intermediate_size": 128, "num_attention_heads": 4,
        "num_key_value_heads": 2, "vocab_size": 128,
        "rms_norm_eps": 1e-5, "rope_theta": 10000.0,
    })


@pytest.fixture
def tiny_deepseek_config():
    from mlx_sharding_amd.config import ModelConfig
    return ModelConfig.from_dict({
        "model_type": "deepseek_v2", "hidden_size": 64, "num_hidden_layers": 3,
        "intermediate_size": 128, "moe_intermediate_size": 32,
        "num_attention_heads": 4, "vocab_size": 128, "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0, "q_lora_rank": None, "kv_lora_rank": 32,
        "qk_nope_head_dim": 16, "qk_rope_head_dim": 8, "v_head_dim": 16,
        "n_routed_experts": 8, "num_experts_per_tok": 2, "n_shared_experts": 1,
        "first_k_dense_replace": 1, "moe_layer_freq": 1,
        "routed_scaling_factor": 1.0, "norm_topk_prob": False,
        "rope_scaling": {"type": "yarn", "factor": 4.0, "beta_fast": 32,
                         "beta_slow": 1, "mscale": 0.707, "mscale_all_dim": 0.707,
                         "original_max_position_embeddings": 4096},
    })


@pytest.fixture
def tiny_gemma2_config():
    from mlx_sharding_amd.config import ModelConfig
    return ModelConfig.from_dict({
        "model_type": "gemma2", "hidden_size": 64, "num_hidden_layers": 4,
        "intermediate_size": 128, "num_attention_heads": 4,
        "num_key_value_heads": 2, "head_dim": 16, "vocab_size": 128,
        "rms_norm_eps": 1e-6, "rope_theta": 10000.0,
        "query_pre_attn_scalar": 16, "attn_logit_softcapping": 50.0,
        "final_logit_softcapping": 30.0, "sliding_window": 3,
    })


def init_model(cls, config, shard, seed=0, quant_for=None):
    torch.manual_seed(seed)
    m = cls(config, shard, quant_for=quant_for)
    for p in m.parameters():
        p.data = p.data.float().normal_(0, 0.05).to(p.dtype) \
            if p.is_floating_point() else p.data
    m.eval()
    return m
