"""Failure-detection paths.

The reference swallows shard errors into ``None`` and crashes on the
next op (/root/reference/shard/utils.py:79-85).  The build fails fast
and cleanly: typed ShardUnavailable from the client, HTTP 502 from the
API, and a liveness probe."""

import json
import threading

import http.client
import pytest
import torch

from mlx_sharding_amd.parallel.grpc_transport import (ShardUnavailable,
                                                      StageClient,
                                                      serve_forward)


DEAD_ADDR = "127.0.0.1:9"  # discard port — nothing listens there


def test_dead_shard_raises_shard_unavailable():
    c = StageClient(DEAD_ADDR, retries=0, timeout_s=2.0)
    with pytest.raises(ShardUnavailable, match="unreachable"):
        c.send_tensor(torch.zeros(1, 2, 4))
    c.close()


def test_healthy_probe():
    c_dead = StageClient(DEAD_ADDR, retries=0, timeout_s=2.0)
    assert c_dead.healthy() is False
    c_dead.close()

    server = serve_forward(lambda t: t, lambda: None, port=0)
    try:
        c = StageClient(f"127.0.0.1:{server._mlxs_port}")
        assert c.healthy() is True
        c.close()
    finally:
        server.stop(0)


def test_forward_error_surfaces_with_type_name():
    def bad_forward(t):
        raise ValueError("boom")

    server = serve_forward(bad_forward, lambda: None, port=0)
    try:
        c = StageClient(f"127.0.0.1:{server._mlxs_port}")
        with pytest.raises(RuntimeError, match="ValueError: boom"):
            c.send_tensor(torch.zeros(1, 1, 4))
        c.close()
    finally:
        server.stop(0)


def test_api_returns_502_on_dead_shard(tmp_path):
    """A completion whose remote stage is dead returns a clean 502 JSON
    error, not a dropped connection."""
    from safetensors.torch import save_file
    from tokenizers import Tokenizer
    from tokenizers.models import WordLevel
    from tokenizers.pre_tokenizers import Whitespace

    from mlx_sharding_amd.config import ModelConfig
    from mlx_sharding_amd.models import get_model_class
    from mlx_sharding_amd.server import openai_api

    ckpt = tmp_path / "ckpt"
    ckpt.mkdir()
    vocab = {"<unk>": 0, "<eos>": 1, "hi": 2}
    vocab.update({f"t{i}": 3 + i for i in range(29)})
    tok = Tokenizer(WordLevel(vocab, unk_token="<unk>"))
    tok.pre_tokenizer = Whitespace()
    tok.save(str(ckpt / "tokenizer.json"))
    with open(ckpt / "tokenizer_config.json", "w") as f:
        json.dump({"tokenizer_class": "PreTrainedTokenizerFast",
                   "eos_token": "<eos>", "unk_token": "<unk>"}, f)
    cfg_raw = {"model_type": "llama", "hidden_size": 32,
               "num_hidden_layers": 2, "intermediate_size": 64,
               "num_attention_heads": 2, "num_key_value_heads": 2,
               "vocab_size": 32, "rms_norm_eps": 1e-5,
               "rope_theta": 10000.0}
    with open(ckpt / "config.json", "w") as f:
        json.dump(cfg_raw, f)
    cfg = ModelConfig.from_dict(cfg_raw)
    m = get_model_class("llama")(cfg, cfg.shard(0, 1))  # first stage only
    for p in m.parameters():
        p.data = p.data.float().normal_(0, 0.05).to(p.dtype)
    sd = {k: v for k, v in m.state_dict().items() if "rope_inv_freq" not in k}
    save_file(sd, str(ckpt / "model.safetensors"))

    class Args:
        model = str(ckpt)
        llm_shard_addresses = DEAD_ADDR  # second stage: dead
        start_layer = 0
        end_layer = 1

    provider = openai_api.ModelProvider(Args())
    for r in provider.remotes:  # fail fast in the test
        r.retries = 0
        r.timeout_s = 2.0
    server = openai_api.run("127.0.0.1", 0, provider)
    t = threading.Thread(target=server.serve_forever, daemon=True)
    t.start()
    try:
        conn = http.client.HTTPConnection("127.0.0.1",
                                          server.server_address[1], timeout=60)
        conn.request("POST", "/v1/completions",
                     json.dumps({"prompt": "hi", "max_tokens": 2,
                                 "temperature": 0}),
                     {"Content-Type": "application/json"})
        resp = conn.getresponse()
        body = json.loads(resp.read())
        conn.close()
        assert resp.status == 502
        assert "generation failed" in body["error"]
    finally:
        server.shutdown()


def test_malformed_payload_yields_clean_error():
    """Garbage bytes on the wire come back as success=False and a
    typed client error — never a server crash (the reference swallows
    this into None, utils.py:79-85)."""
    import grpc

    from mlx_sharding_amd.parallel import wire

    server = serve_forward(lambda t: t, lambda: None, port=0)
    try:
        ch = grpc.insecure_channel(f"127.0.0.1:{server._mlxs_port}")
        send = ch.unary_unary(wire.SEND_TENSOR)
        resp = send(b"\x13\x37 not a tensor message \xff")
        ok, message, tmsg = wire.decode_tensor_response(bytes(resp))
        assert ok is False
        assert message  # carries the exception type/text
        ch.close()
        # the server must still serve valid requests afterwards
        c = StageClient(f"127.0.0.1:{server._mlxs_port}")
        out = c.send_tensor(torch.ones(1, 2, 3))
        assert torch.equal(out, torch.ones(1, 2, 3))
        c.close()
    finally:
        server.stop(0)
