"""API server over the continuous-batching engine (in-process TestClient)."""

import pytest
import torch

fastapi = pytest.importorskip("fastapi")

from colossalai_amd.inference import ContinuousBatchEngine, InferenceConfig
from colossalai_amd.inference.server import create_app
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM


def test_generate_endpoint():
    from starlette.testclient import TestClient

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128)
    model = LlamaForCausalLM(cfg).eval()
    engine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=2, max_input_len=32,
                                                          max_output_len=16), block_size=4)
    client = TestClient(create_app(engine))

    r = client.get("/health")
    assert r.status_code == 200 and r.json()["status"] == "ok"

    r = client.post("/generate", json={"prompt_ids": [5, 17, 42], "max_new_tokens": 6})
    assert r.status_code == 200
    out = r.json()["output_ids"]
    assert out[:3] == [5, 17, 42] and len(out) == 9

    r = client.post("/generate", json={"max_new_tokens": 4})
    assert r.status_code == 400  # no ids and no tokenizer


def test_openai_completions_endpoint():
    from starlette.testclient import TestClient

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128)
    model = LlamaForCausalLM(cfg).eval()
    engine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=2, max_input_len=32,
                                                          max_output_len=16), block_size=4)
    client = TestClient(create_app(engine))
    r = client.post("/v1/completions", json={"prompt_ids": [5, 17, 42], "max_tokens": 6,
                                             "temperature": 0.0})
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "text_completion"
    assert len(body["choices"]) == 1
    assert len(body["choices"][0]["token_ids"]) == 6
    assert body["usage"]["total_tokens"] == 9


def test_openai_chat_endpoint():
    from starlette.testclient import TestClient

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128)
    model = LlamaForCausalLM(cfg).eval()
    engine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=2, max_input_len=64,
                                                          max_output_len=16), block_size=4)

    class ToyTok:
        """char-level stand-in with no chat template"""

        def __call__(self, text):
            return {"input_ids": [min(ord(c), 127) for c in text][:40]}

        def decode(self, ids):
            return "".join(chr(i % 128) for i in ids)

    client = TestClient(create_app(engine, tokenizer=ToyTok()))
    r = client.post("/v1/chat/completions", json={
        "messages": [{"role": "user", "content": "hi"}], "max_tokens": 5, "temperature": 0})
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "chat.completion"
    assert body["choices"][0]["message"]["role"] == "assistant"
    assert body["usage"]["completion_tokens"] == 5

    # no tokenizer -> 400
    client2 = TestClient(create_app(engine))
    r = client2.post("/v1/chat/completions", json={
        "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 400
