"""LLMEngine vs full-recompute oracle (CPU; GPU variant exercises HIP kernels)."""

import pytest
import torch

from colossalai_amd.inference import GenerationConfig, InferenceConfig, LLMEngine
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

from colossalai_amd.inference.config import GenerationConfig  # noqa: F811


def _tiny():
    return LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128)


def _oracle_generate(model, prompt, n_new):
    """Greedy decode by full recomputation each step."""
    seq = list(prompt)
    for _ in range(n_new):
        x = torch.tensor([seq])
        logits = model(x)["logits"][0, -1]
        seq.append(int(logits.argmax()))
    return seq


def test_engine_matches_full_recompute_cpu():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny()).eval()
    engine = LLMEngine(model, InferenceConfig(max_batch_size=2, max_input_len=32, max_output_len=16))
    prompts = [[5, 17, 42, 7], [99, 3]]
    out = engine.generate(prompts, GenerationConfig(max_new_tokens=8))
    for p, o in zip(prompts, out):
        ref = _oracle_generate(model, p, 8)
        assert o == ref, f"engine {o} vs oracle {ref}"


@pytest.mark.gpu
def test_engine_gpu():
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=512, hidden_size=256, intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=2, num_key_value_heads=2, max_position_embeddings=512)
    model_cpu = LlamaForCausalLM(cfg).eval()
    model = LlamaForCausalLM(cfg)
    model.load_state_dict(model_cpu.state_dict())
    model = model.to("cuda").bfloat16().eval()
    engine = LLMEngine(model, InferenceConfig(max_batch_size=2, max_input_len=64, max_output_len=32))
    prompts = [[5, 17, 42, 7, 100, 250], [99, 3, 4]]
    out = engine.generate(prompts, GenerationConfig(max_new_tokens=12))
    for p, o in zip(prompts, out):
        assert len(o) == len(p) + 12
        assert all(0 <= t < cfg.vocab_size for t in o)
    # decode kernel numerics: compare against bf16 CPU full recompute for a few tokens
    ref = _oracle_generate(model_cpu.bfloat16(), prompts[0], 4)
    assert out[0][: len(prompts[0]) + 2] == ref[: len(prompts[0]) + 2], (out[0], ref)


@pytest.mark.gpu
def test_engine_hip_graph_matches_eager():
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=512, hidden_size=256, intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=2, num_key_value_heads=2, max_position_embeddings=512)
    model = LlamaForCausalLM(cfg).to("cuda").bfloat16().eval()
    prompts = [[5, 17, 42, 7, 100, 250], [99, 3, 4]]
    ref = LLMEngine(model, InferenceConfig(max_batch_size=2, max_input_len=64, max_output_len=32)
                    ).generate(prompts, GenerationConfig(max_new_tokens=12))
    out = LLMEngine(model, InferenceConfig(max_batch_size=2, max_input_len=64, max_output_len=32,
                                           use_hip_graph=True)
                    ).generate(prompts, GenerationConfig(max_new_tokens=12))
    assert out == ref, f"graph {out} vs eager {ref}"


def _run_tp_engine(rank, world_size, port):
    """TP-sharded serving: tp2 engine decode must match the unsharded engine
    token-for-token (reference: inference tp via its own plugin/rpc executor)."""
    import torch.distributed as dist

    import colossalai_amd
    from colossalai_amd.shardformer import ShardConfig, ShardFormer

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny()).eval()
    ref_engine = LLMEngine(model, InferenceConfig(max_batch_size=2, max_input_len=32, max_output_len=16))
    prompts = [[5, 17, 42, 7], [99, 3]]
    ref_out = ref_engine.generate(prompts, GenerationConfig(max_new_tokens=8))

    import copy

    sharded, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD,
                                         parallel_output=False)).optimize(copy.deepcopy(model))
    assert sharded.model.layers[0].self_attn.num_heads * 2 == model.model.layers[0].self_attn.num_heads
    engine = LLMEngine(sharded.eval(), InferenceConfig(max_batch_size=2, max_input_len=32, max_output_len=16))
    out = engine.generate(prompts, GenerationConfig(max_new_tokens=8))
    assert out == ref_out, f"tp engine {out} vs ref {ref_out}"
    dist.destroy_process_group()


def test_engine_tp2_matches_unsharded():
    from colossalai_amd.testing import rerun_if_address_is_in_use, spawn

    rerun_if_address_is_in_use()(lambda: spawn(_run_tp_engine, 2))()
