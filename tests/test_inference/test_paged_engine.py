"""Paged continuous-batching engine vs full-recompute oracle (CPU), block
accounting, and mid-stream admission."""

import pytest
import torch

from colossalai_amd.inference import ContinuousBatchEngine, GenerationConfig, InferenceConfig
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM


def _tiny():
    return LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                       num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128)


def _oracle_generate(model, prompt, n_new):
    seq = list(prompt)
    for _ in range(n_new):
        x = torch.tensor([seq])
        logits = model(x)["logits"][0, -1]
        seq.append(int(logits.argmax()))
    return seq


def test_paged_engine_matches_oracle_cpu():
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny()).eval()
    engine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=4, max_input_len=32,
                                                          max_output_len=16), block_size=4)
    prompts = [[5, 17, 42, 7], [99, 3], [1, 2, 3, 4, 5, 6, 7], [88]]
    out = engine.generate(prompts, GenerationConfig(max_new_tokens=8))
    for p, o in zip(prompts, out):
        ref = _oracle_generate(model, p, 8)
        assert o == ref, f"paged engine {o} vs oracle {ref}"
    # all blocks returned after the batch drains
    assert engine.kv.free_blocks == engine.kv.num_blocks


def test_continuous_admission():
    """More requests than batch slots: later requests are admitted as
    earlier ones retire, and all outputs still match the oracle."""
    torch.manual_seed(0)
    model = LlamaForCausalLM(_tiny()).eval()
    engine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=2, max_input_len=32,
                                                          max_output_len=8), block_size=4)
    engine._gen = GenerationConfig(max_new_tokens=4)
    prompts = [[5, 17, 42], [99, 3], [1, 2, 3, 4], [7, 8]]
    ids = [engine.add_request(p, 4) for p in prompts]
    assert len(engine.rm.waiting) == 4
    results = {}
    steps = 0
    while engine.rm.has_work:
        assert len(engine.rm.running) <= 2
        results.update(engine.step())
        steps += 1
        assert steps < 50
    for rid, p in zip(ids, prompts):
        assert results[rid] == _oracle_generate(model, p, 4)


def test_kv_manager_accounting():
    from colossalai_amd.inference import KVCacheManager

    kv = KVCacheManager(num_layers=1, num_kv_heads=2, head_dim=8, num_blocks=8, block_size=4,
                        device="cpu", dtype=torch.float32)
    assert kv.can_allocate(32) and not kv.can_allocate(33)
    kv.allocate(0, 10)  # 3 blocks
    assert kv.free_blocks == 5
    kv.extend(0, 13)  # 4th block
    assert kv.free_blocks == 4
    k = torch.randn(10, 2, 8)
    v = torch.randn(10, 2, 8)
    kv.write_prefill(0, 0, k, v)
    kk, vv = kv.gather_contiguous(0, 0, 10)
    assert torch.equal(kk, k) and torch.equal(vv, v)
    kv.write_token(0, 0, 12, k[0], v[0])
    kk, _ = kv.gather_contiguous(0, 0, 13)
    assert torch.equal(kk[12], k[0])
    kv.free(0)
    assert kv.free_blocks == 8


@pytest.mark.gpu
def test_paged_engine_gpu_matches_contiguous_engine():
    from colossalai_amd.inference import LLMEngine

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=512, hidden_size=256, intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=2, num_key_value_heads=2, max_position_embeddings=512)
    model = LlamaForCausalLM(cfg).to("cuda").bfloat16().eval()
    icfg = InferenceConfig(max_batch_size=4, max_input_len=64, max_output_len=32)
    prompts = [[5, 17, 42, 7, 100, 250], [99, 3, 4], [1, 2, 3, 4, 5], [300]]
    ref = LLMEngine(model, icfg).generate(prompts, GenerationConfig(max_new_tokens=12))
    out = ContinuousBatchEngine(model, icfg, block_size=16).generate(
        prompts, GenerationConfig(max_new_tokens=12))
    assert out == ref, f"paged {out} vs contiguous {ref}"


def test_async_engine_concurrent():
    """Concurrent awaits share the continuous batch; results match the
    synchronous engine token-for-token."""
    import asyncio

    from colossalai_amd.inference import AsyncInferenceEngine, ContinuousBatchEngine, InferenceConfig
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128)
    model = LlamaForCausalLM(cfg).eval()
    prompts = [[5, 17, 42], [99, 3, 4, 7], [1, 2]]

    from colossalai_amd.inference import GenerationConfig

    ref_engine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=4, max_input_len=16,
                                                              max_output_len=8), block_size=4)
    ref = ref_engine.generate(prompts, GenerationConfig(max_new_tokens=8))

    engine = ContinuousBatchEngine(model, InferenceConfig(max_batch_size=4, max_input_len=16,
                                                          max_output_len=8), block_size=4)
    aeng = AsyncInferenceEngine(engine)

    async def main():
        return await asyncio.gather(*(aeng.submit(p, 8) for p in prompts))

    outs = asyncio.run(main())
    assert outs == ref


def _run_tp_paged(rank, world_size, port):
    """TP-sharded continuous batching matches the unsharded engine."""
    import copy

    import torch.distributed as dist

    import colossalai_amd
    from colossalai_amd.inference import ContinuousBatchEngine, GenerationConfig, InferenceConfig
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.shardformer import ShardConfig, ShardFormer

    colossalai_amd.launch(rank, world_size, "127.0.0.1", port, backend="gloo", verbose=False)
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=128)
    model = LlamaForCausalLM(cfg).eval()
    prompts = [[5, 17, 42], [99, 3, 4, 7]]
    icfg = InferenceConfig(max_batch_size=2, max_input_len=16, max_output_len=8)
    ref = ContinuousBatchEngine(model, icfg, block_size=4).generate(
        prompts, GenerationConfig(max_new_tokens=6))

    sharded, _ = ShardFormer(ShardConfig(tensor_parallel_process_group=dist.group.WORLD,
                                         parallel_output=False)).optimize(copy.deepcopy(model))
    eng = ContinuousBatchEngine(sharded.eval(), icfg, block_size=4)
    out = eng.generate(prompts, GenerationConfig(max_new_tokens=6))
    assert out == ref, f"tp paged {out} vs ref {ref}"
    dist.destroy_process_group()


def test_paged_engine_tp2():
    from colossalai_amd.testing import rerun_if_address_is_in_use, spawn

    rerun_if_address_is_in_use()(lambda: spawn(_run_tp_paged, 2))()
