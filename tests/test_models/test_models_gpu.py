"""GPU integration tests: tiny models end-to-end on the HIP kernel path."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _llama_tiny():
    from colossalai_amd.models import LlamaConfig

    return LlamaConfig(vocab_size=512, hidden_size=256, intermediate_size=512, num_hidden_layers=2,
                       num_attention_heads=2, num_key_value_heads=2, max_position_embeddings=512)


def test_llama_fwd_bwd_matches_cpu_ref():
    """GPU bf16 model (HIP kernels) vs the same model run on CPU (torch ref ops)."""
    from colossalai_amd.models import LlamaForCausalLM

    torch.manual_seed(0)
    cfg = _llama_tiny()
    m_cpu = LlamaForCausalLM(cfg).bfloat16()
    m_gpu = LlamaForCausalLM(cfg)
    m_gpu.load_state_dict(m_cpu.state_dict())
    m_gpu = m_gpu.to("cuda").bfloat16()

    x = torch.randint(0, cfg.vocab_size, (2, 256))
    out_cpu = m_cpu(x, labels=x)
    out_gpu = m_gpu(x.cuda(), labels=x.cuda())
    assert torch.isfinite(out_gpu["loss"])
    # bf16 end-to-end: losses should agree to ~1%
    assert abs(out_gpu["loss"].item() - out_cpu["loss"].item()) < 0.05 * max(1.0, out_cpu["loss"].item())

    out_gpu["loss"].backward()
    for n, p in m_gpu.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_llama_zero2_training_loss_decreases():
    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import LowLevelZeroPlugin
    from colossalai_amd.models import LlamaForCausalLM
    from colossalai_amd.nn import FusedAdam

    torch.manual_seed(0)
    model = LlamaForCausalLM(_llama_tiny())
    optimizer = FusedAdam(model.parameters(), lr=3e-4)
    booster = Booster(plugin=LowLevelZeroPlugin(stage=2, precision="bf16"))
    model, optimizer, *_ = booster.boost(model, optimizer)

    x = torch.randint(0, 512, (4, 128), device="cuda")
    losses = []
    for _ in range(10):
        out = model(x, labels=x)
        booster.backward(out["loss"], optimizer)
        optimizer.step()
        losses.append(out["loss"].item())
    assert losses[-1] < losses[0] * 0.8, f"loss did not decrease: {losses}"


def test_mixtral_fwd_bwd_gpu():
    from colossalai_amd.models.mixtral import MIXTRAL_CONFIGS, MixtralConfig, MixtralForCausalLM

    torch.manual_seed(0)
    cfg = MixtralConfig(vocab_size=512, hidden_size=256, intermediate_size=512, num_hidden_layers=2,
                        num_attention_heads=2, num_key_value_heads=2, max_position_embeddings=512,
                        num_local_experts=4, num_experts_per_tok=2)
    m = MixtralForCausalLM(cfg).to("cuda").bfloat16()
    x = torch.randint(0, cfg.vocab_size, (2, 256), device="cuda")
    out = m(x, labels=x)
    assert torch.isfinite(out["loss"])
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_gemini_offload_step():
    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import GeminiPlugin
    from colossalai_amd.models import LlamaForCausalLM
    from colossalai_amd.nn import HybridAdam

    torch.manual_seed(0)
    model = LlamaForCausalLM(_llama_tiny())
    optimizer = HybridAdam(model.parameters(), lr=3e-4)
    booster = Booster(plugin=GeminiPlugin(offload_optim_frac=1.0, precision="bf16"))
    model, optimizer, *_ = booster.boost(model, optimizer)
    x = torch.randint(0, 512, (2, 128), device="cuda")
    l0 = None
    for _ in range(5):
        out = model(x, labels=x)
        booster.backward(out["loss"], optimizer)
        optimizer.step()
        if l0 is None:
            l0 = out["loss"].item()
    assert out["loss"].item() < l0


def test_gpt2_gpu_train_step():
    from colossalai_amd.models.gpt2 import GPT2Config, GPT2LMHeadModel

    torch.manual_seed(0)
    m = GPT2LMHeadModel(GPT2Config(vocab_size=512, n_positions=512, n_embd=256, n_layer=2, n_head=4))
    m = m.to("cuda").bfloat16()
    x = torch.randint(0, 512, (2, 256), device="cuda")
    out = m(x, labels=x)
    assert torch.isfinite(out["loss"])
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
