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


@pytest.mark.gpu
def test_new_families_gpu_fwd_bwd():
    """Every round-1 model family steps on the HIP paths (bf16, head_dim 64):
    finite loss + finite grads."""
    import torch

    from colossalai_amd.models.bert import BertConfig, BertForMaskedLM
    from colossalai_amd.models.bloom import BloomConfig, BloomForCausalLM
    from colossalai_amd.models.falcon import FalconConfig, FalconForCausalLM
    from colossalai_amd.models.opt import OPTConfig, OPTForCausalLM
    from colossalai_amd.models.t5 import T5Config, T5ForConditionalGeneration
    from colossalai_amd.models.vit import ViTConfig, ViTForImageClassification
    from colossalai_amd.models.whisper import WhisperConfig, WhisperForConditionalGeneration

    torch.manual_seed(0)
    x = torch.randint(0, 512, (2, 128), device="cuda")
    y = torch.randint(0, 512, (2, 32), device="cuda")

    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    cases = [
        (LlamaForCausalLM(LlamaConfig(vocab_size=512, hidden_size=256, intermediate_size=512,
                                      num_hidden_layers=2, num_attention_heads=4,
                                      num_key_value_heads=2, head_dim_override=64,
                                      max_position_embeddings=256, qk_norm=True)),
         lambda m: m(x, labels=x)),  # Qwen3-style qk-norm path
        (OPTForCausalLM(OPTConfig(vocab_size=512, hidden_size=256, ffn_dim=512,
                                  num_hidden_layers=2, num_attention_heads=4,
                                  max_position_embeddings=256)),
         lambda m: m(x, labels=x)),
        (BertForMaskedLM(BertConfig(vocab_size=512, hidden_size=256, num_hidden_layers=2,
                                    num_attention_heads=4, intermediate_size=512,
                                    max_position_embeddings=256)),
         lambda m: m(x, labels=x)),
        (FalconForCausalLM(FalconConfig(vocab_size=512, hidden_size=256, num_hidden_layers=2,
                                        num_attention_heads=4, max_position_embeddings=256)),
         lambda m: m(x, labels=x)),
        (BloomForCausalLM(BloomConfig(vocab_size=512, hidden_size=256, num_hidden_layers=2,
                                      num_attention_heads=4)),
         lambda m: m(x, labels=x)),
        (T5ForConditionalGeneration(T5Config(vocab_size=512, d_model=256, d_kv=64, d_ff=512,
                                             num_layers=2, num_decoder_layers=2, num_heads=4)),
         lambda m: m(x, labels=y)),
        (ViTForImageClassification(ViTConfig(image_size=64, patch_size=8, hidden_size=256,
                                             num_hidden_layers=2, num_attention_heads=4,
                                             intermediate_size=512, num_labels=7)),
         lambda m: m(torch.randn(2, 3, 64, 64, device="cuda", dtype=torch.bfloat16),
                     labels=torch.randint(0, 7, (2,), device="cuda"))),
        (WhisperForConditionalGeneration(WhisperConfig(vocab_size=512, num_mel_bins=32,
                                                       d_model=256, encoder_layers=2,
                                                       decoder_layers=2, num_heads=4, d_ff=512,
                                                       max_source_positions=64,
                                                       max_target_positions=64,
                                                       decoder_start_token_id=1, pad_token_id=0)),
         lambda m: m(torch.randn(2, 32, 128, device="cuda", dtype=torch.bfloat16), labels=y)),
    ]
    for model, run in cases:
        name = type(model).__name__
        model = model.to("cuda").bfloat16()
        out = run(model)
        assert out["loss"] is not None and torch.isfinite(out["loss"]), name
        out["loss"].backward()
        for n, p in model.named_parameters():
            assert p.grad is None or torch.isfinite(p.grad).all(), f"{name}.{n}"
