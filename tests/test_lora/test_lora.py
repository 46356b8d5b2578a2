"""LoRA: only adapters train; merge reproduces the adapted forward."""

import torch

from colossalai_amd.lora import LoraConfig, apply_lora, lora_state_dict, merge_lora
from colossalai_amd.models import LlamaConfig, LlamaForCausalLM


def test_lora_injection_and_merge():
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg)
    model = apply_lora(model, LoraConfig(r=4, lora_alpha=8))

    trainable = [n for n, p in model.named_parameters() if p.requires_grad]
    assert trainable and all("lora_" in n for n in trainable), trainable[:5]

    x = torch.randint(0, 128, (2, 16))
    out = model(x, labels=x)
    out["loss"].backward()
    for n, p in model.named_parameters():
        if p.requires_grad:
            assert p.grad is not None, n

    # make adapters non-trivial, then merging must preserve the function
    with torch.no_grad():
        for n, p in model.named_parameters():
            if "lora_B" in n:
                p.normal_(0, 0.05)
    ref_logits = model(x)["logits"]
    sd = lora_state_dict(model)
    assert sd and all("lora_" in k for k in sd)
    merged = merge_lora(model)
    torch.testing.assert_close(merged(x)["logits"], ref_logits, rtol=1e-4, atol=1e-5)


def test_booster_enable_lora():
    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin import TorchDDPPlugin

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=1,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg)
    booster = Booster(plugin=TorchDDPPlugin())
    model = booster.enable_lora(model)
    assert any(p.requires_grad and "lora" in n for n, p in model.named_parameters())


def test_qlora_int8_base():
    """QLoRA flow (reference: bnb-quantized base + LoRA adapters): int8
    frozen base, trainable adapters, loss decreases."""
    from colossalai_amd.lora import LoraConfig, apply_lora
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
    from colossalai_amd.quantization import W8Linear, quantize_model

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = quantize_model(LlamaForCausalLM(cfg), bits=8)
    model = apply_lora(model, LoraConfig(r=4, lora_alpha=8))
    # adapters wrap the quantized projections
    attn = model.model.layers[0].self_attn
    assert isinstance(attn.qkv_proj.base, W8Linear)
    trainable = [p for p in model.parameters() if p.requires_grad]
    assert all(p.shape[0] == 4 or p.shape[1] == 4 for p in trainable)

    opt = torch.optim.AdamW(trainable, lr=1e-2)
    x = torch.randint(0, 128, (2, 16))
    losses = []
    for _ in range(6):
        out = model(x, labels=x)
        out["loss"].backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(out["loss"]))
    assert losses[-1] < losses[0]
