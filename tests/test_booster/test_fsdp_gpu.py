"""TorchFSDPPlugin smoke on a real GPU (world 1, RCCL): boost, step, save,
reload — executes the previously-untested FSDP path."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_fsdp_boost_step(tmp_path):
    if not torch.cuda.is_available():
        pytest.skip("GPU only")
    import torch.distributed as dist

    import colossalai_amd
    from colossalai_amd import Booster
    from colossalai_amd.booster.plugin.torch_fsdp_plugin import TorchFSDPPlugin
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29511")
    if not dist.is_initialized():
        colossalai_amd.launch(0, 1, "127.0.0.1", 29511, backend="nccl", verbose=False)

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=128, hidden_size=256, intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM(cfg)
    plugin = TorchFSDPPlugin(precision="bf16")
    booster = Booster(plugin=plugin)
    optimizer = torch.optim.AdamW(model.parameters(), lr=1e-2)
    model_b, optimizer_b, *_ = booster.boost(model, optimizer)

    x = torch.randint(0, 128, (2, 16), device="cuda")
    losses = []
    for _ in range(3):
        out = model_b(input_ids=x, labels=x)
        loss = out["loss"]
        assert torch.isfinite(loss)
        booster.backward(loss, optimizer_b)
        optimizer_b.step()
        optimizer_b.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0], f"FSDP training did not reduce the loss: {losses}"

    path = str(tmp_path / "fsdp_model.bin")
    booster.save_model(model_b, path)
    if dist.get_rank() == 0:
        sd = torch.load(path, weights_only=False)
        assert any("layers.0" in k for k in sd)
    dist.barrier()
