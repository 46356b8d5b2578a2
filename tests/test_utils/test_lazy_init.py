

def test_materialize_from_state_dict():
    """Meta skeleton + checkpoint weights -> working model with identical
    outputs to the source (the from_pretrained path, no double allocation)."""
    import torch

    from colossalai_amd.lazy import LazyInitContext
    from colossalai_amd.models import LlamaConfig, LlamaForCausalLM

    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    torch.manual_seed(0)
    src = LlamaForCausalLM(cfg).float()
    sd = src.state_dict()
    # pretend the checkpoint came in two shards
    keys = sorted(sd)
    shards = [{k: sd[k] for k in keys[: len(keys) // 2]}, {k: sd[k] for k in keys[len(keys) // 2:]}]

    with LazyInitContext():
        model = LlamaForCausalLM(cfg)
    assert all(p.is_meta for p in model.parameters())
    model = LazyInitContext.materialize_from_state_dict(model, shards, device="cpu",
                                                        dtype=torch.float32, strict=True)
    x = torch.randint(0, 128, (2, 16))
    torch.testing.assert_close(model(x, labels=x)["loss"], src(x, labels=x)["loss"])
