import torch

from colossalai_amd.models import LlamaConfig, LlamaForCausalLM
from colossalai_amd.shardformer.layer.padded_vocab import pad_vocab, unpad_vocab_weight


def test_pad_vocab_roundtrip():
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=100, hidden_size=64, intermediate_size=128, num_hidden_layers=1,
                      num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64)
    m = LlamaForCausalLM(cfg)
    ref_logits = m(torch.randint(0, 100, (2, 8)))["logits"]
    orig, padded = pad_vocab(m, tp_size=2, make_divisible_by=64)
    assert orig == 100 and padded == 128
    assert m.model.embed_tokens.num_embeddings == 128
    assert m.lm_head.out_features == 128
    out = m(torch.randint(0, 100, (2, 8)))["logits"]
    assert out.shape[-1] == 128
    w = unpad_vocab_weight(m.lm_head.weight.data, orig)
    assert w.shape[0] == 100
