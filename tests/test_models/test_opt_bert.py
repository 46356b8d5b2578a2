"""Native OPT + BERT vs HF transformers parity (CPU) and train-step smoke."""

import pytest
import torch

transformers = pytest.importorskip("transformers")


def test_native_opt_matches_hf():
    from transformers import OPTConfig as HFConfig
    from transformers import OPTForCausalLM as HFOPT

    from colossalai_amd.models.opt import OPTConfig, OPTForCausalLM, hf_opt_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, hidden_size=64, ffn_dim=128, num_hidden_layers=2,
                      num_attention_heads=4, max_position_embeddings=64,
                      do_layer_norm_before=True, word_embed_proj_dim=64,
                      attn_implementation="eager", dropout=0.0)
    hf = HFOPT(hf_cfg).eval()
    native = OPTForCausalLM(OPTConfig(vocab_size=256, hidden_size=64, ffn_dim=128,
                                      num_hidden_layers=2, num_attention_heads=4,
                                      max_position_embeddings=64)).eval()
    missing, unexpected = native.load_state_dict(hf_opt_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    x = torch.randint(0, 256, (2, 32))
    with torch.no_grad():
        ref = hf(x).logits
        out = native(x)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)


def test_opt_train_step():
    from colossalai_amd.models.opt import OPTConfig, OPTForCausalLM

    torch.manual_seed(0)
    m = OPTForCausalLM(OPTConfig(vocab_size=256, hidden_size=64, ffn_dim=128,
                                 num_hidden_layers=2, num_attention_heads=4,
                                 max_position_embeddings=64))
    x = torch.randint(0, 256, (2, 32))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_native_bert_mlm_matches_hf():
    from transformers import BertConfig as HFConfig
    from transformers import BertForMaskedLM as HFBert

    from colossalai_amd.models.bert import BertConfig, BertForMaskedLM, hf_bert_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
                      intermediate_size=128, max_position_embeddings=64,
                      attn_implementation="eager", hidden_dropout_prob=0.0,
                      attention_probs_dropout_prob=0.0)
    hf = HFBert(hf_cfg).eval()
    native = BertForMaskedLM(BertConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2,
                                        num_attention_heads=4, intermediate_size=128,
                                        max_position_embeddings=64)).eval()
    missing, unexpected = native.load_state_dict(hf_bert_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    x = torch.randint(0, 256, (2, 32))
    mask = torch.ones(2, 32, dtype=torch.long)
    mask[:, -5:] = 0  # padding exercises the additive-mask path
    with torch.no_grad():
        ref = hf(x, attention_mask=mask).logits
        out = native(x, attention_mask=mask)["logits"]
    torch.testing.assert_close(out, ref, rtol=2e-3, atol=2e-3)


def test_native_bert_cls_matches_hf():
    from transformers import BertConfig as HFConfig
    from transformers import BertForSequenceClassification as HFBert

    from colossalai_amd.models.bert import BertConfig, BertForSequenceClassification, hf_bert_to_native

    torch.manual_seed(0)
    hf_cfg = HFConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
                      intermediate_size=128, max_position_embeddings=64, num_labels=3,
                      attn_implementation="eager", hidden_dropout_prob=0.0,
                      attention_probs_dropout_prob=0.0, classifier_dropout=0.0)
    hf = HFBert(hf_cfg).eval()
    native = BertForSequenceClassification(
        BertConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2, num_attention_heads=4,
                   intermediate_size=128, max_position_embeddings=64, num_labels=3)).eval()
    missing, unexpected = native.load_state_dict(hf_bert_to_native(hf.state_dict()), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected

    x = torch.randint(0, 256, (2, 32))
    y = torch.randint(0, 3, (2,))
    with torch.no_grad():
        ref = hf(x, labels=y)
        out = native(x, labels=y)
    torch.testing.assert_close(out["logits"], ref.logits, rtol=2e-3, atol=2e-3)
    torch.testing.assert_close(out["loss"], ref.loss, rtol=1e-3, atol=1e-4)


def test_bert_train_step():
    from colossalai_amd.models.bert import BertConfig, BertForMaskedLM

    torch.manual_seed(0)
    m = BertForMaskedLM(BertConfig(vocab_size=256, hidden_size=64, num_hidden_layers=2,
                                   num_attention_heads=4, intermediate_size=128,
                                   max_position_embeddings=64))
    x = torch.randint(0, 256, (2, 32))
    out = m(x, labels=x)
    out["loss"].backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n
