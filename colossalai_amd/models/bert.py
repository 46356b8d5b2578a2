"""MI355X-native BERT family (bidirectional encoder).

Post-LN encoder: packed QKV GEMM feeding the native flash-attention kernel
with ``causal=False`` (full attention), HIP LayerNorm, tanh-GELU MLP,
learned position + token-type embeddings. Heads: masked-LM (transform +
tied decoder) and sequence classification (pooler + classifier).
`hf_bert_to_native` maps transformers BertForMaskedLM /
BertForSequenceClassification state dicts.

With an ``attention_mask`` (padding) the attention falls back to the
additive-mask reference path; the unpadded fast path uses the flash
kernel. Reference parity target: transformers Bert* as sharded by
colossalai/shardformer/policies/bert.py.
"""

import math
from dataclasses import dataclass
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import flash_attention, layer_norm

__all__ = [
    "BertConfig", "BertModel", "BertForMaskedLM", "BertForSequenceClassification",
    "BERT_CONFIGS", "hf_bert_to_native",
]


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    intermediate_size: int = 3072
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    layer_norm_eps: float = 1e-12
    initializer_range: float = 0.02
    num_labels: int = 2

    @property
    def head_dim(self) -> int:
        return self.hidden_size // self.num_attention_heads


BERT_CONFIGS = {
    "bert-base": BertConfig(),
    "bert-large": BertConfig(hidden_size=1024, num_hidden_layers=24, num_attention_heads=16,
                             intermediate_size=4096),
}


class BertAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.num_heads = cfg.num_attention_heads
        self.head_dim = cfg.head_dim
        self.qkv = nn.Linear(cfg.hidden_size, 3 * cfg.hidden_size, bias=True)
        self.out = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=True)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, hidden: torch.Tensor, attention_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        B, S, _ = hidden.shape
        H, D = self.num_heads, self.head_dim
        qkv = self.qkv(hidden)
        q = qkv[:, :, : H * D].view(B, S, H, D)
        k = qkv[:, :, H * D : 2 * H * D].view(B, S, H, D)
        v = qkv[:, :, 2 * H * D :].view(B, S, H, D)
        if attention_mask is None and D in (64, 128) and hidden.dtype == torch.bfloat16:
            attn = flash_attention(q, k, v, causal=False, scale=self.scale)
        else:
            from ..ops.attention import attention_ref

            bias = None
            if attention_mask is not None:
                # [B, S] 1=keep 0=pad → additive [B, 1, 1, S]
                bias = (1.0 - attention_mask[:, None, None, :].to(q.dtype)) * -1e9
            attn = attention_ref(q, k, v, causal=False, scale=self.scale, upcast=False, bias=bias)
        return self.out(attn.reshape(B, S, H * D))


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.eps = cfg.layer_norm_eps
        self.attention = BertAttention(cfg)
        self.attn_ln_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.attn_ln_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.intermediate = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=True)
        self.output = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=True)
        self.out_ln_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.out_ln_bias = nn.Parameter(torch.zeros(cfg.hidden_size))

    def forward(self, hidden, attention_mask=None):
        # post-LN: residual add THEN norm (original BERT)
        hidden = layer_norm(hidden + self.attention(hidden, attention_mask),
                            self.attn_ln_weight, self.attn_ln_bias, self.eps)
        mlp = self.output(F.gelu(self.intermediate(hidden), approximate="tanh"))
        return layer_norm(hidden + mlp, self.out_ln_weight, self.out_ln_bias, self.eps)


class BertModel(nn.Module):
    def __init__(self, cfg: BertConfig, add_pooler: bool = True):
        super().__init__()
        self.cfg = cfg
        self.word_embeddings = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.position_embeddings = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        self.token_type_embeddings = nn.Embedding(cfg.type_vocab_size, cfg.hidden_size)
        self.emb_ln_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.emb_ln_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.layers = nn.ModuleList(BertLayer(cfg) for _ in range(cfg.num_hidden_layers))
        self.pooler = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=True) if add_pooler else None
        self.gradient_checkpointing = False

    def forward(self, input_ids, attention_mask=None, token_type_ids=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        hidden = (self.word_embeddings(input_ids) + self.position_embeddings(pos)[None]
                  + self.token_type_embeddings(token_type_ids))
        hidden = layer_norm(hidden, self.emb_ln_weight, self.emb_ln_bias, self.cfg.layer_norm_eps)
        for layer in self.layers:
            if self.gradient_checkpointing and self.training:
                hidden = torch.utils.checkpoint.checkpoint(layer, hidden, attention_mask, use_reentrant=False)
            else:
                hidden = layer(hidden, attention_mask)
        pooled = torch.tanh(self.pooler(hidden[:, 0])) if self.pooler is not None else None
        return hidden, pooled


class BertForMaskedLM(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.config = cfg
        self.bert = BertModel(cfg, add_pooler=False)
        self.transform = nn.Linear(cfg.hidden_size, cfg.hidden_size, bias=True)
        self.transform_ln_weight = nn.Parameter(torch.ones(cfg.hidden_size))
        self.transform_ln_bias = nn.Parameter(torch.zeros(cfg.hidden_size))
        self.decoder = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=True)
        self.decoder.weight = self.bert.word_embeddings.weight  # tied
        self.apply(_bert_init(cfg))

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.bert.gradient_checkpointing = True

    def forward(self, input_ids, attention_mask=None, token_type_ids=None, labels=None):
        hidden, _ = self.bert(input_ids, attention_mask, token_type_ids)
        h = F.gelu(self.transform(hidden), approximate="tanh")
        h = layer_norm(h, self.transform_ln_weight, self.transform_ln_bias, self.config.layer_norm_eps)
        logits = self.decoder(h)
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits.float().view(-1, logits.shape[-1]), labels.view(-1),
                                   ignore_index=-100)
        return {"logits": logits, "loss": loss}


class BertForSequenceClassification(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.config = cfg
        self.bert = BertModel(cfg, add_pooler=True)
        self.classifier = nn.Linear(cfg.hidden_size, cfg.num_labels, bias=True)
        self.apply(_bert_init(cfg))

    def gradient_checkpointing_enable(self, ratio: float = 1.0):
        self.bert.gradient_checkpointing = True

    def forward(self, input_ids, attention_mask=None, token_type_ids=None, labels=None):
        _, pooled = self.bert(input_ids, attention_mask, token_type_ids)
        logits = self.classifier(pooled)
        loss = None
        if labels is not None:
            loss = F.cross_entropy(logits.float(), labels.view(-1))
        return {"logits": logits, "loss": loss}


def _bert_init(cfg: BertConfig):
    def init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            m.weight.data.normal_(0.0, cfg.initializer_range)
            if isinstance(m, nn.Linear) and m.bias is not None:
                m.bias.data.zero_()
    return init


def hf_bert_to_native(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map transformers BertForMaskedLM / BertForSequenceClassification
    state dicts; attention.self.{query,key,value} pack into qkv."""
    out = {}
    qkv: Dict[str, Dict[str, torch.Tensor]] = {}
    for k, v in hf_sd.items():
        nk = k
        nk = nk.replace("bert.embeddings.word_embeddings", "bert.word_embeddings")
        nk = nk.replace("bert.embeddings.position_embeddings", "bert.position_embeddings")
        nk = nk.replace("bert.embeddings.token_type_embeddings", "bert.token_type_embeddings")
        nk = nk.replace("bert.embeddings.LayerNorm.weight", "bert.emb_ln_weight")
        nk = nk.replace("bert.embeddings.LayerNorm.bias", "bert.emb_ln_bias")
        nk = nk.replace("bert.encoder.layer.", "bert.layers.")
        if ".attention.self." in nk:
            layer = nk.split(".attention.self.")[0]
            proj = nk.split(".attention.self.")[1].split(".")[0]  # query/key/value
            which = "weight" if nk.endswith("weight") else "bias"
            qkv.setdefault(layer, {})[f"{proj}.{which}"] = v
            continue
        nk = nk.replace(".attention.output.dense.", ".attention.out.")
        nk = nk.replace(".attention.output.LayerNorm.weight", ".attn_ln_weight")
        nk = nk.replace(".attention.output.LayerNorm.bias", ".attn_ln_bias")
        nk = nk.replace(".intermediate.dense.", ".intermediate.")
        nk = nk.replace(".output.dense.", ".output.")
        nk = nk.replace(".output.LayerNorm.weight", ".out_ln_weight")
        nk = nk.replace(".output.LayerNorm.bias", ".out_ln_bias")
        nk = nk.replace("bert.pooler.dense.", "bert.pooler.")
        nk = nk.replace("cls.predictions.transform.dense.", "transform.")
        nk = nk.replace("cls.predictions.transform.LayerNorm.weight", "transform_ln_weight")
        nk = nk.replace("cls.predictions.transform.LayerNorm.bias", "transform_ln_bias")
        nk = nk.replace("cls.predictions.decoder.", "decoder.")
        if nk == "cls.predictions.bias":
            nk = "decoder.bias"
        out[nk] = v
    for layer, parts in qkv.items():
        out[f"{layer}.attention.qkv.weight"] = torch.cat(
            [parts["query.weight"], parts["key.weight"], parts["value.weight"]], dim=0
        )
        out[f"{layer}.attention.qkv.bias"] = torch.cat(
            [parts["query.bias"], parts["key.bias"], parts["value.bias"]], dim=0
        )
    return out
