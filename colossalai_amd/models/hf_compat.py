"""HF-transformers interop for the native model families.

The native models pack QKV and gate|up into single GEMMs; these helpers
remap state dicts between HF layout (q_proj/k_proj/v_proj, gate_proj/
up_proj) and the packed native layout, in both directions — so HF Llama /
Mixtral checkpoints load into the MI355X-optimized modules and checkpoints
we save can be consumed by HF `from_pretrained`.
"""

from typing import Dict

import torch

__all__ = ["hf_to_native_llama", "native_to_hf_llama", "load_hf_llama"]


def hf_to_native_llama(hf_sd: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Map an HF LlamaForCausalLM state dict to the native packed layout."""
    out = {}
    packs: Dict[str, Dict[str, torch.Tensor]] = {}
    for k, v in hf_sd.items():
        if any(f".self_attn.{p_}_proj.weight" in k for p_ in "qkv"):
            base = k.rsplit(".self_attn.", 1)[0]
            packs.setdefault(base + ".qkv", {})[k.split(".")[-2]] = v
        elif any(f".self_attn.{p_}_proj.bias" in k for p_ in "qkv"):
            base = k.rsplit(".self_attn.", 1)[0]
            packs.setdefault(base + ".qkvbias", {})[k.split(".")[-2]] = v
        elif ".mlp.gate_proj.weight" in k or ".mlp.up_proj.weight" in k:
            base = k.rsplit(".mlp.", 1)[0]
            packs.setdefault(base + ".gateup", {})[k.split(".")[-2]] = v
        elif k.endswith(".self_attn.q_norm.weight"):
            out[k.replace(".self_attn.q_norm.weight", ".self_attn.q_norm_weight")] = v
        elif k.endswith(".self_attn.k_norm.weight"):
            out[k.replace(".self_attn.k_norm.weight", ".self_attn.k_norm_weight")] = v
        elif k.endswith("input_layernorm.weight"):
            out[k.replace("input_layernorm.weight", "input_layernorm_weight")] = v
        elif k.endswith("post_attention_layernorm.weight"):
            out[k.replace("post_attention_layernorm.weight", "post_attention_layernorm_weight")] = v
        elif k == "model.norm.weight":
            out["model.norm_weight"] = v
        elif "rotary_emb" in k:
            continue  # native model builds its own table
        else:
            out[k] = v
    for base, parts in packs.items():
        if base.endswith(".qkvbias"):
            prefix = base[: -len(".qkvbias")]
            out[f"{prefix}.self_attn.qkv_proj.bias"] = torch.cat(
                [parts["q_proj"], parts["k_proj"], parts["v_proj"]], dim=0
            )
        elif base.endswith(".qkv"):
            prefix = base[: -len(".qkv")]
            out[f"{prefix}.self_attn.qkv_proj.weight"] = torch.cat(
                [parts["q_proj"], parts["k_proj"], parts["v_proj"]], dim=0
            )
        else:
            prefix = base[: -len(".gateup")]
            out[f"{prefix}.mlp.gate_up_proj.weight"] = torch.cat([parts["gate_proj"], parts["up_proj"]], dim=0)
    return out


def native_to_hf_llama(native_sd: Dict[str, torch.Tensor], num_heads: int, num_kv_heads: int, head_dim: int) -> Dict[str, torch.Tensor]:
    """Inverse mapping: native packed layout -> HF LlamaForCausalLM keys."""
    out = {}
    qd = num_heads * head_dim
    kd = num_kv_heads * head_dim
    for k, v in native_sd.items():
        if k.endswith(".self_attn.qkv_proj.weight"):
            prefix = k[: -len(".qkv_proj.weight")]
            out[f"{prefix}.q_proj.weight"] = v[:qd]
            out[f"{prefix}.k_proj.weight"] = v[qd : qd + kd]
            out[f"{prefix}.v_proj.weight"] = v[qd + kd :]
        elif k.endswith(".mlp.gate_up_proj.weight"):
            prefix = k[: -len(".gate_up_proj.weight")]
            half = v.shape[0] // 2
            out[f"{prefix}.gate_proj.weight"] = v[:half]
            out[f"{prefix}.up_proj.weight"] = v[half:]
        elif k.endswith("input_layernorm_weight"):
            out[k.replace("input_layernorm_weight", "input_layernorm.weight")] = v
        elif k.endswith("post_attention_layernorm_weight"):
            out[k.replace("post_attention_layernorm_weight", "post_attention_layernorm.weight")] = v
        elif k == "model.norm_weight":
            out["model.norm.weight"] = v
        else:
            out[k] = v
    return out


def load_hf_llama(native_model, hf_state_dict: Dict[str, torch.Tensor], strict: bool = True):
    """Load an HF Llama state dict into a native LlamaForCausalLM."""
    mapped = hf_to_native_llama(hf_state_dict)
    return native_model.load_state_dict(mapped, strict=strict)
