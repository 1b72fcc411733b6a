"""HuggingFace-weights importers (switch path for users arriving with HF
checkpoints; `transformers` is only needed to SOURCE a state_dict --
conversion itself is pure tensor remapping).

Layout differences handled:
  * our Linear stores [in, out] (paddle convention) -- HF stores [out, in]
  * our Llama MLP packs [gate | up] into one gate_up_proj
  * prefix "model."  ->  "llama." / "gpt."
"""
from __future__ import annotations

import torch


def convert_llama_from_hf(hf_state_dict, num_layers=None):
    """HF LlamaForCausalLM state_dict -> our LlamaForCausalLM state_dict."""
    out = {}
    sd = {k: v for k, v in hf_state_dict.items()}
    out["llama.embed_tokens.weight"] = sd["model.embed_tokens.weight"]
    out["llama.norm.weight"] = sd["model.norm.weight"]
    if "lm_head.weight" in sd:
        out["lm_head.weight"] = sd["lm_head.weight"].t().contiguous()
    else:  # tied embeddings
        out["lm_head.weight"] = sd["model.embed_tokens.weight"].t().contiguous()
    n = num_layers or max(int(k.split(".")[2]) for k in sd
                          if k.startswith("model.layers.")) + 1
    for i in range(n):
        src = f"model.layers.{i}."
        dst = f"llama.layers.{i}."
        for nm in ("input_layernorm", "post_attention_layernorm"):
            out[dst + nm + ".weight"] = sd[src + nm + ".weight"]
        for nm in ("q_proj", "k_proj", "v_proj", "o_proj"):
            out[dst + "self_attn." + nm + ".weight"] = \
                sd[src + "self_attn." + nm + ".weight"].t().contiguous()
        gate = sd[src + "mlp.gate_proj.weight"].t()   # [H, I]
        up = sd[src + "mlp.up_proj.weight"].t()       # [H, I]
        out[dst + "mlp.gate_up_proj.weight"] = \
            torch.cat([gate, up], dim=1).contiguous()  # [H, 2I] = [gate|up]
        out[dst + "mlp.down_proj.weight"] = \
            sd[src + "mlp.down_proj.weight"].t().contiguous()
    return out


def load_llama_from_hf(model, hf_state_dict):
    """Copy converted HF weights into a built LlamaForCausalLM."""
    conv = convert_llama_from_hf(hf_state_dict, len(model.llama.layers))
    own = model.state_dict()
    missing = [k for k in own if k not in conv]
    if missing:
        raise KeyError(f"HF conversion missing keys: {missing[:5]}")
    with torch.no_grad():
        for k, v in own.items():
            v.copy_(conv[k].to(v.dtype))
    return model
