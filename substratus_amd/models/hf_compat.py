"""HF-format checkpoint interop for the llama family.

The reference's loader image was `substratusai/model-loader-huggingface` —
users bring checkpoints in HF layout. The native loader exports OUR
state-dict names; this module maps an HF-named safetensors state dict
(model.layers.N.self_attn.q_proj... — meta-llama/Llama-2-* layout) onto the
native CausalLM names so such checkpoints serve directly. HF's q/k layout
is the neox/half-split one our rope kernel uses, so the mapping is pure
renaming + row-concat fusion (qkv, gate+up) — no permutations.

Falcon/OPT HF layouts (fused interleaved qkv, decoder.* prefixes) are not
mapped yet; the native loader covers those families.
"""

from __future__ import annotations

import torch

from .config import ModelConfig


def is_hf_llama_state_dict(sd: dict) -> bool:
    return any(k.startswith("model.layers.") and "self_attn.q_proj" in k
               for k in sd)


def hf_llama_to_native(sd: dict, cfg: ModelConfig) -> dict:
    out = {}
    flat = {"model.embed_tokens.weight": "embed_tokens.weight",
            "model.norm.weight": "final_norm.weight",
            "lm_head.weight": "lm_head.weight"}
    for src, dst in flat.items():
        if src in sd:
            out[dst] = sd[src]
    for i in range(cfg.num_layers):
        h = f"model.layers.{i}."
        p = f"blocks.{i}."
        if h + "input_layernorm.weight" not in sd:
            continue
        out[p + "norm1.weight"] = sd[h + "input_layernorm.weight"]
        out[p + "norm2.weight"] = sd[h + "post_attention_layernorm.weight"]
        out[p + "wqkv.weight"] = torch.cat(
            [sd[h + "self_attn.q_proj.weight"],
             sd[h + "self_attn.k_proj.weight"],
             sd[h + "self_attn.v_proj.weight"]], dim=0)
        out[p + "wo.weight"] = sd[h + "self_attn.o_proj.weight"]
        out[p + "w_gateup.weight"] = torch.cat(
            [sd[h + "mlp.gate_proj.weight"],
             sd[h + "mlp.up_proj.weight"]], dim=0)
        out[p + "w_down.weight"] = sd[h + "mlp.down_proj.weight"]
    return out
