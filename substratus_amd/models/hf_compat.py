"""HF-format checkpoint interop for the llama, falcon and opt families.

The reference's loader image was `substratusai/model-loader-huggingface` —
users bring checkpoints in HF layout. The native loader exports OUR
state-dict names; this module maps an HF-named safetensors state dict
(model.layers.N.self_attn.q_proj... — meta-llama/Llama-2-* layout) onto the
native CausalLM names so such checkpoints serve directly. HF's q/k layout
is the neox/half-split one our rope kernel uses, so the mapping is pure
renaming + row-concat fusion (qkv, gate+up) — no permutations.

Falcon HF layouts map too (hf_falcon_to_native): 7b-style MQA fused qkv
directly, 40b-style per-kv-group interleaved qkv via a row degroup. OPT
(hf_opt_to_native) maps the decoder.* projections with their biases and
the +2-offset embed_positions. All served model families accept their HF
checkpoints.
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


def is_hf_falcon_state_dict(sd: dict) -> bool:
    return any(k.startswith("transformer.h.") and
               "self_attention.query_key_value" in k for k in sd)


def hf_falcon_to_native(sd: dict, cfg: ModelConfig) -> dict:
    """tiiuae/falcon-* HF layout -> native names.

    falcon-7b (multi_query): query_key_value rows are [q0..q(hq-1), k, v] —
    exactly the native wqkv order, pure rename. falcon-40b
    (new_decoder_architecture, GQA): rows are PER-KV-GROUP interleaved
    [q_g0..q_g(gq-1), k_g, v_g | g=0..hkv-1]; the native layout wants all q
    (same g-major head order), then all k, then all v — a row reorder.
    ln_attn/ln_mlp map to norm1/norm2 (two parallel norms)."""
    d = cfg.head_dim
    hq, hkv = cfg.num_heads, cfg.num_kv_heads
    gq = hq // hkv

    def degroup_qkv(w: torch.Tensor) -> torch.Tensor:
        if hkv == 1:
            return w  # multi_query layout == native layout
        per = (gq + 2) * d
        qs, ks, vs = [], [], []
        for g in range(hkv):
            blockk = w[g * per:(g + 1) * per]
            qs.append(blockk[:gq * d])
            ks.append(blockk[gq * d:(gq + 1) * d])
            vs.append(blockk[(gq + 1) * d:])
        return torch.cat(qs + ks + vs, dim=0)

    out = {}
    flat = {"transformer.word_embeddings.weight": "embed_tokens.weight",
            "transformer.ln_f.weight": "final_norm.weight",
            "transformer.ln_f.bias": "final_norm.bias",
            "lm_head.weight": "lm_head.weight"}
    for src, dst in flat.items():
        if src in sd:
            out[dst] = sd[src]
    for i in range(cfg.num_layers):
        h = f"transformer.h.{i}."
        p = f"blocks.{i}."
        qkv = h + "self_attention.query_key_value.weight"
        if qkv not in sd:
            continue
        out[p + "wqkv.weight"] = degroup_qkv(sd[qkv])
        out[p + "wo.weight"] = sd[h + "self_attention.dense.weight"]
        out[p + "w_up.weight"] = sd[h + "mlp.dense_h_to_4h.weight"]
        out[p + "w_down.weight"] = sd[h + "mlp.dense_4h_to_h.weight"]
        for hf_n, our_n in (("input_layernorm", "norm1"),
                            ("ln_attn", "norm1"), ("ln_mlp", "norm2"),
                            ("post_attention_layernorm", "norm2")):
            for suf in ("weight", "bias"):
                k = f"{h}{hf_n}.{suf}"
                if k in sd:
                    out[f"{p}{our_n}.{suf}"] = sd[k]
    return out


def is_hf_opt_state_dict(sd: dict) -> bool:
    return any(k.startswith("model.decoder.layers.") and
               "self_attn.q_proj" in k for k in sd)


def hf_opt_to_native(sd: dict, cfg: ModelConfig) -> dict:
    """facebook/opt-* HF layout -> native names: renames + qkv fusion
    (weights AND biases — OPT projections carry biases). OPT's
    embed_positions has a +2 offset quirk (first two rows reserved); the
    native embed_positions is sized max_position_embeddings, so the HF
    tensor's tail rows are taken."""
    out = {}
    if "model.decoder.embed_tokens.weight" in sd:
        out["embed_tokens.weight"] = sd["model.decoder.embed_tokens.weight"]
    if "model.decoder.embed_positions.weight" in sd:
        pos = sd["model.decoder.embed_positions.weight"]
        want = cfg.max_position_embeddings
        out["embed_positions.weight"] = pos[-want:] if pos.shape[0] > want \
            else pos
    for hf_n, our_n in (("model.decoder.final_layer_norm", "final_norm"),):
        for suf in ("weight", "bias"):
            k = f"{hf_n}.{suf}"
            if k in sd:
                out[f"{our_n}.{suf}"] = sd[k]
    if "lm_head.weight" in sd:
        out["lm_head.weight"] = sd["lm_head.weight"]
    for i in range(cfg.num_layers):
        h = f"model.decoder.layers.{i}."
        p = f"blocks.{i}."
        if h + "self_attn.q_proj.weight" not in sd:
            continue
        for suf in ("weight", "bias"):
            out[p + f"wqkv.{suf}"] = torch.cat(
                [sd[h + f"self_attn.q_proj.{suf}"],
                 sd[h + f"self_attn.k_proj.{suf}"],
                 sd[h + f"self_attn.v_proj.{suf}"]], dim=0)
            out[p + f"wo.{suf}"] = sd[h + f"self_attn.out_proj.{suf}"]
            out[p + f"w_up.{suf}"] = sd[h + f"fc1.{suf}"]
            out[p + f"w_down.{suf}"] = sd[h + f"fc2.{suf}"]
            out[p + f"norm1.{suf}"] = sd[h + f"self_attn_layer_norm.{suf}"]
            out[p + f"norm2.{suf}"] = sd[h + f"final_layer_norm.{suf}"]
    return out
