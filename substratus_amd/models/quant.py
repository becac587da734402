"""Quantized linear layers for GGUF serving (the llama.cpp
serving path rebuilt first-party).

QuantLinear keeps weights resident as repacked GGUF blocks — fp16 scales
[N, K/32] plus packed q4_0 nibbles [N, K/2] or q8_0 int8 [N, K] — and
dispatches:
  * GPU, M <= 8 (single-stream decode): fused block-dequant GEMV HIP kernel
    (~4.25 bits/weight of HBM traffic instead of 16 — the decode-latency win
    quantization exists for).
  * GPU, M > 8 (prefill): dequant kernel to a transient bf16 buffer +
    hipBLASLt.
  * CPU: reference dequant + F.linear (the numerics oracle).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .gguf import (QK, QK_K, Q4_K_BLOCK_BYTES, Q6_K_BLOCK_BYTES,
                   QUANT_CODECS, _pack_scale_min_k4, _unpack_scale_min_k4,
                   dequantize_q4_0, dequantize_q8_0, quantize_q4_0,
                   quantize_q8_0)


def repack(packed: torch.Tensor, k: int, quant: str) -> tuple[torch.Tensor, torch.Tensor]:
    """GGUF block stream [N, K/32*BB] -> (scales fp16 [N, K/32],
    qs uint8 [N, K/2 or K]) for aligned GPU access."""
    n = packed.shape[0]
    if quant == "q4_0":
        bb, payload = 18, 16
    elif quant == "q8_0":
        bb, payload = 34, 32
    else:
        raise ValueError(quant)
    blocks = packed.reshape(n, k // QK, bb)
    scales = blocks[:, :, :2].contiguous().reshape(n, -1).view(torch.float16)
    qs = blocks[:, :, 2:].contiguous().reshape(n, k // QK * payload)
    return scales.contiguous(), qs.contiguous()


def repack_k(packed: torch.Tensor, k: int, quant: str) -> tuple[torch.Tensor, torch.Tensor]:
    """K-quant GGUF block stream -> aligned SoA for the GPU kernels
    (quant_k.hip header documents the layout; llama.cpp's 144/210-byte
    blocks are not naturally aligned for vector loads)."""
    n = packed.shape[0]
    sb = k // QK_K
    if quant == "q4_K":
        blocks = packed.reshape(n, sb, Q4_K_BLOCK_BYTES)
        scales = blocks[:, :, 0:4].contiguous().reshape(n, -1)\
            .view(torch.float16)                       # [N, 2*SB] d,dmin
        sc, mn = _unpack_scale_min_k4(
            blocks[:, :, 4:16].reshape(-1, 12))        # [N*SB, 8]
        qs = torch.cat([sc.to(torch.uint8).reshape(n, sb * 8),
                        mn.to(torch.uint8).reshape(n, sb * 8),
                        blocks[:, :, 16:144].reshape(n, sb * 128)], dim=1)
        return scales.contiguous(), qs.contiguous()
    if quant == "q6_K":
        blocks = packed.reshape(n, sb, Q6_K_BLOCK_BYTES)
        scales = blocks[:, :, 208:210].contiguous().reshape(n, -1)\
            .view(torch.float16)                       # [N, SB] d
        qs = torch.cat([blocks[:, :, 0:128].reshape(n, sb * 128),
                        blocks[:, :, 128:192].reshape(n, sb * 64),
                        blocks[:, :, 192:208].reshape(n, sb * 16)], dim=1)
        return scales.contiguous(), qs.contiguous()
    raise ValueError(quant)


def unrepack_k(scales: torch.Tensor, qs: torch.Tensor, k: int,
               quant: str) -> torch.Tensor:
    """Inverse of repack_k: rebuild the GGUF block stream (CPU dequant)."""
    n = scales.shape[0]
    sb = k // QK_K
    scales = scales.cpu()
    qs = qs.cpu()
    if quant == "q4_K":
        out = torch.zeros(n, sb, Q4_K_BLOCK_BYTES, dtype=torch.uint8)
        out[:, :, 0:4] = scales.view(torch.uint8).reshape(n, sb, 4)
        sc = qs[:, 0:sb * 8].reshape(n * sb, 8)
        mn = qs[:, sb * 8:sb * 16].reshape(n * sb, 8)
        out[:, :, 4:16] = _pack_scale_min_k4(sc, mn).reshape(n, sb, 12)
        out[:, :, 16:144] = qs[:, sb * 16:].reshape(n, sb, 128)
        return out.reshape(n, sb * Q4_K_BLOCK_BYTES)
    out = torch.zeros(n, sb, Q6_K_BLOCK_BYTES, dtype=torch.uint8)
    out[:, :, 0:128] = qs[:, 0:sb * 128].reshape(n, sb, 128)
    out[:, :, 128:192] = qs[:, sb * 128:sb * 192].reshape(n, sb, 64)
    out[:, :, 192:208] = qs[:, sb * 192:sb * 208].reshape(n, sb, 16)
    out[:, :, 208:210] = scales.view(torch.uint8).reshape(n, sb, 2)
    return out.reshape(n, sb * Q6_K_BLOCK_BYTES)


class QuantLinear(nn.Module):
    def __init__(self, scales: torch.Tensor, qs: torch.Tensor,
                 out_features: int, in_features: int, quant: str,
                 bias: Optional[torch.Tensor] = None):
        super().__init__()
        self.out_features = out_features
        self.in_features = in_features
        self.quant = quant
        self.register_buffer("scales", scales)
        self.register_buffer("qs", qs)
        if bias is not None:
            self.register_buffer("bias", bias)
        else:
            self.bias = None

    @classmethod
    def from_linear(cls, lin: nn.Module, quant: str) -> "QuantLinear":
        # quantize on the weight's own device (GPU for big models: a 13b
        # CPU quantize would take minutes)
        w = lin.weight.detach().float()
        qfn = QUANT_CODECS[quant][2]
        packed = qfn(w)
        rp = repack_k if quant.endswith("_K") else repack
        scales, qs = rp(packed, w.shape[1], quant)
        dev = lin.weight.device
        bias = None
        if getattr(lin, "bias", None) is not None:
            bias = lin.bias.detach().clone()
        return cls(scales.to(dev), qs.to(dev), w.shape[0], w.shape[1],
                   quant, bias)

    def _dequant_cpu(self) -> torch.Tensor:
        n, k = self.out_features, self.in_features
        if self.quant.endswith("_K"):
            blocks = unrepack_k(self.scales, self.qs, k, self.quant)
            return QUANT_CODECS[self.quant][3](blocks, k)
        if self.quant == "q4_0":
            blocks = torch.cat([
                self.scales.view(torch.uint8).reshape(n, k // QK, 2).cpu(),
                self.qs.reshape(n, k // QK, 16).cpu()], dim=2)
            return dequantize_q4_0(blocks.reshape(n, -1), k)
        blocks = torch.cat([
            self.scales.view(torch.uint8).reshape(n, k // QK, 2).cpu(),
            self.qs.reshape(n, k // QK, 32).cpu()], dim=2)
        return dequantize_q8_0(blocks.reshape(n, -1), k)

    def dequant_weight(self) -> torch.Tensor:
        """Full-precision weight (bf16 on GPU via the dequant kernel)."""
        if self.scales.is_cuda:
            from substratus_amd import _C
            out = torch.empty(self.out_features, self.in_features,
                              dtype=torch.bfloat16, device=self.scales.device)
            if self.quant.endswith("_K"):
                _C.dequant_k(out, self.scales, self.qs,
                             self.quant == "q6_K")
            else:
                _C.dequant(out, self.scales, self.qs, self.quant == "q8_0")
            return out
        return self._dequant_cpu()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape
        x2 = x.reshape(-1, self.in_features)
        m = x2.shape[0]
        if x.is_cuda and m <= 8:
            from substratus_amd import _C
            y = torch.empty(m, self.out_features, dtype=torch.bfloat16,
                            device=x.device)
            if self.quant.endswith("_K"):
                _C.quant_gemv_k(y, x2.contiguous().bfloat16(), self.scales,
                                self.qs, self.quant == "q6_K")
            else:
                _C.quant_gemv(y, x2.contiguous().bfloat16(), self.scales,
                              self.qs, self.quant == "q8_0")
            y = y.to(x.dtype)
        else:
            w = self.dequant_weight().to(x.dtype)
            y = F.linear(x2, w)
        if self.bias is not None:
            y = y + self.bias
        return y.reshape(*shape[:-1], self.out_features)


def quantize_model(model: nn.Module, quant: str) -> nn.Module:
    """Swap every projection Linear (wqkv/wo/w_gateup/w_up/w_down, lm_head)
    for a QuantLinear; embeddings and norms stay bf16."""
    from ..parallel.tp import Linear
    for name, mod in model.named_children():
        if isinstance(mod, Linear) and mod.in_features % QK == 0:
            setattr(model, name, QuantLinear.from_linear(mod, quant))
        else:
            quantize_model(mod, quant)
    return model


def _llamacpp_to_native(tensors: dict, cfg) -> dict:
    """Map llama.cpp GGUF tensor names (TheBloke-style files — the
    reference's examples/llama2-13b-chat-gguf served exactly these through
    its llama.cpp image) onto the native CausalLM state-dict names.

    Fusions are row-granular so they work on PACKED q4_0/q8_0 data:
      blk.i.attn_q|attn_k|attn_v -> blocks.i.wqkv   (row concat)
      blk.i.ffn_gate|ffn_up      -> blocks.i.w_gateup (row concat)
    llama.cpp stores wq/wk with its rope row-permutation applied
    (convert.py permute: within each head, row j comes from row
    (j%2)*(d/2)+(j//2)); our rope is HF/neox layout, so the inverse
    permutation — also pure row reordering — is applied here."""
    from .gguf import GGUFTensor

    def rows_perm_inverse(t: GGUFTensor, n_head: int) -> GGUFTensor:
        n = t.shape[0]
        dh = n // n_head
        half = dh // 2
        idx = []
        for h in range(n_head):
            for j in range(dh):
                idx.append(h * dh + (j % half) * 2 + (j // half))
        data = t.data[torch.tensor(idx)]
        return GGUFTensor(t.name, t.shape, t.ggml_type, data)

    def cat_rows(parts: list[GGUFTensor], name: str) -> GGUFTensor:
        gt0 = parts[0]
        shape = [sum(p.shape[0] for p in parts)] + list(gt0.shape[1:])
        if any(p.ggml_type != gt0.ggml_type for p in parts):
            # Q4_K_M files mix q4_K and q6_K WITHIN a fusion group (e.g.
            # attn_v is q6_K while attn_q/attn_k are q4_K). The fused
            # QuantLinear needs one codec, so dequantize the parts and
            # re-encode the fused tensor at the widest member format —
            # a one-time re-quantization of already-quantized data
            # (error bounded by the codec step, not the weights).
            from .gguf import GGML_Q6_K, GGML_Q8_0, QUANT_CODECS
            types = {p.ggml_type for p in parts}
            target = GGML_Q8_0 if GGML_Q8_0 in types else GGML_Q6_K
            tname = {v[0]: k for k, v in QUANT_CODECS.items()}[target]
            full = torch.cat([p.dequantize() for p in parts], dim=0)
            return GGUFTensor(name, shape, target,
                              QUANT_CODECS[tname][2](full))
        return GGUFTensor(name, shape, gt0.ggml_type,
                          torch.cat([p.data for p in parts], dim=0))

    out = {}
    hq, hkv = cfg.num_heads, cfg.num_kv_heads
    flat = {"token_embd.weight": "embed_tokens.weight",
            "output_norm.weight": "final_norm.weight",
            "output.weight": "lm_head.weight",
            "output_norm.bias": "final_norm.bias"}
    for name, gt in tensors.items():
        if name in flat:
            out[flat[name]] = gt
            continue
        if not name.startswith("blk."):
            continue
        _, i, rest = name.split(".", 2)
        pre = f"blocks.{i}."
        if rest == "attn_norm.weight":
            out[pre + "norm1.weight"] = gt
        elif rest == "attn_norm.bias":
            out[pre + "norm1.bias"] = gt
        elif rest == "ffn_norm.weight":
            out[pre + "norm2.weight"] = gt
        elif rest == "ffn_norm.bias":
            out[pre + "norm2.bias"] = gt
        elif rest == "attn_output.weight":
            out[pre + "wo.weight"] = gt
        elif rest == "ffn_down.weight":
            out[pre + "w_down.weight"] = gt
        # fused parts collected below
    for i in range(cfg.num_layers):
        qn, kn, vn = (f"blk.{i}.attn_q.weight", f"blk.{i}.attn_k.weight",
                      f"blk.{i}.attn_v.weight")
        if qn in tensors:
            q = rows_perm_inverse(tensors[qn], hq)
            k = rows_perm_inverse(tensors[kn], hkv)
            out[f"blocks.{i}.wqkv.weight"] = cat_rows(
                [q, k, tensors[vn]], f"blocks.{i}.wqkv.weight")
        gn, un = f"blk.{i}.ffn_gate.weight", f"blk.{i}.ffn_up.weight"
        if gn in tensors:
            out[f"blocks.{i}.w_gateup.weight"] = cat_rows(
                [tensors[gn], tensors[un]], f"blocks.{i}.w_gateup.weight")
    return out


def load_gguf_into_model(model: nn.Module, path: str, device,
                         dtype=torch.bfloat16) -> str:
    """Load a GGUF checkpoint: quantized 2-D tensors become QuantLinear
    buffers in place; everything else is dequantized/copied. Returns the
    quant name."""
    from .gguf import (GGML_Q4_0, GGML_Q4_K, GGML_Q6_K, GGML_Q8_0,
                       read_gguf)
    meta, tensors = read_gguf(path)
    gt_names = {GGML_Q4_0: "q4_0", GGML_Q8_0: "q8_0",
                GGML_Q4_K: "q4_K", GGML_Q6_K: "q6_K"}
    quant = meta.get("substratus.quant", "q4_0")
    if any(n.startswith("blk.") or n == "token_embd.weight"
           for n in tensors):
        # llama.cpp-named file (ecosystem GGUFs)
        tensors = _llamacpp_to_native(tensors, model.cfg)
        for gt in tensors.values():
            if gt.ggml_type in gt_names:
                quant = gt_names[gt.ggml_type]
                break
    sd = {}
    qmap = {}
    for name, gt in tensors.items():
        if gt.ggml_type in gt_names:
            qmap[name] = (gt, gt_names[gt.ggml_type])
        else:
            sd[name] = gt.dequantize()
    # first load full-precision leftovers
    model.load_state_dict({k: v.to(device, dtype) for k, v in sd.items()},
                          strict=False)
    # then swap matching Linears for QuantLinears with the stored blocks
    mods = dict(model.named_modules())
    for name, (gt, qname) in qmap.items():
        mod_name = name.rsplit(".", 1)[0]
        mod = mods.get(mod_name)
        from ..parallel.tp import Linear
        if isinstance(mod, Linear):
            rp = repack_k if qname.endswith("_K") else repack
            scales, qs = rp(gt.data, gt.shape[-1], qname)
            parent_name, attr = (mod_name.rsplit(".", 1) + [""])[:2] \
                if "." in mod_name else ("", mod_name)
            parent = mods[parent_name] if parent_name else model
            bias = mod.bias.detach().clone() if mod.bias is not None else None
            setattr(parent, attr, QuantLinear(
                scales.to(device), qs.to(device), gt.shape[0], gt.shape[1],
                qname, bias))
        else:
            # non-Linear target (e.g. embedding): dequantize fully
            model.load_state_dict({name: gt.dequantize().to(device, dtype)},
                                  strict=False)
    return quant
