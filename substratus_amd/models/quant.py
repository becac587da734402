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

from .gguf import (QK, dequantize_q4_0, dequantize_q8_0, quantize_q4_0,
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
        qfn = quantize_q4_0 if quant == "q4_0" else quantize_q8_0
        packed = qfn(w)
        scales, qs = repack(packed, w.shape[1], quant)
        dev = lin.weight.device
        bias = None
        if getattr(lin, "bias", None) is not None:
            bias = lin.bias.detach().clone()
        return cls(scales.to(dev), qs.to(dev), w.shape[0], w.shape[1],
                   quant, bias)

    def _dequant_cpu(self) -> torch.Tensor:
        n, k = self.out_features, self.in_features
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


def load_gguf_into_model(model: nn.Module, path: str, device,
                         dtype=torch.bfloat16) -> str:
    """Load a GGUF checkpoint: quantized 2-D tensors become QuantLinear
    buffers in place; everything else is dequantized/copied. Returns the
    quant name."""
    from .gguf import GGML_Q4_0, GGML_Q8_0, read_gguf
    meta, tensors = read_gguf(path)
    quant = meta.get("substratus.quant", "q4_0")
    sd = {}
    qmap = {}
    for name, gt in tensors.items():
        if gt.ggml_type in (GGML_Q4_0, GGML_Q8_0):
            qname = "q4_0" if gt.ggml_type == GGML_Q4_0 else "q8_0"
            qmap[name] = (gt, qname)
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
            scales, qs = repack(gt.data, gt.shape[-1], qname)
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
