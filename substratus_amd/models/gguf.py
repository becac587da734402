"""GGUF quantized model format: reader, writer, and quantization codecs.

The reference serves GGUF models through `substratusai/model-server-llama-cpp`
(examples/llama2-13b-chat-gguf/server-gpu.yaml — SURVEY.md §2.5); this module
is the first-party replacement's format layer. Supported tensor codecs:
F32, F16, Q8_0 (fp16 scale + 32×int8 per block) and Q4_0 (fp16 scale +
32×4-bit per block) — the llama.cpp baseline quantizations.

The GPU compute path does NOT dequantize weights to bf16 at load: QuantLinear
(substratus_amd/models/quant.py) keeps the packed blocks resident and decode
runs a fused block-dequant GEMV HIP kernel (SURVEY.md §2.5 ledger item 8).
The offline loader can also EXPORT random-init models to GGUF so the whole
path is testable with no network.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import BinaryIO, Optional

import torch

GGUF_MAGIC = 0x46554747  # "GGUF"
GGUF_VERSION = 3

# ggml tensor types (subset)
GGML_F32 = 0
GGML_F16 = 1
GGML_Q4_0 = 2
GGML_Q8_0 = 8
GGML_Q4_K = 12
GGML_Q6_K = 14

QK = 32  # block size (weights per quant block)
Q4_0_BLOCK_BYTES = 2 + QK // 2   # fp16 scale + 16 nibbles
Q8_0_BLOCK_BYTES = 2 + QK        # fp16 scale + 32 int8
QK_K = 256  # K-quant super-block size
Q4_K_BLOCK_BYTES = 2 + 2 + 12 + QK_K // 2       # d, dmin, 6-bit scales, qs
Q6_K_BLOCK_BYTES = QK_K // 2 + QK_K // 4 + QK_K // 16 + 2  # ql, qh, sc, d

# gguf metadata value types
_T_U8, _T_I8, _T_U16, _T_I16, _T_U32, _T_I32 = 0, 1, 2, 3, 4, 5
_T_F32, _T_BOOL, _T_STR, _T_ARR, _T_U64, _T_I64, _T_F64 = 6, 7, 8, 9, 10, 11, 12


# ---------------------------------------------------------------- quantize
def quantize_q8_0(w: torch.Tensor) -> torch.Tensor:
    """w: [..., K] float -> packed uint8 [..., K/32 * 34]."""
    shape = w.shape
    assert shape[-1] % QK == 0
    blocks = w.float().reshape(-1, QK)
    amax = blocks.abs().amax(dim=1, keepdim=True)
    scale = (amax / 127.0).clamp(min=1e-12)
    qs = torch.round(blocks / scale).clamp(-127, 127).to(torch.int8)
    out = torch.empty(blocks.shape[0], Q8_0_BLOCK_BYTES, dtype=torch.uint8,
                      device=w.device)
    out[:, :2] = scale.to(torch.float16).view(torch.uint8).reshape(-1, 2)
    out[:, 2:] = qs.view(torch.uint8)
    return out.reshape(*shape[:-1], (shape[-1] // QK) * Q8_0_BLOCK_BYTES)


def dequantize_q8_0(packed: torch.Tensor, k: int) -> torch.Tensor:
    rows = packed.reshape(-1, Q8_0_BLOCK_BYTES)
    scale = rows[:, :2].contiguous().view(torch.float16).float()
    qs = rows[:, 2:].contiguous().view(torch.int8).float()
    vals = qs * scale
    return vals.reshape(*packed.shape[:-1], k)


def quantize_q4_0(w: torch.Tensor) -> torch.Tensor:
    """llama.cpp Q4_0: per 32-block, scale = max|w|/(-8); q = round(w/scale)+8
    in [0,15]; nibble i packs weights i and i+16 (lo=i, hi=i+16)."""
    shape = w.shape
    assert shape[-1] % QK == 0
    blocks = w.float().reshape(-1, QK)
    absmax_idx = blocks.abs().argmax(dim=1)
    maxval = blocks.gather(1, absmax_idx[:, None]).squeeze(1)
    scale = (maxval / -8.0)
    scale = torch.where(scale.abs() < 1e-12,
                        torch.full_like(scale, 1e-12), scale)
    q = torch.round(blocks / scale[:, None]) + 8
    q = q.clamp(0, 15).to(torch.uint8)
    lo = q[:, :16]
    hi = q[:, 16:]
    packed = (lo | (hi << 4))
    out = torch.empty(blocks.shape[0], Q4_0_BLOCK_BYTES, dtype=torch.uint8,
                      device=w.device)
    out[:, :2] = scale.to(torch.float16).view(torch.uint8).reshape(-1, 2)
    out[:, 2:] = packed
    return out.reshape(*shape[:-1], (shape[-1] // QK) * Q4_0_BLOCK_BYTES)


def dequantize_q4_0(packed: torch.Tensor, k: int) -> torch.Tensor:
    rows = packed.reshape(-1, Q4_0_BLOCK_BYTES)
    scale = rows[:, :2].contiguous().view(torch.float16).float()
    nib = rows[:, 2:]
    lo = (nib & 0xF).float() - 8.0
    hi = (nib >> 4).float() - 8.0
    vals = torch.cat([lo, hi], dim=1) * scale
    return vals.reshape(*packed.shape[:-1], k)


# ---------------------------------------------------------------- K-quants
# llama.cpp's Q4_K / Q6_K super-block codecs (ggml-quants.c semantics) —
# the formats real ecosystem *.Q4_K_M.gguf files use: 256-weight blocks,
# 8 (q4_K) / 16 (q6_K) sub-scales.

def _pack_scale_min_k4(sc: torch.Tensor, mn: torch.Tensor) -> torch.Tensor:
    """sc/mn: [B, 8] uint8 (6-bit) -> scales[12] per llama.cpp layout:
    j<4:  q[j]   = sc&63 | (upper 2 bits of sc[j+4] later)
    The exact inverse of _unpack_scale_min_k4."""
    B = sc.shape[0]
    out = torch.zeros(B, 12, dtype=torch.uint8, device=sc.device)
    out[:, 0:4] = (sc[:, 0:4] & 63) | (((sc[:, 4:8] >> 4) & 0x3) << 6)
    out[:, 4:8] = (mn[:, 0:4] & 63) | (((mn[:, 4:8] >> 4) & 0x3) << 6)
    out[:, 8:12] = (sc[:, 4:8] & 0xF) | ((mn[:, 4:8] & 0xF) << 4)
    return out


def _unpack_scale_min_k4(scales: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """scales: [B, 12] uint8 -> (sc, mn) [B, 8] uint8 — llama.cpp
    get_scale_min_k4 for all 8 sub-blocks."""
    q = scales.to(torch.int32)
    sc = torch.empty(q.shape[0], 8, dtype=torch.int32, device=scales.device)
    mn = torch.empty_like(sc)
    sc[:, 0:4] = q[:, 0:4] & 63
    mn[:, 0:4] = q[:, 4:8] & 63
    sc[:, 4:8] = (q[:, 8:12] & 0xF) | ((q[:, 0:4] >> 6) << 4)
    mn[:, 4:8] = (q[:, 8:12] >> 4) | ((q[:, 4:8] >> 6) << 4)
    return sc, mn


def quantize_q4_K(w: torch.Tensor) -> torch.Tensor:
    """w: [..., K] float -> packed uint8 [..., K/256 * 144]. Encoder picks
    per-sub-block (32) scale/min quantized to 6 bits against block-level
    fp16 d/dmin — simpler than llama.cpp's iterative fit, same format."""
    shape = w.shape
    assert shape[-1] % QK_K == 0
    blocks = w.float().reshape(-1, 8, 32)   # [B, sub, 32]
    B = blocks.shape[0]
    wmin = (-blocks.amin(dim=2)).clamp(min=0.0)      # stored min is -min
    wmax = blocks.amax(dim=2)
    scale = ((wmax + wmin) / 15.0).clamp(min=1e-10)  # q in [0,15]
    d = (scale.amax(dim=1, keepdim=True) / 63.0).clamp(min=1e-10)
    dmin = (wmin.amax(dim=1, keepdim=True) / 63.0).clamp(min=1e-10)
    sc = torch.round(scale / d).clamp(0, 63).to(torch.uint8)
    mn = torch.round(wmin / dmin).clamp(0, 63).to(torch.uint8)
    eff_d = d * sc.float()                   # [B, 8]
    eff_m = dmin * mn.float()
    q = torch.round((blocks + eff_m[:, :, None]) / eff_d[:, :, None])
    q = q.clamp(0, 15).to(torch.uint8)       # [B, 8, 32]
    out = torch.zeros(B, Q4_K_BLOCK_BYTES, dtype=torch.uint8,
                      device=w.device)
    out[:, 0:2] = d.to(torch.float16).view(torch.uint8).reshape(B, 2)
    out[:, 2:4] = dmin.to(torch.float16).view(torch.uint8).reshape(B, 2)
    out[:, 4:16] = _pack_scale_min_k4(sc, mn)
    # qs layout: per 64-chunk c (2 sub-blocks), byte l of qs[32c:32c+32]
    # packs elem 64c+l (lo nibble) and 64c+32+l (hi nibble)
    q64 = q.reshape(B, 4, 64)
    out[:, 16:144] = (q64[:, :, 0:32] | (q64[:, :, 32:64] << 4))        .reshape(B, 128)
    return out.reshape(*shape[:-1], (shape[-1] // QK_K) * Q4_K_BLOCK_BYTES)


def dequantize_q4_K(packed: torch.Tensor, k: int) -> torch.Tensor:
    rows = packed.reshape(-1, Q4_K_BLOCK_BYTES)
    B = rows.shape[0]
    d = rows[:, 0:2].contiguous().view(torch.float16).float()       # [B,1]
    dmin = rows[:, 2:4].contiguous().view(torch.float16).float()
    sc, mn = _unpack_scale_min_k4(rows[:, 4:16])
    qs = rows[:, 16:144].reshape(B, 4, 32).to(torch.int32)
    lo = qs & 0xF
    hi = qs >> 4
    q = torch.cat([lo, hi], dim=2).reshape(B, 8, 32).float()  # [B, sub, 32]
    vals = d[:, :, None] * sc.float()[:, :, None] * q         - dmin[:, :, None] * mn.float()[:, :, None]
    return vals.reshape(*packed.shape[:-1], k)


def quantize_q6_K(w: torch.Tensor) -> torch.Tensor:
    """w: [..., K] float -> packed uint8 [..., K/256 * 210]. q in [-32,31],
    per-16 int8 scales against a block fp16 d."""
    shape = w.shape
    assert shape[-1] % QK_K == 0
    blocks = w.float().reshape(-1, 16, 16)   # [B, group, 16]
    B = blocks.shape[0]
    amax = blocks.abs().amax(dim=2)
    gscale = (amax / 31.0).clamp(min=1e-10)
    d = (gscale.amax(dim=1, keepdim=True) / 127.0).clamp(min=1e-10)
    sc = torch.round(gscale / d).clamp(-128, 127).to(torch.int8)
    eff = d * sc.float().clamp(min=1.0)      # sc >= 0 by construction
    q = torch.round(blocks / eff[:, :, None]).clamp(-32, 31).to(torch.int32)
    q = (q + 32).reshape(B, 2, 128)          # [B, half, 128] in [0, 63]
    out = torch.zeros(B, Q6_K_BLOCK_BYTES, dtype=torch.uint8,
                      device=w.device)
    # per 128-half: ql = low 4 bits of (l, l+64) pairs? llama.cpp layout:
    # element groups within a half: e[l], e[l+32], e[l+64], e[l+96] for
    # l in [0,32) pack as ql[l]=lo4(e0)|lo4(e2)<<4, ql[l+32]=lo4(e1)|lo4(e3)<<4,
    # qh[l] = hi2(e0) | hi2(e1)<<2 | hi2(e2)<<4 | hi2(e3)<<6
    e = q.reshape(B, 2, 4, 32)               # [B, half, quarter, 32]
    ql = torch.zeros(B, 2, 64, dtype=torch.int32, device=w.device)
    ql[:, :, 0:32] = (e[:, :, 0] & 0xF) | ((e[:, :, 2] & 0xF) << 4)
    ql[:, :, 32:64] = (e[:, :, 1] & 0xF) | ((e[:, :, 3] & 0xF) << 4)
    qh = (e[:, :, 0] >> 4) | ((e[:, :, 1] >> 4) << 2) |          ((e[:, :, 2] >> 4) << 4) | ((e[:, :, 3] >> 4) << 6)
    out[:, 0:128] = ql.reshape(B, 128).to(torch.uint8)
    out[:, 128:192] = qh.reshape(B, 64).to(torch.uint8)
    out[:, 192:208] = sc.view(torch.uint8).reshape(B, 16)
    out[:, 208:210] = d.to(torch.float16).view(torch.uint8).reshape(B, 2)
    return out.reshape(*shape[:-1], (shape[-1] // QK_K) * Q6_K_BLOCK_BYTES)


def dequantize_q6_K(packed: torch.Tensor, k: int) -> torch.Tensor:
    rows = packed.reshape(-1, Q6_K_BLOCK_BYTES)
    B = rows.shape[0]
    ql = rows[:, 0:128].reshape(B, 2, 64).to(torch.int32)
    qh = rows[:, 128:192].reshape(B, 2, 32).to(torch.int32)
    sc = rows[:, 192:208].contiguous().view(torch.int8).to(torch.int32)
    d = rows[:, 208:210].contiguous().view(torch.float16).float()   # [B,1]
    e = torch.empty(B, 2, 4, 32, dtype=torch.int32)
    e[:, :, 0] = (ql[:, :, 0:32] & 0xF) | ((qh & 0x3) << 4)
    e[:, :, 1] = (ql[:, :, 32:64] & 0xF) | (((qh >> 2) & 0x3) << 4)
    e[:, :, 2] = (ql[:, :, 0:32] >> 4) | (((qh >> 4) & 0x3) << 4)
    e[:, :, 3] = (ql[:, :, 32:64] >> 4) | (((qh >> 6) & 0x3) << 4)
    q = e.reshape(B, 16, 16).float() - 32.0
    vals = d[:, :, None] * sc.reshape(B, 16, 1).float() * q
    return vals.reshape(*packed.shape[:-1], k)


QUANT_CODECS = {
    "q4_0": (GGML_Q4_0, Q4_0_BLOCK_BYTES, quantize_q4_0, dequantize_q4_0),
    "q8_0": (GGML_Q8_0, Q8_0_BLOCK_BYTES, quantize_q8_0, dequantize_q8_0),
    "q4_K": (GGML_Q4_K, Q4_K_BLOCK_BYTES, quantize_q4_K, dequantize_q4_K),
    "q6_K": (GGML_Q6_K, Q6_K_BLOCK_BYTES, quantize_q6_K, dequantize_q6_K),
}


# ------------------------------------------------------------------ writer
def _w_str(f: BinaryIO, s: str) -> None:
    b = s.encode()
    f.write(struct.pack("<Q", len(b)))
    f.write(b)


def _w_kv(f: BinaryIO, key: str, vtype: int, packfmt: str, value) -> None:
    _w_str(f, key)
    f.write(struct.pack("<I", vtype))
    if vtype == _T_STR:
        _w_str(f, value)
    else:
        f.write(struct.pack(packfmt, value))


def write_gguf(path: str, tensors: dict[str, tuple[torch.Tensor, int, list[int]]],
               metadata: Optional[dict] = None, alignment: int = 32) -> None:
    """tensors: name -> (raw bytes tensor uint8 or float tensor, ggml_type,
    logical_shape). metadata: str -> (str|int|float). alignment: data-section
    alignment; recorded as general.alignment when non-default."""
    metadata = dict(metadata or {})
    if alignment != 32:
        metadata["general.alignment"] = alignment
    with open(path, "wb") as f:
        f.write(struct.pack("<IIQQ", GGUF_MAGIC, GGUF_VERSION,
                            len(tensors), len(metadata)))
        for k, v in metadata.items():
            if isinstance(v, str):
                _w_kv(f, k, _T_STR, "", v)
            elif isinstance(v, bool):
                _w_kv(f, k, _T_BOOL, "<?", v)
            elif isinstance(v, int):
                _w_kv(f, k, _T_U64, "<Q", v)
            elif isinstance(v, float):
                _w_kv(f, k, _T_F32, "<f", v)
            else:
                raise TypeError(f"unsupported metadata type for {k}")
        # tensor infos
        offset = 0
        infos = []
        for name, (data, gtype, shape) in tensors.items():
            nbytes = data.numel() * data.element_size()
            infos.append((name, gtype, shape, offset, data))
            offset += -(-nbytes // alignment) * alignment
        for name, gtype, shape, off, _ in infos:
            _w_str(f, name)
            f.write(struct.pack("<I", len(shape)))
            # gguf dims are innermost-first
            for dim in reversed(shape):
                f.write(struct.pack("<Q", dim))
            f.write(struct.pack("<IQ", gtype, off))
        # align data section
        pos = f.tell()
        pad = (alignment - pos % alignment) % alignment
        f.write(b"\0" * pad)
        base = f.tell()
        for name, gtype, shape, off, data in infos:
            cur = f.tell() - base
            if cur < off:
                f.write(b"\0" * (off - cur))
            f.write(data.contiguous().numpy().tobytes())


# ------------------------------------------------------------------ reader
def _r_str(f: BinaryIO) -> str:
    (n,) = struct.unpack("<Q", f.read(8))
    return f.read(n).decode()


_SCALAR_FMT = {_T_U8: "<B", _T_I8: "<b", _T_U16: "<H", _T_I16: "<h",
               _T_U32: "<I", _T_I32: "<i", _T_F32: "<f", _T_BOOL: "<?",
               _T_U64: "<Q", _T_I64: "<q", _T_F64: "<d"}


def _r_value(f: BinaryIO, vtype: int):
    if vtype == _T_STR:
        return _r_str(f)
    if vtype == _T_ARR:
        (etype, n) = struct.unpack("<IQ", f.read(12))
        return [_r_value(f, etype) for _ in range(n)]
    fmt = _SCALAR_FMT[vtype]
    return struct.unpack(fmt, f.read(struct.calcsize(fmt)))[0]


@dataclass
class GGUFTensor:
    name: str
    shape: list[int]       # logical shape, outermost-first
    ggml_type: int
    data: torch.Tensor     # raw uint8 (quantized) or float tensor

    def dequantize(self) -> torch.Tensor:
        k = self.shape[-1]
        if self.ggml_type == GGML_F32:
            return self.data
        if self.ggml_type == GGML_F16:
            return self.data.float()
        if self.ggml_type == GGML_Q4_0:
            return dequantize_q4_0(self.data, k)
        if self.ggml_type == GGML_Q8_0:
            return dequantize_q8_0(self.data, k)
        if self.ggml_type == GGML_Q4_K:
            return dequantize_q4_K(self.data, k)
        if self.ggml_type == GGML_Q6_K:
            return dequantize_q6_K(self.data, k)
        raise ValueError(f"unsupported ggml type {self.ggml_type}")


def read_gguf(path: str) -> tuple[dict, dict[str, GGUFTensor]]:
    with open(path, "rb") as f:
        magic, version, n_tensors, n_kv = struct.unpack("<IIQQ", f.read(24))
        if magic != GGUF_MAGIC:
            raise ValueError(f"not a GGUF file: {path}")
        metadata = {}
        for _ in range(n_kv):
            key = _r_str(f)
            (vtype,) = struct.unpack("<I", f.read(4))
            metadata[key] = _r_value(f, vtype)
        infos = []
        for _ in range(n_tensors):
            name = _r_str(f)
            (ndim,) = struct.unpack("<I", f.read(4))
            dims = [struct.unpack("<Q", f.read(8))[0] for _ in range(ndim)]
            shape = list(reversed(dims))  # to outermost-first
            gtype, off = struct.unpack("<IQ", f.read(12))
            infos.append((name, shape, gtype, off))
        pos = f.tell()
        # the data section and per-tensor offsets are aligned to
        # general.alignment (llama.cpp default 32) — honor non-default
        # alignments from ecosystem files
        align = int(metadata.get("general.alignment", 32))
        base = pos + (align - pos % align) % align
        tensors = {}
        for name, shape, gtype, off in infos:
            f.seek(base + off)
            numel = 1
            for dim in shape:
                numel *= dim
            if gtype == GGML_F32:
                raw = f.read(numel * 4)
                data = torch.frombuffer(bytearray(raw), dtype=torch.float32)\
                    .reshape(shape).clone()
            elif gtype == GGML_F16:
                raw = f.read(numel * 2)
                data = torch.frombuffer(bytearray(raw), dtype=torch.float16)\
                    .reshape(shape).clone()
            elif gtype in (GGML_Q4_0, GGML_Q8_0):
                bb = Q4_0_BLOCK_BYTES if gtype == GGML_Q4_0 else Q8_0_BLOCK_BYTES
                nbytes = numel // QK * bb
                raw = f.read(nbytes)
                data = torch.frombuffer(bytearray(raw), dtype=torch.uint8)\
                    .reshape(*shape[:-1], shape[-1] // QK * bb).clone()
            elif gtype in (GGML_Q4_K, GGML_Q6_K):
                bb = Q4_K_BLOCK_BYTES if gtype == GGML_Q4_K \
                    else Q6_K_BLOCK_BYTES
                nbytes = numel // QK_K * bb
                raw = f.read(nbytes)
                data = torch.frombuffer(bytearray(raw), dtype=torch.uint8)\
                    .reshape(*shape[:-1], shape[-1] // QK_K * bb).clone()
            else:
                raise ValueError(f"unsupported ggml type {gtype} for {name}")
            tensors[name] = GGUFTensor(name, shape, gtype, data)
        return metadata, tensors


def export_model_gguf(path: str, state_dict: dict, model_name: str,
                      quant: str = "q4_0") -> None:
    """Write a CausalLM state dict as GGUF: 2-D weights quantized with
    `quant`, 1-D tensors as F32."""
    gtype, _bb, qfn, _ = QUANT_CODECS[quant]
    blk = QK_K if quant.endswith("_K") else QK
    tensors: dict = {}
    for name, t in state_dict.items():
        t = t.detach().float().cpu()
        # projections quantize; embeddings/norms stay F32 (matches the
        # in-memory quantize_model policy). K-quants need K % 256 == 0;
        # non-divisible projections fall back to q8_0 like llama.cpp's
        # mixed-type files do for awkward shapes.
        if t.dim() >= 2 and "embed" not in name:
            if t.shape[-1] % blk == 0:
                tensors[name] = (qfn(t), gtype, list(t.shape))
                continue
            if quant.endswith("_K") and t.shape[-1] % QK == 0:
                tensors[name] = (quantize_q8_0(t), GGML_Q8_0,
                                 list(t.shape))
                continue
        tensors[name] = (t, GGML_F32, list(t.shape))
    write_gguf(path, tensors, metadata={
        "general.architecture": "substratus-causallm",
        "general.name": model_name,
        "substratus.quant": quant,
    })
