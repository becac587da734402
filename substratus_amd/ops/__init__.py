"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, reference PyTorch on CPU.

Policy (task contract): on a ROCm GPU the HIP extension is MANDATORY — if a
tensor lives on a CUDA/HIP device and `substratus_amd._C` is not importable the
op raises instead of silently falling back to eager PyTorch. CPU tensors run
the fp32 reference implementations (substratus_amd/ops/ref.py), which are also
the parity oracle for the GPU tests.
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from . import ref

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from substratus_amd import _C  # built in-tree by setup.py build_ext --inplace
        _EXT = _C
    except ImportError as e:  # pragma: no cover - exercised only without the .so
        _EXT_ERR = str(e)
    return _EXT


def have_ext() -> bool:
    return _load_ext() is not None


def _dispatch(t: torch.Tensor):
    """Return the extension for GPU tensors (raising loudly if absent), None for CPU."""
    if t.is_cuda:
        ext = _load_ext()
        if ext is None:
            raise RuntimeError(
                "substratus_amd HIP extension (substratus_amd/_C*.so) is not built "
                f"but a GPU tensor was passed — no eager fallback on GPU. Build it "
                f"with `python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
                f"Import error: {_EXT_ERR}")
        return ext
    return None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    ext = _dispatch(x)
    if ext is None:
        return ref.rms_norm(x, weight, eps)
    out = torch.empty_like(x)
    ext.rms_norm(out, x.contiguous(), weight, eps)
    return out


def rms_norm_residual(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float) -> tuple[torch.Tensor, torch.Tensor]:
    """Fused: residual += x; y = rms_norm(residual). Returns (y, residual).

    The producing kernel fuses the residual add so the hidden state is read
    once from HBM (guide: fuse elementwise work into the producing kernel).
    """
    ext = _dispatch(x)
    if ext is None:
        residual = residual + x
        return ref.rms_norm(residual, weight, eps), residual
    y = torch.empty_like(x)
    ext.rms_norm_residual(y, residual, x.contiguous(), weight, eps)
    return y, residual


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float) -> torch.Tensor:
    ext = _dispatch(x)
    if ext is None:
        return ref.layer_norm(x, weight, bias, eps)
    out = torch.empty_like(x)
    ext.layer_norm(out, x.contiguous(), weight, bias, eps)
    return out


def _rows_contiguous(t: torch.Tensor) -> bool:
    """[n, h, d] with contiguous inner rows; the leading (token) stride may
    be larger — a qkv-split view."""
    return t.stride(2) == 1 and t.stride(1) == t.shape[2]


def rope_kv_append(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   positions: torch.Tensor, k_cache: torch.Tensor,
                   v_cache: torch.Tensor, slots: torch.Tensor,
                   theta: float) -> torch.Tensor:
    ext = _dispatch(q)
    if ext is None:
        return ref.rope_kv_append(q, k, v, positions, k_cache, v_cache, slots, theta)
    # The kernel takes a token-row stride, so qkv-split VIEWS flow straight
    # through (decode saves three elementwise copies per layer; rotation
    # lands in-place inside the fused qkv buffer, which nothing re-reads).
    # Only reshape if the inner [h, d] rows themselves aren't contiguous.
    if not _rows_contiguous(q):
        q = q.contiguous()
    if not _rows_contiguous(k):
        k = k.contiguous()
    if not _rows_contiguous(v):
        v = v.contiguous()
    ext.rope_kv_append(q, k, v, positions, k_cache, v_cache, slots, theta)
    return q


def paged_attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, block_table: torch.Tensor,
                      seq_lens: torch.Tensor, scale: float) -> torch.Tensor:
    ext = _dispatch(q)
    if ext is None:
        return ref.paged_attn_decode(q, k_cache, v_cache, block_table, seq_lens, scale)
    if not _rows_contiguous(q):
        q = q.contiguous()
    out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
    ext.paged_attn_decode(out, q, k_cache, v_cache,
                          block_table, seq_lens, scale)
    return out


def attn_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 seq_start: torch.Tensor, scale: float) -> torch.Tensor:
    ext = _dispatch(q)
    if ext is None:
        return ref.attn_prefill(q, k, v, seq_start, scale)
    out = torch.empty_like(q)
    max_len = int((seq_start[1:] - seq_start[:-1]).max())
    if os.environ.get("SUB_ATTN_V5", "1") != "0":
        # V5: 8-wave swapped-QK^T in-register-softmax structure —
        # 293-516 TF vs V4's 167-250 (scripts/microbench.py --which attnv5)
        ext.attn_v5(out, q.contiguous(), k.contiguous(), v.contiguous(),
                    seq_start, max_len, scale, None)
        return out
    ext.attn_prefill(out, q.contiguous(), k.contiguous(), v.contiguous(),
                     seq_start, max_len, scale)
    return out


class _FlashAttnTrain(torch.autograd.Function):
    """Training attention on the first-party MFMA kernels: varlen causal
    forward (attention_prefill.hip, LSE saved) + FA2-style backward
    (attention_bwd.hip). GQA native — K/V stay at n_kv_heads; no
    repeat_interleave materialization."""

    @staticmethod
    def forward(ctx, q, k, v, seq_start, max_len, scale):
        ext = _dispatch(q)
        assert ext is not None, "flash_attn_train needs the HIP extension"
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        out = torch.empty_like(q)
        lse = torch.empty(q.shape[0], q.shape[1], dtype=torch.float32,
                          device=q.device)
        if os.environ.get("SUB_ATTN_V5", "1") != "0":
            ext.attn_v5(out, q, k, v, seq_start, max_len, scale, lse)
        else:
            ext.attn_train_fwd(out, lse, q, k, v, seq_start, max_len, scale)
        ctx.save_for_backward(q, k, v, out, lse, seq_start)
        ctx.max_len = max_len
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse, seq_start = ctx.saved_tensors
        ext = _dispatch(q)
        dout = dout.contiguous()
        dq = torch.empty_like(q)
        # f32 accumulators: under GQA every q head of a group atomically
        # adds its dK/dV contribution (attention_bwd.hip dkv kernel)
        dk32 = torch.zeros(k.shape, dtype=torch.float32, device=k.device)
        dv32 = torch.zeros(v.shape, dtype=torch.float32, device=v.device)
        delta = torch.empty_like(lse)
        if os.environ.get("SUB_ATTN_V5", "1") != "0":
            ext.attn_train_bwd_v5(dq, dk32, dv32, q, k, v, out, dout, lse,
                                  delta, seq_start, ctx.max_len, ctx.scale)
        else:
            ext.attn_train_bwd(dq, dk32, dv32, q, k, v, out, dout, lse,
                               delta, seq_start, ctx.max_len, ctx.scale)
        return dq, dk32.to(k.dtype), dv32.to(v.dtype), None, None, None


def flash_attn_train(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     scale: float) -> torch.Tensor:
    """q: [B, S, HQ, D] bf16 cuda; k/v: [B, S, HKV, D]. Returns [B, S, HQ, D].
    Equal-length sequences are packed varlen (seq_start = i*S)."""
    b, s, hq, d = q.shape
    seq_start = torch.arange(0, (b + 1) * s, s, dtype=torch.int32,
                             device=q.device)
    out = _FlashAttnTrain.apply(
        q.reshape(b * s, hq, d), k.reshape(b * s, k.shape[2], d),
        v.reshape(b * s, v.shape[2], d), seq_start, s, scale)
    return out.view(b, s, hq, d)


def flash_attn_train_available(d: int) -> bool:
    return have_ext() and d in (64, 128)


# decode-projection GEMM (mfma_gemm.hip): split-K fp32 workspaces, cached
# per (splitk, m, n) so hipGraph capture replays into stable buffers
_GEMM_WS: dict = {}


def linear(x: torch.Tensor, weight: torch.Tensor,
           bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """F.linear, with an opt-in route (SUB_MGEMM=1) to the first-party
    256x256 glds GEMM (mfma_gemm.hip) on decode-shaped inference cases.

    NOT on by default: the round-3 measurement (profiles/mgemm_r02.md)
    shows hipBLASLt's MT192/MT256 solutions at 1.3-4.1 TB/s of weight
    stream on the llama2-7b decode shapes vs our kernel's 1.0-2.25 — the
    library already implements the deep-pipelined schedule whose measured
    ceiling on this chip (~1.33-1.47 PF, guide 8-phase template) equals
    hipBLASLt's best here. The kernel stays for measurement and as the
    base for a future counted-vmcnt schedule."""
    ext = _dispatch(x)
    if (ext is None or torch.is_grad_enabled() and (
            x.requires_grad or weight.requires_grad)
            or x.dtype != torch.bfloat16
            or os.environ.get("SUB_MGEMM") != "1"):
        return torch.nn.functional.linear(x, weight, bias)
    x2 = x.reshape(-1, x.shape[-1])
    m, k = x2.shape
    n = weight.shape[0]
    if not (256 <= m <= 1024):
        return torch.nn.functional.linear(x, weight, bias)
    splitk = int(ext.mfma_gemm_pick_splitk(m, n, k))
    if splitk == 0:   # unsupported shape
        return torch.nn.functional.linear(x, weight, bias)
    y = x.new_empty(*x.shape[:-1], n)
    part = None
    if splitk > 1:
        key = (x.device.index, splitk, m, n)
        part = _GEMM_WS.get(key)
        if part is None:
            part = torch.empty(splitk * m * n, dtype=torch.float32,
                               device=x.device)
            _GEMM_WS[key] = part
    ext.mfma_gemm(y.view(m, n), x2.contiguous(), weight, bias, part, splitk)
    return y


def silu_mul(x: torch.Tensor) -> torch.Tensor:
    ext = _dispatch(x)
    if ext is None:
        return ref.silu_mul(x)
    out = x.new_empty(*x.shape[:-1], x.shape[-1] // 2)
    ext.silu_mul(out, x.contiguous())
    return out


def gelu(x: torch.Tensor) -> torch.Tensor:
    ext = _dispatch(x)
    if ext is None:
        return ref.gelu(x)
    out = torch.empty_like(x)
    ext.gelu_tanh(out, x.contiguous())
    return out


def softmax_sample(logits: torch.Tensor, temperature: torch.Tensor,
                   top_k: torch.Tensor, top_p: torch.Tensor,
                   uniform: torch.Tensor) -> torch.Tensor:
    ext = _dispatch(logits)
    if ext is None:
        return ref.softmax_sample(logits, temperature, top_k, top_p, uniform)
    out = torch.empty(logits.shape[0], dtype=torch.int64, device=logits.device)
    ext.softmax_sample(out, logits.contiguous(), temperature, top_k, top_p, uniform)
    return out


def cross_entropy_fwd_bwd(logits: torch.Tensor, targets: torch.Tensor,
                          ignore_index: int = -100) -> tuple[torch.Tensor, torch.Tensor]:
    ext = _dispatch(logits)
    if ext is None:
        return ref.cross_entropy_fwd_bwd(logits, targets, ignore_index)
    loss = torch.zeros((), dtype=torch.float32, device=logits.device)
    dlogits = torch.empty_like(logits)
    ext.cross_entropy_fwd_bwd(loss, dlogits, logits.contiguous(), targets, ignore_index)
    return loss, dlogits


def adamw_step(param: torch.Tensor, grad: torch.Tensor, exp_avg: torch.Tensor,
               exp_avg_sq: torch.Tensor, lr: float, beta1: float, beta2: float,
               eps: float, weight_decay: float, step: int,
               grad_scale: float = 1.0) -> None:
    ext = _dispatch(param)
    if ext is None:
        ref.adamw_step(param, grad, exp_avg, exp_avg_sq, lr, beta1, beta2,
                       eps, weight_decay, step, grad_scale)
        return
    ext.adamw_step(param, grad, exp_avg, exp_avg_sq, lr, beta1, beta2,
                   eps, weight_decay, step, grad_scale)


# ---- weight-streaming decode GEMM (ops/hip/wstream_gemm.hip) -------------

def wstream_repack(w: torch.Tensor) -> torch.Tensor:
    """Repack a torch Linear weight W[N, K] (bf16) into the MFMA
    B-fragment-linear layout [N/16][K/32][64 lanes][8 bf16] the wstream
    decode GEMM streams sequentially. Lane l = lhi*16+llo holds
    W[n16*16 + llo][t*32 + lhi*8 + j] — one contiguous uint4 per fragment."""
    n, k = w.shape
    assert n % 16 == 0 and k % 32 == 0, (n, k)
    return (w.view(n // 16, 16, k // 32, 4, 8)
             .permute(0, 2, 3, 1, 4).contiguous().view(-1))


def wstream_supported(m: int, n: int, k: int) -> bool:
    return m <= 256 and n % 128 == 0 and k % 64 == 0


def wstream_linear(x: torch.Tensor, wp: torch.Tensor, n: int,
                   k: int) -> torch.Tensor:
    """Y[M, N] = X[M, K] @ W^T with W pre-repacked by wstream_repack.
    GPU-only (decode path); callers fall back to F.linear when unsupported."""
    ext = _dispatch(x)
    assert ext is not None, "wstream_linear is GPU-only"
    m = x.shape[0]
    y = torch.empty(m, n, dtype=x.dtype, device=x.device)
    ext.wstream_gemm(y, x.contiguous(), wp, n, k)
    return y
