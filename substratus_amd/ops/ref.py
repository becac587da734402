"""Reference (plain PyTorch, fp32-accumulating) implementations of every op the
HIP kernel library provides.

These are the numerics oracle for the kernel parity tests (a HIP
kernel's test compares it against a plain PyTorch fp32 reference of the same op)
and the CPU execution path for control-plane/system tests on machines without a
GPU. On a ROCm GPU the dispatch layer (substratus_amd/ops/__init__.py) refuses
to fall back to these: the HIP extension is mandatory there.
"""

from __future__ import annotations

import math
from typing import Optional

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """y = x / rms(x) * weight, accumulating in fp32."""
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * weight.float()
    return y.to(x.dtype)


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float) -> torch.Tensor:
    return torch.nn.functional.layer_norm(
        x.float(), (x.shape[-1],), weight.float(), bias.float(), eps).to(x.dtype)


def rope_cos_sin(positions: torch.Tensor, head_dim: int, theta: float,
                 dtype: torch.dtype = torch.float32) -> tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables [N, head_dim/2] for the given positions (fp32)."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2,
                      dtype=torch.float32, device=positions.device) / head_dim))
    ang = positions.float()[:, None] * inv_freq[None, :]
    return ang.cos(), ang.sin()


def apply_rope(x: torch.Tensor, positions: torch.Tensor, theta: float) -> torch.Tensor:
    """Neox-style rotary embedding. x: [N, H, D]; positions: [N].
    theta <= 0 disables rotation (opt-style models)."""
    if theta <= 0:
        return x
    n, h, d = x.shape
    cos, sin = rope_cos_sin(positions, d, theta)
    xf = x.float()
    x1, x2 = xf[..., : d // 2], xf[..., d // 2:]
    cos = cos[:, None, :]
    sin = sin[:, None, :]
    out = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)
    return out.to(x.dtype)


def rope_kv_append(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   positions: torch.Tensor, k_cache: torch.Tensor,
                   v_cache: torch.Tensor, slots: torch.Tensor,
                   theta: float) -> torch.Tensor:
    """Apply RoPE to q and k (in fp32), then scatter k,v into the paged cache.

    q: [N, Hq, D]; k, v: [N, Hkv, D]; positions: [N] int32 (absolute position,
    drives RoPE); slots: [N] int32/int64 flat slot index = block_id * block_size
    + offset into k_cache/v_cache of shape [num_blocks, Hkv, block_size, D].
    Returns the rotated q. k_cache/v_cache are written in place.
    """
    q_out = apply_rope(q, positions, theta)
    k_rot = apply_rope(k, positions, theta)
    k.copy_(k_rot)  # k is rotated IN PLACE too (prefill consumes it dense)
    num_blocks, hkv, block_size, d = k_cache.shape
    blk = torch.div(slots, block_size, rounding_mode="floor").long()
    off = (slots % block_size).long()
    k_cache[blk, :, off, :] = k_rot.to(k_cache.dtype)
    v_cache[blk, :, off, :] = v.to(v_cache.dtype)
    return q_out


def kv_append(k: torch.Tensor, v: torch.Tensor, k_cache: torch.Tensor,
              v_cache: torch.Tensor, slots: torch.Tensor) -> None:
    """Scatter pre-rotated k,v into the paged cache (no RoPE)."""
    block_size = k_cache.shape[2]
    blk = torch.div(slots, block_size, rounding_mode="floor").long()
    off = (slots % block_size).long()
    k_cache[blk, :, off, :] = k.to(k_cache.dtype)
    v_cache[blk, :, off, :] = v.to(v_cache.dtype)


def paged_attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                      v_cache: torch.Tensor, block_table: torch.Tensor,
                      seq_lens: torch.Tensor, scale: float) -> torch.Tensor:
    """Single-token decode attention over a block-paged KV cache.

    q: [B, Hq, D]; k_cache/v_cache: [num_blocks, Hkv, block_size, D];
    block_table: [B, max_blocks] int32; seq_lens: [B] int32 (length INCLUDING
    the current token, whose k/v are already in the cache).
    Returns [B, Hq, D] in q.dtype; softmax/accumulation in fp32.
    """
    bsz, hq, d = q.shape
    num_blocks, hkv, block_size, _ = k_cache.shape
    group = hq // hkv
    out = torch.empty_like(q)
    for b in range(bsz):
        n = int(seq_lens[b])
        nblk = (n + block_size - 1) // block_size
        blocks = block_table[b, :nblk].long()
        k = k_cache[blocks].permute(1, 0, 2, 3).reshape(hkv, nblk * block_size, d)[:, :n]
        v = v_cache[blocks].permute(1, 0, 2, 3).reshape(hkv, nblk * block_size, d)[:, :n]
        qb = q[b].float().view(hkv, group, d)
        scores = torch.einsum("hgd,hnd->hgn", qb, k.float()) * scale
        p = torch.softmax(scores, dim=-1)
        ob = torch.einsum("hgn,hnd->hgd", p, v.float())
        out[b] = ob.reshape(hq, d).to(q.dtype)
    return out


def attn_prefill(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 seq_start: torch.Tensor, scale: float,
                 causal: bool = True) -> torch.Tensor:
    """Varlen causal prefill attention over packed sequences.

    q: [T, Hq, D]; k, v: [T, Hkv, D]; seq_start: [num_seqs+1] int32 cumulative
    token offsets. fp32 softmax/accum.
    """
    t, hq, d = q.shape
    hkv = k.shape[1]
    group = hq // hkv
    out = torch.empty_like(q)
    for s in range(seq_start.numel() - 1):
        a, b = int(seq_start[s]), int(seq_start[s + 1])
        n = b - a
        qs = q[a:b].float().view(n, hkv, group, d)
        ks = k[a:b].float()
        vs = v[a:b].float()
        scores = torch.einsum("ihgd,jhd->hgij", qs, ks) * scale
        if causal:
            mask = torch.full((n, n), float("-inf"), device=q.device).triu(1)
            scores = scores + mask
        p = torch.softmax(scores, dim=-1)
        o = torch.einsum("hgij,jhd->ihgd", p, vs)
        out[a:b] = o.reshape(n, hq, d).to(q.dtype)
    return out


def silu_mul(x: torch.Tensor) -> torch.Tensor:
    """x: [..., 2F] (gate ‖ up) -> silu(gate) * up, fp32 math."""
    gate, up = x.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(gate) * up).to(x.dtype)


def gelu(x: torch.Tensor) -> torch.Tensor:
    return torch.nn.functional.gelu(x.float(), approximate="tanh").to(x.dtype)


def softmax_sample(logits: torch.Tensor, temperature: torch.Tensor,
                   top_k: torch.Tensor, top_p: torch.Tensor,
                   uniform: torch.Tensor) -> torch.Tensor:
    """Temperature / top-k / top-p sampling of one token per row.

    logits: [B, V] (any float dtype); temperature/top_p: [B] fp32 (temperature
    <= 0 means greedy); top_k: [B] int32 (0 = disabled); uniform: [B] fp32 in
    [0,1) — the random draw, passed in so GPU and reference sample identically.
    Returns [B] int64 token ids.
    """
    bsz, vocab = logits.shape
    out = torch.empty(bsz, dtype=torch.int64, device=logits.device)
    lf = logits.float()
    for b in range(bsz):
        row = lf[b]
        t = float(temperature[b])
        if t <= 0.0:
            out[b] = int(torch.argmax(row))
            continue
        row = row / t
        k = int(top_k[b])
        if 0 < k < vocab:
            kth = torch.topk(row, k).values[-1]
            row = torch.where(row < kth, torch.full_like(row, float("-inf")), row)
        p = torch.softmax(row, dim=-1)
        tp = float(top_p[b])
        if 0.0 < tp < 1.0:
            sp, idx = torch.sort(p, descending=True)
            cum = torch.cumsum(sp, 0)
            keep = cum - sp < tp  # keep tokens until cumulative prob >= top_p
            sp = torch.where(keep, sp, torch.zeros_like(sp))
            sp = sp / sp.sum()
            # inverse-cdf draw on the sorted distribution
            j = int(torch.searchsorted(torch.cumsum(sp, 0), uniform[b], right=False).clamp(max=vocab - 1))
            out[b] = int(idx[j])
        else:
            j = int(torch.searchsorted(torch.cumsum(p, 0), uniform[b], right=False).clamp(max=vocab - 1))
            out[b] = j
    return out


def cross_entropy_fwd_bwd(logits: torch.Tensor, targets: torch.Tensor,
                          ignore_index: int = -100) -> tuple[torch.Tensor, torch.Tensor]:
    """Fused log-softmax + NLL + dlogits (mean over non-ignored targets).

    logits: [N, V] (bf16/fp32); targets: [N] int64. Returns (scalar fp32 loss,
    dlogits in logits.dtype). dlogits = (softmax - onehot)/n_valid.
    """
    lf = logits.float()
    valid = targets != ignore_index
    n_valid = int(valid.sum())
    lse = torch.logsumexp(lf, dim=-1)
    tclamped = targets.clamp(min=0)
    tlogit = lf.gather(1, tclamped[:, None]).squeeze(1)
    losses = (lse - tlogit) * valid
    loss = losses.sum() / max(n_valid, 1)
    p = torch.softmax(lf, dim=-1)
    p.scatter_add_(1, tclamped[:, None],
                   -torch.ones_like(tlogit)[:, None])
    dlogits = p * (valid[:, None] / max(n_valid, 1))
    return loss, dlogits.to(logits.dtype)


def adamw_step(param: torch.Tensor, grad: torch.Tensor, exp_avg: torch.Tensor,
               exp_avg_sq: torch.Tensor, lr: float, beta1: float, beta2: float,
               eps: float, weight_decay: float, step: int,
               grad_scale: float = 1.0) -> None:
    """One fused AdamW update, in place. param may be bf16 (master-less);
    exp_avg/exp_avg_sq are fp32. Decoupled weight decay. grad_scale folds
    gradient clipping into the read of g."""
    g = grad.float() * grad_scale
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    pf = param.float()
    pf.mul_(1 - lr * weight_decay)
    denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    pf.addcdiv_(exp_avg / bc1, denom, value=-lr)
    param.copy_(pf.to(param.dtype))
