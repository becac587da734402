"""Unified decoder-only transformer for the llama / falcon / opt families.

One implementation serves all model configs the reference's examples name
(SURVEY.md §2.5): llama2-7b/13b/70b (RMSNorm + RoPE + GQA + SwiGLU),
falcon-7b-instruct (LayerNorm + RoPE + MQA + parallel attn/MLP + GELU),
facebook/opt-125m (LayerNorm + learned positions + GELU).

Three execution paths:
  * forward_prefill — varlen packed prompt ingestion writing the paged KV cache
  * forward_decode  — one token per sequence over the paged KV cache (the serve
    hot loop; every op here is a hand-written HIP kernel on GPU)
  * forward_train   — dense differentiable forward (trainer; autograd +
    torch SDPA; loss/gradients via the fused cross-entropy kernel)

Tensor-parallel: heads are sharded column-parallel for QKV, row-parallel for
the output/down projections (2 RCCL all-reduces per layer — SURVEY.md §2.5
ledger items 9).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..parallel.comm import state
from ..parallel.tp import ColumnParallelLinear, Linear, RowParallelLinear
from .config import ModelConfig


def _norm(x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor],
          eps: float, rms: bool) -> torch.Tensor:
    if rms:
        return ops.rms_norm(x, weight, eps)
    return ops.layer_norm(x, weight, bias, eps)


def fused_norm_residual(delta: torch.Tensor, h: torch.Tensor,
                        norm: "NormParams",
                        eps: float) -> tuple[torch.Tensor, torch.Tensor]:
    """y = norm(h + delta), h = h + delta — fused on the RMSNorm path so the
    hidden state makes one HBM round trip per norm (profiles/
    serve_decode_r01_baseline.txt: unfused residual adds were ~2.2 ms of a
    10.3 ms decode step)."""
    if norm.rms:
        return ops.rms_norm_residual(delta, h, norm.weight, eps)
    h = h + delta
    return norm(h, eps), h


class NormParams(nn.Module):
    def __init__(self, size: int, rms: bool, device=None, dtype=None):
        super().__init__()
        self.rms = rms
        self.weight = nn.Parameter(torch.ones(size, device=device, dtype=dtype))
        self.bias = None if rms else nn.Parameter(
            torch.zeros(size, device=device, dtype=dtype))

    def forward(self, x: torch.Tensor, eps: float) -> torch.Tensor:
        return _norm(x, self.weight, self.bias, eps, self.rms)


class Block(nn.Module):
    def __init__(self, cfg: ModelConfig, device=None, dtype=None):
        super().__init__()
        st = state()
        tp = st.tp_size
        self.cfg = cfg
        self.tp = tp
        assert cfg.num_heads % tp == 0, "num_heads must divide TP world size"
        self.n_heads = cfg.num_heads // tp
        # KV heads: shard when possible, replicate when fewer than TP ranks
        # (falcon MQA, llama2-70b GQA at high TP).
        if cfg.num_kv_heads >= tp:
            assert cfg.num_kv_heads % tp == 0
            self.n_kv_heads = cfg.num_kv_heads // tp
            self.kv_replicated = False
        else:
            self.n_kv_heads = 1
            self.kv_replicated = True
        d = cfg.head_dim
        rms = cfg.family == "llama"
        bias = cfg.family == "opt"
        eps = cfg.rms_norm_eps if rms else cfg.layer_norm_eps
        self.eps = eps
        self.norm1 = NormParams(cfg.hidden_size, rms, device, dtype)
        # Fused QKV, column-parallel by construction: this rank's slice is
        # (n_heads + 2*n_kv_heads) * head_dim output features.
        self.wqkv = Linear(cfg.hidden_size,
                           (self.n_heads + 2 * self.n_kv_heads) * d,
                           bias=bias, device=device, dtype=dtype)
        self.wo = RowParallelLinear(cfg.num_heads * d, cfg.hidden_size,
                                    bias=bias, device=device, dtype=dtype)
        self.parallel_attn = cfg.parallel_attn
        self.parallel_norms = cfg.parallel_norms
        # norm2 is the post-attention norm (sequential blocks) OR falcon-40b's
        # ln_mlp (parallel blocks with parallel_norms: norm1 = ln_attn feeds
        # attention, norm2 = ln_mlp feeds the MLP, both of the same residual)
        if not cfg.parallel_attn or cfg.parallel_norms:
            self.norm2 = NormParams(cfg.hidden_size, rms, device, dtype)
        if cfg.activation == "silu":
            self.w_gateup = ColumnParallelLinear(
                cfg.hidden_size, 2 * cfg.intermediate_size, bias=bias,
                device=device, dtype=dtype)
        else:
            self.w_up = ColumnParallelLinear(
                cfg.hidden_size, cfg.intermediate_size, bias=bias,
                device=device, dtype=dtype)
        self.w_down = RowParallelLinear(cfg.intermediate_size, cfg.hidden_size,
                                        bias=bias, device=device, dtype=dtype)
        self.scale = 1.0 / math.sqrt(d)
        self.rope_theta = cfg.rope_theta if cfg.family in ("llama", "falcon") else 0.0

    # -- shared pieces ------------------------------------------------------
    def _split_qkv(self, qkv: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        d = self.cfg.head_dim
        n = qkv.shape[0]
        q, k, v = qkv.split([self.n_heads * d, self.n_kv_heads * d,
                             self.n_kv_heads * d], dim=-1)
        return (q.view(n, self.n_heads, d), k.view(n, self.n_kv_heads, d),
                v.view(n, self.n_kv_heads, d))

    def _mlp(self, y: torch.Tensor) -> torch.Tensor:
        if self.cfg.activation == "silu":
            return self.w_down(ops.silu_mul(self.w_gateup(y)))
        return self.w_down(ops.gelu(self.w_up(y)))

    # -- decode -------------------------------------------------------------
    def decode_ops(self, y: torch.Tensor, h: torch.Tensor,
                   positions: torch.Tensor,
                   k_cache: torch.Tensor, v_cache: torch.Tensor,
                   slots: torch.Tensor, block_table: torch.Tensor,
                   seq_lens: torch.Tensor):
        """Generator form of the decode block: yields after each op group
        (qkv GEMM | rope+attention | o GEMM | up GEMM | down GEMM) so the
        dual-stream runner (serve/graph_runner.py) can interleave two
        half-batches op-by-op — the trailing half's GEMMs re-read weights
        the leading half just streamed (LLC-hot) while its attention
        overlaps the leader's compute. forward_decode drives this
        generator to completion, so single-stream numerics are by
        construction identical."""
        qkv = self.wqkv(y)
        yield
        q, k, v = self._split_qkv(qkv)
        q = ops.rope_kv_append(q, k, v, positions, k_cache, v_cache, slots,
                               self.rope_theta)
        a = ops.paged_attn_decode(q, k_cache, v_cache, block_table, seq_lens,
                                  self.scale)
        yield
        attn_out = self.wo(a.reshape(a.shape[0], -1))
        yield
        if self.parallel_attn:
            mlp_in = self.norm2(h, self.eps) if self.parallel_norms else y
            if self.cfg.activation == "silu":
                gu = self.w_gateup(mlp_in)
                yield
                out = attn_out + self.w_down(ops.silu_mul(gu))
            else:
                u = self.w_up(mlp_in)
                yield
                out = attn_out + self.w_down(ops.gelu(u))
            return out, h
        y2, h = fused_norm_residual(attn_out, h, self.norm2, self.eps)
        if self.cfg.activation == "silu":
            gu = self.w_gateup(y2)
            yield
            out = self.w_down(ops.silu_mul(gu))
        else:
            u = self.w_up(y2)
            yield
            out = self.w_down(ops.gelu(u))
        return out, h

    def forward_decode(self, y: torch.Tensor, h: torch.Tensor,
                       positions: torch.Tensor,
                       k_cache: torch.Tensor, v_cache: torch.Tensor,
                       slots: torch.Tensor, block_table: torch.Tensor,
                       seq_lens: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        """y: normalized input (this block's norm1 already applied by the
        caller's fused residual+norm); h: the residual stream. Returns
        (delta, h) where delta is this block's output to be added by the
        NEXT norm's fused residual."""
        gen = self.decode_ops(y, h, positions, k_cache, v_cache, slots,
                              block_table, seq_lens)
        while True:
            try:
                next(gen)
            except StopIteration as e:
                return e.value

    # -- prefill ------------------------------------------------------------
    def forward_prefill(self, h: torch.Tensor, positions: torch.Tensor,
                        seq_start: torch.Tensor, k_cache: torch.Tensor,
                        v_cache: torch.Tensor, slots: torch.Tensor) -> torch.Tensor:
        y = self.norm1(h, self.eps)
        q, k, v = self._split_qkv(self.wqkv(y))
        # rope_kv_append rotates q AND k in place (k also lands rotated in
        # the paged cache); the dense varlen attention consumes them directly
        k = k.contiguous()
        q = ops.rope_kv_append(q, k, v, positions, k_cache, v_cache, slots,
                               self.rope_theta)
        a = ops.attn_prefill(q, k, v, seq_start, self.scale)
        attn_out = self.wo(a.reshape(a.shape[0], -1))
        if self.parallel_attn:
            mlp_in = self.norm2(h, self.eps) if self.parallel_norms else y
            return h + attn_out + self._mlp(mlp_in)
        h = h + attn_out
        return h + self._mlp(self.norm2(h, self.eps))

    # -- train (dense, differentiable) --------------------------------------
    def forward_train(self, h: torch.Tensor, cos: torch.Tensor,
                      sin: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        if self.norm1.rms:
            y = F.rms_norm(h.float(), (cfg.hidden_size,),
                           self.norm1.weight.float(), self.eps).to(h.dtype)
        else:
            y = F.layer_norm(h.float(), (cfg.hidden_size,),
                             self.norm1.weight.float(),
                             self.norm1.bias.float(), self.eps).to(h.dtype)
        b, s, _ = y.shape
        d = cfg.head_dim
        q, k, v = self.wqkv(y).split(
            [self.n_heads * d, self.n_kv_heads * d, self.n_kv_heads * d], dim=-1)
        if (h.is_cuda and h.dtype == torch.bfloat16
                and ops.flash_attn_train_available(d)):
            # first-party MFMA flash fwd+bwd (attention_prefill.hip /
            # attention_bwd.hip): BSHD layout end-to-end — no transposes,
            # no GQA head materialization, no torch-SDPA/aotriton on the
            # trainer hot path
            q = q.view(b, s, self.n_heads, d)
            k = k.view(b, s, self.n_kv_heads, d)
            v = v.view(b, s, self.n_kv_heads, d)
            if self.rope_theta > 0:
                q = _rope_train(q, cos, sin, bshd=True)
                k = _rope_train(k, cos, sin, bshd=True)
            a = ops.flash_attn_train(q, k, v, self.scale)
            a = a.reshape(b, s, self.n_heads * d)
        else:
            q = q.view(b, s, self.n_heads, d).transpose(1, 2)
            k = k.view(b, s, self.n_kv_heads, d).transpose(1, 2)
            v = v.view(b, s, self.n_kv_heads, d).transpose(1, 2)
            if self.rope_theta > 0:
                q = _rope_train(q, cos, sin)
                k = _rope_train(k, cos, sin)
            if self.n_kv_heads != self.n_heads:
                rep = self.n_heads // self.n_kv_heads
                k = k.repeat_interleave(rep, dim=1)
                v = v.repeat_interleave(rep, dim=1)
            a = F.scaled_dot_product_attention(q, k, v, is_causal=True)
            a = a.transpose(1, 2).reshape(b, s, self.n_heads * d)
        attn_out = self.wo(a)
        if self.parallel_attn:
            if self.parallel_norms:
                y_mlp = F.layer_norm(h.float(), (cfg.hidden_size,),
                                     self.norm2.weight.float(),
                                     self.norm2.bias.float(),
                                     self.eps).to(h.dtype)
            else:
                y_mlp = y
            return h + attn_out + self._mlp_train(y_mlp)
        h = h + attn_out
        if self.norm2.rms:
            y2 = F.rms_norm(h.float(), (cfg.hidden_size,),
                            self.norm2.weight.float(), self.eps).to(h.dtype)
        else:
            y2 = F.layer_norm(h.float(), (cfg.hidden_size,),
                              self.norm2.weight.float(),
                              self.norm2.bias.float(), self.eps).to(h.dtype)
        return h + self._mlp_train(y2)

    def _mlp_train(self, y: torch.Tensor) -> torch.Tensor:
        if self.cfg.activation == "silu":
            gu = self.w_gateup(y)
            g, u = gu.chunk(2, dim=-1)
            return self.w_down(F.silu(g) * u)
        return self.w_down(F.gelu(self.w_up(y), approximate="tanh"))


def _rope_train(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                bshd: bool = False) -> torch.Tensor:
    # x: [B, H, S, D] (or [B, S, H, D] when bshd); cos/sin: [S, D/2]
    d = x.shape[-1]
    x1, x2 = x[..., : d // 2], x[..., d // 2:]
    if bshd:
        c = cos[None, :, None, :]
        s = sin[None, :, None, :]
    else:
        c = cos[None, None, :, :]
        s = sin[None, None, :, :]
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


class CausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig, device=None, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        st = state()
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                         device=device, dtype=dtype)
        if cfg.learned_pos_emb:
            self.embed_positions = nn.Embedding(
                cfg.max_position_embeddings, cfg.hidden_size, device=device,
                dtype=dtype)
        self.blocks = nn.ModuleList(
            Block(cfg, device=device, dtype=dtype) for _ in range(cfg.num_layers))
        rms = cfg.family == "llama"
        self.eps = cfg.rms_norm_eps if rms else cfg.layer_norm_eps
        self.final_norm = NormParams(cfg.hidden_size, rms, device, dtype)
        if cfg.tie_word_embeddings:
            self.lm_head = None
        else:
            self.lm_head = Linear(cfg.hidden_size, cfg.vocab_size, bias=False,
                                  device=device, dtype=dtype)
        self.n_kv_heads = self.blocks[0].n_kv_heads  # per-rank
        self.gradient_checkpointing = False

    def _logits(self, y: torch.Tensor) -> torch.Tensor:
        if self.lm_head is None:
            return F.linear(y, self.embed_tokens.weight)
        return self.lm_head(y)  # module call: works for Linear AND QuantLinear

    @torch.no_grad()
    def init_random(self, seed: int = 0) -> None:
        """Initialize EVERY parameter (models are built on meta + to_empty,
        so nothing may be left untouched): matrices ~ N(0, 0.02), norm
        weights = 1, biases = 0."""
        dev = next(self.parameters()).device
        g = torch.Generator(device=dev).manual_seed(seed)
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                p.normal_(0.0, 0.02, generator=g)
            elif name.endswith("bias"):
                p.zero_()
            else:  # norm weights
                p.fill_(1.0)

    # -- decode --------------------------------------------------------------
    def decode_ops(self, tokens: torch.Tensor, positions: torch.Tensor,
                   kv_caches: list[tuple[torch.Tensor, torch.Tensor]],
                   slots: torch.Tensor, block_table: torch.Tensor,
                   seq_lens: torch.Tensor):
        """Whole-model decode as an op-group generator (see
        TransformerBlock.decode_ops); returns the logits."""
        h = self.embed_tokens(tokens)
        if self.cfg.learned_pos_emb:
            h = h + self.embed_positions(positions.long())
        # fused residual+norm chain: each block's output delta is added to
        # the residual stream inside the NEXT norm kernel
        y = self.blocks[0].norm1(h, self.eps)
        yield
        for i, (blk, (kc, vc)) in enumerate(zip(self.blocks, kv_caches)):
            delta, h = yield from blk.decode_ops(y, h, positions, kc, vc,
                                                 slots, block_table,
                                                 seq_lens)
            nxt = (self.blocks[i + 1].norm1 if i + 1 < len(self.blocks)
                   else self.final_norm)
            y, h = fused_norm_residual(delta, h, nxt, self.eps)
            yield
        return self._logits(y)

    def forward_decode(self, tokens: torch.Tensor, positions: torch.Tensor,
                       kv_caches: list[tuple[torch.Tensor, torch.Tensor]],
                       slots: torch.Tensor, block_table: torch.Tensor,
                       seq_lens: torch.Tensor) -> torch.Tensor:
        gen = self.decode_ops(tokens, positions, kv_caches, slots,
                              block_table, seq_lens)
        while True:
            try:
                next(gen)
            except StopIteration as e:
                return e.value

    # -- prefill -------------------------------------------------------------
    def forward_prefill(self, tokens: torch.Tensor, positions: torch.Tensor,
                        seq_start: torch.Tensor,
                        kv_caches: list[tuple[torch.Tensor, torch.Tensor]],
                        slots: torch.Tensor) -> torch.Tensor:
        """Returns logits for the LAST token of each sequence: [num_seqs, V]."""
        h = self.embed_tokens(tokens)
        if self.cfg.learned_pos_emb:
            h = h + self.embed_positions(positions.long())
        for blk, (kc, vc) in zip(self.blocks, kv_caches):
            h = blk.forward_prefill(h, positions, seq_start, kc, vc, slots)
        last = (seq_start[1:] - 1).long()
        h = self.final_norm(h[last], self.eps)
        return self._logits(h)

    # -- train ---------------------------------------------------------------
    def forward_train(self, tokens: torch.Tensor) -> torch.Tensor:
        """tokens: [B, S] -> logits [B, S, V], differentiable."""
        b, s = tokens.shape
        h = self.embed_tokens(tokens)
        pos = torch.arange(s, device=tokens.device)
        if self.cfg.learned_pos_emb:
            h = h + self.embed_positions(pos)
        if self.cfg.family in ("llama", "falcon"):
            from ..ops.ref import rope_cos_sin
            cos, sin = rope_cos_sin(pos, self.cfg.head_dim, self.cfg.rope_theta)
            cos, sin = cos.to(h.device), sin.to(h.device)
        else:
            cos = sin = None
        for blk in self.blocks:
            if self.gradient_checkpointing and h.requires_grad is not False \
                    and torch.is_grad_enabled():
                h = torch.utils.checkpoint.checkpoint(
                    blk.forward_train, h, cos, sin, use_reentrant=False)
            else:
                h = blk.forward_train(h, cos, sin)
        if self.final_norm.rms:
            h = F.rms_norm(h.float(), (self.cfg.hidden_size,),
                           self.final_norm.weight.float(), self.eps).to(h.dtype)
        else:
            h = F.layer_norm(h.float(), (self.cfg.hidden_size,),
                             self.final_norm.weight.float(),
                             self.final_norm.bias.float(), self.eps).to(h.dtype)
        return self._logits(h)
