"""Tensor-parallel weight sharding: full checkpoint ↔ per-rank shards.

Sharding map for CausalLM (SURVEY.md §2.5 ledger item 9 — TP over RCCL/xGMI):
  * wqkv.weight  [(<Hq+2Hkv>)·d, h] — the q/k/v row segments are sharded BY
    HEAD independently (k/v replicated when Hkv < tp, e.g. falcon MQA,
    llama2-70b GQA at TP=8) and re-fused per rank.
  * wo.weight    [h, Hq·d]  — input-column shard by the same head slice.
  * w_gateup.weight [2F, h] — gate and up row segments sharded separately and
    re-fused (a naive contiguous row shard would split gate/up unevenly).
  * w_up.weight  [F, h]     — row shard.
  * w_down.weight [h, F]    — input-column shard.
  * everything else (norms, embeddings, lm_head, biases of row-parallel
    outputs) — replicated.
"""

from __future__ import annotations

import torch

from .config import ModelConfig


def _kv_slice(cfg: ModelConfig, rank: int, tp: int) -> tuple[int, int]:
    """(start_head, n_heads) of this rank's KV heads; replicate when scarce."""
    if cfg.num_kv_heads >= tp:
        per = cfg.num_kv_heads // tp
        return rank * per, per
    return 0, 1  # replicated single head


def shard_param(name: str, t: torch.Tensor, cfg: ModelConfig, rank: int,
                tp: int) -> torch.Tensor:
    if tp <= 1:
        return t
    d = cfg.head_dim
    hq, hkv = cfg.num_heads, cfg.num_kv_heads
    nq = hq // tp
    kv0, nkv = _kv_slice(cfg, rank, tp)
    base = name.rsplit(".", 2)[-2] if "." in name else name

    if base == "wqkv":
        if name.endswith("weight") or name.endswith("bias"):
            qpart = t[:hq * d]
            kpart = t[hq * d: (hq + hkv) * d]
            vpart = t[(hq + hkv) * d:]
            return torch.cat([
                qpart[rank * nq * d: (rank + 1) * nq * d],
                kpart[kv0 * d: (kv0 + nkv) * d],
                vpart[kv0 * d: (kv0 + nkv) * d],
            ], dim=0)
    if base == "wo" and name.endswith("weight"):
        return t[:, rank * nq * d: (rank + 1) * nq * d]
    if base == "w_gateup":
        f = cfg.intermediate_size
        per = f // tp
        if name.endswith("weight") or name.endswith("bias"):
            gate = t[:f]
            up = t[f:]
            return torch.cat([gate[rank * per: (rank + 1) * per],
                              up[rank * per: (rank + 1) * per]], dim=0)
    if base == "w_up" and (name.endswith("weight") or name.endswith("bias")):
        per = cfg.intermediate_size // tp
        return t[rank * per: (rank + 1) * per]
    if base == "w_down" and name.endswith("weight"):
        per = cfg.intermediate_size // tp
        return t[:, rank * per: (rank + 1) * per]
    return t  # replicated


def shard_state_dict(full: dict, cfg: ModelConfig, rank: int,
                     tp: int) -> dict:
    return {k: shard_param(k, v, cfg, rank, tp) for k, v in full.items()}
