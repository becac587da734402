"""Block-paged KV cache sized for 288 GB HBM3E per MI355X GPU.

Layout per layer: k_cache/v_cache of shape [num_blocks, n_kv_heads, block_size,
head_dim] — one block's K rows for one head are a contiguous
block_size × head_dim panel (16×128 bf16 = 4 KiB), the unit the decode
attention kernel streams through LDS/registers with wide coalesced loads.
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch

from ..models.config import ModelConfig

DEFAULT_BLOCK_SIZE = 16


class OutOfBlocks(RuntimeError):
    pass


class BlockAllocator:
    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self._free = list(range(num_blocks - 1, -1, -1))

    @property
    def free_count(self) -> int:
        return len(self._free)

    def alloc(self, n: int = 1) -> list[int]:
        if len(self._free) < n:
            raise OutOfBlocks(f"need {n} KV blocks, {len(self._free)} free")
        return [self._free.pop() for _ in range(n)]

    def free(self, blocks: list[int]) -> None:
        self._free.extend(blocks)


class PagedKVCache:
    """Per-rank paged KV cache for every layer of one model."""

    def __init__(self, cfg: ModelConfig, n_kv_heads_local: int, num_blocks: int,
                 block_size: int = DEFAULT_BLOCK_SIZE, device="cpu",
                 dtype=torch.bfloat16):
        self.cfg = cfg
        self.block_size = block_size
        self.num_blocks = num_blocks
        self.layers: list[tuple[torch.Tensor, torch.Tensor]] = []
        for _ in range(cfg.num_layers):
            k = torch.zeros(num_blocks, n_kv_heads_local, block_size,
                            cfg.head_dim, device=device, dtype=dtype)
            v = torch.zeros_like(k)
            self.layers.append((k, v))
        self.allocator = BlockAllocator(num_blocks)

    @staticmethod
    def blocks_for_memory(cfg: ModelConfig, n_kv_heads_local: int,
                          bytes_budget: int,
                          block_size: int = DEFAULT_BLOCK_SIZE,
                          dtype_bytes: int = 2) -> int:
        per_block = (2 * cfg.num_layers * n_kv_heads_local * block_size *
                     cfg.head_dim * dtype_bytes)
        return max(bytes_budget // per_block, 1)


@dataclass(slots=True)
class SequenceState:
    """Server-side state of one running request."""
    seq_id: int
    tokens: list[int]               # prompt + generated
    num_prompt: int
    max_new_tokens: int
    temperature: float = 0.0
    top_k: int = 0
    top_p: float = 1.0
    blocks: list[int] = field(default_factory=list)
    batch_slot: int = -1            # position in the decode batch
    finished: bool = False
    eos_token: int = -1

    @property
    def length(self) -> int:
        return len(self.tokens)

    @property
    def num_generated(self) -> int:
        return self.length - self.num_prompt

    def needed_blocks(self, block_size: int, extra_tokens: int = 0) -> int:
        total = self.length + extra_tokens
        return (total + block_size - 1) // block_size

    def slot_of(self, pos: int, block_size: int) -> int:
        return self.blocks[pos // block_size] * block_size + pos % block_size
