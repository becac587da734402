"""Direct PagedKVCache / BlockAllocator / SequenceState coverage. The
engine tests exercise these indirectly; these pin the primitives'
contracts (allocation accounting, memory sizing arithmetic, slot
mapping) so engine refactors cannot silently bend them."""

import pytest
import torch

from substratus_amd.models.config import get_config
from substratus_amd.serve.kv_cache import (BlockAllocator, OutOfBlocks,
                                           PagedKVCache, SequenceState)


def test_allocator_accounting():
    a = BlockAllocator(8)
    assert a.free_count == 8
    got = a.alloc(3)
    assert len(got) == 3 and len(set(got)) == 3
    assert a.free_count == 5
    with pytest.raises(OutOfBlocks):
        a.alloc(6)
    assert a.free_count == 5          # failed alloc takes nothing
    a.free(got)
    assert a.free_count == 8
    # every block id stays in [0, n)
    assert all(0 <= b < 8 for b in a.alloc(8))


def test_blocks_for_memory_matches_layout():
    cfg = get_config("llama-tiny")
    n_local = cfg.num_kv_heads
    nb = 6
    cache = PagedKVCache(cfg, n_local, nb, block_size=16, device="cpu",
                         dtype=torch.bfloat16)
    actual = sum(k.numel() * k.element_size() + v.numel() * v.element_size()
                 for k, v in cache.layers)
    # sizing must invert exactly: the budget that buys nb blocks is the
    # bytes nb blocks actually occupy
    assert PagedKVCache.blocks_for_memory(cfg, n_local, actual,
                                          block_size=16) == nb
    assert PagedKVCache.blocks_for_memory(cfg, n_local, actual - 1,
                                          block_size=16) == nb - 1
    # fp8 cache holds twice the blocks for the same budget
    assert PagedKVCache.blocks_for_memory(cfg, n_local, actual,
                                          block_size=16,
                                          dtype_bytes=1) == 2 * nb


def test_sequence_slot_mapping():
    s = SequenceState(seq_id=1, tokens=list(range(40)), num_prompt=40,
                      max_new_tokens=8, blocks=[7, 2, 9])
    bs = 16
    assert s.needed_blocks(bs) == 3
    assert s.needed_blocks(bs, extra_tokens=9) == 4    # 49 tokens
    # position -> physical slot: block_id * bs + offset
    assert s.slot_of(0, bs) == 7 * 16
    assert s.slot_of(15, bs) == 7 * 16 + 15
    assert s.slot_of(16, bs) == 2 * 16
    assert s.slot_of(39, bs) == 9 * 16 + 7
    assert s.num_generated == 0
    s.tokens.append(99)
    assert s.num_generated == 1


def test_tunable_loader_cpu_noop(monkeypatch):
    """load_tunables is a strict no-op without a GPU, under SUB_TUNABLEOP=0,
    and when the user drives TunableOp explicitly."""
    import substratus_amd.utils.tunable as tun
    monkeypatch.setattr(tun, "_loaded", None)
    assert tun.load_tunables() is False          # no CUDA here
    monkeypatch.setattr(tun, "_loaded", None)
    monkeypatch.setenv("PYTORCH_TUNABLEOP_ENABLED", "1")
    assert tun.load_tunables() is False          # explicit user control
    monkeypatch.delenv("PYTORCH_TUNABLEOP_ENABLED")
    monkeypatch.setattr(tun, "_loaded", None)
    monkeypatch.setenv("SUB_TUNABLEOP", "0")
    assert tun.load_tunables() is False
    # the shipped table exists and parses: validators + gemm entries
    import os
    path = os.path.join(tun.tunables_dir(), "gfx950.csv")
    assert os.path.exists(path)
    lines = open(path).read().strip().split("\n")
    assert any(line.startswith("Validator,GCN_ARCH_NAME,gfx950")
               for line in lines)
    assert sum(1 for line in lines if line.startswith("GemmTunableOp")) >= 20
