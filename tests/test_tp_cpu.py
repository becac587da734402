"""Tensor-parallel CPU tests over gloo (world_size 2): a TP=2 decode must
match the single-process full model bit-for-tolerance — the correctness-by-
construction evidence for the multi-GPU Server path the driver never runs
here (SURVEY.md §4 MI355X test plan)."""

import io
import os

import pytest
import torch

from substratus_amd.models import CausalLM, get_config
from substratus_amd.models.sharding import shard_state_dict


def _tp_worker(rank, world, q):
    import traceback
    try:
        _tp_worker_inner(rank, world, q)
    except Exception:
        q.put((f"ERR{rank}", traceback.format_exc()))


def _tp_worker_inner(rank, world, q):
    import torch.distributed as dist
    os.environ.update({"RANK": str(rank), "WORLD_SIZE": str(world),
                       "LOCAL_RANK": str(rank),
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29771"})
    from substratus_amd.parallel import comm
    comm._STATE = comm.ParallelState()
    comm.init_distributed(tp_size=world, backend="gloo")
    torch.manual_seed(0)
    mcfg = get_config("llama-tiny")
    # full reference weights (built OUTSIDE the TP context), then sharded
    with comm.single_process():
        full = CausalLM(mcfg, device="cpu", dtype=torch.float32)
        full.init_random(seed=11)
    model = CausalLM(mcfg, device="cpu", dtype=torch.float32)
    model.load_state_dict(shard_state_dict(full.state_dict(), mcfg, rank,
                                           world))
    from substratus_amd.serve.engine import Engine, EngineConfig
    eng = Engine(EngineConfig(model="llama-tiny", dtype=torch.float32,
                              max_batch=2, max_seq_len=64, kv_blocks=32),
                 device="cpu", model=model)
    sid = eng.add_request([5, 9, 2, 7, 1, 3, 8, 11], max_new_tokens=6)
    toks = []
    while eng.has_work():
        toks.extend(o.token for o in eng.step())
    q.put((rank, toks))
    dist.destroy_process_group()


def test_tp2_decode_matches_full_model():
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_tp_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, toks = q.get(timeout=180)
        assert not str(rank).startswith("ERR"), toks
        results[rank] = toks
    for p in procs:
        p.join(timeout=60)
    assert results[0] == results[1], "TP ranks diverged"
    # single-process full model reference
    torch.manual_seed(0)
    mcfg = get_config("llama-tiny")
    full = CausalLM(mcfg, device="cpu", dtype=torch.float32)
    full.init_random(seed=11)
    from substratus_amd.serve.engine import Engine, EngineConfig
    eng = Engine(EngineConfig(model="llama-tiny", dtype=torch.float32,
                              max_batch=2, max_seq_len=64, kv_blocks=32),
                 device="cpu", model=full)
    sid = eng.add_request([5, 9, 2, 7, 1, 3, 8, 11], max_new_tokens=6)
    ref = []
    while eng.has_work():
        ref.extend(o.token for o in eng.step())
    assert results[0] == ref, f"TP tokens {results[0]} != full-model {ref}"


def test_shard_shapes_llama70b_tp8():
    """llama2-70b at TP=8: 8 q heads + 1 kv head per rank, fused segments."""
    cfg = get_config("llama2-70b")
    d = cfg.head_dim
    full_wqkv = torch.arange((cfg.num_heads + 2 * cfg.num_kv_heads) * d *
                             2, dtype=torch.float32).reshape(-1, 2)
    # use a skinny fake [rows, 2] weight to check row bookkeeping
    from substratus_amd.models.sharding import shard_param
    for rank in range(8):
        local = shard_param("blocks.0.wqkv.weight", full_wqkv, cfg, rank, 8)
        assert local.shape[0] == (cfg.num_heads // 8 + 2 * 1) * d
    gateup = torch.zeros(2 * cfg.intermediate_size, 3)
    local = shard_param("blocks.0.w_gateup.weight", gateup, cfg, 0, 8)
    assert local.shape[0] == 2 * cfg.intermediate_size // 8
    down = torch.zeros(4, cfg.intermediate_size)
    local = shard_param("blocks.0.w_down.weight", down, cfg, 3, 8)
    assert local.shape[1] == cfg.intermediate_size // 8


def test_shard_reconstruction_roundtrip():
    """Concatenating every rank's shards reproduces the full tensors exactly
    (column-parallel rows re-fuse; row-parallel cols re-fuse; replicated
    params are identical on all ranks)."""
    import torch
    from substratus_amd.models.config import get_config
    from substratus_amd.models.model import CausalLM
    from substratus_amd.models.sharding import shard_state_dict
    from substratus_amd.parallel.comm import single_process

    cfg = get_config("llama-tiny")
    tp = 2
    with single_process():
        model = CausalLM(cfg, device="meta", dtype=torch.float32)
        model = model.to_empty(device="cpu")
        model.init_random(3)
    full = {k: v.clone() for k, v in model.state_dict().items()}
    shards = [shard_state_dict(full, cfg, r, tp) for r in range(tp)]
    d = cfg.head_dim
    hq, hkv = cfg.num_heads, cfg.num_kv_heads
    nq = hq // tp
    for name, t in full.items():
        base = name.rsplit(".", 2)[-2] if "." in name else name
        parts = [shards[r][name] for r in range(tp)]
        if base == "wqkv" and name.endswith("weight"):
            # q segments re-fuse in rank order; kv replicated when hkv < tp
            qs = torch.cat([p[:nq * d] for p in parts], dim=0)
            assert torch.equal(qs, t[:hq * d])
            for p in parts:  # kv identical (replicated, hkv=2, tp=2 -> 1 each)
                assert p.shape[0] == (nq + 2 * max(hkv // tp, 1)) * d
        elif base == "wo" and name.endswith("weight"):
            assert torch.equal(torch.cat(parts, dim=1), t)
        elif base == "w_gateup" and name.endswith("weight"):
            f = cfg.intermediate_size
            per = f // tp
            gate = torch.cat([p[:per] for p in parts], dim=0)
            up = torch.cat([p[per:] for p in parts], dim=0)
            assert torch.equal(torch.cat([gate, up], dim=0), t)
        elif base == "w_down" and name.endswith("weight"):
            assert torch.equal(torch.cat(parts, dim=1), t)
        else:
            for p in parts:
                assert torch.equal(p, t), name


def test_tp_forces_eager_decode(monkeypatch):
    """With tp_size>1 the engine must NOT take the hipGraph path (RCCL
    collectives inside stream capture are unvalidated on ROCm) — the
    guard is in Engine._decode."""
    import torch
    from substratus_amd.parallel import comm
    from substratus_amd.serve.engine import Engine, EngineConfig
    eng = Engine(EngineConfig(model="llama-tiny", dtype=torch.float32,
                              max_batch=2, max_seq_len=64, kv_blocks=16,
                              use_graphs=True), device="cpu")
    eng.add_request([1, 2, 3], max_new_tokens=3)
    # pretend we're TP rank 0 of 2 (no process group: stub the collective)
    monkeypatch.setattr(comm, "_STATE",
                        comm.ParallelState(initialized=True, tp_size=2))
    from substratus_amd.parallel import tp as tpmod
    monkeypatch.setattr(tpmod, "tp_all_reduce_", lambda t: t)
    called = {"graph": False}
    monkeypatch.setattr(eng, "_decode_graph",
                        lambda: called.__setitem__("graph", True) or [])
    while eng.has_work():
        eng.step()
    assert called["graph"] is False
