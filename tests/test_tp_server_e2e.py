"""End-to-end TP=2 serving over gloo: the FULL server path — leader
broadcasts its resolved EngineConfig, followers build identical engines and
mirror the schedule, an HTTP completion round-trips through the leader's
FastAPI app while rank 1 runs follower_loop (VERDICT r01 next-round #1:
TP must be survivable on first contact, proven through the server, not just
the raw engine)."""

import os

import pytest
import torch


def _rank_worker(rank, world, q, tmpdir):
    import traceback
    try:
        _rank_worker_inner(rank, world, q, tmpdir)
    except Exception:
        q.put((f"ERR{rank}", traceback.format_exc()))


def _rank_worker_inner(rank, world, q, tmpdir):
    import torch.distributed as dist
    os.environ.update({"RANK": str(rank), "WORLD_SIZE": str(world),
                       "LOCAL_RANK": str(rank),
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29781",
                       "PARAM_NAME": "llama-tiny",
                       "PARAM_MAX_BATCH": "2", "PARAM_MAX_LENGTH": "64",
                       "MODEL_DIR": tmpdir, "PARAMS_PATH": "/nonexistent"})
    from substratus_amd.parallel import comm
    comm._STATE = comm.ParallelState()
    comm.init_distributed(tp_size=world, backend="gloo")
    from substratus_amd.serve import server as srv
    from substratus_amd.serve.engine import EngineConfig
    from substratus_amd.serve.tp import TPCoordinator, follower_loop
    coord = TPCoordinator()
    if rank == 0:
        ecfg, gguf_path = srv.build_engine_config(srv.load_params())
        coord.broadcast_config([ecfg.to_wire(), gguf_path])
        engine = srv.build_engine(ecfg, gguf_path)
        app = srv.create_app(engine=engine)
        from fastapi.testclient import TestClient
        with TestClient(app) as c:
            r = c.post("/v1/completions", json={
                "prompt": [5, 9, 2, 7], "max_tokens": 4,
                "temperature": 0.0})
            q.put((rank, r.status_code, r.json()))
        coord.publish([], stop=True)
    else:
        wire, gguf_path = coord.broadcast_config()
        ecfg = EngineConfig.from_wire(wire)
        assert ecfg.model == "llama-tiny" and ecfg.max_batch == 2, \
            "follower did not receive the leader's resolved config"
        engine = srv.build_engine(ecfg, gguf_path)
        follower_loop(engine)
        q.put((rank, "follower-done",
               {"steps": engine.steps, "tokens": engine.tokens_generated}))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp2_http_server_end_to_end(tmp_path):
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_worker,
                         args=(r, 2, q, str(tmp_path))) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, status, body = q.get(timeout=240)
        assert not str(rank).startswith("ERR"), status
        results[rank] = (status, body)
    for p in procs:
        p.join(timeout=60)
    status, body = results[0]
    assert status == 200, body
    assert body["usage"]["completion_tokens"] == 4
    # follower engine stepped the same schedule (prefill+3 decode steps,
    # 4 tokens emitted)
    _, fstats = results[1]
    assert fstats["tokens"] == 4, fstats


def _abort_rank_worker(rank, world, q):
    import traceback
    try:
        _abort_rank_inner(rank, world, q)
    except Exception:
        q.put((f"ERR{rank}", traceback.format_exc()))


def _abort_rank_inner(rank, world, q):
    import torch
    import torch.distributed as dist
    os.environ.update({"RANK": str(rank), "WORLD_SIZE": str(world),
                       "LOCAL_RANK": str(rank),
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29783"})
    from substratus_amd.parallel import comm
    comm._STATE = comm.ParallelState()
    comm.init_distributed(tp_size=world, backend="gloo")
    from substratus_amd.serve.engine import Engine, EngineConfig
    from substratus_amd.serve.tp import TPCoordinator, follower_loop
    engine = Engine(EngineConfig(model="llama-tiny", dtype=torch.float32,
                                 max_batch=2, max_seq_len=64, kv_blocks=16),
                    device="cpu")
    coord = TPCoordinator()
    add = dict(prompt_tokens=[3, 1, 4], max_new_tokens=40, temperature=0.0,
               top_k=0, top_p=1.0, eos_token=-1)
    if rank == 0:
        coord.publish([add])
        sid = engine.add_request(**add)
        engine.step()
        for _ in range(3):
            coord.publish([])
            engine.step()
        # abort mid-generation; follower must mirror it or the next
        # collective hangs / the schedules desync
        coord.publish([], aborts=[sid])
        assert engine.abort(sid)
        assert not engine.has_work()
        coord.publish([], stop=True)
    else:
        follower_loop(engine)
    q.put((rank, {"steps": engine.steps,
                  "free": engine.kv.allocator.free_count,
                  "blocks": engine.kv.num_blocks,
                  "has_work": engine.has_work()}))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp2_abort_propagates_to_follower():
    """Per-request abort rides the iteration command: after the leader
    cancels, BOTH ranks drain, free every KV block, and agree on the step
    count (a desync here would hang the next all-reduce)."""
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_abort_rank_worker, args=(r, 2, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, stats = q.get(timeout=240)
        assert not str(rank).startswith("ERR"), stats
        results[rank] = stats
    for p in procs:
        p.join(timeout=60)
    assert results[0]["steps"] == results[1]["steps"] == 4
    for r in (0, 1):
        assert not results[r]["has_work"]
        assert results[r]["free"] == results[r]["blocks"], \
            f"rank {r} leaked KV blocks after abort"


def test_llama2_70b_tp8_shardable():
    """BASELINE config 5 sanity without weights: llama2-70b head counts and
    FFN divide by tp=8, and the bf16 shard + KV fit comfortably in 8×288 GB
    (the HBM sizing admission the controller applies)."""
    from substratus_amd.models.config import get_config
    cfg = get_config("llama2-70b")
    for tp in (2, 4, 8):
        assert cfg.num_heads % tp == 0
        assert cfg.num_kv_heads % tp == 0
        assert cfg.intermediate_size % tp == 0
    # rough param count (weights dominate): embed + layers + head
    p = (cfg.vocab_size * cfg.hidden_size * 2 +
         cfg.num_layers * (
             cfg.hidden_size * cfg.head_dim *
             (cfg.num_heads + 2 * cfg.num_kv_heads) +
             cfg.num_heads * cfg.head_dim * cfg.hidden_size +
             3 * cfg.hidden_size * cfg.intermediate_size))
    bf16_bytes = 2 * p
    assert 120e9 < bf16_bytes < 160e9  # ≈140 GB — needs >1 GPU at bf16
    assert bf16_bytes / 8 < 288e9 * 0.25  # TP=8 shard leaves ≥75% for KV
