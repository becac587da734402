"""HTTP inference server — replaces the reference's external
`substratusai/model-server-basaran` image.

Container contract (reference docs/container-contract.md:50-55 +
test/system.sh:73-77): listens on 8080; readiness = GET / → 200; OpenAI-style
POST /v1/completions with {prompt, max_tokens, temperature, top_p, top_k,
stream}. Loads the model from /content/model (loader artifacts: config.json +
model.safetensors) or from PARAM_NAME with random-init weights.

Observability: /metrics (Prometheus) exposes served tokens,
request latency p50/p95, and per-GPU HBM usage — the metrics the reference's
control plane never had from its external images.

Engine loop runs in a background thread; requests enqueue into the continuous
batcher and stream tokens back through per-request queues.
"""

from __future__ import annotations

import json
import os
import queue
import threading
import time
from typing import Optional

import torch
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, StreamingResponse

from . import tokenizer as tokenizer_mod
from .engine import Engine, EngineConfig, StepOutput


def load_params() -> dict:
    params = {}
    path = os.environ.get("PARAMS_PATH", "/content/params.json")
    if os.path.exists(path):
        with open(path) as f:
            params.update(json.load(f))
    for k, v in os.environ.items():
        if k.startswith("PARAM_"):
            params[k[len("PARAM_"):].lower()] = v
    return params


def resolve_model_name(params: dict, model_dir: str | None = None) -> str:
    model_dir = model_dir or os.environ.get("MODEL_DIR", "/content/model")
    cfg_path = os.path.join(model_dir, "config.json")
    if os.path.exists(cfg_path):
        with open(cfg_path) as f:
            return json.load(f)["model"]
    return params.get("name", params.get("model", "llama2-7b"))


class InferenceWorker:
    """Owns the engine; steps continuously while there is work. Under TP
    (torchrun, tp_size>1) the worker is the scheduling leader: it broadcasts
    each iteration's admissions to follower ranks before stepping
    (serve/tp.py)."""

    def __init__(self, engine: Engine):
        from .tp import TPCoordinator
        self.coord = TPCoordinator()
        self.engine = engine
        self.inbox: queue.Queue = queue.Queue()
        self.abort_inbox: queue.Queue = queue.Queue()
        self.outputs: dict[int, queue.Queue] = {}
        self.lock = threading.Lock()
        self.total_tokens = 0
        self.latencies: list[float] = []
        self._stop = False
        self.thread = threading.Thread(target=self._run, daemon=True)
        self.thread.start()

    def submit(self, prompt_tokens, max_new_tokens, temperature, top_k,
               top_p, eos_token=-1) -> tuple[int, queue.Queue]:
        out_q: queue.Queue = queue.Queue()
        done = threading.Event()
        holder = {}
        self.inbox.put((prompt_tokens, max_new_tokens, temperature, top_k,
                        top_p, eos_token, out_q, holder, done))
        done.wait()
        return holder["sid"], out_q

    def cancel(self, sid: int) -> None:
        """Drop a request whose client went away. Queued to the engine
        thread; a no-op if the sequence already finished. cancel() can
        only be called after submit() returned sid, so on TP followers
        the add always lands in an earlier publish than the abort."""
        self.abort_inbox.put(sid)

    def _run(self) -> None:
        while not self._stop:
            moved = False
            pending = []
            while True:
                try:
                    pending.append(self.inbox.get_nowait())
                except queue.Empty:
                    break
            abort_ids = []
            while True:
                try:
                    abort_ids.append(self.abort_inbox.get_nowait())
                except queue.Empty:
                    break
            if self.coord.active and (pending or abort_ids or
                                      self.engine.has_work()):
                # lockstep TP: publish this iteration's admissions first
                self.coord.publish([
                    dict(prompt_tokens=pt, max_new_tokens=mnt,
                         temperature=temp, top_k=tk, top_p=tp, eos_token=eos)
                    for (pt, mnt, temp, tk, tp, eos, *_rest) in pending],
                    aborts=abort_ids)
            for (pt, mnt, temp, tk, tp, eos, out_q, holder, done) in pending:
                sid = self.engine.add_request(pt, max_new_tokens=mnt,
                                              temperature=temp, top_k=tk,
                                              top_p=tp, eos_token=eos)
                with self.lock:
                    self.outputs[sid] = out_q
                holder["sid"] = sid
                done.set()
                moved = True
            for sid in abort_ids:     # after adds: same order as followers
                self.engine.abort(sid)
                with self.lock:
                    self.outputs.pop(sid, None)
                moved = True
            if self.engine.has_work():
                t0 = time.time()
                try:
                    outs = self.engine.step()
                except Exception as e:  # noqa: BLE001
                    # whole-engine failure (per-request problems like an
                    # unfittable prompt are returned as error StepOutputs by
                    # engine.step, not raised): fail all in-flight requests
                    # WITH an error marker — handlers answer 5xx, never a
                    # fake completion — and keep the worker thread serving
                    import sys as _sys
                    print(f"[server] engine.step failed: {e}",
                          file=_sys.stderr, flush=True)
                    with self.lock:
                        for sid, q_ in list(self.outputs.items()):
                            q_.put(StepOutput(sid, 0, True,
                                              error=f"engine failure: {e}"))
                            self.outputs.pop(sid, None)
                    self.engine.abort_all()
                    continue
                dt = time.time() - t0
                with self.lock:
                    self.latencies.append(dt)
                    if len(self.latencies) > 1000:
                        self.latencies = self.latencies[-1000:]
                    for o in outs:
                        self.total_tokens += 1
                        q_ = self.outputs.get(o.seq_id)
                        if q_ is not None:
                            q_.put(o)
                            if o.finished:
                                self.outputs.pop(o.seq_id, None)
                moved = True
            if not moved:
                time.sleep(0.002)

    def stop(self) -> None:
        self._stop = True


def build_engine_config(params: dict,
                        model_name: Optional[str] = None
                        ) -> tuple[EngineConfig, Optional[str]]:
    """Resolve the serving configuration (model, dtype, quant, GGUF path)
    from params/env. Under TP only the leader runs this; followers receive
    the result over the wire so every rank's engine is identical."""
    name = model_name or resolve_model_name(params)
    on_gpu = torch.cuda.is_available()
    quant = None
    # reference knob: MODEL_LOAD_IN_8BIT / _4BIT env
    # (examples/llama2-70b/server.yaml)
    if params.get("load_in_4bit") or os.environ.get("MODEL_LOAD_IN_4BIT"):
        quant = "q4_0"
    if params.get("load_in_8bit") or os.environ.get("MODEL_LOAD_IN_8BIT"):
        quant = "q8_0"
    gguf_path = _find_gguf()
    ecfg = EngineConfig(
        model=name,
        dtype=torch.bfloat16 if on_gpu else torch.float32,
        max_batch=int(params.get("max_batch", 64)),
        max_seq_len=int(params.get("max_length", 2048)),
        quant=None if gguf_path else quant,
    )
    return ecfg, gguf_path


def build_engine(ecfg: EngineConfig,
                 gguf_path: Optional[str] = None) -> Engine:
    """Construct the engine + load weights. Identical on every TP rank."""
    engine = Engine(ecfg)
    if gguf_path:
        from ..models.quant import load_gguf_into_model
        from ..parallel.comm import state as pstate
        if pstate().tp_size > 1:
            raise RuntimeError(
                "GGUF-quantized serving is single-rank for now: remove the "
                "*.gguf from the model dir or set gpu.count to 1")
        load_gguf_into_model(engine.model, gguf_path, engine.device,
                             engine.cfg.dtype)
    else:
        _maybe_load_weights(engine)
    return engine


def create_app(engine: Optional[Engine] = None,
               model_name: Optional[str] = None) -> FastAPI:
    params = load_params()
    if engine is None:
        ecfg, gguf_path = build_engine_config(params, model_name)
        engine = build_engine(ecfg, gguf_path)
    worker = InferenceWorker(engine)
    app = FastAPI(title="substratus-amd model server")
    app.state.worker = worker
    app.state.engine = engine
    vocab = engine.mcfg.vocab_size
    model_id = engine.cfg.model
    # real tokenizer.json from the artifacts when present (loader exports
    # one), byte-level fallback otherwise
    tok_obj = tokenizer_mod.load(os.environ.get("MODEL_DIR",
                                                "/content/model"), vocab)

    try:
        from prometheus_client import CONTENT_TYPE_LATEST, generate_latest
        from prometheus_client.core import (CollectorRegistry,
                                            CounterMetricFamily,
                                            GaugeMetricFamily)

        class _ServerCollector:
            """Reads live server state at scrape time — public collector
            API only, and a per-app registry so app re-creation in one
            process (tests, embedded restarts) never collides."""

            def collect(self):
                c = CounterMetricFamily("served_tokens", "tokens generated")
                c.add_metric([], worker.total_tokens)
                yield c
                g = GaugeMetricFamily("hbm_allocated_bytes",
                                      "GPU memory allocated")
                g.add_metric([], torch.cuda.memory_allocated()
                             if torch.cuda.is_available() else 0)
                yield g

        prom_registry = CollectorRegistry()
        prom_registry.register(_ServerCollector())
        app.state.prom_registry = prom_registry
        have_prom = True
    except ImportError:  # pragma: no cover
        have_prom = False

    @app.get("/")
    def root():
        # readiness contract: GET / -> 200
        return {"status": "ok", "model": model_id}

    @app.get("/v1/models")
    def models():
        return {"object": "list",
                "data": [{"id": model_id, "object": "model"}]}

    @app.get("/stats")
    def stats():
        lat = sorted(worker.latencies) or [0.0]
        return {
            "served_tokens_total": worker.total_tokens,
            "p50_step_latency_ms": lat[len(lat) // 2] * 1000,
            "p95_step_latency_ms": lat[int(len(lat) * 0.95)] * 1000,
            "running": len(engine.running),
            "waiting": len(engine.waiting),
            "kv_blocks_free": engine.kv.allocator.free_count,
            "hbm_allocated_bytes": (torch.cuda.memory_allocated()
                                    if torch.cuda.is_available() else 0),
        }

    if have_prom:
        @app.get("/metrics")
        def metrics():
            from fastapi import Response
            return Response(generate_latest(prom_registry),
                            media_type=CONTENT_TYPE_LATEST)

    def _bad_request(msg: str) -> JSONResponse:
        # OpenAI-style error body, 400 instead of a 500 from deep asserts
        return JSONResponse({"error": {"message": msg,
                                       "type": "invalid_request_error"}},
                            status_code=400)

    @app.post("/v1/completions")
    async def completions(req: Request):
        try:
            body = await req.json()
        except Exception:
            return _bad_request("request body is not valid JSON")
        if not isinstance(body, dict):
            return _bad_request("request body must be a JSON object")
        prompt = body.get("prompt", "")
        try:
            if isinstance(prompt, list):
                prompt_tokens = [int(t) for t in prompt]
            else:
                prompt_tokens = tok_obj.encode(str(prompt))
            max_tokens = int(body.get("max_tokens", 16))
            temperature = float(body.get("temperature", 1.0))
            top_p = float(body.get("top_p", 1.0))
            top_k = int(body.get("top_k", 0))
            stream = bool(body.get("stream", False))
        except (TypeError, ValueError) as e:
            return _bad_request(f"invalid parameter: {e}")
        if not prompt_tokens:
            return _bad_request("prompt must not be empty")
        if max_tokens < 1:
            return _bad_request("max_tokens must be >= 1")
        limit = engine.cfg.max_seq_len
        if len(prompt_tokens) + max_tokens > limit:
            return _bad_request(
                f"prompt ({len(prompt_tokens)} tokens) + max_tokens "
                f"({max_tokens}) exceeds the server max_length ({limit})")
        t_start = time.time()
        sid, out_q = worker.submit(prompt_tokens, max_tokens, temperature,
                                   top_k, top_p, eos_token=tok_obj.eos)

        def collect() -> tuple[list[int], Optional[str]]:
            toks = []
            while True:
                o: StepOutput = out_q.get(timeout=600)
                if o.error:
                    return toks, o.error
                toks.append(o.token)
                if o.finished:
                    return toks, None

        if stream:
            def gen():
                # a dropped client closes the generator (GeneratorExit at
                # the yield); the finally cancels the request so it stops
                # occupying batch slots and KV blocks
                from .tokenizer import StreamDecoder
                dec = StreamDecoder(tok_obj)

                def chunk_for(text, done):
                    return {
                        "id": f"cmpl-{sid}", "object": "text_completion",
                        "model": model_id,
                        "choices": [{"index": 0, "text": text,
                                     "finish_reason": "stop" if done
                                     else None}],
                    }
                try:
                    while True:
                        o: StepOutput = out_q.get(timeout=600)
                        if o.error:
                            err = {"error": {"message": o.error,
                                             "type": "server_error"}}
                            yield f"data: {json.dumps(err)}\n\n"
                            yield "data: [DONE]\n\n"
                            return
                        text = dec.push(o.token)
                        if o.finished:
                            text += dec.flush()
                        # one chunk per token (empty delta while a
                        # multi-byte char is still incomplete)
                        yield ("data: " +
                               json.dumps(chunk_for(text, o.finished)) +
                               "\n\n")
                        if o.finished:
                            yield "data: [DONE]\n\n"
                            return
                finally:
                    worker.cancel(sid)    # no-op if already finished
            return StreamingResponse(gen(), media_type="text/event-stream")

        toks, err = collect()
        if err is not None:
            return JSONResponse({"error": {"message": err,
                                           "type": "server_error"}},
                                status_code=500)
        return JSONResponse({
            "id": f"cmpl-{sid}",
            "object": "text_completion",
            "created": int(t_start),
            "model": model_id,
            "choices": [{
                "index": 0,
                "text": tok_obj.decode(toks),
                "finish_reason": "length" if len(toks) >= max_tokens
                                 else "stop",
            }],
            "usage": {
                "prompt_tokens": len(prompt_tokens),
                "completion_tokens": len(toks),
                "total_tokens": len(prompt_tokens) + len(toks),
            },
        })

    @app.post("/v1/chat/completions")
    async def chat_completions(req: Request):
        """OpenAI chat API on the same engine (beyond the reference's
        basaran surface, which was /v1/completions only): messages render
        through a minimal chat template and the completion flow runs
        unchanged."""
        try:
            body = await req.json()
        except Exception:
            return _bad_request("request body is not valid JSON")
        msgs = body.get("messages")
        if not isinstance(msgs, list) or not msgs:
            return _bad_request("messages must be a non-empty list")
        parts = []
        for m_ in msgs:
            if not isinstance(m_, dict) or "content" not in m_:
                return _bad_request("each message needs role and content")
            parts.append(f"{m_.get('role', 'user')}: {m_['content']}")
        parts.append("assistant:")
        prompt = "\n".join(parts)
        prompt_tokens = tok_obj.encode(prompt)
        try:
            max_tokens = int(body.get("max_tokens", 64))
            temperature = float(body.get("temperature", 1.0))
            top_p = float(body.get("top_p", 1.0))
            stream = bool(body.get("stream", False))
        except (TypeError, ValueError) as e:
            return _bad_request(f"invalid parameter: {e}")
        limit = engine.cfg.max_seq_len
        if len(prompt_tokens) + max_tokens > limit:
            return _bad_request("messages exceed the server max_length")
        t_start = time.time()
        sid, out_q = worker.submit(prompt_tokens, max_tokens, temperature,
                                   0, top_p, eos_token=tok_obj.eos)
        if stream:
            def gen():
                from .tokenizer import StreamDecoder
                dec = StreamDecoder(tok_obj)
                try:
                    while True:
                        o: StepOutput = out_q.get(timeout=600)
                        if o.error:
                            yield ("data: " + json.dumps(
                                {"error": {"message": o.error,
                                           "type": "server_error"}}) + "\n\n")
                            yield "data: [DONE]\n\n"
                            return
                        text = dec.push(o.token)
                        if o.finished:
                            text += dec.flush()
                        chunk = {
                            "id": f"chatcmpl-{sid}",
                            "object": "chat.completion.chunk",
                            "model": model_id,
                            "choices": [{"index": 0, "delta": {
                                "content": text},
                                "finish_reason": "stop" if o.finished
                                else None}],
                        }
                        yield f"data: {json.dumps(chunk)}\n\n"
                        if o.finished:
                            yield "data: [DONE]\n\n"
                            return
                finally:
                    worker.cancel(sid)    # no-op if already finished
            return StreamingResponse(gen(), media_type="text/event-stream")
        toks = []
        while True:
            o: StepOutput = out_q.get(timeout=600)
            if o.error:
                return JSONResponse({"error": {"message": o.error,
                                               "type": "server_error"}},
                                    status_code=500)
            toks.append(o.token)
            if o.finished:
                break
        return JSONResponse({
            "id": f"chatcmpl-{sid}",
            "object": "chat.completion",
            "created": int(t_start),
            "model": model_id,
            "choices": [{
                "index": 0,
                "message": {"role": "assistant",
                            "content": tok_obj.decode(toks)},
                "finish_reason": "length" if len(toks) >= max_tokens
                                 else "stop",
            }],
            "usage": {
                "prompt_tokens": len(prompt_tokens),
                "completion_tokens": len(toks),
                "total_tokens": len(prompt_tokens) + len(toks),
            },
        })

    return app


def _find_gguf(model_dir: str | None = None) -> Optional[str]:
    model_dir = model_dir or os.environ.get("MODEL_DIR", "/content/model")
    if not os.path.isdir(model_dir):
        return None
    for f in sorted(os.listdir(model_dir)):
        if f.endswith(".gguf"):
            return os.path.join(model_dir, f)
    return None


def _maybe_load_weights(engine: Engine,
                        model_dir: str | None = None) -> None:
    model_dir = model_dir or os.environ.get("MODEL_DIR", "/content/model")
    path = os.path.join(model_dir, "model.safetensors")
    if not os.path.exists(path):
        # search newest checkpoint (trainer artifacts layout)
        if os.path.isdir(model_dir):
            ckpts = sorted((d for d in os.listdir(model_dir)
                            if d.startswith("checkpoint-")),
                           key=lambda d: int(d.split("-")[1]))
            if ckpts:
                path = os.path.join(model_dir, ckpts[-1], "model.safetensors")
    if os.path.exists(path):
        from safetensors.torch import load_file
        from ..models.hf_compat import (hf_falcon_to_native,
                                        hf_llama_to_native,
                                        hf_opt_to_native,
                                        is_hf_falcon_state_dict,
                                        is_hf_llama_state_dict,
                                        is_hf_opt_state_dict)
        from ..parallel.comm import state as pstate
        sd = load_file(path)
        if is_hf_llama_state_dict(sd):
            # HF-layout checkpoint (meta-llama/Llama-2-* naming)
            sd = hf_llama_to_native(sd, engine.mcfg)
        elif is_hf_falcon_state_dict(sd):
            sd = hf_falcon_to_native(sd, engine.mcfg)
        elif is_hf_opt_state_dict(sd):
            sd = hf_opt_to_native(sd, engine.mcfg)
        st = pstate()
        if st.tp_size > 1:
            from ..models.sharding import shard_state_dict
            sd = shard_state_dict(sd, engine.mcfg, st.tp_rank, st.tp_size)
        engine.model.load_state_dict(
            {k: v.to(engine.device, engine.cfg.dtype) for k, v in sd.items()})


def main() -> None:
    import uvicorn
    from ..parallel.comm import init_distributed
    st = init_distributed()  # tp over all ranks of the Pod (amd.com/gpu: N)
    if st.initialized and st.tp_size > 1:
        # All ranks build their engine from the LEADER's exact resolved
        # configuration (one startup broadcast): a follower that guessed
        # its own quant/GGUF/kv_dtype settings would desync the first
        # collective of the first forward.
        from .tp import TPCoordinator, follower_loop
        coord = TPCoordinator()
        if st.tp_rank == 0:
            ecfg, gguf_path = build_engine_config(load_params())
            coord.broadcast_config([ecfg.to_wire(), gguf_path])
        else:
            wire, gguf_path = coord.broadcast_config()
            ecfg = EngineConfig.from_wire(wire)
        engine = build_engine(ecfg, gguf_path)
        if st.tp_rank != 0:
            follower_loop(engine)
            return
        app = create_app(engine=engine)
    else:
        app = create_app()
    uvicorn.run(app, host="0.0.0.0", port=int(os.environ.get("PORT", 8080)))


if __name__ == "__main__":
    main()
