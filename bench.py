#!/usr/bin/env python3
"""Flagship benchmark: llama2-7b bf16 serving throughput (served tokens/sec).

This measures BASELINE.json's headline metric — served tokens/sec for the
llama2-7b Server config — on synthetic prompts and random-init weights (no
network for datasets/checkpoints). One step = one continuous-batching decode
iteration over the whole running batch, every op on the hand-written CDNA4
HIP path.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for N>1
launched via torch.distributed.run with one rank per GPU over RCCL. Each rank
runs an independent engine replica (weak scaling, data-parallel serving —
replicated Servers are the reference's scale-out shape; TP=8 for llama2-70b is
exercised separately). Prefill + W decode steps are untimed warmup; exactly K
decode steps are timed between barrier+synchronize pairs; the slowest rank's
time is used; rank 0 prints ONE JSON line.

Also supports `--mode train` (fine-tune tokens/s, falcon-7b-instruct config —
BASELINE config 3) once the trainer lands.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=64)
    p.add_argument("--warmup", type=int, default=16)
    p.add_argument("--mode", choices=["serve", "train"], default="serve")
    p.add_argument("--model", default=None, help="override model config")
    p.add_argument("--batch", type=int, default=512)
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--micro-batch", type=int, default=16,
                   help="per-GPU micro batch (train mode; 16 is the "
                        "measured 1xMI355X sweet spot — 20.5k tok/s vs "
                        "18.3k at mb4; mb32 exceeds 288 GB without "
                        "gradient checkpointing)")
    p.add_argument("--seq-len", type=int, default=None)
    p.add_argument("--prefill-chunk", type=int, default=None,
                   help="max packed prefill tokens per step (default: "
                        "batch*prompt_len, one chunk; cap it for models "
                        "whose prefill activations would not fit beside "
                        "weights+KV, e.g. 32768 for llama2-70b bf16)")
    p.add_argument("--gpu-mem-fraction", type=float, default=None,
                   help="KV pool share of free HBM after weights (engine "
                        "default 0.85; raise toward 0.92 for max-batch "
                        "experiments — llama2-7b batch>512 needs it)")
    p.add_argument("--kv-dtype", default=None, choices=[None, "fp8"],
                   help="KV-cache dtype override (fp8 e4m3; reduced-precision"
                        " secondary config, NOT the headline)")
    p.add_argument("--quant", default=None,
                   help="GGUF weight quantization: q4_0 | q8_0 (config 4)")
    p.add_argument("--tp", type=int, default=1,
                   help="tensor-parallel degree (serve); default DP replicas")
    p.add_argument("--profile", default=None, metavar="PATH",
                   help="profile the timed region with torch.profiler and "
                        "write a kernel table + chrome trace to PATH "
                        "(SURVEY.md §5: bake a --profile mode into the data "
                        "plane; complements rocprofv3)")
    return p.parse_args()


def profiled(path):
    """Context wrapping the timed loop in torch.profiler when --profile is
    set (no-op otherwise). Writes PATH.txt (kernel table by device time) and
    PATH.json (chrome trace) on exit — rank 0 only."""
    import contextlib
    if path is None or int(os.environ.get("RANK", "0")) != 0:
        return contextlib.nullcontext()

    @contextlib.contextmanager
    def _ctx():
        acts = [torch.profiler.ProfilerActivity.CPU]
        if torch.cuda.is_available():
            acts.append(torch.profiler.ProfilerActivity.CUDA)
        with torch.profiler.profile(activities=acts) as prof:
            yield
        table = prof.key_averages().table(
            sort_by="self_cuda_time_total" if torch.cuda.is_available()
            else "self_cpu_time_total", row_limit=40)
        with open(path + ".txt", "w") as f:
            f.write(table)
        prof.export_chrome_trace(path + ".json")
        print(f"[bench] profile written to {path}.txt / {path}.json",
              flush=True)
    return _ctx()


def log(msg: str) -> None:
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, flush=True)


def bench_serve(args) -> None:
    from substratus_amd.parallel.comm import init_distributed, state, barrier
    import torch.distributed as dist
    from substratus_amd.serve.engine import Engine, EngineConfig

    st = init_distributed(tp_size=args.tp)
    n_gpus = max(st.world_size, 1)
    model_name = args.model or "llama2-7b"
    on_gpu = torch.cuda.is_available()
    device = f"cuda:{st.local_rank}" if on_gpu else "cpu"
    dtype = torch.bfloat16 if on_gpu else torch.float32

    max_seq = args.prompt_len + args.warmup + args.steps + 64
    cfg = EngineConfig(model=model_name, dtype=dtype, max_batch=args.batch,
                       max_seq_len=max_seq,
                       prefill_chunk=(args.prefill_chunk or
                                      args.batch * args.prompt_len),
                       quant=args.quant,
                       kv_dtype=args.kv_dtype,
                       **({"gpu_mem_fraction": args.gpu_mem_fraction}
                          if args.gpu_mem_fraction else {}),
                       seed=1234 + st.dp_rank)
    log(f"[bench] building {model_name} engine on {device} ...")
    t_build = time.time()
    eng = Engine(cfg, device=device)
    log(f"[bench] engine ready in {time.time()-t_build:.1f}s; "
        f"kv_blocks={eng.kv.num_blocks}")

    g = torch.Generator().manual_seed(7 + st.rank)
    vocab = eng.mcfg.vocab_size
    for _ in range(args.batch):
        prompt = torch.randint(0, vocab, (args.prompt_len,), generator=g).tolist()
        # never finishes inside the timed region (+63 headroom < max_seq)
        eng.add_request(prompt, max_new_tokens=args.warmup + args.steps + 32,
                        temperature=0.0)

    # prefill all requests (untimed region, but measured: prefill
    # throughput + TTFT — time to the first sampled token of the first
    # admitted chunk — go into config for the multi-context table)
    if on_gpu:
        torch.cuda.synchronize()
    t_pref = time.time()
    t_first = None
    stuck = 0
    while eng.waiting:
        n_wait = len(eng.waiting)
        outs = eng.step()
        # a too-small KV pool can preempt-admit-preempt forever: fail
        # loudly instead of spinning until an external timeout
        stuck = stuck + 1 if len(eng.waiting) >= n_wait else 0
        assert stuck < 2000, (
            f"prefill stalled: {len(eng.waiting)} waiting / "
            f"{len(eng.running)} running, kv free "
            f"{eng.kv.allocator.free_count}/{eng.kv.num_blocks} — "
            f"batch x max_seq_len exceeds the KV pool; lower --batch")
        if outs and t_first is None:
            if on_gpu:
                torch.cuda.synchronize()
            t_first = time.time() - t_pref
    if on_gpu:
        torch.cuda.synchronize()
    prefill_s = time.time() - t_pref
    prefill_tok_s = args.batch * args.prompt_len / prefill_s
    log(f"[bench] prefill of {args.batch}x{args.prompt_len} tokens: "
        f"{prefill_s:.2f}s ({prefill_tok_s/1e3:.1f}k tok/s, "
        f"ttft {t_first*1e3:.0f} ms)")
    assert len(eng.running) == args.batch, (
        f"only {len(eng.running)}/{args.batch} sequences running after "
        f"prefill (kv_blocks={eng.kv.num_blocks}, "
        f"free={eng.kv.allocator.free_count}); the KV pool cannot hold "
        f"batch x max_seq_len at this model size — lower --batch")
    for _ in range(args.warmup):
        eng.decode_only_step()

    if on_gpu:
        torch.cuda.synchronize()
    barrier()
    t0 = time.time()
    with profiled(args.profile):
        for _ in range(args.steps):
            eng.decode_only_step()
        if on_gpu:
            torch.cuda.synchronize()
    barrier()
    elapsed = time.time() - t0
    # max over ranks
    if st.initialized:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if on_gpu:
            t = t.to(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    assert len(eng.running) == args.batch, "sequences finished mid-bench"
    # TP ranks serve ONE shared batch; DP replicas each serve their own
    n_replicas = n_gpus // max(args.tp, 1)
    tokens = args.batch * args.steps * n_replicas
    value = tokens / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    parallelism = (f"tp{args.tp}" if args.tp > 1 else f"dp{n_gpus}")
    result = {
        "metric": "served_tokens_per_s",
        "value": round(value, 2),
        "unit": "tokens/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,   # reference publishes no numbers
        "dtype": "bf16" if on_gpu else "fp32",
        "data": "synthetic",
        "config": {
            "model": model_name,
            "global_batch": args.batch * n_replicas,
            "seq_len": args.prompt_len + args.warmup + args.steps,
            "prompt_len": args.prompt_len,
            "parallelism": parallelism,
            "p50_decode_latency_ms": round(ms_per_step, 3),
            "prefill_tokens_per_s": round(prefill_tok_s, 1),
            "ttft_ms": round((t_first or 0.0) * 1000, 1),
            **({"weights": args.quant} if args.quant else {}),
            **({"kv_cache": "fp8_e4m3"} if args.kv_dtype == "fp8" else {}),
        },
    }
    if st.rank == 0:
        print(json.dumps(result), flush=True)


def bench_train(args) -> None:
    from substratus_amd.train.bench import run_train_bench
    run_train_bench(args)


def main() -> None:
    args = parse_args()
    if args.mode == "serve":
        bench_serve(args)
    else:
        bench_train(args)


if __name__ == "__main__":
    main()
