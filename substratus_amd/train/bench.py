"""Fine-tune throughput bench (BASELINE config 3: falcon-7b-instruct on
MI355X; metric fine-tune tokens/sec aggregated over GPUs, weak scaling —
fixed per-GPU micro-batch, DDP over RCCL)."""

from __future__ import annotations

import json
import os
import time

import torch
import torch.distributed as dist

from ..parallel.comm import barrier, init_distributed, state
from .data import SyntheticDataset
from .trainer import TrainConfig, Trainer


def run_train_bench(args) -> None:
    st = init_distributed(tp_size=1)  # trainer is pure DP
    n_gpus = max(st.world_size, 1)
    on_gpu = torch.cuda.is_available()
    model_name = args.model or "falcon-7b-instruct"
    micro_batch = getattr(args, "micro_batch", 4)
    seq_len = args.seq_len or 2048
    cfg = TrainConfig(model=model_name, micro_batch=micro_batch,
                      seq_len=seq_len,
                      dtype=torch.bfloat16 if on_gpu else torch.float32,
                      artifacts_dir="/tmp/substratus-bench-artifacts",
                      resume=False, save_steps=0)
    if st.rank == 0:
        print(f"[bench] building {model_name} trainer ...", flush=True)
    trainer = Trainer(cfg)
    data = SyntheticDataset(trainer.mcfg.vocab_size, seq_len)
    it = data.batches(micro_batch, st.dp_rank, max(st.dp_size, 1))

    for _ in range(args.warmup):
        trainer.train_step(next(it))
    if on_gpu:
        torch.cuda.synchronize()
    barrier()
    t0 = time.time()
    try:
        from bench import profiled  # repo-root bench.py (the CLI entry)
    except ImportError:  # imported outside the repo root
        import contextlib
        def profiled(path):
            return contextlib.nullcontext()
    with profiled(getattr(args, "profile", None)):
        for _ in range(args.steps):
            loss = trainer.train_step(next(it))
        if on_gpu:
            torch.cuda.synchronize()
    barrier()
    elapsed = time.time() - t0
    if st.initialized:
        t = torch.tensor([elapsed], dtype=torch.float64)
        if on_gpu:
            t = t.to(trainer.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
    tokens = micro_batch * seq_len * args.steps * n_gpus
    value = tokens / elapsed
    result = {
        "metric": "fine_tune_tokens_per_s",
        "value": round(value, 2),
        "unit": "tokens/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1000.0, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16" if on_gpu else "fp32",
        "data": "synthetic",
        "config": {
            "model": model_name,
            "global_batch": micro_batch * n_gpus,
            "seq_len": seq_len,
            "parallelism": f"dp{n_gpus}",
            "final_loss": round(float(loss), 4),
        },
    }
    if st.rank == 0:
        print(json.dumps(result), flush=True)
