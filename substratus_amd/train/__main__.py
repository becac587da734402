"""Trainer container entrypoint — replaces
`substratusai/model-trainer-huggingface`.

Reads the container contract mounts (/content/model RO, /content/data RO,
/content/artifacts RW) and params (num_train_epochs, save_steps, lr,
prompt_template, micro_batch, seq_len — reference
examples/falcon-7b-instruct/finetuned-model.yaml,
examples/llama2-7b/finetuned-model.yaml:10-17). Multi-GPU: launched under
torchrun with one rank per GPU of the Pod's amd.com/gpu count (DDP over
RCCL). Resumes from the newest checkpoint; writes completed.json.
"""

from __future__ import annotations

import json
import os

import torch

from ..loader.model_loader import load_params
from ..parallel.comm import init_distributed, state
from ..serve.server import resolve_model_name
from .data import JsonlDataset, SyntheticDataset
from .trainer import TrainConfig, Trainer


def main() -> None:
    init_distributed(tp_size=1)
    st = state()
    params = load_params()
    model_name = resolve_model_name(params)
    artifacts = os.environ.get("ARTIFACTS_DIR", "/content/artifacts")
    data_dir = os.environ.get("DATA_DIR", "/content/data")
    model_dir = os.environ.get("MODEL_DIR", "/content/model")
    on_gpu = torch.cuda.is_available()
    cfg = TrainConfig(
        model=model_name,
        lr=float(params.get("lr", params.get("learning_rate", 2e-5))),
        warmup_steps=int(params.get("warmup_steps", 0)),
        lr_schedule=str(params.get("lr_scheduler_type",
                                   params.get("lr_schedule", "constant"))),
        min_lr=float(params.get("min_lr", 0.0)),
        micro_batch=int(params.get("micro_batch", 4)),
        grad_accum=int(params.get("gradient_accumulation_steps",
                                  params.get("grad_accum", 1))),
        seq_len=int(params.get("seq_len", params.get("max_length", 2048))),
        num_train_epochs=int(params.get("num_train_epochs", 1)),
        save_steps=int(params.get("save_steps", 0)),
        gradient_checkpointing=bool(int(params.get(
            "gradient_checkpointing", 0))),
        artifacts_dir=artifacts,
        dtype=torch.bfloat16 if on_gpu else torch.float32,
    )
    trainer = Trainer(cfg)
    # base weights from /content/model if the loader provided them
    base = os.path.join(model_dir, "model.safetensors")
    if os.path.exists(base):
        from safetensors.torch import load_file
        sd = load_file(base)
        trainer.model.load_state_dict(
            {k: v.to(trainer.device, cfg.dtype) for k, v in sd.items()})
    resumed = trainer.maybe_resume()
    if resumed and st.rank == 0:
        print(f"resumed from step {resumed}")

    if os.path.isdir(data_dir) and any(
            f.endswith(".jsonl") for f in os.listdir(data_dir)):
        ds = JsonlDataset(data_dir, trainer.mcfg.vocab_size, cfg.seq_len,
                          params.get("prompt_template",
                                     "{prompt}\n{completion}"))
        per_epoch = None
    else:
        ds = SyntheticDataset(trainer.mcfg.vocab_size, cfg.seq_len)
        per_epoch = int(params.get("steps_per_epoch", 16))

    step = trainer.global_step
    for epoch in range(cfg.num_train_epochs):
        if isinstance(ds, SyntheticDataset):
            it = ds.batches(cfg.micro_batch, st.dp_rank, max(st.dp_size, 1))
            batches = (next(it) for _ in range(per_epoch))
        else:
            batches = ds.batches(cfg.micro_batch, st.dp_rank,
                                 max(st.dp_size, 1), seed=epoch)
        for batch in batches:
            if cfg.max_steps and step >= cfg.max_steps:
                break
            loss = trainer.train_step(batch)
            step = trainer.global_step
            if st.rank == 0 and step % 10 == 0:
                print(f"step {step} loss {loss:.4f}", flush=True)
            if cfg.save_steps and step % cfg.save_steps == 0:
                trainer.save_checkpoint()
    trainer.finish()
    if st.rank == 0:
        print(json.dumps({"final_step": trainer.global_step,
                          "tokens_seen": trainer.tokens_seen}))


if __name__ == "__main__":
    main()
