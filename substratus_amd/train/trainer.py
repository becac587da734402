"""Trainer — replaces the reference's external
`substratusai/model-trainer-huggingface` image.

Container contract (reference docs/container-contract.md + examples/
llama2-7b/finetuned-model.yaml:10-21): base model RO at /content/model,
dataset RO at /content/data, artifacts RW at /content/artifacts; params
(num_train_epochs, save_steps, prompt_template, lr, ...) arrive via
/content/params.json / PARAM_* env. Writes checkpoint-{step}/ directories
into artifacts and resumes from the newest on restart; writes completed.json
when done (docs/design.md "Reconcile Logic").

Compute path: dense differentiable forward (torch autograd + SDPA on ROCm),
loss + dlogits from the fused cross-entropy HIP kernel (backward starts from
dlogits — no autograd through the loss), first-party DDP gradient buckets
over RCCL, fused AdamW HIP kernel for the update.
"""

from __future__ import annotations

import json
import math
import os
import time
from dataclasses import dataclass, field
from typing import Optional

import torch

from .. import ops
from ..models.config import get_config
from ..models.model import CausalLM
from ..parallel.comm import barrier, state
from .ddp import GradBuckets, ZeroShardedDDP


@dataclass
class TrainConfig:
    model: str = "falcon-7b-instruct"
    lr: float = 2e-5
    beta1: float = 0.9
    beta2: float = 0.95
    eps: float = 1e-8
    weight_decay: float = 0.0
    micro_batch: int = 4
    seq_len: int = 2048
    grad_accum: int = 1
    num_train_epochs: int = 1
    max_steps: int = 0          # 0 = derive from epochs × data
    save_steps: int = 0         # 0 = only final checkpoint
    artifacts_dir: str = "/content/artifacts"
    resume: bool = True
    gradient_checkpointing: bool = False   # recompute activations in backward
    seed: int = 0
    dtype: torch.dtype = torch.bfloat16
    grad_clip: float = 1.0
    # LR schedule (HF TrainingArguments parity: warmup_steps +
    # lr_scheduler_type): linear warmup to lr, then constant | cosine |
    # linear decay toward min_lr over max_steps (or endlessly constant
    # when max_steps is 0)
    warmup_steps: int = 0
    lr_schedule: str = "constant"   # constant | cosine | linear
    min_lr: float = 0.0
    # ZeRO-1 sharded optimizer (reduce-scatter grads, shard AdamW,
    # all-gather params) whenever dp_size > 1; False = plain bucketed
    # all-reduce DDP (train/ddp.py GradBuckets)
    zero: bool = True


class AdamWState:
    """fp32 m/v per parameter, driven through the fused HIP kernel."""

    def __init__(self, params: list[torch.Tensor]):
        self.params = params
        self.m = [torch.zeros(p.shape, dtype=torch.float32, device=p.device)
                  for p in params]
        self.v = [torch.zeros_like(m) for m in self.m]
        self.step_num = 0

    def step(self, cfg: TrainConfig, grad_scale: float = 1.0,
             lr: Optional[float] = None) -> None:
        self.step_num += 1
        lr = cfg.lr if lr is None else lr
        for p, m, v in zip(self.params, self.m, self.v):
            if p.grad is None:
                continue
            ops.adamw_step(p.data, p.grad, m, v, lr, cfg.beta1, cfg.beta2,
                           cfg.eps, cfg.weight_decay, self.step_num,
                           grad_scale)

    def zero_grad(self) -> None:
        for p in self.params:
            p.grad = None

    def state_dict(self) -> dict:
        return {"step": self.step_num,
                "m": [t.cpu() for t in self.m],
                "v": [t.cpu() for t in self.v]}

    def load_state_dict(self, sd: dict, device) -> None:
        self.step_num = sd["step"]
        for dst, src in zip(self.m, sd["m"]):
            dst.copy_(src.to(device))
        for dst, src in zip(self.v, sd["v"]):
            dst.copy_(src.to(device))


class Trainer:
    def __init__(self, cfg: TrainConfig, device: Optional[str] = None,
                 model: Optional[CausalLM] = None):
        self.cfg = cfg
        self.st = state()
        if device is None:
            device = (f"cuda:{self.st.local_rank}"
                      if torch.cuda.is_available() else "cpu")
        self.device = torch.device(device)
        if self.device.type == "cuda":
            from ..utils.tunable import load_tunables
            load_tunables()     # pre-tuned GEMM solutions (no-op sans table)
        self.mcfg = get_config(cfg.model)
        if model is None:
            model = CausalLM(self.mcfg, device="meta", dtype=cfg.dtype)
            model = model.to_empty(device=self.device)
            model.init_random(cfg.seed)  # all DP ranks: same seed = same init
        self.model = model.train()
        self.model.gradient_checkpointing = cfg.gradient_checkpointing
        params = [p for p in self.model.parameters()]
        for p in params:
            p.requires_grad_(True)
        self.use_zero = bool(cfg.zero) and self.st.dp_size > 1
        if self.use_zero:
            self.zero = ZeroShardedDDP(params)
            self.opt = AdamWState([])   # kept for interface; state in zero
            self.buckets = None
        else:
            self.zero = None
            self.opt = AdamWState(params)
            self.buckets = GradBuckets(params)
        self.global_step = 0
        self.tokens_seen = 0

    def lr_at(self, step: int) -> float:
        """Scheduled LR for 1-based step `step`."""
        cfg = self.cfg
        if cfg.warmup_steps > 0 and step <= cfg.warmup_steps:
            return cfg.lr * step / cfg.warmup_steps
        if cfg.lr_schedule == "constant" or cfg.max_steps <= 0:
            return cfg.lr
        total = max(cfg.max_steps - cfg.warmup_steps, 1)
        t = min(max(step - cfg.warmup_steps, 0) / total, 1.0)
        if cfg.lr_schedule == "cosine":
            return cfg.min_lr + (cfg.lr - cfg.min_lr) * 0.5 * (
                1.0 + math.cos(math.pi * t))
        if cfg.lr_schedule == "linear":
            return cfg.min_lr + (cfg.lr - cfg.min_lr) * (1.0 - t)
        raise ValueError(f"unknown lr_schedule {cfg.lr_schedule!r}")

    # ------------------------------------------------------------- one step
    def train_step(self, tokens: torch.Tensor) -> float:
        """tokens: [micro_batch, seq_len+1] int64 (inputs ‖ shifted targets).
        Returns the scalar loss (mean over accumulation chunks).

        cfg.grad_accum > 1 (HF gradient_accumulation_steps): the batch is
        split into that many forward/backward chunks whose gradients
        accumulate locally (the DDP hooks hold their reduce until the
        final chunk), then one optimizer step runs."""
        accum = max(int(self.cfg.grad_accum), 1)
        chunks = tokens.chunk(accum) if accum > 1 else [tokens]
        ddp = self.zero if self.use_zero else self.buckets
        losses = []
        for ci, chunk in enumerate(chunks):
            if ddp is not None:
                ddp.accumulating = ci < len(chunks) - 1
            inputs = chunk[:, :-1].to(self.device)
            targets = chunk[:, 1:].to(self.device).reshape(-1)
            logits = self.model.forward_train(inputs)
            flat = logits.reshape(-1, logits.shape[-1])
            loss, dlogits = ops.cross_entropy_fwd_bwd(flat.detach(), targets)
            flat.backward(gradient=dlogits / len(chunks))
            losses.append(float(loss))
        loss = sum(losses) / len(losses)
        lr = self.lr_at(self.global_step + 1)
        if self.use_zero:
            self.zero.finalize()
            gscale = 1.0
            if self.cfg.grad_clip > 0:
                total = self.zero.global_grad_norm()
                gscale = min(1.0, self.cfg.grad_clip / (total + 1e-6))
            self.zero.step(lr, self.cfg.beta1, self.cfg.beta2,
                           self.cfg.eps, self.cfg.weight_decay, gscale)
            self.zero.zero_grad()
        else:
            self.buckets.finalize()
            gscale = 1.0
            if self.cfg.grad_clip > 0:
                # norm only; the clip SCALE is folded into the fused AdamW
                # kernel's gradient read (saves a foreach_mul pass over
                # every grad — ~10 ms/step at falcon-7b scale)
                grads = [p.grad for p in self.opt.params
                         if p.grad is not None]
                norms = torch._foreach_norm(grads, 2.0)
                total = torch.linalg.vector_norm(torch.stack(norms))
                gscale = min(1.0, self.cfg.grad_clip / (float(total) + 1e-6))
            self.opt.step(self.cfg, grad_scale=gscale, lr=lr)
            self.opt.zero_grad()
        self.global_step += 1
        self.tokens_seen += (tokens.shape[0] * (tokens.shape[1] - 1) *
                             max(self.st.dp_size, 1))
        return float(loss)

    # ---------------------------------------------------------- checkpoints
    def save_checkpoint(self, step: Optional[int] = None) -> str:
        step = step if step is not None else self.global_step
        out = os.path.join(self.cfg.artifacts_dir, f"checkpoint-{step}")
        os.makedirs(out, exist_ok=True)
        if self.use_zero:
            # each DP rank owns 1/world of the optimizer state — every
            # rank persists its own shard (resume requires the same world)
            torch.save(self.zero.state_dict(),
                       os.path.join(out,
                                    f"optimizer-rank{self.st.dp_rank}.pt"))
        if self.st.rank == 0:
            from safetensors.torch import save_file
            tensors = {k: v.detach().cpu().contiguous()
                       for k, v in self.model.state_dict().items()}
            save_file(tensors, os.path.join(out, "model.safetensors"))
            if not self.use_zero:
                torch.save(self.opt.state_dict(),
                           os.path.join(out, "optimizer.pt"))
            with open(os.path.join(out, "trainer_state.json"), "w") as f:
                json.dump({"global_step": step,
                           "tokens_seen": self.tokens_seen,
                           "model": self.cfg.model}, f)
        barrier()
        return out

    def maybe_resume(self) -> int:
        """Resume from the newest checkpoint-{N} under artifacts_dir
        (reference behavior: training-level resume is the trainer's job —
        SURVEY.md §5 'Checkpoint / resume')."""
        if not self.cfg.resume or not os.path.isdir(self.cfg.artifacts_dir):
            return 0
        ckpts = []
        for d in os.listdir(self.cfg.artifacts_dir):
            if d.startswith("checkpoint-"):
                # trainer_state.json is written LAST by save_checkpoint, so
                # its presence is the completion marker: a crash mid-save
                # leaves a partial dir that resume must skip (falling back
                # to the previous complete checkpoint)
                if not os.path.exists(os.path.join(
                        self.cfg.artifacts_dir, d, "trainer_state.json")):
                    continue
                try:
                    ckpts.append((int(d.split("-")[1]), d))
                except ValueError:
                    pass
        if not ckpts:
            return 0
        _, newest = max(ckpts)
        path = os.path.join(self.cfg.artifacts_dir, newest)
        from safetensors.torch import load_file
        sd = load_file(os.path.join(path, "model.safetensors"))
        self.model.load_state_dict(
            {k: v.to(self.device) for k, v in sd.items()})
        if self.use_zero:
            opt_path = os.path.join(
                path, f"optimizer-rank{self.st.dp_rank}.pt")
            if os.path.exists(opt_path):
                self.zero.load_state_dict(torch.load(opt_path), self.device)
        else:
            opt_path = os.path.join(path, "optimizer.pt")
            if os.path.exists(opt_path):
                self.opt.load_state_dict(torch.load(opt_path), self.device)
        with open(os.path.join(path, "trainer_state.json")) as f:
            st = json.load(f)
        self.global_step = st["global_step"]
        self.tokens_seen = st.get("tokens_seen", 0)
        return self.global_step

    def finish(self) -> None:
        self.save_checkpoint()
        if self.st.rank == 0:
            with open(os.path.join(self.cfg.artifacts_dir,
                                   "completed.json"), "w") as f:
                json.dump({"completed": True, "global_step": self.global_step,
                           "ts": time.time()}, f)
        barrier()
