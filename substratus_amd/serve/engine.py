"""Continuous-batching generation engine (the Server data plane's core).

Replaces the reference's external `substratusai/model-server-basaran` image
(SURVEY.md §2.5): token-level continuous batching over a paged KV cache, with
prefill admission and per-step decode across all running sequences. On GPU
every decode op is a hand-written CDNA4 HIP kernel; GEMMs go to hipBLASLt via
torch.nn.functional.linear.

Scheduling: each step() admits waiting requests while KV blocks and batch
slots are available (prefilling them in one packed varlen batch), then runs
one decode step for every running sequence. Preemption: when the allocator
runs out of blocks mid-decode, the youngest sequence is evicted back to the
wait queue (its blocks freed, prompt re-prefilled later).
"""

from __future__ import annotations

import itertools
import time
from dataclasses import dataclass, field
from typing import Optional

import torch

from .. import ops
from ..models.config import ModelConfig, get_config
from ..models.model import CausalLM
from ..parallel.comm import state
from .kv_cache import (DEFAULT_BLOCK_SIZE, BlockAllocator, OutOfBlocks,
                       PagedKVCache, SequenceState)


@dataclass
class EngineConfig:
    model: str = "llama2-7b"
    dtype: torch.dtype = torch.bfloat16
    max_batch: int = 64
    max_seq_len: int = 4096
    block_size: int = DEFAULT_BLOCK_SIZE
    kv_blocks: Optional[int] = None        # None = size from gpu_mem_fraction
    gpu_mem_fraction: float = 0.85         # of free HBM after weights
    prefill_chunk: int = 8192              # max packed prefill tokens per step
    use_graphs: bool = True                # hipGraph-captured decode on GPU
    quant: Optional[str] = None            # "q4_0" | "q8_0" GGUF weights
    kv_dtype: Optional[str] = None         # None=model dtype | "fp8" (e4m3)
    seed: int = 0

    def to_wire(self) -> dict:
        """Plain-dict form for broadcast_object_list: the TP leader ships its
        EXACT config to follower ranks at startup so every rank builds an
        identical engine (quant/kv_dtype/graph settings included)."""
        d = self.__dict__.copy()
        d["dtype"] = str(self.dtype).removeprefix("torch.")
        return d

    @classmethod
    def from_wire(cls, d: dict) -> "EngineConfig":
        d = dict(d)
        d["dtype"] = getattr(torch, d["dtype"])
        return cls(**d)


@dataclass(slots=True)
class StepOutput:
    seq_id: int
    token: int
    finished: bool
    text_done: bool = False
    error: Optional[str] = None     # set => request failed, token is invalid


class Engine:
    def __init__(self, cfg: EngineConfig, device: Optional[str] = None,
                 model: Optional[CausalLM] = None):
        self.cfg = cfg
        self.mcfg: ModelConfig = get_config(cfg.model)
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        if self.device.type == "cuda":
            from ..utils.tunable import load_tunables
            load_tunables()     # pre-tuned GEMM solutions (no-op sans table)
        if model is None:
            with torch.no_grad():
                model = CausalLM(self.mcfg, device="meta", dtype=cfg.dtype)
                model = model.to_empty(device=self.device)
                model.init_random(cfg.seed)
        if cfg.quant:
            from ..models.quant import quantize_model
            from ..parallel.comm import state as _pstate
            assert _pstate().tp_size <= 1, \
                "quantized serving is single-rank for now"
            quantize_model(model, cfg.quant)
        self.model = model.eval()
        n_kv_local = self.model.n_kv_heads
        kv_dtype = cfg.dtype
        if cfg.kv_dtype == "fp8":
            assert self.device.type == "cuda", "fp8 KV cache is GPU-only"
            kv_dtype = torch.float8_e4m3fn
        elif cfg.kv_dtype is not None:
            raise ValueError(f"unknown kv_dtype {cfg.kv_dtype!r}")
        kv_bytes_per_el = 1 if kv_dtype == torch.float8_e4m3fn else 2
        if cfg.kv_blocks is not None:
            num_blocks = cfg.kv_blocks
        elif self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
            budget = int(free * cfg.gpu_mem_fraction)
            num_blocks = PagedKVCache.blocks_for_memory(
                self.mcfg, n_kv_local, budget, cfg.block_size,
                dtype_bytes=kv_bytes_per_el)
            max_needed = (cfg.max_batch *
                          ((cfg.max_seq_len + cfg.block_size - 1) // cfg.block_size))
            num_blocks = min(num_blocks, max_needed)
            if state().tp_size > 1:
                # TP ranks must agree on the pool size: schedules are
                # lockstep and preemption fires when the allocator runs
                # out — a rank with less free HBM (fragmentation, other
                # allocations) would preempt at a different step and
                # desync the first collective after it. Take the MIN.
                import torch.distributed as dist
                t = torch.tensor([num_blocks], dtype=torch.int64,
                                 device=self.device)
                dist.all_reduce(t, op=dist.ReduceOp.MIN,
                                group=state().tp_group)
                num_blocks = int(t.item())
        else:
            num_blocks = cfg.max_batch * (
                (cfg.max_seq_len + cfg.block_size - 1) // cfg.block_size)
        self.kv = PagedKVCache(self.mcfg, n_kv_local, num_blocks,
                               cfg.block_size, device=self.device,
                               dtype=kv_dtype)
        self.max_blocks_per_seq = (cfg.max_seq_len + cfg.block_size - 1) // cfg.block_size
        self._next_id = itertools.count()
        self.waiting: list[SequenceState] = []
        self.running: list[SequenceState] = []
        self._gen = torch.Generator(device="cpu").manual_seed(cfg.seed)
        import os as _os
        # sanitizer tier (scripts/sanitize_gpu.sh) forces eager: serialized
        # launches invalidate hipGraph stream capture
        self._force_eager = _os.environ.get("SUB_EAGER_DECODE") == "1"
        self.graph_runner = None
        self.steps = 0
        self.tokens_generated = 0

    # ------------------------------------------------------------------ API
    def add_request(self, prompt_tokens: list[int], max_new_tokens: int = 16,
                    temperature: float = 0.0, top_k: int = 0,
                    top_p: float = 1.0, eos_token: int = -1) -> int:
        assert len(prompt_tokens) >= 1
        assert len(prompt_tokens) + max_new_tokens <= self.cfg.max_seq_len, \
            "request exceeds max_seq_len"
        sid = next(self._next_id)
        self.waiting.append(SequenceState(
            seq_id=sid, tokens=list(prompt_tokens),
            num_prompt=len(prompt_tokens), max_new_tokens=max_new_tokens,
            temperature=temperature, top_k=top_k, top_p=top_p,
            eos_token=eos_token))
        return sid

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    # ------------------------------------------------------------ scheduling
    def _try_admit(self, out: list[StepOutput]) -> list[SequenceState]:
        admitted = []
        budget = self.cfg.prefill_chunk
        while self.waiting and len(self.running) + len(admitted) < self.cfg.max_batch:
            seq = self.waiting[0]
            nblk = seq.needed_blocks(self.cfg.block_size, extra_tokens=1)
            if admitted and seq.length > budget:
                # budget gates only the 2nd+ admission: a prompt longer than
                # prefill_chunk must still go through alone or it would
                # never be admitted (livelock)
                break
            if nblk > self.kv.num_blocks:
                # can never fit even an empty cache — fail ONLY this request
                # (error StepOutput) and keep scheduling the rest; one
                # poisoned request must not take down the batch
                self.waiting.pop(0)
                out.append(StepOutput(
                    seq.seq_id, 0, True,
                    error=(f"request needs {nblk} KV blocks but the cache "
                           f"has only {self.kv.num_blocks}")))
                continue
            if self.kv.allocator.free_count < nblk + 2 * len(self.running):
                break  # keep headroom for running seqs to grow
            seq.blocks = self.kv.allocator.alloc(nblk)
            admitted.append(self.waiting.pop(0))
            budget -= seq.length
        return admitted

    def _preempt_youngest(self) -> None:
        assert self.running, "out of KV blocks with nothing to preempt"
        seq = self.running.pop()  # youngest = most recently admitted
        self.kv.allocator.free(seq.blocks)
        seq.blocks = []
        # keep generated tokens: readmission re-prefills the whole token list
        # (prompt + generated so far). num_prompt stays at the true prompt
        # length so the max_new_tokens budget is not reset by preemption.
        self.waiting.insert(0, seq)

    # --------------------------------------------------------------- prefill
    def _prefill(self, seqs: list[SequenceState]) -> list[StepOutput]:
        bs = self.cfg.block_size
        dev = self.device
        tokens, positions, slots, starts = [], [], [], [0]
        for seq in seqs:
            tokens.extend(seq.tokens)
            positions.extend(range(seq.length))
            slots.extend(seq.slot_of(p, bs) for p in range(seq.length))
            starts.append(starts[-1] + seq.length)
        t = torch.tensor(tokens, dtype=torch.int64, device=dev)
        pos = torch.tensor(positions, dtype=torch.int32, device=dev)
        sl = torch.tensor(slots, dtype=torch.int32, device=dev)
        ss = torch.tensor(starts, dtype=torch.int32, device=dev)
        with torch.no_grad():
            logits = self.model.forward_prefill(t, pos, ss, self.kv.layers, sl)
        return self._emit(seqs, logits)

    # ---------------------------------------------------------------- decode
    def _decode(self) -> list[StepOutput]:
        if not self.running:
            return []
        # hipGraph capture of a forward containing RCCL collectives
        # (dist.all_reduce inside the row-parallel layers) is unvalidated on
        # ROCm — TP decodes eagerly until graph+RCCL capture is proven.
        if (self.cfg.use_graphs and self.device.type == "cuda"
                and state().tp_size <= 1 and not self._force_eager):
            return self._decode_graph()
        return self._decode_eager()

    def _decode_graph(self) -> list[StepOutput]:
        """hipGraph-captured decode (see serve/graph_runner.py): the token/
        position/slot state lives on device; the host only grows block
        tables and reads sampled tokens back. Sampling uses the device RNG
        (not the engine's CPU generator)."""
        from .kv_cache import OutOfBlocks
        from .graph_runner import GraphDecodeRunner
        if self.graph_runner is None:
            self.graph_runner = GraphDecodeRunner(self)
        runner = self.graph_runner
        ids = [s.seq_id for s in self.running]
        try:
            if [s.seq_id for s in runner.active] != ids:
                runner.prime(self.running)
            toks = runner.step()
        except OutOfBlocks:
            runner.invalidate()
            self._preempt_youngest()
            return self._decode() if self.running else []
        out = []
        for seq, tok in zip(self.running, toks):
            self.tokens_generated += 1
            done = (seq.num_generated >= seq.max_new_tokens or
                    (seq.eos_token >= 0 and tok == seq.eos_token) or
                    seq.length >= self.cfg.max_seq_len)
            if done:
                seq.finished = True
            out.append(StepOutput(seq.seq_id, int(tok), done))
        if any(o.finished for o in out):
            runner.invalidate()
        return out

    def _decode_eager(self) -> list[StepOutput]:
        seqs = self.running
        bs = self.cfg.block_size
        dev = self.device
        bsz = len(seqs)
        # grow block tables for the incoming token
        for seq in seqs:
            need = seq.needed_blocks(bs, extra_tokens=0)
            while len(seq.blocks) < need:
                try:
                    seq.blocks.extend(self.kv.allocator.alloc(1))
                except OutOfBlocks:
                    self._preempt_youngest()
                    return self._decode() if self.running else []
        tokens = torch.tensor([s.tokens[-1] for s in seqs], dtype=torch.int64,
                              device=dev)
        positions = torch.tensor([s.length - 1 for s in seqs],
                                 dtype=torch.int32, device=dev)
        slots = torch.tensor([s.slot_of(s.length - 1, bs) for s in seqs],
                             dtype=torch.int32, device=dev)
        seq_lens = torch.tensor([s.length for s in seqs], dtype=torch.int32,
                                device=dev)
        maxb = max(len(s.blocks) for s in seqs)
        bt = torch.zeros(bsz, maxb, dtype=torch.int32, device=dev)
        for i, s in enumerate(seqs):
            bt[i, :len(s.blocks)] = torch.tensor(s.blocks, dtype=torch.int32)
        with torch.no_grad():
            logits = self.model.forward_decode(tokens, positions,
                                               self.kv.layers, slots, bt,
                                               seq_lens)
        return self._emit(seqs, logits)

    # ----------------------------------------------------------------- emit
    def _emit(self, seqs: list[SequenceState], logits: torch.Tensor) -> list[StepOutput]:
        dev = self.device
        n = len(seqs)
        temp = torch.tensor([s.temperature for s in seqs], dtype=torch.float32,
                            device=dev)
        top_k = torch.tensor([s.top_k for s in seqs], dtype=torch.int32,
                             device=dev)
        top_p = torch.tensor([s.top_p for s in seqs], dtype=torch.float32,
                             device=dev)
        uniform = torch.rand(n, generator=self._gen).to(dev)
        next_tokens = ops.softmax_sample(logits, temp, top_k, top_p, uniform)
        out = []
        next_list = next_tokens.tolist()
        for seq, tok in zip(seqs, next_list):
            seq.tokens.append(int(tok))
            self.tokens_generated += 1
            done = (seq.num_generated >= seq.max_new_tokens or
                    (seq.eos_token >= 0 and tok == seq.eos_token) or
                    seq.length >= self.cfg.max_seq_len)
            if done:
                seq.finished = True
            out.append(StepOutput(seq.seq_id, int(tok), done))
        return out

    # ----------------------------------------------------------------- step
    def step(self) -> list[StepOutput]:
        """One engine iteration: admit+prefill, then decode everyone running."""
        out: list[StepOutput] = []
        admitted = self._try_admit(out)
        if admitted:
            out.extend(self._prefill(admitted))
            self.running.extend(admitted)
            self._retire(admitted, out)
        if self.running:
            res = self._decode()
            out.extend(res)
            self._retire(self.running, res)
            self.steps += 1
        return out

    def decode_only_step(self) -> list[StepOutput]:
        """Pure decode step (bench hot loop — no admission logic)."""
        res = self._decode()
        self._retire(self.running, res)
        self.steps += 1
        return res

    def abort(self, seq_id: int) -> bool:
        """Drop ONE request (client disconnect / cancellation). Safe
        between steps: a running sequence is removed exactly the way
        _retire removes a finished one (free blocks, compact running), so
        the next decode step just sees a smaller batch. Returns False for
        an unknown id (already finished — the common disconnect race).
        Under TP the leader broadcasts aborts with the iteration command
        (serve/tp.py) and every rank applies them identically."""
        for i, s in enumerate(self.waiting):
            if s.seq_id == seq_id:
                del self.waiting[i]
                if s.blocks:        # waiting seqs hold none today (fresh
                    self.kv.allocator.free(s.blocks)   # and preempted both
                    s.blocks = []                      # freed) — stay safe
                return True
        for s in self.running:
            if s.seq_id == seq_id:
                if s.blocks:
                    self.kv.allocator.free(s.blocks)
                    s.blocks = []
                self.running = [r for r in self.running
                                if r.seq_id != seq_id]
                return True
        return False

    def abort_all(self) -> None:
        """Drop every waiting/running request and free their KV blocks.
        Whole-engine failure recovery; deterministic, so TP leader and
        followers can apply it independently and stay lockstep."""
        self.waiting.clear()
        for s in self.running:
            if s.blocks:
                self.kv.allocator.free(s.blocks)
                s.blocks = []
        self.running.clear()
        if self.graph_runner is not None:
            self.graph_runner.invalidate()

    def _retire(self, pool: list[SequenceState], outs: list[StepOutput]) -> None:
        finished = {o.seq_id for o in outs if o.finished}
        if not finished:
            return
        for seq in [s for s in pool if s.seq_id in finished and s.blocks]:
            self.kv.allocator.free(seq.blocks)
            seq.blocks = []
        self.running = [s for s in self.running if s.seq_id not in finished]
