"""hipGraph-captured decode step with device-resident batch state.

The eager decode step is launch-bound: ~260 kernel launches plus Python
bookkeeping per step (profiles/serve_decode_r01_baseline.txt: ~1.5-2 ms of a
10.3 ms step was launch/host gap). This runner captures ONE decode iteration
— embedding → 32 fused-norm blocks → logits → fused sampling → device-side
position/seq_len/slot advance — into a hipGraph per batch-size bucket and
replays it (guide §launch costs: graph replay ≈10-16 µs vs ~3.5 µs × 260
eager launches).

Device-autonomous state: the sampling kernel writes next tokens INTO the
token buffer the embedding reads, and positions/seq_lens/slots advance on
device at the end of the captured region, so consecutive replays need no
host→device traffic at all. The host mirrors sequence lengths to (a) append
fresh KV blocks to the block table the step BEFORE a sequence crosses a
16-token block boundary (a single 4-byte H2D per crossing), and (b) read the
sampled tokens back for emission/finish handling.

Batch buckets: the active set is padded to the next bucket size; inactive
slots decode against a reserved dummy KV block (wasted lanes, no hazards).
Any admission/finish re-primes the buffers; a new bucket size captures its
graph lazily.
"""

from __future__ import annotations

from typing import Optional

import torch

import os

from .kv_cache import SequenceState

BUCKETS = (1, 2, 4, 8, 16, 32, 64, 96, 128, 192, 256)


def bucket_for(n: int) -> int:
    for b in BUCKETS:
        if n <= b:
            return b
    return n


class GraphDecodeRunner:
    def __init__(self, engine):
        self.eng = engine
        self.device = engine.device
        self.model = engine.model
        B = engine.cfg.max_batch
        maxb = engine.max_blocks_per_seq
        self.max_batch = B
        self.maxb = maxb
        self.bs = engine.cfg.block_size
        dev = self.device
        self.tokens = torch.zeros(B, dtype=torch.int64, device=dev)
        self.positions = torch.zeros(B, dtype=torch.int32, device=dev)
        self.seq_lens = torch.ones(B, dtype=torch.int32, device=dev)
        self.slots = torch.zeros(B, dtype=torch.int32, device=dev)
        self.block_table = torch.zeros(B, maxb, dtype=torch.int32, device=dev)
        self.temp = torch.zeros(B, dtype=torch.float32, device=dev)
        self.top_k = torch.zeros(B, dtype=torch.int32, device=dev)
        self.top_p = torch.ones(B, dtype=torch.float32, device=dev)
        self.graphs: dict[int, torch.cuda.CUDAGraph] = {}
        # Dual-stream decode is OPT-IN (SUB_SPLIT_STREAMS=1): both the
        # free-running split (-5.5%) and the op-skewed interleave (-10%)
        # measured SLOWER than single-stream at B=512 — the KV stream
        # (multi-GB per step) flushes the 256 MiB LLC between the halves'
        # weight reads, so split GEMMs pay double HBM weight traffic, and
        # the halves' attention kernels contend for the same HBM the
        # single-stream schedule already saturates. Kept (with the
        # op-generator plumbing in models/model.py decode_ops) as the
        # harness for future overlap experiments.
        self.split_streams = os.environ.get("SUB_SPLIT_STREAMS",
                                            "0") == "1"
        self.side = torch.cuda.Stream() if engine.device.type == "cuda" \
            else None
        self.active: list[SequenceState] = []
        self.host_lens: list[int] = []
        self.primed_bucket = 0
        # one reserved block absorbs the padded (inactive) lanes' KV writes
        self.dummy_block = engine.kv.allocator.alloc(1)[0]

    # ------------------------------------------------------------- priming
    def prime(self, seqs: list[SequenceState]) -> None:
        """(Re)load device state from the engine's sequence states; called on
        any change of the active set."""
        n = len(seqs)
        assert n <= self.max_batch
        b = min(bucket_for(n), self.max_batch)
        self.active = list(seqs)
        self.host_lens = [s.length for s in seqs]
        bs = self.bs
        tok = [s.tokens[-1] for s in seqs] + [0] * (b - n)
        pos = [s.length - 1 for s in seqs] + [0] * (b - n)
        lens = [s.length for s in seqs] + [1] * (b - n)
        slots = [s.slot_of(s.length - 1, bs) for s in seqs] + \
                [self.dummy_block * bs] * (b - n)
        bt = torch.full((b, self.maxb), self.dummy_block, dtype=torch.int32)
        for i, s in enumerate(seqs):
            bt[i, :len(s.blocks)] = torch.tensor(s.blocks, dtype=torch.int32)
        self.tokens[:b] = torch.tensor(tok, dtype=torch.int64)
        self.positions[:b] = torch.tensor(pos, dtype=torch.int32)
        self.seq_lens[:b] = torch.tensor(lens, dtype=torch.int32)
        self.slots[:b] = torch.tensor(slots, dtype=torch.int32)
        self.block_table[:b] = bt.to(self.device)
        self.temp[:b] = torch.tensor(
            [s.temperature for s in seqs] + [0.0] * (b - n))
        self.top_k[:b] = torch.tensor(
            [s.top_k for s in seqs] + [0] * (b - n), dtype=torch.int32)
        self.top_p[:b] = torch.tensor(
            [s.top_p for s in seqs] + [1.0] * (b - n))
        self.primed_bucket = b

    # ------------------------------------------------------------- capture
    def _decode_gen(self, lo: int, hi: int):
        sl = slice(lo, hi)
        return self.model.decode_ops(
            self.tokens[sl], self.positions[sl], self.eng.kv.layers,
            self.slots[sl], self.block_table[sl], self.seq_lens[sl])

    def _step_body(self, b: int) -> None:
        # Op-skewed dual-stream decode: split the batch in half and
        # interleave the two forwards op-group by op-group on two HIP
        # streams, the trailer waiting on the leader's matching op event.
        # The skew makes every overlapped pair COMPLEMENTARY: the
        # trailer's qkv/up/down GEMM runs right after the leader streamed
        # those weights (LLC-hot — a free-running split doubles the HBM
        # weight traffic and measured -5.5%), while the trailer's
        # HBM-bound paged attention overlaps the leader's MFMA GEMMs.
        # Halves touch disjoint sequences/KV blocks — no data hazards;
        # the RNG draw happens once on the main stream before the fork.
        from substratus_amd import _C
        uniform = torch.rand(self.max_batch, device=self.device)

        def sample(lo, hi, logits):
            sl = slice(lo, hi)
            _C.softmax_sample(self.tokens[sl], logits.contiguous(),
                              self.temp[sl], self.top_k[sl],
                              self.top_p[sl], uniform[sl])

        if self.split_streams and b >= 32:
            h = b // 2
            cur = torch.cuda.current_stream()
            self.side.wait_stream(cur)
            ga, gb = self._decode_gen(0, h), self._decode_gen(h, b)
            logits_a = logits_b = None
            while logits_a is None or logits_b is None:
                ev = None
                if logits_a is None:
                    try:
                        next(ga)
                    except StopIteration as e:
                        logits_a = e.value
                        sample(0, h, logits_a)
                    ev = torch.cuda.Event()
                    ev.record(cur)
                if logits_b is None:
                    with torch.cuda.stream(self.side):
                        if ev is not None:
                            self.side.wait_event(ev)
                        try:
                            next(gb)
                        except StopIteration as e:
                            logits_b = e.value
                            sample(h, b, logits_b)
            cur.wait_stream(self.side)
        else:
            gen = self._decode_gen(0, b)
            while True:
                try:
                    next(gen)
                except StopIteration as e:
                    sample(0, b, e.value)
                    break
        # device-side advance for the next iteration
        pos = self.positions[:b]
        lens = self.seq_lens[:b]
        bt = self.block_table[:b]
        pos += 1
        lens += 1
        blk_col = torch.div(pos, self.bs, rounding_mode="floor").long()
        blk = bt.gather(1, blk_col.unsqueeze(1)).squeeze(1)
        self.slots[:b].copy_(blk * self.bs + pos % self.bs)

    def _get_graph(self, b: int) -> torch.cuda.CUDAGraph:
        if b in self.graphs:
            return self.graphs[b]
        torch.cuda.synchronize()
        # Warmup executes REAL steps on the live state (tokens/pos/lens/slots
        # advance; KV slots for the next 2 positions get scribbled) — save
        # and restore the small state. The scribbled KV positions are always
        # rewritten by rope_kv_append before attention ever reads them (a
        # position's KV is written in the same step that first includes it),
        # so the cache needs no restore.
        saved = [t.clone() for t in (self.tokens, self.positions,
                                     self.seq_lens, self.slots)]
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._step_body(b)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        # The op-generator step creates collectible cycles (suspended
        # frames + StopIteration tracebacks); a Python GC firing MID-
        # capture frees earlier tensors into foreign mempools (hipFree
        # during stream capture aborts the process on ROCm). Drain garbage
        # first, then hold GC off for the capture.
        import gc
        gc.collect()
        gc.disable()
        try:
            with torch.cuda.graph(g):
                self._step_body(b)
        finally:
            gc.enable()
        for dst, src in zip((self.tokens, self.positions, self.seq_lens,
                             self.slots), saved):
            dst.copy_(src)
        torch.cuda.synchronize()
        self.graphs[b] = g
        return g

    # ---------------------------------------------------------------- step
    def step(self) -> list[int]:
        """One captured decode step for the primed active set. Returns the
        sampled token per active sequence (synchronous)."""
        b = self.primed_bucket
        n = len(self.active)
        assert b > 0 and n > 0
        # grow block tables one step AHEAD of the device-side slot advance
        for i, seq in enumerate(self.active):
            nl = self.host_lens[i] + 1
            need = (nl + self.bs - 1) // self.bs
            while len(seq.blocks) < need:
                newb = self.eng.kv.allocator.alloc(1)[0]
                seq.blocks.append(newb)
                self.block_table[i, len(seq.blocks) - 1] = newb
            self.host_lens[i] = nl
        graph = self._get_graph(b)
        graph.replay()
        toks = self.tokens[:n].tolist()  # one sync per step
        for seq, t in zip(self.active, toks):
            seq.tokens.append(int(t))
        return toks

    def invalidate(self) -> None:
        self.primed_bucket = 0
        self.active = []
