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
    def _step_body(self, b: int) -> None:
        from substratus_amd import _C
        tokens = self.tokens[:b]
        pos = self.positions[:b]
        lens = self.seq_lens[:b]
        slots = self.slots[:b]
        bt = self.block_table[:b]
        logits = self.model.forward_decode(tokens, pos, self.eng.kv.layers,
                                           slots, bt, lens)
        uniform = torch.rand(b, device=self.device)
        _C.softmax_sample(tokens, logits.contiguous(), self.temp[:b],
                          self.top_k[:b], self.top_p[:b], uniform)
        # device-side advance for the next iteration
        pos += 1
        lens += 1
        blk_col = torch.div(pos, self.bs, rounding_mode="floor").long()
        blk = bt.gather(1, blk_col.unsqueeze(1)).squeeze(1)
        slots.copy_(blk * self.bs + pos % self.bs)

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
        with torch.cuda.graph(g):
            self._step_body(b)
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
