"""Serving soak: continuous batching under churn on the full llama2-7b —
staggered arrivals, mixed prompt/output lengths and sampling settings,
graph-captured decode, admission + completion + slot recycling. Asserts
every request finishes with exactly its requested token count and prints a
throughput/latency summary (evidence: profiles/soak_r01.txt).

Usage: python scripts/soak.py [--model llama2-7b] [--requests 400]
"""

import argparse
import os
import random
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from substratus_amd.serve.engine import Engine, EngineConfig


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama2-7b")
    ap.add_argument("--requests", type=int, default=400)
    ap.add_argument("--max-batch", type=int, default=128)
    ap.add_argument("--max-seq-len", type=int, default=1024)
    ap.add_argument("--abort-frac", type=float, default=0.1,
                    help="fraction of requests cancelled mid-generation "
                         "(client-disconnect simulation; 0 disables)")
    a = ap.parse_args()

    on_gpu = torch.cuda.is_available()
    eng = Engine(EngineConfig(
        model=a.model, dtype=torch.bfloat16 if on_gpu else torch.float32,
        max_batch=a.max_batch, max_seq_len=a.max_seq_len, seed=7))
    rng = random.Random(1234)
    want: dict[int, int] = {}
    got: dict[int, int] = {}
    t_first: dict[int, float] = {}
    t_sub: dict[int, float] = {}
    t_done: dict[int, float] = {}
    pending = a.requests
    to_abort: dict[int, int] = {}
    aborted: set[int] = set()
    steps = 0
    t0 = time.time()
    while pending > 0 or eng.has_work():
        # staggered arrivals: top the engine up a few requests at a time
        while (pending > 0 and
               len(eng.waiting) + len(eng.running) < a.max_batch):
            n_new = rng.randint(1, 8)
            for _ in range(n_new):
                if pending == 0:
                    break
                plen = rng.randint(4, max(4, a.max_seq_len // 2))
                mnew = rng.randint(8, max(8, min(128, a.max_seq_len // 4)))
                sid = eng.add_request(
                    [rng.randrange(eng.mcfg.vocab_size)
                     for _ in range(plen)],
                    max_new_tokens=mnew,
                    temperature=rng.choice([0.0, 0.7, 1.0]),
                    top_k=rng.choice([0, 40]),
                    top_p=rng.choice([1.0, 0.95]))
                want[sid] = mnew
                t_sub[sid] = time.time()
                if rng.random() < a.abort_frac:
                    # cancel after a random share of its tokens
                    to_abort[sid] = max(1, int(mnew * rng.random()))
                pending -= 1
        for out in eng.step():
            now = time.time()
            t_first.setdefault(out.seq_id, now)
            t_done[out.seq_id] = now
            got[out.seq_id] = got.get(out.seq_id, 0) + 1
            if (out.seq_id in to_abort and
                    got[out.seq_id] >= to_abort[out.seq_id]):
                eng.abort(out.seq_id)
                aborted.add(out.seq_id)
                del to_abort[out.seq_id]
        steps += 1
    if on_gpu:
        torch.cuda.synchronize()
    wall = time.time() - t0

    bad = [(s, want[s], got.get(s, 0)) for s in want
           if s not in aborted and got.get(s, 0) != want[s]]
    assert not bad, f"requests with wrong token counts: {bad[:5]}"
    over = [(s, got.get(s, 0)) for s in aborted if got.get(s, 0) > want[s]]
    assert not over, f"aborted requests kept generating: {over[:5]}"
    assert eng.kv.allocator.free_count == eng.kv.num_blocks, \
        "KV blocks leaked across the soak"
    total_tokens = sum(got.values())
    ttfts = sorted(t_first[s] - t_sub[s] for s in want)
    lats = sorted(t_done[s] - t_sub[s] for s in want)
    print(f"SOAK OK: {len(want)} requests ({len(aborted)} aborted), "
          f"{total_tokens} tokens, "
          f"{steps} engine steps, {wall:.1f}s wall, "
          f"{total_tokens / wall:.0f} tok/s")
    print(f"  TTFT p50/p95: {ttfts[len(ttfts) // 2] * 1000:.0f}/"
          f"{ttfts[int(len(ttfts) * 0.95)] * 1000:.0f} ms; "
          f"request latency p50/p95: {lats[len(lats) // 2] * 1000:.0f}/"
          f"{lats[int(len(lats) * 0.95)] * 1000:.0f} ms")


if __name__ == "__main__":
    main()
