"""Kernel microbenchmarks at llama2-7b decode shapes (run on MI355X).

Reports achieved bytes/s against the HBM ceiling (guide: 8 TB/s peak,
~6.3 TB/s achievable) for the decode hot kernels.
"""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse
import time

import torch

from substratus_amd import ops


def timeit(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / iters


def bench_attn(B=64, hq=32, hkv=32, d=128, ctx=512, bs=16):
    torch.manual_seed(0)
    nblk_per = (ctx + bs - 1) // bs
    nb = B * nblk_per + 1
    kc = torch.randn(nb, hkv, bs, d, device="cuda").bfloat16()
    vc = torch.randn_like(kc)
    bt = torch.arange(B * nblk_per, dtype=torch.int32,
                      device="cuda").reshape(B, nblk_per)
    q = torch.randn(B, hq, d, device="cuda").bfloat16()
    lens = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
    scale = d ** -0.5
    dt = timeit(lambda: ops.paged_attn_decode(q, kc, vc, bt, lens, scale))
    kv_bytes = 2 * B * ctx * hkv * d * 2
    print(f"paged_attn_decode B={B} ctx={ctx} hq={hq} hkv={hkv} d={d}: "
          f"{dt*1e6:.1f} us, {kv_bytes/dt/1e12:.2f} TB/s")


def bench_rope(B=64, hq=32, hkv=32, d=128, bs=16):
    torch.manual_seed(0)
    nb = B * 4
    q = torch.randn(B, hq, d, device="cuda").bfloat16()
    k = torch.randn(B, hkv, d, device="cuda").bfloat16()
    v = torch.randn(B, hkv, d, device="cuda").bfloat16()
    pos = torch.full((B,), 500, dtype=torch.int32, device="cuda")
    slots = torch.arange(B, dtype=torch.int32, device="cuda")
    kc = torch.zeros(nb, hkv, bs, d, device="cuda").bfloat16()
    vc = torch.zeros_like(kc)
    dt = timeit(lambda: ops.rope_kv_append(q, k, v, pos, kc, vc, slots,
                                           10000.0))
    print(f"rope_kv_append B={B}: {dt*1e6:.1f} us")


def bench_norm(B=64, h=4096):
    x = torch.randn(B, h, device="cuda").bfloat16()
    r = torch.randn(B, h, device="cuda").bfloat16()
    w = torch.randn(h, device="cuda").bfloat16()
    dt = timeit(lambda: ops.rms_norm_residual(x, r, w, 1e-5))
    print(f"rms_norm_residual B={B} h={h}: {dt*1e6:.1f} us")


def bench_sample(B=64, V=32000):
    logits = torch.randn(B, V, device="cuda").bfloat16()
    temp = torch.zeros(B, device="cuda")
    tk = torch.zeros(B, dtype=torch.int32, device="cuda")
    tp = torch.ones(B, device="cuda")
    u = torch.rand(B, device="cuda")
    dt = timeit(lambda: ops.softmax_sample(logits, temp, tk, tp, u))
    print(f"softmax_sample greedy B={B} V={V}: {dt*1e6:.1f} us")


def bench_prefill(B=8, hq=32, hkv=32, d=128, S=512):
    torch.manual_seed(0)
    t = B * S
    q = torch.randn(t, hq, d, device="cuda").bfloat16()
    k = torch.randn(t, hkv, d, device="cuda").bfloat16()
    v = torch.randn(t, hkv, d, device="cuda").bfloat16()
    ss = torch.arange(0, t + 1, S, dtype=torch.int32, device="cuda")
    scale = d ** -0.5
    dt = timeit(lambda: ops.attn_prefill(q, k, v, ss, scale), iters=50)
    flops = 4.0 * B * hq * (S * (S + 1) / 2) * d
    print(f"attn_prefill B={B} S={S} hq={hq} d={d}: {dt*1e6:.1f} us, "
          f"{flops/dt/1e12:.0f} TF/s")


def bench_prefill_ablation(B=8, S=512, hq=32, hkv=32, d=128):
    """Within-probe variant ablation (guide rule 8): where does prefill time
    go? Variants are timing-only (wrong numerics)."""
    from substratus_amd import _C
    torch.manual_seed(0)
    t = B * S
    q = torch.randn(t, hq, d, device="cuda").bfloat16()
    k = torch.randn(t, hkv, d, device="cuda").bfloat16()
    v = torch.randn(t, hkv, d, device="cuda").bfloat16()
    out = torch.empty_like(q)
    ss = torch.arange(0, t + 1, S, dtype=torch.int32, device="cuda")
    names = {0: "full", 1: "no-softmax-valu", 2: "no-pv-mfma",
             3: "no-v-staging", 4: "no-qk-mfma"}
    flops = 4.0 * B * hq * (S * (S + 1) / 2) * d
    for variant in (0, 1, 2, 3, 4):
        dt = timeit(lambda: _C.attn_prefill_probe(variant, out, q, k, v, ss,
                                                  S, d ** -0.5), iters=50)
        print(f"prefill variant {variant} ({names[variant]:16s}): "
              f"{dt*1e6:7.1f} us  ({flops/dt/1e12:4.0f} TF-equiv)")


def bench_decode_gemm(m=256):
    import torch.nn.functional as F
    from substratus_amd import _C
    shapes = [(12288, 4096), (4096, 4096), (22016, 4096), (4096, 11008)]
    for n, k in shapes:
        torch.manual_seed(0)
        x = torch.randn(m, k, device="cuda").bfloat16()
        w = torch.randn(n, k, device="cuda").bfloat16() * 0.02
        y = torch.empty(m, n, device="cuda").bfloat16()
        dt_ours = timeit(lambda: _C.decode_gemm(y, x, w), iters=100)
        dt_blas = timeit(lambda: F.linear(x, w), iters=100)
        wb = n * k * 2
        print(f"decode_gemm M={m} N={n} K={k}: ours {dt_ours*1e6:.1f} us "
              f"({wb/dt_ours/1e12:.2f} TB/s W) vs hipBLASLt "
              f"{dt_blas*1e6:.1f} us ({wb/dt_blas/1e12:.2f} TB/s)")


def bench_wstream(m=256):
    """wstream GEMM v2 vs hipBLASLt, interleaved rounds (guide rule 24),
    plus a parity check against fp32 matmul. Shapes: llama2-7b decode
    projections + the logits GEMM."""
    import torch.nn.functional as F
    from substratus_amd import ops
    from substratus_amd import _C
    shapes = [(12288, 4096), (4096, 4096), (22016, 4096), (4096, 11008),
              (32000, 4096)]
    for n, k in shapes:
        torch.manual_seed(0)
        x = (torch.randn(m, k, device="cuda") * 0.3).bfloat16()
        w = (torch.randn(n, k, device="cuda") * 0.02).bfloat16()
        wp = ops.wstream_repack(w)
        y = ops.wstream_linear(x, wp, n, k)
        refy = (x.float() @ w.float().T)
        err = (y.float() - refy).abs().max().item()
        scale = refy.abs().max().item()
        ok = err <= 0.02 * max(scale, 1.0)
        # interleaved timing rounds
        ours, blas = [], []
        for _ in range(5):
            ours.append(timeit(lambda: ops.wstream_linear(x, wp, n, k),
                               iters=50))
            blas.append(timeit(lambda: F.linear(x, w), iters=50))
        do, db = min(ours), min(blas)
        wb = n * k * 2
        print(f"wstream M={m} N={n} K={k}: ours {do*1e6:6.1f} us "
              f"({wb/do/1e12:.2f} TB/s W) vs hipBLASLt {db*1e6:6.1f} us "
              f"({wb/db/1e12:.2f} TB/s)  x{db/do:.2f}  "
              f"parity={'OK' if ok else f'FAIL err={err:.4f}/{scale:.1f}'}")


def bench_attn_v5():
    """V4 vs V5 prefill attention, interleaved rounds (guide rule 24), on
    the serving (llama2-7b) and training (falcon) shapes."""
    from substratus_amd import _C
    shapes = [("llama S512", 8, 32, 32, 128, 512),
              ("llama S2048", 4, 32, 32, 128, 2048),
              ("falcon S2048", 4, 71, 1, 64, 2048)]
    for name, B, hq, hkv, d, S in shapes:
        torch.manual_seed(0)
        t = B * S
        q = (torch.rand(t, hq, d, device="cuda") * 2 - 1).bfloat16()
        k = (torch.rand(t, hkv, d, device="cuda") * 2 - 1).bfloat16()
        v = (torch.rand(t, hkv, d, device="cuda") * 2 - 1).bfloat16()
        ss = torch.arange(0, (B + 1) * S, S, dtype=torch.int32,
                          device="cuda")
        out = torch.empty_like(q)
        scale = d ** -0.5
        t4, t5 = [], []
        for _ in range(5):
            t4.append(timeit(lambda: _C.attn_prefill(
                out, q, k, v, ss, S, scale), iters=30))
            t5.append(timeit(lambda: _C.attn_v5(
                out, q, k, v, ss, S, scale, None), iters=30))
        flops = 2 * 2 * B * hq * S * S * d * 0.5
        print(f"attn {name}: v4 {min(t4)*1e6:.0f} us "
              f"({flops/min(t4)/1e12:.0f} TF) vs v5 {min(t5)*1e6:.0f} us "
              f"({flops/min(t5)/1e12:.0f} TF)")


def bench_attn_bwd():
    """Backward kernels old-vs-V5, interleaved (falcon MQA + llama
    shapes) — the dkv_v5 slowdown diagnosis harness."""
    from substratus_amd import _C
    shapes = [("falcon", 4, 71, 1, 64, 2048), ("llama", 4, 32, 32, 128, 2048)]
    for name, B, hq, hkv, d, S in shapes:
        torch.manual_seed(0)
        t = B * S
        q = (torch.rand(t, hq, d, device="cuda") * 2 - 1).bfloat16()
        k = (torch.rand(t, hkv, d, device="cuda") * 2 - 1).bfloat16()
        v = (torch.rand(t, hkv, d, device="cuda") * 2 - 1).bfloat16()
        do = torch.randn_like(q)
        ss = torch.arange(0, (B + 1) * S, S, dtype=torch.int32,
                          device="cuda")
        out = torch.empty_like(q)
        lse = torch.empty(t, hq, dtype=torch.float32, device="cuda")
        scale = d ** -0.5
        _C.attn_v5(out, q, k, v, ss, S, scale, lse)
        dq = torch.empty_like(q)
        dk = torch.zeros(t, hkv, d, dtype=torch.float32, device="cuda")
        dv = torch.zeros_like(dk)
        delta = torch.empty_like(lse)
        import os as _os
        for tag, env in (("mixed", None), ("dkv_v5", "1")):
            if env: _os.environ["SUB_ATTN_DKV_V5"] = env
            else: _os.environ.pop("SUB_ATTN_DKV_V5", None)
            dt = timeit(lambda: _C.attn_train_bwd_v5(
                dq, dk, dv, q, k, v, out, do, lse, delta, ss, S, scale),
                iters=20)
            print(f"bwd {name} [{tag}]: {dt*1e6:.0f} us")


def bench_mgemm(m=512):
    """mfma_gemm v3 (256x256 glds tile) vs hipBLASLt, interleaved rounds
    (guide rule 24) + parity vs fp32 matmul. Shapes: llama2-7b B=512 decode
    projections + logits."""
    import torch.nn.functional as F
    from substratus_amd import _C, ops
    shapes = [("qkv", 12288, 4096), ("o", 4096, 4096),
              ("gateup", 22016, 4096), ("down", 4096, 11008),
              ("logits", 32000, 4096)]
    for name, n, k in shapes:
        torch.manual_seed(0)
        x = (torch.rand(m, k, device="cuda") * 2 - 1).bfloat16()
        w = ((torch.rand(n, k, device="cuda") * 2 - 1) * 0.02).bfloat16()
        y = torch.empty(m, n, device="cuda").bfloat16()
        splitk = int(_C.mfma_gemm_pick_splitk(m, n, k))
        part = (torch.empty(splitk * m * n, dtype=torch.float32,
                            device="cuda") if splitk > 1 else None)
        _C.mfma_gemm(y, x, w, None, part, splitk)
        ref = (x.float() @ w.float().t())
        err = (y.float() - ref).abs().max().item()
        rel = err / ref.abs().max().item()
        # interleaved A/B rounds
        ours, blas = [], []
        for _ in range(5):
            ours.append(timeit(
                lambda: _C.mfma_gemm(y, x, w, None, part, splitk), iters=50))
            blas.append(timeit(lambda: F.linear(x, w), iters=50))
        dt_o, dt_b = min(ours), min(blas)
        wb = n * k * 2
        print(f"mgemm {name} M={m} N={n} K={k} splitk={splitk}: "
              f"ours {dt_o*1e6:.1f} us ({wb/dt_o/1e12:.2f} TB/s W, "
              f"{2*m*n*k/dt_o/1e12:.0f} TF) vs hipBLASLt {dt_b*1e6:.1f} us "
              f"({wb/dt_b/1e12:.2f} TB/s) | maxerr {err:.4f} rel {rel:.2e}")


def bench_quant_gemv(n=13824, k=5120, quant="q4_0"):
    from substratus_amd.parallel.tp import Linear
    from substratus_amd.models.quant import QuantLinear
    torch.manual_seed(0)
    lin = Linear(k, n, bias=False, dtype=torch.float32)
    lin.weight.data.normal_(0, 0.1)
    ql = QuantLinear.from_linear(lin, quant).to("cuda")
    for m in (1, 4):
        x = torch.randn(m, k, device="cuda").bfloat16()
        dt = timeit(lambda: ql(x))
        wb = ql.qs.numel() + ql.scales.numel() * 2
        print(f"quant_gemv {quant} N={n} K={k} M={m}: {dt*1e6:.1f} us, "
              f"{wb/dt/1e12:.2f} TB/s (weights)")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--which", default="all")
    a = ap.parse_args()
    if a.which in ("all", "attn"):
        bench_attn()
        bench_attn(ctx=2048)
        bench_attn(hq=64, hkv=8, ctx=512)  # llama2-70b-style GQA
    if a.which == "attngqa":           # GQA decode sweep (LDS-staged path)
        bench_attn(B=512, hq=32, hkv=32, ctx=512)   # MHA reference point
        bench_attn(B=512, hq=64, hkv=8, ctx=512)    # llama2-70b
        bench_attn(B=256, hq=64, hkv=8, ctx=512)
        bench_attn(B=256, hq=64, hkv=8, ctx=128)    # short-ctx regime
        bench_attn(B=512, hq=64, hkv=8, ctx=2048)
        bench_attn(B=512, hq=128, hkv=8, d=64, ctx=512)  # GQA-16 d=64
        bench_attn(B=64, hq=64, hkv=8, ctx=512)     # nsplit>1 regime
    if a.which in ("all", "prefill"):
        bench_prefill()
        bench_prefill(S=2048, B=4)
    if a.which == "prefill_ablate":
        bench_prefill_ablation()
    if a.which in ("all", "rope"):
        bench_rope()
    if a.which in ("all", "norm"):
        bench_norm()
    if a.which in ("all", "sample"):
        bench_sample()
    if a.which in ("all", "dgemm"):
        bench_decode_gemm()
        bench_decode_gemm(m=64)
    if a.which in ("all", "wstream"):
        bench_wstream()
        bench_wstream(m=64)
    if a.which in ("all", "attnv5"):
        bench_attn_v5()
    if a.which == "attnbwd":
        bench_attn_bwd()
    if a.which in ("all", "mgemm"):
        m = int(os.environ.get("MGEMM_M", "512"))
        bench_mgemm(m)
    if a.which in ("all", "quant"):
        bench_quant_gemv()
        bench_quant_gemv(quant="q8_0")
