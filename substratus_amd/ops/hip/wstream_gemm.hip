// Weight-streaming decode GEMM v2: Y[M<=256, N] = X[M, K] · W^T with W
// pre-repacked into MFMA-B-fragment-linear order — the second attempt at
// the skinny-M decode projections (v1 in decode_gemm.hip
// measured 0.43-0.70 TB/s and documents why: per-32-k barriers + 8-way
// duplicated W fragment loads).
//
// What v1 got wrong, fixed here:
//   * W is STATIC at serving time, so repack it once on load into
//     [N/16][K/32][64 lanes][8 bf16] (ops/__init__.py wstream_repack): each
//     wave's B-fragment for one 32-k MFMA step is ONE contiguous uint4 per
//     lane, and consecutive steps are sequential in memory — the weight
//     stream becomes a perfectly coalesced, nt-tagged linear read. v1
//     gathered fragments from torch's [N, K] layout (16 rows × 16 B
//     scatter per fragment).
//   * Each of the 8 waves owns its OWN 16-col n-tile (WG covers 128 cols):
//     W bytes are fetched exactly once per grid. v1's 8 waves all loaded
//     the same 64-col fragments (8× L2 traffic, 8× issue slots).
//   * X chunks are 64 k deep (one barrier per 32 MFMA per wave, vs v1's
//     barrier per 8) and the W prefetch pipeline runs two chunks (4
//     fragments) ahead, so the stream never waits a full HBM latency.
//   * split-K slices partition the CHUNK RANGE (no divisibility demands);
//     fp32 partials + the v1 reduce kernel only when the base grid
//     underfills the 256-CU chip, direct bf16 store otherwise.
//
// X rows are padded to 144 B in LDS (128 data + one 16-B access width) so
// the 16-lane A-fragment column reads are bank-conflict-free — same rule
// as v1's 80-B rows.
//
// STATUS: correct (parity vs fp32 matmul on all llama2-7b decode shapes +
// the logits GEMM, scripts/microbench.py --which wstream) and 2-3x faster
// than v1 (1.3-1.6 TB/s of weight stream at M=256), but still 0.3-0.55x
// hipBLASLt (2.5-4.0 TB/s) — NOT wired into the model. Measured why, in
// order of discovery (each variant A/B'd on MI355X):
//   1. This schedule (compiler-serialized ds_read→lgkmcnt(0)→MFMA chain,
//      120 VGPR → 2 workgroups/CU) is the FASTEST of three tried. Explicit
//      A-fragment register ping-pong (pipelined reads, partial lgkmcnt
//      waits in the ISA) pushed VGPR to 139-154, halving occupancy —
//      measured 0.6-0.9x of THIS version. Occupancy beats static
//      scheduling here.
//   2. The structural ceiling is LDS read bandwidth: with one n16 tile
//      per wave, every 16-B W fragment needs 16x16-B A-fragment LDS reads
//      (A:W = 16:1), capping the W stream at ~LDS_BW/16 per CU. Breaking
//      it needs A-reuse across n-tiles per wave (acc > 128 VGPRs → AGPR
//      accumulators), lds-dma staging and counted-vmcnt phases — the
//      library-grade schedule hipBLASLt already has (its M=256 solutions
//      run 40-65% of the 6.3 TB/s roofline; TunableOp confirms +2% is all
//      the library has left). hipBLASLt stays the production path; the
//      remaining decode-GEMM headroom (~3.8 ms/step at B=256) is a
//      known-open item, quantified in profiles/decode_step_profile_r01.txt.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float float4v;

#define WS_WAVES 8
#define WS_BK 64                  // k per staged X chunk
#define WS_ROWB 144               // padded LDS row bytes (128 data + 16)

__global__ __launch_bounds__(WS_WAVES * WAVE)
void wstream_gemm_kernel(float* __restrict__ part,   // [splitk, 256, N]
                         bf16_t* __restrict__ y,     // [M, N] (splitk == 1)
                         const bf16_t* __restrict__ x,   // [M, K]
                         const uint4v* __restrict__ wp,  // repacked W
                         int m, int n, int k, int splitk) {
  const int wg_nt = blockIdx.x;        // tile of 8 n16 tiles (128 cols)
  const int slice = blockIdx.y;        // K slice
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lhi = lane >> 4, llo = lane & 15;
  const int n16 = wg_nt * WS_WAVES + wave;   // this wave's 16-col tile
  const int k32 = k / 32;

  const int nchunks = k / WS_BK;
  const int cpk = nchunks / splitk, rem = nchunks % splitk;
  const int c0 = slice * cpk + min(slice, rem);
  const int c1 = c0 + cpk + (slice < rem ? 1 : 0);

  __shared__ unsigned char xbuf[2][256 * WS_ROWB];

  float4v acc[16];
  #pragma unroll
  for (int mb = 0; mb < 16; ++mb) acc[mb] = (float4v){0.f, 0.f, 0.f, 0.f};

  // ---- cooperative X chunk stage (256 rows x 64 k, 4 uint4/thread) ------
  auto stage_x = [&](int c, int buf) {
    const int tid = (int)threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int flat = i * WS_WAVES * WAVE + tid;  // uint4 index, 8 per row
      const int row = flat / 8, kc = flat % 8;     // kc: 8-elem group
      const int grow = min(row, m - 1);
      const uint4 v4 = *reinterpret_cast<const uint4*>(
          x + (long)grow * k + c * WS_BK + kc * 8);
      *reinterpret_cast<uint4*>(&xbuf[buf][row * WS_ROWB + kc * 16]) = v4;
    }
  };
  // ---- W fragment load for global 32-k step t (one uint4, nt stream) ----
  auto load_w = [&](int t) -> uint4v {
    return nt_load_u4(reinterpret_cast<const bf16_t*>(
        wp + ((long)n16 * k32 + t) * WAVE + lane));
  };

  stage_x(c0, 0);
  uint4v wf[4];                        // 2-chunk (4-step) prefetch pipeline
  wf[0] = load_w(c0 * 2);
  wf[1] = load_w(c0 * 2 + 1);
  if (c0 + 1 < c1) {
    wf[2] = load_w(c0 * 2 + 2);
    wf[3] = load_w(c0 * 2 + 3);
  }
  __syncthreads();

  for (int c = c0; c < c1; ++c) {
    const int buf = (c - c0) & 1;
    if (c + 1 < c1) stage_x(c + 1, buf ^ 1);
    uint4v wn[2];
    if (c + 2 < c1) {
      wn[0] = load_w((c + 2) * 2);
      wn[1] = load_w((c + 2) * 2 + 1);
    }
    #pragma unroll
    for (int t = 0; t < 2; ++t) {
      const short8 bf = *reinterpret_cast<const short8*>(&wf[t]);
      #pragma unroll
      for (int mb = 0; mb < 16; ++mb) {
        const short8 af = *reinterpret_cast<const short8*>(
            &xbuf[buf][(mb * 16 + llo) * WS_ROWB + t * 64 + lhi * 16]);
        acc[mb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[mb],
                                                          0, 0, 0);
      }
    }
    wf[0] = wf[2]; wf[1] = wf[3];
    wf[2] = wn[0]; wf[3] = wn[1];
    __syncthreads();
  }

  // ---- epilogue ---------------------------------------------------------
  const int col = n16 * 16 + llo;
  if (splitk == 1) {
    #pragma unroll
    for (int mb = 0; mb < 16; ++mb) {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = mb * 16 + lhi * 4 + i;
        if (row < m) y[(long)row * n + col] = f2bf(acc[mb][i]);
      }
    }
  } else {
    float* out = part + (long)slice * 256 * n;
    #pragma unroll
    for (int mb = 0; mb < 16; ++mb) {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = mb * 16 + lhi * 4 + i;
        out[(long)row * n + col] = acc[mb][i];
      }
    }
  }
}

// split-K combine (same partial layout as v1's; duplicated here because the
// build is -fno-gpu-rdc — device symbols don't cross TUs)
__global__ __launch_bounds__(256)
void ws_reduce_kernel(bf16_t* __restrict__ y, const float* __restrict__ part,
                      int m, long n, int splitk) {
  const long i = blockIdx.x * 256 + threadIdx.x;   // over m*n/2 (bf16x2)
  const long total = (long)m * n;
  const long i2 = i * 2;
  if (i2 >= total) return;
  const long row = i2 / n, col = i2 % n;
  float a = 0.f, b = 0.f;
  for (int s = 0; s < splitk; ++s) {
    const float* p = part + ((long)s * 256 + row) * n + col;
    a += p[0];
    b += (col + 1 < n) ? p[1] : 0.f;
  }
  if (col + 1 < n) {
    reinterpret_cast<unsigned int*>(y)[i] = pack_bf16x2(a, b);
  } else {
    y[i2] = f2bf(a);
  }
}

int wstream_splitk(int n, int k) {
  const int base = n / 128;
  if (base >= 230) return 1;            // grid already ~fills 256 CUs
  int s = (256 + base - 1) / base;
  if (s > 16) s = 16;
  const int nchunks = k / WS_BK;
  if (s > nchunks) s = nchunks;
  return s;
}

bool wstream_supported(int m, int n, int k) {
  return m <= 256 && n % 128 == 0 && k % WS_BK == 0;
}

void launch_wstream_gemm(bf16_t* y, const bf16_t* x, const bf16_t* wp,
                         float* part, int m, int n, int k, int splitk,
                         hipStream_t stream) {
  dim3 grid(n / 128, splitk);
  wstream_gemm_kernel<<<grid, dim3(WS_WAVES * WAVE), 0, stream>>>(
      part, y, x, reinterpret_cast<const uint4v*>(wp), m, n, k, splitk);
  if (splitk > 1) {
    const long total = (long)m * n;
    const long nthreads = (total + 1) / 2;
    ws_reduce_kernel<<<dim3((unsigned)((nthreads + 255) / 256)),
                       dim3(256), 0, stream>>>(y, part, m, n, splitk);
  }
}
