// Prefill/train-forward attention V5 — the guide's 8-wave swapped-QK^T
// structure (§B "8-warp 32×32 ladder"): per wave ONE 32-row q-block with
// Q resident in registers, K streamed as MFMA A-fragments
// (v_mfma_f32_32x32x16_bf16, swapped: S^T = K·Q^T so each lane's scores
// sit in its own q-column), softmax fully in-register (no P round-trip
// through LDS), P redistributed to PV A-fragments with
// v_cvt_pk_bf16_f32 + v_permlane32_swap (guide T12), V consumed through
// ds_read_b64_tr_b16 images (T10).
//
// V4 (attention_prefill.hip) measured 204-240 TF: its per-16-row-wave
// C-layout softmax round-trips P through LDS and exposes the serial
// VALU chain. This structure removes both; the guide's ladder reaches
// 667-1000 TF on it at N=2048 D=128.
//
// mfma_f32_32x32x16_bf16 fragment maps (probe32 kernel below, verified
// on gfx950 against torch.matmul on asymmetric data):
//   A: lane l holds A[row = l&31][k = (l>>5)*8 + j]   (j = 0..7, short8)
//   B: lane l holds B[k = (l>>5)*8 + j][col = l&31]
//   C/D: lane l holds C[row = (reg&3) + 8*(reg>>2) + 4*(l>>5)][col = l&31]
//        (reg = 0..15, float16v)
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(16))) float float16v;

namespace attnv5 {

DEV_INLINE short4v tr16_read(const short* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) short4v*)p);
}

DEV_INLINE unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

}  // namespace attnv5

// ---- 32x32x16 fragment-map probe (GPU test: test_mfma32_fragment_layout)
__global__ void mfma32_probe_kernel(float* __restrict__ dout,  // [32, 32]
                                    const bf16_t* __restrict__ a,  // [32,16]
                                    const bf16_t* __restrict__ b) {  // [16,32]
  const int lane = threadIdx.x % WAVE;
  const int lo5 = lane & 31, hi = lane >> 5;
  short8 af, bf;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = *reinterpret_cast<const short*>(&a[lo5 * 16 + hi * 8 + j]);
    bf[j] = *reinterpret_cast<const short*>(&b[(hi * 8 + j) * 32 + lo5]);
  }
  float16v acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    dout[row * 32 + lo5] = acc[r];
  }
}

void launch_mfma32_probe(float* dout, const bf16_t* a, const bf16_t* b,
                         hipStream_t stream) {
  mfma32_probe_kernel<<<1, 64, 0, stream>>>(dout, a, b);
}

namespace attnv5 {

#define V5_WAVES 8
#define V5_QTILE 256   // q rows per workgroup (32 per wave)
#define V5_KCHUNK 64   // keys staged per LDS buffer (2 online 32-subtiles)

// XOR-swizzled 16-B-chunk address inside a row-major LDS tile
template <int ROW_BYTES>
DEV_INLINE int swz(int row, int chunk) {
  constexpr int NCH = ROW_BYTES / 16;
  return row * NCH + (chunk ^ (row & (NCH - 1)));
}

template <int D, int CH = V5_KCHUNK>
__global__ __launch_bounds__(V5_WAVES * WAVE)
void attn_v5_kernel(bf16_t* __restrict__ out,      // [T, HQ, D]
                    const bf16_t* __restrict__ q,  // [T, HQ, D]
                    const bf16_t* __restrict__ k,  // [T, HKV, D]
                    const bf16_t* __restrict__ v,  // [T, HKV, D]
                    const int* __restrict__ seq_start,  // [S+1]
                    int hq, int hkv, float scale,
                    float* __restrict__ lse) {     // [T, HQ] or null
  constexpr int KK = D / 16;     // QK^T contraction steps per 32-key tile
  constexpr int DC = D / 32;     // 32-d output tiles
  const int s = blockIdx.x;
  const int head = blockIdx.y;
  const int tile = blockIdx.z;
  const int kvh = head / (hq / hkv);
  const int base = seq_start[s];
  const int n = seq_start[s + 1] - base;
  const int q0_tile = tile * V5_QTILE;
  if (q0_tile >= n) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lo5 = lane & 31;     // this lane's q row (both halves)
  const int hi = lane >> 5;      // d/k half
  const int q0 = q0_tile + wave * 32;
  const bool wave_live = q0 < n;
  const int qpos = q0 + lo5;     // this lane's q position (may be >= n)

  const long qrow_stride = (long)hq * D;
  const long krow_stride = (long)hkv * D;

  // ---- Q resident in registers: chunk kk holds Q[qpos][kk*16+hi*8 .. +8]
  short8 qreg[KK];
  {
    const int qrow = min(qpos, n - 1);
    const bf16_t* qp = q + (long)(base + qrow) * qrow_stride +
                       (long)head * D;
    #pragma unroll
    for (int kk = 0; kk < KK; ++kk)
      qreg[kk] = *reinterpret_cast<const short8*>(qp + kk * 16 + hi * 8);
  }

  float m = -INFINITY, l = 0.f;
  float16v o_t[DC];              // O^T[32 d rows][q col = lo5]
  #pragma unroll
  for (int c = 0; c < DC; ++c) o_t[c] = (float16v){};

  // LDS: K rows (QK^T A-frags) + V tr16 images (PV A-frags), dbuf
  //   v_img[c(32d)][sc][g2(16k)][sel][16][16]:
  //   img[( (kk>=8)*8 + dhalf*4 + kk%4 )][d%16], sel = (kk%8)>=4
  __shared__ uint4 k_tile[2][CH * D * 2 / 16];
  __shared__ short v_img[2][DC][CH / 32][2][2][16][16];

  const int kmax_tile = min(q0_tile + V5_QTILE, n);
  const int nchunks = (kmax_tile + CH - 1) / CH;
  const float scale2 = scale * 1.44269504088896340736f;

  // T14 async-STAGE split (guide G15, +17% on attention): ISSUE the next
  // chunk's global loads before the compute phase, WRITE them to LDS
  // after it — HBM latency hides under the MFMA stream.
  constexpr int LOADS = CH * D / 8 / (V5_WAVES * WAVE);
  uint4 st_k[LOADS], st_v[LOADS];

  auto stage_load = [&](int kc2) {
    if (kc2 >= nchunks) return;
    const int kbase = kc2 * CH;
    #pragma unroll
    for (int i = 0; i < LOADS; ++i) {
      const int flat = (i * V5_WAVES * WAVE + threadIdx.x) * 8;
      const int row = flat / D, col = flat % D;
      const bool ok = kbase + row < n;
      const long goff = (long)(base + kbase + row) * krow_stride +
                        (long)kvh * D + col;
      st_k[i] = ok ? *reinterpret_cast<const uint4*>(k + goff)
                   : make_uint4(0, 0, 0, 0);
      st_v[i] = ok ? *reinterpret_cast<const uint4*>(v + goff)
                   : make_uint4(0, 0, 0, 0);
    }
  };

  auto stage_write = [&](int kc2, int buf) {
    if (kc2 >= nchunks) return;
    #pragma unroll
    for (int i = 0; i < LOADS; ++i) {
      const int flat = (i * V5_WAVES * WAVE + threadIdx.x) * 8;
      const int row = flat / D, col = flat % D;
      k_tile[buf][swz<D * 2>(row, col * 2 / 16)] = st_k[i];
      const int sc = row / 32, g2 = (row % 32) / 16, kk = row % 16;
      const int sel = (kk % 8) >= 4;
      const int r = ((kk >= 8) ? 8 : 0) + ((col % 32) / 16) * 4 + (kk % 4);
      *reinterpret_cast<uint4*>(
          &v_img[buf][col / 32][sc][g2][sel][r][col % 16]) = st_v[i];
    }
  };

  // T5 static form: the second-dispatched half loses VALU arbitration on
  // every segment; one setprio for it, no per-cluster flips (guide T5 —
  // the condition must be provably wave-uniform or s_setprio lands in
  // every wave)
  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  stage_load(0);
  stage_write(0, 0);
  __syncthreads();

  for (int kc = 0; kc < nchunks; ++kc) {
    const int cur = kc & 1;
    stage_load(kc + 1);              // loads fly over this chunk's MFMAs
    const int kbase = kc * CH;
    if (wave_live && kbase <= min(q0 + 31, n - 1)) {
      // ---- S^T = K·Q^T over the WHOLE 64-key chunk (both 32-subtiles
      // back-to-back: one online-softmax step per chunk instead of two —
      // halves the rescale/stat passes and gives the scheduler two
      // independent MFMA chains)
      float16v sacc[CH / 32];
      #pragma unroll
      for (int sc = 0; sc < CH / 32; ++sc) {
        sacc[sc] = (float16v){};
        #pragma unroll
        for (int kk = 0; kk < KK; ++kk) {
          const int key = sc * 32 + lo5;
          const uint4 ka4 = k_tile[cur][swz<D * 2>(key, kk * 2 + hi)];
          sacc[sc] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<const short8*>(&ka4), qreg[kk], sacc[sc],
              0, 0, 0);
        }
      }
      // ---- one in-register online-softmax step over 64 keys ------------
      float sv[CH / 32][16];
      float mloc = -INFINITY;
      #pragma unroll
      for (int sc = 0; sc < CH / 32; ++sc) {
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kpos = kbase + sc * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          const bool live = (kpos <= qpos) && (kpos < n) && (qpos < n);
          sv[sc][r] = live ? sacc[sc][r] * scale2 : -INFINITY;
          mloc = fmaxf(mloc, sv[sc][r]);
        }
      }
      mloc = fmaxf(mloc, __shfl_xor(mloc, 32, WAVE));
      // wave-uniform control flow only (see the divergent-MFMA note in
      // the git history of the two-subtile version); defer-max as before
      constexpr float THR = 8.0f;
      {
        float alpha = 1.0f;
        if (!__all((mloc == -INFINITY) || (mloc - m <= THR))) {
          const float mn = fmaxf(m, mloc);
          alpha = (m == -INFINITY) ? 0.f
                                   : __builtin_amdgcn_exp2f(m - mn);
          m = mn;
          #pragma unroll
          for (int c = 0; c < DC; ++c)
            #pragma unroll
            for (int r = 0; r < 16; ++r) o_t[c][r] *= alpha;
        }
        float rowsum = 0.f;
        unsigned w[CH / 32][8];
        #pragma unroll
        for (int sc = 0; sc < CH / 32; ++sc) {
          #pragma unroll
          for (int i = 0; i < 8; ++i) {
            float p0 = (sv[sc][2 * i] == -INFINITY) ? 0.f
                : __builtin_amdgcn_exp2f(sv[sc][2 * i] - m);
            float p1 = (sv[sc][2 * i + 1] == -INFINITY) ? 0.f
                : __builtin_amdgcn_exp2f(sv[sc][2 * i + 1] - m);
            rowsum += p0 + p1;
            w[sc][i] = cvt_pk_bf16(p0, p1);
          }
        }
        rowsum += __shfl_xor(rowsum, 32, WAVE);
        l = l * alpha + rowsum;
        // ---- O^T += V^T · P over the chunk's four 16-key steps ---------
        #pragma unroll
        for (int sc = 0; sc < CH / 32; ++sc) {
          #pragma unroll
          for (int g = 0; g < 2; ++g) {
            auto r02 = __builtin_amdgcn_permlane32_swap(
                w[sc][4 * g + 0], w[sc][4 * g + 2], false, false);
            auto r13 = __builtin_amdgcn_permlane32_swap(
                w[sc][4 * g + 1], w[sc][4 * g + 3], false, false);
            uint4 pw = make_uint4(r02[0], r13[0], r02[1], r13[1]);
            const short8 pb = *reinterpret_cast<short8*>(&pw);
            #pragma unroll
            for (int c = 0; c < DC; ++c) {
              const short4v lo_ =
                  tr16_read(&v_img[cur][c][sc][g][0][0][0] + lane * 4);
              const short4v hi_ =
                  tr16_read(&v_img[cur][c][sc][g][1][0][0] + lane * 4);
              short8 vt;
              vt[0] = lo_[0]; vt[1] = lo_[1]; vt[2] = lo_[2]; vt[3] = lo_[3];
              vt[4] = hi_[0]; vt[5] = hi_[1]; vt[6] = hi_[2]; vt[7] = hi_[3];
              o_t[c] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                  vt, pb, o_t[c], 0, 0, 0);
            }
          }
        }
      }
    }
    stage_write(kc + 1, cur ^ 1);  // vmcnt waits land HERE, post-compute
    __syncthreads();
  }

  // ---- epilogue: out[qpos][head][d] = O^T[d][q] / l ----------------------
  if (wave_live && qpos < n) {
    const float invl = (l > 0.f) ? 1.f / l : 0.f;
    bf16_t* op = out + (long)(base + qpos) * qrow_stride + (long)head * D;
    #pragma unroll
    for (int c = 0; c < DC; ++c) {
      #pragma unroll
      for (int rr = 0; rr < 4; ++rr) {   // 4 runs of 4 consecutive d
        const int d0 = c * 32 + 8 * rr + 4 * hi;
        const unsigned w0 = pack_bf16x2(o_t[c][rr * 4 + 0] * invl,
                                        o_t[c][rr * 4 + 1] * invl);
        const unsigned w1 = pack_bf16x2(o_t[c][rr * 4 + 2] * invl,
                                        o_t[c][rr * 4 + 3] * invl);
        *reinterpret_cast<uint2*>(op + d0) = make_uint2(w0, w1);
      }
    }
    if (lse != nullptr && hi == 0)
      lse[(long)(base + qpos) * hq + head] =
          m + __builtin_amdgcn_logf(l);
  }
}

}  // namespace attnv5

void launch_attn_v5(bf16_t* out, const bf16_t* q, const bf16_t* k,
                    const bf16_t* v, const int* seq_start, int num_seqs,
                    int max_len, int hq, int hkv, int d, float scale,
                    hipStream_t stream, float* lse) {
  const int tiles = (max_len + V5_QTILE - 1) / V5_QTILE;
  dim3 grid(num_seqs, hq, tiles);
  dim3 block(V5_WAVES * WAVE);
  if (d == 128) {
    attnv5::attn_v5_kernel<128, 64><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, lse);
  } else if (d == 64) {
    // CH=128 measured -21% at d=64 (fewer barriers did not pay for the
    // longer serial softmax span); CH stays 64
    attnv5::attn_v5_kernel<64, 64><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, lse);
  }
}

bool attn_v5_supported(int d) { return d == 128 || d == 64; }

namespace attnv5 {

// STATUS of the V5 backward pair: dq_v5 WINS (393 vs 504 us/call on the
// falcon MQA train shape) and is the default; dkv_v5 below is correct but
// measured 2.3x SLOWER than the 16-row dkv (2104 vs 918 us/call) despite
// the same structure winning on fwd/dq — its four chains (S, dP, dV, dK)
// push the packing/tr16 VALU volume past what the removed P round-trip
// buys, and occupancy drops to 3 waves/SIMD. Kept behind
// SUB_ATTN_DKV_V5=1 as a measured negative result; the shipped backward
// is delta + dq_v5 + 16-row dkv (bindings attn_train_bwd_v5).

// ---------------------------------------------------------------- dq V5
// Per wave ONE 32-row q block (like the forward): Q and dO register-
// resident, S^T = K·Q^T / dP^T = V·dO^T on 32x32x16 MFMA (per-lane
// q-column -> lse/delta are lane scalars), dS^T assembled to B-fragments
// in-register (cvt_pk + permlane32_swap), dQ^T accumulated transposed
// with K^T via tr16 images — no P/dS round-trip through LDS.
template <int D>
__global__ __launch_bounds__(V5_WAVES * WAVE)
void attn_bwd_dq_v5_kernel(bf16_t* __restrict__ dq,       // [T, HQ, D]
                           const bf16_t* __restrict__ q,
                           const bf16_t* __restrict__ k,
                           const bf16_t* __restrict__ v,
                           const bf16_t* __restrict__ dout,
                           const float* __restrict__ lse,   // [T, HQ]
                           const float* __restrict__ delta, // [T, HQ]
                           const int* __restrict__ seq_start,
                           int hq, int hkv, float scale) {
  constexpr int KK = D / 16;
  constexpr int DC = D / 32;
  const int s = blockIdx.x;
  const int head = blockIdx.y;
  const int tile = blockIdx.z;
  const int kvh = head / (hq / hkv);
  const int base = seq_start[s];
  const int n = seq_start[s + 1] - base;
  const int q0_tile = tile * V5_QTILE;
  if (q0_tile >= n) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lo5 = lane & 31;
  const int hi = lane >> 5;
  const int q0 = q0_tile + wave * 32;
  const bool wave_live = q0 < n;
  const int qpos = q0 + lo5;

  const long qrow_stride = (long)hq * D;
  const long krow_stride = (long)hkv * D;

  short8 qreg[KK], doreg[KK];
  float lse2, dlt;
  {
    const int qrow = min(qpos, n - 1);
    const long roff = (long)(base + qrow) * qrow_stride + (long)head * D;
    #pragma unroll
    for (int kk = 0; kk < KK; ++kk) {
      qreg[kk] = *reinterpret_cast<const short8*>(q + roff + kk * 16 + hi * 8);
      doreg[kk] = *reinterpret_cast<const short8*>(
          dout + roff + kk * 16 + hi * 8);
    }
    lse2 = lse[(long)(base + qrow) * hq + head];
    dlt = delta[(long)(base + qrow) * hq + head];
  }

  float16v dq_t[DC];     // dQ^T[32 d rows][q col = lo5]
  #pragma unroll
  for (int c = 0; c < DC; ++c) dq_t[c] = (float16v){};

  __shared__ uint4 k_tile[V5_KCHUNK * D * 2 / 16];
  __shared__ uint4 v_tile[V5_KCHUNK * D * 2 / 16];
  __shared__ short k_img[DC][2][2][2][16][16];   // tr16 image of K

  const int kmax_tile = min(q0_tile + V5_QTILE, n);
  const int nchunks = (kmax_tile + V5_KCHUNK - 1) / V5_KCHUNK;
  const float scale2 = scale * 1.44269504088896340736f;

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  for (int kc = 0; kc < nchunks; ++kc) {
    const int kbase = kc * V5_KCHUNK;
    {
      constexpr int LOADS = V5_KCHUNK * D / 8 / (V5_WAVES * WAVE);
      #pragma unroll
      for (int i = 0; i < LOADS; ++i) {
        const int flat = (i * V5_WAVES * WAVE + threadIdx.x) * 8;
        const int row = flat / D, col = flat % D;
        const bool ok = kbase + row < n;
        const long goff = (long)(base + kbase + row) * krow_stride +
                          (long)kvh * D + col;
        uint4 k4 = ok ? *reinterpret_cast<const uint4*>(k + goff)
                      : make_uint4(0, 0, 0, 0);
        k_tile[swz<D * 2>(row, col * 2 / 16)] = k4;
        const int sc = row / 32, g2 = (row % 32) / 16, kk = row % 16;
        const int sel = (kk % 8) >= 4;
        const int r = ((kk >= 8) ? 8 : 0) + ((col % 32) / 16) * 4 + (kk % 4);
        *reinterpret_cast<uint4*>(
            &k_img[col / 32][sc][g2][sel][r][col % 16]) = k4;
        uint4 v4 = ok ? *reinterpret_cast<const uint4*>(v + goff)
                      : make_uint4(0, 0, 0, 0);
        v_tile[swz<D * 2>(row, col * 2 / 16)] = v4;
      }
    }
    __syncthreads();
    if (wave_live && kbase <= min(q0 + 31, n - 1)) {
      #pragma unroll
      for (int sc = 0; sc < 2; ++sc) {
        float16v sacc = (float16v){};
        float16v dpacc = (float16v){};
        #pragma unroll
        for (int kk = 0; kk < KK; ++kk) {
          const int key = sc * 32 + lo5;
          const uint4 ka4 = k_tile[swz<D * 2>(key, kk * 2 + hi)];
          sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<const short8*>(&ka4), qreg[kk], sacc,
              0, 0, 0);
          const uint4 va4 = v_tile[swz<D * 2>(key, kk * 2 + hi)];
          dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<const short8*>(&va4), doreg[kk], dpacc,
              0, 0, 0);
        }
        // dS^T = P ∘ (dP − Δ)·scale, packed straight into B-frag dwords
        unsigned w[8];
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
          float dsv[2];
          #pragma unroll
          for (int t = 0; t < 2; ++t) {
            const int r = 2 * i + t;
            const int kpos = kbase + sc * 32 + (r & 3) + 8 * (r >> 2) +
                             4 * hi;
            const bool live = (kpos <= qpos) && (kpos < n) && (qpos < n);
            const float p = live
                ? __builtin_amdgcn_exp2f(sacc[r] * scale2 - lse2) : 0.f;
            dsv[t] = p * (dpacc[r] - dlt) * scale;
          }
          w[i] = cvt_pk_bf16(dsv[0], dsv[1]);
        }
        #pragma unroll
        for (int g = 0; g < 2; ++g) {
          auto r02 = __builtin_amdgcn_permlane32_swap(
              w[4 * g + 0], w[4 * g + 2], false, false);
          auto r13 = __builtin_amdgcn_permlane32_swap(
              w[4 * g + 1], w[4 * g + 3], false, false);
          uint4 pw = make_uint4(r02[0], r13[0], r02[1], r13[1]);
          const short8 pb = *reinterpret_cast<short8*>(&pw);
          #pragma unroll
          for (int c = 0; c < DC; ++c) {
            const short4v lo_ =
                tr16_read(&k_img[c][sc][g][0][0][0] + lane * 4);
            const short4v hi_ =
                tr16_read(&k_img[c][sc][g][1][0][0] + lane * 4);
            short8 kt;
            kt[0] = lo_[0]; kt[1] = lo_[1]; kt[2] = lo_[2]; kt[3] = lo_[3];
            kt[4] = hi_[0]; kt[5] = hi_[1]; kt[6] = hi_[2]; kt[7] = hi_[3];
            dq_t[c] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                kt, pb, dq_t[c], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  if (wave_live && qpos < n) {
    bf16_t* op = dq + (long)(base + qpos) * qrow_stride + (long)head * D;
    #pragma unroll
    for (int c = 0; c < DC; ++c) {
      #pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int d0 = c * 32 + 8 * rr + 4 * hi;
        const unsigned w0 = pack_bf16x2(dq_t[c][rr * 4 + 0],
                                        dq_t[c][rr * 4 + 1]);
        const unsigned w1 = pack_bf16x2(dq_t[c][rr * 4 + 2],
                                        dq_t[c][rr * 4 + 3]);
        *reinterpret_cast<uint2*>(op + d0) = make_uint2(w0, w1);
      }
    }
  }
}

}  // namespace attnv5

void launch_attn_bwd_dq_v5(bf16_t* dq, const bf16_t* q, const bf16_t* k,
                           const bf16_t* v, const bf16_t* dout,
                           const float* lse, const float* delta,
                           const int* seq_start, int num_seqs, int max_len,
                           int hq, int hkv, int d, float scale,
                           hipStream_t stream) {
  const int tiles = (max_len + V5_QTILE - 1) / V5_QTILE;
  dim3 grid(num_seqs, hq, tiles);
  dim3 block(V5_WAVES * WAVE);
  if (d == 128)
    attnv5::attn_bwd_dq_v5_kernel<128><<<grid, block, 0, stream>>>(
        dq, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
  else
    attnv5::attn_bwd_dq_v5_kernel<64><<<grid, block, 0, stream>>>(
        dq, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
}

namespace attnv5 {

// --------------------------------------------------------------- dkv V5
// Per wave ONE 32-row kv block: K and V register-resident (lane's key =
// l&31), q chunks streamed. S = Q·K^T / dP = dO·V^T (per-lane KEY-column;
// lse/delta come per C-ROW from an LDS-staged slice), P and dS packed
// in-register to B-fragments, dV^T = dO^T·P and dK^T = Q^T·dS accumulated
// transposed (per-lane key-scalar epilogue; GQA heads atomically add into
// the fp32 dk/dv buffers, grid is per Q head like the 16-row dkv).
template <int D, bool NOATOMIC = false>  // NOATOMIC: timing-only probe
__global__ __launch_bounds__(V5_WAVES * WAVE)
void attn_bwd_dkv_v5_kernel(float* __restrict__ dk,       // [T, HKV, D] f32
                            float* __restrict__ dv,
                            const bf16_t* __restrict__ q,
                            const bf16_t* __restrict__ kk_,
                            const bf16_t* __restrict__ v,
                            const bf16_t* __restrict__ dout,
                            const float* __restrict__ lse,
                            const float* __restrict__ delta,
                            const int* __restrict__ seq_start,
                            int hq, int hkv, float scale) {
  constexpr int KK = D / 16;
  constexpr int DC = D / 32;
  const int s = blockIdx.x;
  const int head = blockIdx.y;
  const int tile = blockIdx.z;
  const int grp = hq / hkv;
  const int kvh = head / grp;
  const int base = seq_start[s];
  const int n = seq_start[s + 1] - base;
  const int kv0_tile = tile * V5_QTILE;
  if (kv0_tile >= n) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lo5 = lane & 31;
  const int hi = lane >> 5;
  const int kv0 = kv0_tile + wave * 32;
  const bool wave_live = kv0 < n;
  const int kpos = kv0 + lo5;       // this lane's kv position

  const long qrow_stride = (long)hq * D;
  const long krow_stride = (long)hkv * D;

  short8 kreg[KK], vreg[KK];
  {
    const int krow = min(kpos, n - 1);
    const long roff = (long)(base + krow) * krow_stride + (long)kvh * D;
    #pragma unroll
    for (int c = 0; c < KK; ++c) {
      kreg[c] = *reinterpret_cast<const short8*>(kk_ + roff + c * 16 + hi * 8);
      vreg[c] = *reinterpret_cast<const short8*>(v + roff + c * 16 + hi * 8);
    }
  }

  float16v dk_t[DC], dv_t[DC];   // [32 d rows][key col = lo5]
  #pragma unroll
  for (int c = 0; c < DC; ++c) {
    dk_t[c] = (float16v){};
    dv_t[c] = (float16v){};
  }

  __shared__ uint4 q_tile[V5_KCHUNK * D * 2 / 16];
  __shared__ uint4 do_tile[V5_KCHUNK * D * 2 / 16];
  __shared__ short q_img[DC][2][2][2][16][16];
  __shared__ short do_img[DC][2][2][2][16][16];
  __shared__ float lse_s[V5_KCHUNK], dlt_s[V5_KCHUNK];

  const int qc_first = kv0_tile / V5_KCHUNK;   // 256-row tiles, 64-chunks
  const int nqchunks = (n + V5_KCHUNK - 1) / V5_KCHUNK;
  const float scale2 = scale * 1.44269504088896340736f;

  if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  for (int qc = qc_first; qc < nqchunks; ++qc) {
    const int qbase = qc * V5_KCHUNK;
    {
      constexpr int LOADS = V5_KCHUNK * D / 8 / (V5_WAVES * WAVE);
      #pragma unroll
      for (int i = 0; i < LOADS; ++i) {
        const int flat = (i * V5_WAVES * WAVE + threadIdx.x) * 8;
        const int row = flat / D, col = flat % D;
        const bool ok = qbase + row < n;
        const long goff = (long)(base + qbase + row) * qrow_stride +
                          (long)head * D + col;
        const int sc = row / 32, g2 = (row % 32) / 16, kk = row % 16;
        const int sel = (kk % 8) >= 4;
        const int r = ((kk >= 8) ? 8 : 0) + ((col % 32) / 16) * 4 + (kk % 4);
        uint4 q4 = ok ? *reinterpret_cast<const uint4*>(q + goff)
                      : make_uint4(0, 0, 0, 0);
        q_tile[swz<D * 2>(row, col * 2 / 16)] = q4;
        *reinterpret_cast<uint4*>(
            &q_img[col / 32][sc][g2][sel][r][col % 16]) = q4;
        uint4 d4 = ok ? *reinterpret_cast<const uint4*>(dout + goff)
                      : make_uint4(0, 0, 0, 0);
        do_tile[swz<D * 2>(row, col * 2 / 16)] = d4;
        *reinterpret_cast<uint4*>(
            &do_img[col / 32][sc][g2][sel][r][col % 16]) = d4;
      }
      // lse/delta slice for this q chunk (read per C-row later)
      if (threadIdx.x < V5_KCHUNK) {
        const int qr = min(qbase + threadIdx.x, n - 1);
        lse_s[threadIdx.x] = lse[(long)(base + qr) * hq + head];
        dlt_s[threadIdx.x] = delta[(long)(base + qr) * hq + head];
      }
    }
    __syncthreads();
    // causal: this wave's keys are kv0..kv0+31 — a q chunk entirely
    // below them is all-masked; skip it (wave-uniform)
    if (wave_live && qbase + V5_KCHUNK > kv0) {
      #pragma unroll
      for (int sc = 0; sc < 2; ++sc) {
        float16v sacc = (float16v){};
        float16v dpacc = (float16v){};
        #pragma unroll
        for (int kk = 0; kk < KK; ++kk) {
          const int qrow = sc * 32 + lo5;
          const uint4 qa4 = q_tile[swz<D * 2>(qrow, kk * 2 + hi)];
          sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<const short8*>(&qa4), kreg[kk], sacc,
              0, 0, 0);
          const uint4 da4 = do_tile[swz<D * 2>(qrow, kk * 2 + hi)];
          dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<const short8*>(&da4), vreg[kk], dpacc,
              0, 0, 0);
        }
        // C[q row][key col]: P and dS per reg, packed to B-frag dwords
        unsigned wp[8], wd[8];
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
          float pv[2], dsv[2];
          #pragma unroll
          for (int t = 0; t < 2; ++t) {
            const int r = 2 * i + t;
            const int qoff = sc * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
            const int qpos = qbase + qoff;
            const bool live = (qpos >= kpos) && (qpos < n) && (kpos < n);
            const float p = live
                ? __builtin_amdgcn_exp2f(sacc[r] * scale2 - lse_s[qoff])
                : 0.f;
            pv[t] = p;
            dsv[t] = p * (dpacc[r] - dlt_s[qoff]) * scale;
          }
          wp[i] = cvt_pk_bf16(pv[0], pv[1]);
          wd[i] = cvt_pk_bf16(dsv[0], dsv[1]);
        }
        #pragma unroll
        for (int g = 0; g < 2; ++g) {
          auto p02 = __builtin_amdgcn_permlane32_swap(
              wp[4 * g + 0], wp[4 * g + 2], false, false);
          auto p13 = __builtin_amdgcn_permlane32_swap(
              wp[4 * g + 1], wp[4 * g + 3], false, false);
          uint4 ppw = make_uint4(p02[0], p13[0], p02[1], p13[1]);
          const short8 pbP = *reinterpret_cast<short8*>(&ppw);
          auto d02 = __builtin_amdgcn_permlane32_swap(
              wd[4 * g + 0], wd[4 * g + 2], false, false);
          auto d13 = __builtin_amdgcn_permlane32_swap(
              wd[4 * g + 1], wd[4 * g + 3], false, false);
          uint4 dpw = make_uint4(d02[0], d13[0], d02[1], d13[1]);
          const short8 pbD = *reinterpret_cast<short8*>(&dpw);
          #pragma unroll
          for (int c = 0; c < DC; ++c) {
            short8 dot, qot;
            {
              const short4v lo_ =
                  tr16_read(&do_img[c][sc][g][0][0][0] + lane * 4);
              const short4v hi_ =
                  tr16_read(&do_img[c][sc][g][1][0][0] + lane * 4);
              dot[0] = lo_[0]; dot[1] = lo_[1]; dot[2] = lo_[2];
              dot[3] = lo_[3]; dot[4] = hi_[0]; dot[5] = hi_[1];
              dot[6] = hi_[2]; dot[7] = hi_[3];
            }
            dv_t[c] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                dot, pbP, dv_t[c], 0, 0, 0);
            {
              const short4v lo_ =
                  tr16_read(&q_img[c][sc][g][0][0][0] + lane * 4);
              const short4v hi_ =
                  tr16_read(&q_img[c][sc][g][1][0][0] + lane * 4);
              qot[0] = lo_[0]; qot[1] = lo_[1]; qot[2] = lo_[2];
              qot[3] = lo_[3]; qot[4] = hi_[0]; qot[5] = hi_[1];
              qot[6] = hi_[2]; qot[7] = hi_[3];
            }
            dk_t[c] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                qot, pbD, dk_t[c], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  if (wave_live && kpos < n) {
    const long roff = (long)(base + kpos) * krow_stride + (long)kvh * D;
    #pragma unroll
    for (int c = 0; c < DC; ++c) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = c * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        if (NOATOMIC || grp == 1) {
          dk[roff + d] = dk_t[c][r];
          dv[roff + d] = dv_t[c][r];
        } else {
          unsafeAtomicAdd(&dk[roff + d], dk_t[c][r]);
          unsafeAtomicAdd(&dv[roff + d], dv_t[c][r]);
        }
      }
    }
  }
}

}  // namespace attnv5

void launch_attn_bwd_dkv_v5(float* dk, float* dv, const bf16_t* q,
                            const bf16_t* k, const bf16_t* v,
                            const bf16_t* dout, const float* lse,
                            const float* delta, const int* seq_start,
                            int num_seqs, int max_len, int hq, int hkv,
                            int d, float scale, hipStream_t stream) {
  const int tiles = (max_len + V5_QTILE - 1) / V5_QTILE;
  dim3 grid(num_seqs, hq, tiles);
  dim3 block(V5_WAVES * WAVE);
  const char* na = getenv("SUB_DKV_NOATOMIC");   // timing-only probe
  const bool noat = na && na[0] == '1';
  if (d == 128) {
    if (noat)
      attnv5::attn_bwd_dkv_v5_kernel<128, true><<<grid, block, 0,
          stream>>>(dk, dv, q, k, v, dout, lse, delta, seq_start, hq,
                    hkv, scale);
    else
      attnv5::attn_bwd_dkv_v5_kernel<128><<<grid, block, 0, stream>>>(
          dk, dv, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
  } else {
    if (noat)
      attnv5::attn_bwd_dkv_v5_kernel<64, true><<<grid, block, 0,
          stream>>>(dk, dv, q, k, v, dout, lse, delta, seq_start, hq,
                    hkv, scale);
    else
      attnv5::attn_bwd_dkv_v5_kernel<64><<<grid, block, 0, stream>>>(
          dk, dv, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
  }
}
