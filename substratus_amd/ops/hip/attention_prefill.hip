// Varlen causal prefill attention (SURVEY.md §2.5 ledger item 2) —
// flash-style tiled attention on MFMA (mfma_f32_16x16x32_bf16) for prompt
// ingestion.
//
// Geometry: one workgroup (4 waves) per (sequence, q-head, 64-row q tile);
// wave w owns q rows [tile*64 + 16w, +16). Keys stream in chunks of 32:
//   * QK^T: Q A-fragments preloaded in registers (D/32 chunks × short8);
//     K^T B-fragments loaded straight from global (k is dense [T, Hkv, D]).
//     D/32 MFMAs per 16-key subchunk → S[16q × 16k] fp32 C-fragments.
//   * Online softmax in C-layout registers (row = (lane>>4)*4+i, col =
//     lane&15): 16-lane shfl row-max/row-sum, exp, rescale (guide §5
//     common-mistake #6: wave-parallel softmax).
//   * P transposed C-layout → A-fragment through a per-wave LDS tile;
//     V[32, D] staged cooperatively into LDS with coalesced uint4 loads;
//     P·V accumulates O[16q × D] as D/16 C-fragments.
//
// MFMA fragment maps (gfx950, 16x16x32 bf16; verified by the mfma_probe
// GPU test against torch.matmul on asymmetric data):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j]  (j = 0..7, short8)
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C/D: lane l holds C[row = (l>>4)*4 + i][col = l&15]  (i = 0..3, float4)
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float float4v;

#define PF_WAVES 4
#define QTILE 64   // q rows per workgroup (16 per wave)
#define KCHUNK 32  // keys per iteration

template <int D>
__global__ __launch_bounds__(PF_WAVES * WAVE)
void attn_prefill_kernel(bf16_t* __restrict__ out,      // [T, HQ, D]
                         const bf16_t* __restrict__ q,  // [T, HQ, D]
                         const bf16_t* __restrict__ k,  // [T, HKV, D]
                         const bf16_t* __restrict__ v,  // [T, HKV, D]
                         const int* __restrict__ seq_start,  // [S+1]
                         int hq, int hkv, float scale) {
  constexpr int DC32 = D / 32;   // Q/K fragment chunks (contraction dim)
  constexpr int DC16 = D / 16;   // O fragment chunks (output dim)
  const int s = blockIdx.x;
  const int head = blockIdx.y;
  const int tile = blockIdx.z;
  const int kvh = head / (hq / hkv);
  const int base = seq_start[s];
  const int n = seq_start[s + 1] - base;
  const int q0_tile = tile * QTILE;
  if (q0_tile >= n) return;  // whole tile beyond this sequence: all waves exit

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lhi = lane >> 4;      // 0..3
  const int llo = lane & 15;      // 0..15
  const int q0 = q0_tile + wave * 16;  // this wave's first q row (may be >= n)

  const long qrow_stride = (long)hq * D;
  const long krow_stride = (long)hkv * D;

  // ---- preload Q A-fragments ------------------------------------------
  short8 qa[DC32];
  const bool wave_live = q0 < n;
  {
    const int qrow = min(q0 + llo, n - 1);  // clamp; masked rows never used
    const bf16_t* qp = q + (long)(base + qrow) * qrow_stride + (long)head * D;
    #pragma unroll
    for (int c = 0; c < DC32; ++c) {
      // 8 contiguous bf16 at d = c*32 + lhi*8
      qa[c] = *reinterpret_cast<const short8*>(qp + c * 32 + lhi * 8);
    }
  }

  // ---- state ------------------------------------------------------------
  float m[4], l[4];             // per held q row (i = 0..3 → q = lhi*4+i)
  float4v o_acc[DC16];
  #pragma unroll
  for (int i = 0; i < 4; ++i) { m[i] = -INFINITY; l[i] = 0.f; }
  #pragma unroll
  for (int c = 0; c < DC16; ++c) o_acc[c] = (float4v){0.f, 0.f, 0.f, 0.f};

  __shared__ bf16_t v_tile[KCHUNK][D];                 // cooperative stage
  __shared__ bf16_t p_tile[PF_WAVES][16][KCHUNK];      // per-wave P transpose

  const int kmax_tile = min(q0_tile + QTILE, n);       // causal bound of tile
  const int nchunks = (kmax_tile + KCHUNK - 1) / KCHUNK;
  const int kmax_wave = min(q0 + 16, n);               // this wave's bound

  for (int kc = 0; kc < nchunks; ++kc) {
    const int kbase = kc * KCHUNK;
    // ---- cooperative V stage: 32 rows × D, uint4 per lane --------------
    __syncthreads();
    {
      constexpr int VLOADS = KCHUNK * D / 8 / (PF_WAVES * WAVE);  // uint4 each
      #pragma unroll
      for (int i = 0; i < VLOADS; ++i) {
        const int flat = (i * PF_WAVES * WAVE + threadIdx.x) * 8;  // bf16 idx
        const int row = flat / D, col = flat % D;
        uint4 val = make_uint4(0, 0, 0, 0);
        if (kbase + row < n) {
          val = *reinterpret_cast<const uint4*>(
              v + (long)(base + kbase + row) * krow_stride + (long)kvh * D + col);
        }
        *reinterpret_cast<uint4*>(&v_tile[row][col]) = val;
      }
    }
    __syncthreads();

    const bool compute = wave_live && kbase < kmax_wave;
    if (compute) {
      // ---- QK^T: two 16-key subchunks -------------------------------
      float4v sfrag[2];
      #pragma unroll
      for (int sc = 0; sc < 2; ++sc) {
        float4v acc = {0.f, 0.f, 0.f, 0.f};
        const int krow = kbase + sc * 16 + llo;  // B col = llo
        const bf16_t* kp = k + (long)(base + min(krow, n - 1)) * krow_stride +
                           (long)kvh * D;
        #pragma unroll
        for (int c = 0; c < DC32; ++c) {
          short8 kb = *reinterpret_cast<const short8*>(kp + c * 32 + lhi * 8);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa[c], kb, acc, 0, 0, 0);
        }
        sfrag[sc] = acc;
      }
      // ---- mask + online softmax (C layout) -------------------------
      float pm[2][4];
      float rowmax[4];
      #pragma unroll
      for (int i = 0; i < 4; ++i) rowmax[i] = -INFINITY;
      #pragma unroll
      for (int sc = 0; sc < 2; ++sc) {
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int qpos = q0 + lhi * 4 + i;
          const int kpos = kbase + sc * 16 + llo;
          float sv = (kpos <= qpos && kpos < n && qpos < n)
                         ? sfrag[sc][i] * scale : -INFINITY;
          pm[sc][i] = sv;
          rowmax[i] = fmaxf(rowmax[i], sv);
        }
      }
      // reduce rowmax across the 16 lanes holding this q row
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        #pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          rowmax[i] = fmaxf(rowmax[i], __shfl_xor(rowmax[i], off, WAVE));
      }
      float alpha[4];
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const float mn = fmaxf(m[i], rowmax[i]);
        alpha[i] = (m[i] == -INFINITY) ? 0.f : __expf(m[i] - mn);
        m[i] = mn;
      }
      // P = exp(S - m); row-sum; write transposed into p_tile
      float rowsum[4] = {0.f, 0.f, 0.f, 0.f};
      #pragma unroll
      for (int sc = 0; sc < 2; ++sc) {
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
          float pv = (pm[sc][i] == -INFINITY || m[i] == -INFINITY)
                         ? 0.f : __expf(pm[sc][i] - m[i]);
          rowsum[i] += pv;
          p_tile[wave][lhi * 4 + i][sc * 16 + llo] = f2bf(pv);
        }
      }
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        #pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          rowsum[i] += __shfl_xor(rowsum[i], off, WAVE);
        l[i] = l[i] * alpha[i] + rowsum[i];
      }
      // ---- rescale O ------------------------------------------------
      #pragma unroll
      for (int c = 0; c < DC16; ++c) {
        #pragma unroll
        for (int i = 0; i < 4; ++i) o_acc[c][i] *= alpha[i];
      }
      // ---- P·V: A = P [16q × 32k] from LDS, B = V [32k × 16d] from LDS
      short8 pa;  // A-fragment of P: row llo, k = lhi*8 + j
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        ((short*)&pa)[j] = *reinterpret_cast<const short*>(
            &p_tile[wave][llo][lhi * 8 + j]);
      #pragma unroll
      for (int c = 0; c < DC16; ++c) {
        short8 vb;  // B-fragment: k = lhi*8 + j, col = llo (d = c*16 + llo)
        #pragma unroll
        for (int j = 0; j < 8; ++j)
          ((short*)&vb)[j] = *reinterpret_cast<const short*>(
              &v_tile[lhi * 8 + j][c * 16 + llo]);
        o_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vb, o_acc[c],
                                                           0, 0, 0);
      }
    }
  }

  // ---- epilogue: out[q][head][d] = O / l --------------------------------
  if (wave_live) {
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int qpos = q0 + lhi * 4 + i;
      if (qpos >= n) continue;
      const float invl = (l[i] > 0.f) ? 1.f / l[i] : 0.f;
      bf16_t* op = out + (long)(base + qpos) * qrow_stride + (long)head * D;
      #pragma unroll
      for (int c = 0; c < DC16; ++c)
        op[c * 16 + llo] = f2bf(o_acc[c][i] * invl);
    }
  }
}

void launch_attn_prefill(bf16_t* out, const bf16_t* q, const bf16_t* k,
                         const bf16_t* v, const int* seq_start, int num_seqs,
                         int max_len, int hq, int hkv, int d, float scale,
                         hipStream_t stream) {
  const int tiles = (max_len + QTILE - 1) / QTILE;
  dim3 grid(num_seqs, hq, tiles);
  dim3 block(PF_WAVES * WAVE);
  if (d == 128) {
    attn_prefill_kernel<128><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale);
  } else if (d == 64) {
    attn_prefill_kernel<64><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale);
  }
}

bool attn_prefill_supported(int d) { return d == 128 || d == 64; }

// ---------------------------------------------------------------------------
// mfma_probe: D[16,16] = A[16,32]·B[32,16] via one mfma_f32_16x16x32_bf16 —
// the fragment-layout oracle (guide §3: always A=I-check with ASYMMETRIC B).
__global__ void mfma_probe_kernel(float* __restrict__ dout,
                                  const bf16_t* __restrict__ a,
                                  const bf16_t* __restrict__ b) {
  const int lane = threadIdx.x % WAVE;
  const int lhi = lane >> 4, llo = lane & 15;
  short8 af, bf;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    ((short*)&af)[j] = *reinterpret_cast<const short*>(&a[llo * 32 + lhi * 8 + j]);
    ((short*)&bf)[j] = *reinterpret_cast<const short*>(&b[(lhi * 8 + j) * 16 + llo]);
  }
  float4v acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
  #pragma unroll
  for (int i = 0; i < 4; ++i) dout[(lhi * 4 + i) * 16 + llo] = acc[i];
}

void launch_mfma_probe(float* dout, const bf16_t* a, const bf16_t* b,
                       hipStream_t stream) {
  mfma_probe_kernel<<<dim3(1), dim3(WAVE), 0, stream>>>(dout, a, b);
}
