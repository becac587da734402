// Paged-KV decode attention (SURVEY.md §2.5 ledger item 1) — the serving hot
// kernel: per sequence and KV head, softmax(q·K^T/√d)·V over a block-paged
// cache, one new token per sequence.
//
// Decode at batch B is HBM-bound on the K/V stream (e.g. llama2-7b @ctx 512,
// B=64: ~16 GB of KV per step vs 13.5 GB of weights). Structure:
//
//   * grid = (num_seqs, n_kv_heads, ceil(group/GC) × nsplit); 4 waves per
//     workgroup; each wave owns KV blocks stride NW×nsplit (flash-decoding
//     split over the sequence, combined by a reduce kernel when nsplit > 1).
//   * K panel row [16, D]: D/8 lanes per token (uint4 of bf16, or uint2 of
//     fp8 — same fragment map at half the bytes). K tiles are PREFETCHED one
//     block ahead into registers (guide T14 issue-early/consume-late), so
//     the score phase runs entirely from registers; nt loads keep the
//     read-once stream out of L1.
//   * Scores reduced via shfl within the lane group; online softmax per
//     wave; P·V with dims distributed over lanes; cross-wave merge via LDS.
//
// KV8 (opt-in fp8 e4m3 cache): gfx950's v_cvt_f32_fp8 decodes inline —
// halves the dominant KV stream at a documented precision cost (the cache is
// written by rope_kv_append's fp8 path; scale 1.0).
//
// Cache layout: [num_blocks, n_kv_heads, BS=16, D]; block_table [B, maxb]
// int32; seq_lens [B] int32 counts INCLUDING the current token.
#include "common.h"

#define BS 16   // tokens per KV block
#define NW 4    // waves per workgroup

template <bool KV8>
struct KTile {  // one lane's 8 dims of one token row
  uint4v u;
  DEV_INLINE void load(const void* base, long off_elems) {
    u = nt_load_u4(reinterpret_cast<const bf16_t*>(base) + off_elems);
  }
  DEV_INLINE void unpack(float f[8]) const {
    float2 f0 = unpack_bf16x2(u[0]), f1 = unpack_bf16x2(u[1]);
    float2 f2 = unpack_bf16x2(u[2]), f3 = unpack_bf16x2(u[3]);
    f[0] = f0.x; f[1] = f0.y; f[2] = f1.x; f[3] = f1.y;
    f[4] = f2.x; f[5] = f2.y; f[6] = f3.x; f[7] = f3.y;
  }
};

template <>
struct KTile<true> {
  uint2v u;
  DEV_INLINE void load(const void* base, long off_elems) {
    u = nt_load_u2(reinterpret_cast<const unsigned char*>(base) + off_elems);
  }
  DEV_INLINE void unpack(float f[8]) const {
    f[0] = fp8_to_f32<0>(u[0]); f[1] = fp8_to_f32<1>(u[0]);
    f[2] = fp8_to_f32<2>(u[0]); f[3] = fp8_to_f32<3>(u[0]);
    f[4] = fp8_to_f32<0>(u[1]); f[5] = fp8_to_f32<1>(u[1]);
    f[6] = fp8_to_f32<2>(u[1]); f[7] = fp8_to_f32<3>(u[1]);
  }
};

// one lane's DPL V dims of one token (vbase is a BYTE pointer at the block's
// head panel; t rows of d_total elements)
template <bool KV8, int DPL>
DEV_INLINE void load_v(const char* vbase, int t, int d_total, int lane,
                       float out[DPL]) {
  if (!KV8) {
    const bf16_t* p = reinterpret_cast<const bf16_t*>(vbase) +
                      (long)t * d_total;
    if (DPL == 2) {
      const unsigned int uv = __builtin_nontemporal_load(
          reinterpret_cast<const unsigned int*>(p) + lane);
      float2 vf = unpack_bf16x2(uv);
      out[0] = vf.x;
      out[DPL - 1] = vf.y;
    } else {
      out[0] = bf2f(p[lane]);
    }
  } else {
    const unsigned char* p = reinterpret_cast<const unsigned char*>(vbase) +
                             (long)t * d_total;
    if (DPL == 2) {
      const unsigned short us = *reinterpret_cast<const unsigned short*>(
          p + lane * 2);
      out[0] = fp8_to_f32<0>(us);
      out[DPL - 1] = fp8_to_f32<1>(us);
    } else {
      out[0] = fp8_to_f32<0>(p[lane]);
    }
  }
}

template <int D, int GC, bool KV8>
__global__ __launch_bounds__(NW * WAVE)
void paged_attn_decode_kernel(
    bf16_t* __restrict__ out,            // [B, HQ, D]
    const bf16_t* __restrict__ q,        // [B, HQ, D]
    const void* __restrict__ k_cache,    // [NB, HKV, BS, D] bf16 | fp8
    const void* __restrict__ v_cache,
    const int* __restrict__ block_table, // [B, max_blocks]
    const int* __restrict__ seq_lens,    // [B]
    int hq, int hkv, int max_blocks, float scale,
    long q_rstride,                       // token-row stride in elements
    int nsplit,
    float* __restrict__ part_o,           // [B, HQ, NS, D]  (nsplit > 1)
    float* __restrict__ part_ml) {        // [B, HQ, NS, 2]
  constexpr int LPT = D / 8;        // lanes per token (8 dims each)
  constexpr int TPP = WAVE / LPT;   // tokens per pass
  constexpr int PASSES = BS / TPP;
  constexpr int DPL = D / WAVE;     // accum dims per lane (1 or 2)
  static_assert(DPL >= 1 && DPL <= 2, "head_dim must be 64..128");

  const int b = blockIdx.x;
  const int h = blockIdx.y;           // kv head
  const int split = blockIdx.z % nsplit;
  const int gchunk = blockIdx.z / nsplit;
  const int group = hq / hkv;
  const int g0 = gchunk * GC;
  const int ng = min(GC, group - g0);
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int sub = lane / LPT;         // token-within-pass
  const int ll = lane % LPT;          // lane-within-token

  const int n = seq_lens[b];
  const int nblk = (n + BS - 1) / BS;

  // ---- q fragments ------------------------------------------------------
  float qf[GC][8];
  #pragma unroll
  for (int g = 0; g < GC; ++g) {
    #pragma unroll
    for (int j = 0; j < 8; ++j) qf[g][j] = 0.f;
    if (g < ng) {
      const uint4 u = reinterpret_cast<const uint4*>(
          q + (long)b * q_rstride + (long)(h * group + g0 + g) * D)[ll];
      float2 f0 = unpack_bf16x2(u.x), f1 = unpack_bf16x2(u.y);
      float2 f2 = unpack_bf16x2(u.z), f3 = unpack_bf16x2(u.w);
      qf[g][0] = f0.x; qf[g][1] = f0.y; qf[g][2] = f1.x; qf[g][3] = f1.y;
      qf[g][4] = f2.x; qf[g][5] = f2.y; qf[g][6] = f3.x; qf[g][7] = f3.y;
    }
  }

  float m[GC], l[GC], acc[GC][DPL];
  #pragma unroll
  for (int g = 0; g < GC; ++g) {
    m[g] = -INFINITY; l[g] = 0.f;
    #pragma unroll
    for (int j = 0; j < DPL; ++j) acc[g][j] = 0.f;
  }

  __shared__ float s_scores[NW][GC][BS];
  const int* bt = block_table + (long)b * max_blocks;
  const long head_off = (long)h * BS * D;       // elems
  const long blk_stride = (long)hkv * BS * D;   // elems

  // ---- K prefetch pipeline ---------------------------------------------
  const int wstride = NW * nsplit;
  const int w0 = split * NW + wave;
  KTile<KV8> kt[PASSES];
  int blk_cur = (w0 < nblk) ? bt[w0] : 0;
  if (w0 < nblk) {
    const long base = (long)blk_cur * blk_stride + head_off;
    #pragma unroll
    for (int p = 0; p < PASSES; ++p)
      kt[p].load(k_cache, base + (long)(p * TPP + sub) * D + ll * 8);
  }

  for (int bi = w0; bi < nblk; bi += wstride) {
    KTile<KV8> kt_next[PASSES];
    const int bnext = bi + wstride;
    int blk_next = 0;
    if (bnext < nblk) {
      blk_next = bt[bnext];
      const long base = (long)blk_next * blk_stride + head_off;
      #pragma unroll
      for (int p = 0; p < PASSES; ++p)
        kt_next[p].load(k_cache, base + (long)(p * TPP + sub) * D + ll * 8);
    }
    const int blk = blk_cur;

    // ---- scores from registers ----------------------------------------
    #pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      const int t = p * TPP + sub;
      const int tok_idx = bi * BS + t;
      float kf[8];
      kt[p].unpack(kf);
      #pragma unroll
      for (int g = 0; g < GC; ++g) {
        float s = qf[g][0] * kf[0] + qf[g][1] * kf[1] + qf[g][2] * kf[2] +
                  qf[g][3] * kf[3] + qf[g][4] * kf[4] + qf[g][5] * kf[5] +
                  qf[g][6] * kf[6] + qf[g][7] * kf[7];
        #pragma unroll
        for (int off = LPT / 2; off > 0; off >>= 1)
          s += __shfl_xor(s, off, WAVE);
        if (ll == 0)
          s_scores[wave][g][t] = (tok_idx < n) ? s * scale : -INFINITY;
      }
    }
    __builtin_amdgcn_wave_barrier();

    // ---- online softmax + P·V -----------------------------------------
    const char* vbase = reinterpret_cast<const char*>(v_cache) +
                        ((long)blk * blk_stride + head_off) * (KV8 ? 1 : 2);
    const int valid = min(n - bi * BS, BS);
    if (valid == BS) {
      float vv[BS][DPL];
      #pragma unroll
      for (int t = 0; t < BS; ++t)
        load_v<KV8, DPL>(vbase, t, D, lane, vv[t]);
      #pragma unroll
      for (int g = 0; g < GC; ++g) {
        float bm = -INFINITY;
        #pragma unroll
        for (int t = 0; t < BS; ++t) bm = fmaxf(bm, s_scores[wave][g][t]);
        const float mn = fmaxf(m[g], bm);
        const float alpha = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - mn);
        m[g] = mn;
        l[g] *= alpha;
        #pragma unroll
        for (int j = 0; j < DPL; ++j) acc[g][j] *= alpha;
        #pragma unroll
        for (int t = 0; t < BS; ++t) {
          const float p = __expf(s_scores[wave][g][t] - mn);
          l[g] += p;
          #pragma unroll
          for (int j = 0; j < DPL; ++j) acc[g][j] += p * vv[t][j];
        }
      }
    } else {
      #pragma unroll
      for (int g = 0; g < GC; ++g) {
        float bm = -INFINITY;
        #pragma unroll
        for (int t = 0; t < BS; ++t) bm = fmaxf(bm, s_scores[wave][g][t]);
        if (bm == -INFINITY) continue;
        const float mn = fmaxf(m[g], bm);
        const float alpha = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - mn);
        m[g] = mn;
        l[g] *= alpha;
        #pragma unroll
        for (int j = 0; j < DPL; ++j) acc[g][j] *= alpha;
        for (int t = 0; t < valid; ++t) {
          const float p = __expf(s_scores[wave][g][t] - mn);
          l[g] += p;
          float vvt[DPL];
          load_v<KV8, DPL>(vbase, t, D, lane, vvt);
          #pragma unroll
          for (int j = 0; j < DPL; ++j) acc[g][j] += p * vvt[j];
        }
      }
    }
    blk_cur = blk_next;
    #pragma unroll
    for (int p = 0; p < PASSES; ++p) kt[p] = kt_next[p];
  }

  // ---- cross-wave merge via LDS ----------------------------------------
  __shared__ float red_o[NW][GC][D];
  __shared__ float red_m[NW][GC], red_l[NW][GC];
  #pragma unroll
  for (int g = 0; g < GC; ++g) {
    if (lane == 0) { red_m[wave][g] = m[g]; red_l[wave][g] = l[g]; }
    #pragma unroll
    for (int j = 0; j < DPL; ++j)
      red_o[wave][g][lane * DPL + j] = acc[g][j];
  }
  __syncthreads();
  if (wave == 0) {
    #pragma unroll
    for (int g = 0; g < GC; ++g) {
      if (g >= ng) continue;
      float M = -INFINITY;
      #pragma unroll
      for (int w = 0; w < NW; ++w) M = fmaxf(M, red_m[w][g]);
      float L = 0.f;
      float o[DPL];
      #pragma unroll
      for (int j = 0; j < DPL; ++j) o[j] = 0.f;
      #pragma unroll
      for (int w = 0; w < NW; ++w) {
        const float f = (red_m[w][g] == -INFINITY) ? 0.f
                                                   : __expf(red_m[w][g] - M);
        L += f * red_l[w][g];
        #pragma unroll
        for (int j = 0; j < DPL; ++j)
          o[j] += f * red_o[w][g][lane * DPL + j];
      }
      const int qhead = h * group + g0 + g;
      if (nsplit == 1) {
        const float invl = (L > 0.f) ? 1.f / L : 0.f;
        bf16_t* orow = out + ((long)b * hq + qhead) * D;
        if (DPL == 2) {
          reinterpret_cast<unsigned int*>(orow)[lane] =
              pack_bf16x2(o[0] * invl, o[1] * invl);
        } else {
          orow[lane] = f2bf(o[0] * invl);
        }
      } else {
        float* po = part_o + (((long)b * hq + qhead) * nsplit + split) * D;
        #pragma unroll
        for (int j = 0; j < DPL; ++j) po[lane * DPL + j] = o[j];
        if (lane == 0) {
          float* pm = part_ml + (((long)b * hq + qhead) * nsplit + split) * 2;
          pm[0] = M;
          pm[1] = L;
        }
      }
    }
  }
}

// one lane's DPL V dims of token t from an LDS tile (raw cache layout)
template <bool KV8, int DPL>
DEV_INLINE void lds_read_v(const char* vb, int t, int d, int lane,
                           float out[DPL]) {
  if (!KV8) {
    if (DPL == 2) {
      const float2 v2 = unpack_bf16x2(
          *reinterpret_cast<const unsigned int*>(vb + (t * d + lane * 2) * 2));
      out[0] = v2.x;
      out[DPL - 1] = v2.y;
    } else {
      out[0] = bf2f(*reinterpret_cast<const bf16_t*>(vb + (t * d + lane) * 2));
    }
  } else {
    if (DPL == 2) {
      const unsigned short us =
          *reinterpret_cast<const unsigned short*>(vb + t * d + lane * 2);
      out[0] = fp8_to_f32<0>(us);
      out[DPL - 1] = fp8_to_f32<1>(us);
    } else {
      out[0] = fp8_to_f32<0>(
          *reinterpret_cast<const unsigned char*>(vb + t * d + lane));
    }
  }
}

// GQA decode variant: one workgroup per (seq, kv head, split); K/V staged
// ONCE per 16-token block through double-buffered LDS tiles; each wave owns
// group/NW q heads over the FULL (split-strided) context, so there is no
// cross-wave merge at all. Replaces the GC=8 register-tiled path for GQA
// groups 4/8/16: that variant's qf/vv/kt register footprint compiled to
// 256 VGPRs + 75 AGPRs = occupancy 1 wave/SIMD (-Rpass-analysis), and 4
// resident waves/CU cannot hide HBM latency — measured 0.5 TB/s of KV
// stream on llama2-70b decode vs 6.4 TB/s on the MHA path. This shape
// keeps the staging registers to one uint4 pair per thread (occupancy >=5)
// and still reads each KV byte exactly once per workgroup.
//
// LDS bank behavior (no swizzle needed): ds_read/write phase groups of 16
// lanes always touch 16 distinct 16B chunks of the SAME 256B token row
// (D=128 bf16), covering all 64 banks; the P*V phase reads 4B per lane
// across one row = conflict-free by construction.
template <int D, int HPW, bool KV8>
__global__ __launch_bounds__(NW * WAVE)
// NOTE: amdgpu_waves_per_eu(5) was measured: it caps to 96 VGPRs but
// spills 13 VGPRs to scratch, and the spill fills in the hot loop cost
// far more than the 5th wave bought (2.45 -> 1.53 TB/s on llama2-70b
// shapes). 112 VGPRs / 4 waves/SIMD is the measured optimum.
void paged_attn_decode_gqa_kernel(
    bf16_t* __restrict__ out,            // [B, HQ, D]
    const bf16_t* __restrict__ q,        // [B, HQ, D]
    const void* __restrict__ k_cache,    // [NB, HKV, BS, D] bf16 | fp8
    const void* __restrict__ v_cache,
    const int* __restrict__ block_table, // [B, max_blocks]
    const int* __restrict__ seq_lens,    // [B]
    int hq, int hkv, int max_blocks, float scale,
    long q_rstride, int nsplit,
    float* __restrict__ part_o,           // [B, HQ, NS, D]  (nsplit > 1)
    float* __restrict__ part_ml) {        // [B, HQ, NS, 2]
  constexpr int ESZ = KV8 ? 1 : 2;       // cache element bytes
  constexpr int TILE = BS * D * ESZ;     // bytes per K (or V) tile
  constexpr int CHUNKS = TILE / 16;      // uint4 staging chunks
  constexpr int CPR = D * ESZ / 16;      // chunks per token row
  constexpr int LPT = D / 8;             // lanes per token (8 dims each)
  constexpr int TPP = WAVE / LPT;        // tokens per pass
  constexpr int PASSES = BS / TPP;
  constexpr int DPL = D / WAVE;          // accum dims per lane (1 or 2)

  const int b = blockIdx.x;
  const int h = blockIdx.y;              // kv head
  const int split = blockIdx.z;
  const int group = hq / hkv;            // == NW * HPW
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int sub = lane / LPT;
  const int ll = lane % LPT;
  const int tid = threadIdx.x;

  const int n = seq_lens[b];
  const int nblk = (n + BS - 1) / BS;

  // q fragments: lane ll's 8-dim slice of each owned head
  float qf[HPW][8];
  #pragma unroll
  for (int i = 0; i < HPW; ++i) {
    const uint4v u = *reinterpret_cast<const uint4v*>(
        q + (long)b * q_rstride + (long)(h * group + wave * HPW + i) * D +
        ll * 8);
    const float2 f0 = unpack_bf16x2(u[0]), f1 = unpack_bf16x2(u[1]);
    const float2 f2 = unpack_bf16x2(u[2]), f3 = unpack_bf16x2(u[3]);
    qf[i][0] = f0.x; qf[i][1] = f0.y; qf[i][2] = f1.x; qf[i][3] = f1.y;
    qf[i][4] = f2.x; qf[i][5] = f2.y; qf[i][6] = f3.x; qf[i][7] = f3.y;
  }
  float m[HPW], l[HPW], acc[HPW][DPL];
  #pragma unroll
  for (int i = 0; i < HPW; ++i) {
    m[i] = -INFINITY; l[i] = 0.f;
    #pragma unroll
    for (int j = 0; j < DPL; ++j) acc[i][j] = 0.f;
  }

  __shared__ uint4v kts[2][CHUNKS];
  __shared__ uint4v vts[2][CHUNKS];
  __shared__ float s_sc[NW][HPW][BS];

  const int* bt = block_table + (long)b * max_blocks;
  const long head_off = (long)h * BS * D;       // elems
  const long blk_stride = (long)hkv * BS * D;   // elems

  // stage the first block (V rows past n zero-filled: their P weight is 0
  // via the -inf score, but 0 * garbage-decoded-NaN would still poison acc)
  if (split < nblk && tid < CHUNKS) {
    const long boff = ((long)bt[split] * blk_stride + head_off) * ESZ;
    kts[0][tid] = nt_load_u4(
        reinterpret_cast<const char*>(k_cache) + boff + tid * 16);
    uint4v rv = nt_load_u4(
        reinterpret_cast<const char*>(v_cache) + boff + tid * 16);
    if (split * BS + tid / CPR >= n) rv = 0;
    vts[0][tid] = rv;
  }
  __syncthreads();

  int cur = 0;
  for (int bi = split; bi < nblk; bi += nsplit) {
    // prefetch the next block into registers while computing this one
    const int bn = bi + nsplit;
    uint4v rk = 0, rv = 0;
    if (bn < nblk && tid < CHUNKS) {
      const long boff = ((long)bt[bn] * blk_stride + head_off) * ESZ;
      rk = nt_load_u4(
          reinterpret_cast<const char*>(k_cache) + boff + tid * 16);
      if (bn * BS + tid / CPR < n)
        rv = nt_load_u4(
            reinterpret_cast<const char*>(v_cache) + boff + tid * 16);
    }

    // ---- scores from the LDS K tile -----------------------------------
    const char* kb = reinterpret_cast<const char*>(kts[cur]);
    #pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      const int t = p * TPP + sub;
      const int tok_idx = bi * BS + t;
      float kf[8];
      if (!KV8) {
        const uint4v u = *reinterpret_cast<const uint4v*>(
            kb + (t * D + ll * 8) * 2);
        const float2 f0 = unpack_bf16x2(u[0]), f1 = unpack_bf16x2(u[1]);
        const float2 f2 = unpack_bf16x2(u[2]), f3 = unpack_bf16x2(u[3]);
        kf[0] = f0.x; kf[1] = f0.y; kf[2] = f1.x; kf[3] = f1.y;
        kf[4] = f2.x; kf[5] = f2.y; kf[6] = f3.x; kf[7] = f3.y;
      } else {
        const uint2v u = *reinterpret_cast<const uint2v*>(
            kb + (t * D + ll * 8));
        kf[0] = fp8_to_f32<0>(u[0]); kf[1] = fp8_to_f32<1>(u[0]);
        kf[2] = fp8_to_f32<2>(u[0]); kf[3] = fp8_to_f32<3>(u[0]);
        kf[4] = fp8_to_f32<0>(u[1]); kf[5] = fp8_to_f32<1>(u[1]);
        kf[6] = fp8_to_f32<2>(u[1]); kf[7] = fp8_to_f32<3>(u[1]);
      }
      #pragma unroll
      for (int i = 0; i < HPW; ++i) {
        float s = qf[i][0] * kf[0] + qf[i][1] * kf[1] + qf[i][2] * kf[2] +
                  qf[i][3] * kf[3] + qf[i][4] * kf[4] + qf[i][5] * kf[5] +
                  qf[i][6] * kf[6] + qf[i][7] * kf[7];
        #pragma unroll
        for (int off = LPT / 2; off > 0; off >>= 1)
          s += __shfl_xor(s, off, WAVE);
        if (ll == 0)
          s_sc[wave][i][t] = (tok_idx < n) ? s * scale : -INFINITY;
      }
    }
    __builtin_amdgcn_wave_barrier();

    // ---- online softmax + P·V from the LDS V tile ---------------------
    // even/odd partial accumulators: the naive l/acc updates are 16-deep
    // serial expf->FMA chains per head; two chains of 8 double the ILP.
    const char* vb = reinterpret_cast<const char*>(vts[cur]);
    #pragma unroll
    for (int i = 0; i < HPW; ++i) {
      float bm = -INFINITY;
      #pragma unroll
      for (int t = 0; t < BS; ++t) bm = fmaxf(bm, s_sc[wave][i][t]);
      if (bm == -INFINITY) continue;
      const float mn = fmaxf(m[i], bm);
      const float alpha = (m[i] == -INFINITY) ? 0.f : __expf(m[i] - mn);
      m[i] = mn;
      float l0 = 0.f, l1 = 0.f;
      float a0[DPL], a1[DPL];
      #pragma unroll
      for (int j = 0; j < DPL; ++j) { a0[j] = 0.f; a1[j] = 0.f; }
      #pragma unroll
      for (int t = 0; t < BS; t += 2) {
        const float p0 = __expf(s_sc[wave][i][t] - mn);
        const float p1 = __expf(s_sc[wave][i][t + 1] - mn);
        l0 += p0;
        l1 += p1;
        float vf0[DPL], vf1[DPL];
        lds_read_v<KV8, DPL>(vb, t, D, lane, vf0);
        lds_read_v<KV8, DPL>(vb, t + 1, D, lane, vf1);
        #pragma unroll
        for (int j = 0; j < DPL; ++j) {
          a0[j] += p0 * vf0[j];
          a1[j] += p1 * vf1[j];
        }
      }
      l[i] = l[i] * alpha + l0 + l1;
      #pragma unroll
      for (int j = 0; j < DPL; ++j)
        acc[i][j] = acc[i][j] * alpha + a0[j] + a1[j];
    }

    // writes target the OTHER buffer, so no barrier is needed between the
    // compute above and these stores; one barrier orders them against the
    // next iteration's reads.
    if (bn < nblk && tid < CHUNKS) {
      kts[cur ^ 1][tid] = rk;
      vts[cur ^ 1][tid] = rv;
    }
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: each q head lives entirely in one wave ----------------
  #pragma unroll
  for (int i = 0; i < HPW; ++i) {
    const int qhead = h * group + wave * HPW + i;
    if (nsplit == 1) {
      const float invl = (l[i] > 0.f) ? 1.f / l[i] : 0.f;
      bf16_t* orow = out + ((long)b * hq + qhead) * D;
      if (DPL == 2) {
        reinterpret_cast<unsigned int*>(orow)[lane] =
            pack_bf16x2(acc[i][0] * invl, acc[i][DPL - 1] * invl);
      } else {
        orow[lane] = f2bf(acc[i][0] * invl);
      }
    } else {
      float* po = part_o + (((long)b * hq + qhead) * nsplit + split) * D;
      #pragma unroll
      for (int j = 0; j < DPL; ++j) po[lane * DPL + j] = acc[i][j];
      if (lane == 0) {
        float* pm = part_ml + (((long)b * hq + qhead) * nsplit + split) * 2;
        pm[0] = m[i];
        pm[1] = l[i];
      }
    }
  }
}

// GQA heads-per-wave for the LDS-staged decode variant; 0 = use the
// register-tiled kernel (MHA group 1, falcon's prime group 71, d oddballs).
static inline int gqa_hpw(int hq, int hkv, int d) {
  if (d != 64 && d != 128) return 0;
  const int group = hq / hkv;
  if (group == 4) return 1;
  if (group == 8) return 2;
  if (group == 16) return 4;
  return 0;
}

// flash-decoding combine: one wave per (b, qhead), merge nsplit partials.
template <int D>
__global__ __launch_bounds__(WAVE)
void paged_attn_reduce_kernel(bf16_t* __restrict__ out,      // [B, HQ, D]
                              const float* __restrict__ part_o,
                              const float* __restrict__ part_ml,
                              int hq, int nsplit) {
  constexpr int DPL = D / WAVE;
  const int b = blockIdx.x;
  const int qh = blockIdx.y;
  const int lane = threadIdx.x;
  const long base = ((long)b * hq + qh) * nsplit;
  float M = -INFINITY;
  for (int s = 0; s < nsplit; ++s) M = fmaxf(M, part_ml[(base + s) * 2]);
  float L = 0.f;
  float o[DPL];
  #pragma unroll
  for (int j = 0; j < DPL; ++j) o[j] = 0.f;
  for (int s = 0; s < nsplit; ++s) {
    const float pm = part_ml[(base + s) * 2];
    const float f = (pm == -INFINITY) ? 0.f : __expf(pm - M);
    L += f * part_ml[(base + s) * 2 + 1];
    const float* po = part_o + (base + s) * D;
    #pragma unroll
    for (int j = 0; j < DPL; ++j) o[j] += f * po[lane * DPL + j];
  }
  const float invl = (L > 0.f) ? 1.f / L : 0.f;
  bf16_t* orow = out + ((long)b * hq + qh) * D;
  if (DPL == 2) {
    reinterpret_cast<unsigned int*>(orow)[lane] =
        pack_bf16x2(o[0] * invl, o[1] * invl);
  } else {
    orow[lane] = f2bf(o[0] * invl);
  }
}

int paged_attn_nsplit(int bsz, int hq, int hkv, int d) {
  // flash-decoding split factor: fill >=1024 workgroups (256 CUs x 4
  // resident WGs) when the (B, Hkv, group-chunk) grid alone is too small
  // (llama2-70b at TP=8: B x 1 x 1). Shape-only -> hipGraph-stable.
  const int group = hq / hkv;
  int gchunks;
  if (gqa_hpw(hq, hkv, d) > 0) {
    gchunks = 1;                       // whole group lives in one block
  } else {
    const int gc = (group % 8 == 0) ? 8 : 1;
    gchunks = (group + gc - 1) / gc;
  }
  const long base = (long)bsz * hkv * gchunks;
  if (base >= 1024) return 1;
  int ns = (int)((1024 + base - 1) / base);
  if (ns > 16) ns = 16;
  return ns;
}

void launch_paged_attn_decode(bf16_t* out, const bf16_t* q,
                              const void* k_cache, const void* v_cache,
                              const int* block_table, const int* seq_lens,
                              int bsz, int hq, int hkv, int d, int max_blocks,
                              int block_size, float scale, long q_rstride,
                              int nsplit, float* part_o, float* part_ml,
                              bool kv8, hipStream_t stream) {
  const int group = hq / hkv;
  const int hpw = gqa_hpw(hq, hkv, d);
  dim3 block(NW * WAVE);
  if (hpw > 0) {
    // LDS-staged GQA path: groups 4/8/16 (llama2/3-70b, GQA-16 d=64)
    dim3 grid(bsz, hkv, nsplit);
    #define LAUNCHG(DD, HH, K8) \
      paged_attn_decode_gqa_kernel<DD, HH, K8><<<grid, block, 0, stream>>>( \
          out, q, k_cache, v_cache, block_table, seq_lens, hq, hkv, \
          max_blocks, scale, q_rstride, nsplit, part_o, part_ml)
    if (!kv8) {
      if (d == 128 && hpw == 1) LAUNCHG(128, 1, false);
      else if (d == 128 && hpw == 2) LAUNCHG(128, 2, false);
      else if (d == 128) LAUNCHG(128, 4, false);
      else if (hpw == 1) LAUNCHG(64, 1, false);
      else if (hpw == 2) LAUNCHG(64, 2, false);
      else LAUNCHG(64, 4, false);
    } else {
      if (d == 128 && hpw == 1) LAUNCHG(128, 1, true);
      else if (d == 128 && hpw == 2) LAUNCHG(128, 2, true);
      else if (d == 128) LAUNCHG(128, 4, true);
      else if (hpw == 1) LAUNCHG(64, 1, true);
      else if (hpw == 2) LAUNCHG(64, 2, true);
      else LAUNCHG(64, 4, true);
    }
    #undef LAUNCHG
  } else {
  // GC=8 when the GQA group tiles evenly by 8 but the LDS path is out
  // (non-power-of-two d); GC=1 otherwise (MHA group=1; falcon's prime
  // group=71 — its MQA KV is tiny and L3-resident, so per-head K
  // re-reads are cheap).
  const int gc = (group % 8 == 0) ? 8 : 1;
  const int gchunks = (group + gc - 1) / gc;
  dim3 grid(bsz, hkv, gchunks * nsplit);
  #define LAUNCH(DD, GG, K8) \
    paged_attn_decode_kernel<DD, GG, K8><<<grid, block, 0, stream>>>( \
        out, q, k_cache, v_cache, block_table, seq_lens, hq, hkv, \
        max_blocks, scale, q_rstride, nsplit, part_o, part_ml)
  if (!kv8) {
    if (d == 128 && gc == 8) LAUNCH(128, 8, false);
    else if (d == 128) LAUNCH(128, 1, false);
    else if (d == 64 && gc == 8) LAUNCH(64, 8, false);
    else if (d == 64) LAUNCH(64, 1, false);
  } else {
    if (d == 128 && gc == 8) LAUNCH(128, 8, true);
    else if (d == 128) LAUNCH(128, 1, true);
    else if (d == 64 && gc == 8) LAUNCH(64, 8, true);
    else if (d == 64) LAUNCH(64, 1, true);
  }
  #undef LAUNCH
  }
  if (nsplit > 1) {
    dim3 rgrid(bsz, hq);
    if (d == 128) {
      paged_attn_reduce_kernel<128><<<rgrid, dim3(WAVE), 0, stream>>>(
          out, part_o, part_ml, hq, nsplit);
    } else {
      paged_attn_reduce_kernel<64><<<rgrid, dim3(WAVE), 0, stream>>>(
          out, part_o, part_ml, hq, nsplit);
    }
  }
}

bool paged_attn_decode_supported(int d, int block_size) {
  return (d == 128 || d == 64) && block_size == BS;
}
