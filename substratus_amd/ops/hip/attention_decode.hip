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

// ---------------------------------------------------------- MFMA GQA path
// The scalar GQA kernel above is ISSUE/LATENCY-bound, not memory-bound: it
// spends ~400 VALU+LDS instructions per wave per 8 KB KV block (8 q heads
// of dot/softmax/PV per byte) and measures 2.45 TB/s at occupancy 4 where
// the 1-head MHA path reaches 6.5. This variant puts the GQA group on the
// MFMA M dimension instead: S[group<=16, 32 tok] = Q·K^T from an
// XOR-swizzled row-major K tile (B-frags read conflict-free straight from
// LDS, same scheme as attention_bwd.hip's k_tile), P relayed through a
// per-wave tr16 image (ds_read_b64_tr_b16 A-frags, same as its ds_img),
// and O[group, D] += P·V with V staged as tr16 B-frag images. Waves split
// the D columns, so every wave computes the SAME S/softmax (8 redundant
// MFMAs — free next to the byte stream) and owns its O slice outright: no
// cross-wave merge, no stats exchange. ~10 MFMA + ~60 VALU per wave per
// 16 KB chunk.
typedef __attribute__((ext_vector_type(8))) short short8d;
typedef __attribute__((ext_vector_type(4))) short short4d;
typedef __attribute__((ext_vector_type(4))) float float4d;

DEV_INLINE short4d dec_tr16_read(const short* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) short4d*)p);
}

DEV_INLINE unsigned dec_cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

template <int ROW_BYTES>
DEV_INLINE int dec_swz(int row, int chunk) {
  constexpr int NCH = ROW_BYTES / 16;
  return row * NCH + (chunk ^ (row & (NCH - 1)));
}

#define DEC_CH 32   // tokens per chunk (2 KV blocks)
constexpr float DEC_LOG2E = 1.44269504088896340736f;

// load one 8-element source granule (16B of bf16, or 8B of fp8 converted
// through the hardware cvt) as a short8 of bf16 for the LDS tiles
template <bool KV8>
DEV_INLINE short8d dec_load_granule(const char* p, bool live) {
  if (!KV8) {
    uint4v g = live ? nt_load_u4(p) : (uint4v)0;
    return *reinterpret_cast<short8d*>(&g);
  }
  const uint2v g = live ? nt_load_u2(p) : (uint2v)0;
  short8d r;
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const unsigned w = g[i];
    const unsigned a = dec_cvt_pk_bf16(fp8_to_f32<0>(w), fp8_to_f32<1>(w));
    const unsigned b = dec_cvt_pk_bf16(fp8_to_f32<2>(w), fp8_to_f32<3>(w));
    r[i * 4 + 0] = (short)(a & 0xffff);
    r[i * 4 + 1] = (short)(a >> 16);
    r[i * 4 + 2] = (short)(b & 0xffff);
    r[i * 4 + 3] = (short)(b >> 16);
  }
  return r;
}

template <int D, bool KV8>
__global__ __launch_bounds__(NW * WAVE)
void paged_attn_decode_mfma_kernel(
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
  constexpr int DC32 = D / 32;           // 32-elem k-steps of a row
  constexpr int DC16 = D / 16;           // 16-col subtiles of a row
  constexpr int CPW = DC16 / NW;         // out col-subtiles per wave
  constexpr int GRAN = DEC_CH * D / 8;   // 8-elem source granules per tile
  constexpr int GPT = GRAN / (NW * WAVE); // granules per thread (K or V)
  constexpr int ESZ = KV8 ? 1 : 2;

  const int b = blockIdx.x;
  const int h = blockIdx.y;              // kv head
  const int split = blockIdx.z;
  const int group = hq / hkv;            // <= 16
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lhi = lane >> 4;
  const int llo = lane & 15;
  const int tid = threadIdx.x;

  const int n = seq_lens[b];
  const int nch = (n + DEC_CH - 1) / DEC_CH;
  const int nblk = (n + BS - 1) / BS;
  const float scale2 = scale * DEC_LOG2E;

  // Q A-fragments: row llo = q head (zero-padded past the group)
  short8d qa[DC32];
  {
    const bool live = llo < group;
    const bf16_t* qrow =
        q + (long)b * q_rstride + (long)(h * group + (live ? llo : 0)) * D;
    #pragma unroll
    for (int c = 0; c < DC32; ++c) {
      short8d v = *reinterpret_cast<const short8d*>(qrow + c * 32 + lhi * 8);
      if (!live)
        #pragma unroll
        for (int j = 0; j < 8; ++j) v[j] = 0;
      qa[c] = v;
    }
  }

  float m2[4], l[4];                     // per C row (lhi*4+i), log2 domain
  float4d oacc[CPW];
  #pragma unroll
  for (int i = 0; i < 4; ++i) { m2[i] = -INFINITY; l[i] = 0.f; }
  #pragma unroll
  for (int c = 0; c < CPW; ++c) oacc[c] = (float4d){0.f, 0.f, 0.f, 0.f};

  __shared__ uint4v k_tile[2][DEC_CH * D * 2 / 16];      // swizzled rows
  __shared__ short v_img[2][DC16][DEC_CH / 32][2][16][16]; // tr16 B-frags
  __shared__ short p_img[NW][DEC_CH / 32][2][16][16];      // tr16 A-frags

  const int* bt = block_table + (long)b * max_blocks;
  const long head_off = (long)h * BS * D;       // elems
  const long blk_stride = (long)hkv * BS * D;   // elems

  // staging is split T14-style: loads into registers issue BEFORE the
  // chunk's compute (a full compute phase of latency slack), LDS writes
  // land after it. granule g -> row g*8/D, col (g*8)%D.
  short8d nk[GPT], nv[GPT];
  auto stage_load = [&](int ci) {
    #pragma unroll
    for (int i = 0; i < GPT; ++i) {
      const int g = i * NW * WAVE + tid;
      const int row = g * 8 / D, col = (g * 8) % D;
      const int tok = ci * DEC_CH + row;
      const int bl = 2 * ci + row / BS;
      const bool live = tok < n && bl < nblk;
      const long base =
          live ? ((long)bt[bl] * blk_stride + head_off +
                  (long)(row % BS) * D + col) * ESZ
               : 0;
      nk[i] = dec_load_granule<KV8>(
          reinterpret_cast<const char*>(k_cache) + base, live);
      nv[i] = dec_load_granule<KV8>(
          reinterpret_cast<const char*>(v_cache) + base, live);
    }
  };
  auto stage_write = [&](int buf) {
    #pragma unroll
    for (int i = 0; i < GPT; ++i) {
      const int g = i * NW * WAVE + tid;
      const int row = g * 8 / D, col = (g * 8) % D;
      *reinterpret_cast<short8d*>(
          &k_tile[buf][dec_swz<D * 2>(row, col / 8)]) = nk[i];
      const int cc = col / 16, off = col % 16;
      const int kk = row % 32;
      const int sel = (kk % 8) >= 4;
      const int rr = (kk / 8) * 4 + (kk % 4);
      *reinterpret_cast<short8d*>(
          &v_img[buf][cc][row / 32][sel][rr][off]) = nv[i];
    }
  };

  if (split < nch) {
    stage_load(split);
    stage_write(0);
  }
  __syncthreads();

  int cur = 0;
  for (int ci = split; ci < nch; ci += nsplit) {
    const int cn = ci + nsplit;
    if (cn < nch) stage_load(cn);

    // ---- S = Q·K^T (two 16-col subchunks), softmax in log2 domain ------
    float4d sf[DEC_CH / 16];
    #pragma unroll
    for (int sc = 0; sc < DEC_CH / 16; ++sc) {
      sf[sc] = (float4d){0.f, 0.f, 0.f, 0.f};
      const int key = sc * 16 + llo;
      #pragma unroll
      for (int c = 0; c < DC32; ++c) {
        const uint4v kb = k_tile[cur][dec_swz<D * 2>(key, c * 4 + lhi)];
        sf[sc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qa[c], *reinterpret_cast<const short8d*>(&kb), sf[sc], 0, 0, 0);
      }
    }
    float bm[4];
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      float v = -INFINITY;
      #pragma unroll
      for (int sc = 0; sc < DEC_CH / 16; ++sc) {
        const int tok = ci * DEC_CH + sc * 16 + llo;
        if (tok < n) v = fmaxf(v, sf[sc][i] * scale2);
      }
      #pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_xor(v, off, WAVE));
      bm[i] = v;
    }
    // P (bf16) into the per-wave tr16 image + row sums
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const float mn = fmaxf(m2[i], bm[i]);
      if (mn == -INFINITY) continue;
      const float alpha =
          (m2[i] == -INFINITY) ? 0.f : __builtin_amdgcn_exp2f(m2[i] - mn);
      m2[i] = mn;
      l[i] *= alpha;
      #pragma unroll
      for (int c = 0; c < CPW; ++c) oacc[c][i] *= alpha;
    }
    float pv[DEC_CH / 16][4];
    #pragma unroll
    for (int sc = 0; sc < DEC_CH / 16; ++sc) {
      const int tok = ci * DEC_CH + sc * 16 + llo;
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const float p =
            (tok < n && m2[i] != -INFINITY)
                ? __builtin_amdgcn_exp2f(sf[sc][i] * scale2 - m2[i])
                : 0.f;
        pv[sc][i] = p;
      }
    }
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      float rs = 0.f;
      #pragma unroll
      for (int sc = 0; sc < DEC_CH / 16; ++sc) rs += pv[sc][i];
      #pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        rs += __shfl_xor(rs, off, WAVE);
      l[i] += rs;
    }
    #pragma unroll
    for (int sc = 0; sc < DEC_CH / 16; ++sc) {
      const int key = sc * 16 + llo;
      const int kk = key & 31;
      *reinterpret_cast<uint2*>(
          &p_img[wave][key >> 5][(kk % 8) >= 4]
                [(kk / 8) * 4 + (kk % 4)][lhi * 4]) =
          make_uint2(dec_cvt_pk_bf16(pv[sc][0], pv[sc][1]),
                     dec_cvt_pk_bf16(pv[sc][2], pv[sc][3]));
    }
    __builtin_amdgcn_wave_barrier();

    // ---- O += P·V over the chunk's 32 tokens ---------------------------
    #pragma unroll
    for (int ks = 0; ks < DEC_CH / 32; ++ks) {
      const short4d alo = dec_tr16_read(&p_img[wave][ks][0][0][0] + lane * 4);
      const short4d ahi = dec_tr16_read(&p_img[wave][ks][1][0][0] + lane * 4);
      short8d pa;
      pa[0] = alo[0]; pa[1] = alo[1]; pa[2] = alo[2]; pa[3] = alo[3];
      pa[4] = ahi[0]; pa[5] = ahi[1]; pa[6] = ahi[2]; pa[7] = ahi[3];
      #pragma unroll
      for (int c = 0; c < CPW; ++c) {
        const int cc = wave * CPW + c;
        const short4d lo =
            dec_tr16_read(&v_img[cur][cc][ks][0][0][0] + lane * 4);
        const short4d hi =
            dec_tr16_read(&v_img[cur][cc][ks][1][0][0] + lane * 4);
        short8d vb;
        vb[0] = lo[0]; vb[1] = lo[1]; vb[2] = lo[2]; vb[3] = lo[3];
        vb[4] = hi[0]; vb[5] = hi[1]; vb[6] = hi[2]; vb[7] = hi[3];
        oacc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pa, vb, oacc[c], 0, 0, 0);
      }
    }

    // stage the next chunk into the other buffer; one barrier orders the
    // writes against the next iteration's reads (see scalar GQA kernel)
    if (cn < nch) stage_write(cur ^ 1);
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: C row = q head, col = wave's D slice ------------------
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int row = lhi * 4 + i;
    if (row >= group) continue;
    const int qhead = h * group + row;
    if (nsplit == 1) {
      const float invl = (l[i] > 0.f) ? 1.f / l[i] : 0.f;
      bf16_t* orow = out + ((long)b * hq + qhead) * D;
      #pragma unroll
      for (int c = 0; c < CPW; ++c)
        orow[(wave * CPW + c) * 16 + llo] = f2bf(oacc[c][i] * invl);
    } else {
      float* po = part_o + (((long)b * hq + qhead) * nsplit + split) * D;
      #pragma unroll
      for (int c = 0; c < CPW; ++c)
        po[(wave * CPW + c) * 16 + llo] = oacc[c][i];
      if (llo == 0 && wave == 0) {
        float* pm = part_ml + (((long)b * hq + qhead) * nsplit + split) * 2;
        // reduce kernel merges in the natural-log domain
        pm[0] = (m2[i] == -INFINITY) ? -INFINITY : m2[i] / DEC_LOG2E;
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
    dim3 grid(bsz, hkv, nsplit);
    // default: the MFMA variant (group on the M dim); SUB_DECODE_GQA=scalar
    // selects the scalar LDS-staged kernel for A/B measurement
    static const char* ge = getenv("SUB_DECODE_GQA");
    static const bool scalar_gqa = ge && ge[0] == 's';
    if (!scalar_gqa) {
      #define LAUNCHM(DD, K8) \
        paged_attn_decode_mfma_kernel<DD, K8><<<grid, block, 0, stream>>>( \
            out, q, k_cache, v_cache, block_table, seq_lens, hq, hkv, \
            max_blocks, scale, q_rstride, nsplit, part_o, part_ml)
      if (!kv8) {
        if (d == 128) LAUNCHM(128, false);
        else LAUNCHM(64, false);
      } else {
        if (d == 128) LAUNCHM(128, true);
        else LAUNCHM(64, true);
      }
      #undef LAUNCHM
    } else {
    // LDS-staged GQA path: groups 4/8/16 (llama2/3-70b, GQA-16 d=64)
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
    }
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
