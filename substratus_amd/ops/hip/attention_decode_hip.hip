#include "hip/hip_runtime.h"
// Paged-KV decode attention (SURVEY.md §2.5 ledger item 1) — the serving hot
// kernel: per sequence and KV head, softmax(q·K^T/√d)·V over a block-paged
// cache, one new token per sequence.
//
// Decode at batch B is HBM-bound on the K/V stream (e.g. llama2-7b @ctx 512,
// B=64: ~16 GB of KV per step vs 13.5 GB of weights). Structure:
//
//   * grid = (num_seqs, n_kv_heads, ceil(group/GC)); 4 waves per workgroup;
//     each wave owns KV blocks stride-4 (flash-decoding split over the
//     sequence).
//   * K panel row [16, D] bf16: D/8 lanes × uint4 = one full 256 B row per
//     token. K tiles are PREFETCHED one block ahead into registers (guide
//     T14 issue-early/consume-late: the next block's loads are issued before
//     this block's score+PV compute, hiding the ~900-cycle HBM latency), so
//     the score phase runs entirely from registers.
//   * Scores reduced via shfl within the lane group (guide §5
//     common-mistake #6: wave-parallel softmax, never serial).
//   * Online softmax per wave; P·V with dims distributed over lanes; for
//     full blocks all 16 V row-loads are issued before the FMA chain.
//   * Cross-wave merge through LDS at the end.
//
// Cache layout: [num_blocks, n_kv_heads, BS=16, D]; block_table [B, maxb]
// int32; seq_lens [B] int32 counts INCLUDING the current token.
#include "common.h"

#define BS 16   // tokens per KV block
#define NW 4    // waves per workgroup

template <int D, int GC>
__global__ __launch_bounds__(NW * WAVE)
void paged_attn_decode_kernel(
    bf16_t* __restrict__ out,            // [B, HQ, D]
    const bf16_t* __restrict__ q,        // [B, HQ, D]
    const bf16_t* __restrict__ k_cache,  // [NB, HKV, BS, D]
    const bf16_t* __restrict__ v_cache,
    const int* __restrict__ block_table, // [B, max_blocks]
    const int* __restrict__ seq_lens,    // [B]
    int hq, int hkv, int max_blocks, float scale,
    int nsplit,                           // flash-decoding sequence splits
    float* __restrict__ part_o,           // [B, HQ, NS, D]  (nsplit > 1)
    float* __restrict__ part_ml) {        // [B, HQ, NS, 2]
  constexpr int LPT = D / 8;        // lanes per token for K rows (uint4 each)
  constexpr int TPP = WAVE / LPT;   // tokens per pass
  constexpr int PASSES = BS / TPP;
  constexpr int DPL = D / WAVE;     // accum dims per lane (1 or 2)
  static_assert(DPL >= 1 && DPL <= 2, "head_dim must be 64..128");

  const int b = blockIdx.x;
  const int h = blockIdx.y;           // kv head
  const int split = blockIdx.z % nsplit;   // sequence split (flash-decoding)
  const int gchunk = blockIdx.z / nsplit;  // q-head chunk within the group
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
          q + ((long)b * hq + h * group + g0 + g) * D)[ll];
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
  const long head_off = (long)h * BS * D;
  const long blk_stride = (long)hkv * BS * D;

  // ---- K prefetch pipeline: tile bi held in regs while bi+NW streams ----
  // K/V stream is read-once per step (each (seq, kv-head) block is owned by
  // one workgroup): non-temporal loads keep it out of L1 (guide nt-weights:
  // -18% issued-to-landed on one-reader streams).
  const int wstride = NW * nsplit;        // total concurrent waves per (b,h,g)
  const int w0 = split * NW + wave;       // this wave's first block
  uint4v kt[PASSES];
  int blk_cur = (w0 < nblk) ? bt[w0] : 0;
  if (w0 < nblk) {
    const bf16_t* kbase = k_cache + (long)blk_cur * blk_stride + head_off;
    #pragma unroll
    for (int p = 0; p < PASSES; ++p)
      kt[p] = nt_load_u4(
          reinterpret_cast<const uint4*>(kbase + (p * TPP + sub) * D) + ll);
  }

  for (int bi = w0; bi < nblk; bi += wstride) {
    // issue next tile's loads before any compute on the current tile
    uint4v kt_next[PASSES];
    const int bnext = bi + wstride;
    int blk_next = 0;
    if (bnext < nblk) {
      blk_next = bt[bnext];
      const bf16_t* kb = k_cache + (long)blk_next * blk_stride + head_off;
      #pragma unroll
      for (int p = 0; p < PASSES; ++p)
        kt_next[p] = nt_load_u4(
            reinterpret_cast<const uint4*>(kb + (p * TPP + sub) * D) + ll);
    }
    const int blk = blk_cur;

    // ---- scores from registers ----------------------------------------
    #pragma unroll
    for (int p = 0; p < PASSES; ++p) {
      const int t = p * TPP + sub;
      const int tok_idx = bi * BS + t;
      float2 f0 = unpack_bf16x2(kt[p][0]), f1 = unpack_bf16x2(kt[p][1]);
      float2 f2 = unpack_bf16x2(kt[p][2]), f3 = unpack_bf16x2(kt[p][3]);
      #pragma unroll
      for (int g = 0; g < GC; ++g) {
        float s = qf[g][0] * f0.x + qf[g][1] * f0.y + qf[g][2] * f1.x +
                  qf[g][3] * f1.y + qf[g][4] * f2.x + qf[g][5] * f2.y +
                  qf[g][6] * f3.x + qf[g][7] * f3.y;
        #pragma unroll
        for (int off = LPT / 2; off > 0; off >>= 1)
          s += __shfl_xor(s, off, WAVE);
        if (ll == 0)
          s_scores[wave][g][t] = (tok_idx < n) ? s * scale : -INFINITY;
      }
    }
    __builtin_amdgcn_wave_barrier();

    // ---- online softmax + P·V -----------------------------------------
    const bf16_t* vbase = v_cache + (long)blk * blk_stride + head_off;
    const int valid = min(n - bi * BS, BS);
    if (valid == BS) {
      // full block: hoist all 16 V loads ahead of the FMA chains
      unsigned int vv[BS];
      bf16_t vs[BS];
      #pragma unroll
      for (int t = 0; t < BS; ++t) {
        if (DPL == 2)
          vv[t] = __builtin_nontemporal_load(
              reinterpret_cast<const unsigned int*>(vbase + t * D) + lane);
        else
          vs[t] = vbase[t * D + lane];
      }
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
          if (DPL == 2) {
            float2 vf = unpack_bf16x2(vv[t]);
            acc[g][0] += p * vf.x;
            acc[g][1] += p * vf.y;
          } else {
            acc[g][0] += p * bf2f(vs[t]);
          }
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
          if (DPL == 2) {
            const unsigned int uv = reinterpret_cast<const unsigned int*>(
                vbase + t * D)[lane];
            float2 vf = unpack_bf16x2(uv);
            acc[g][0] += p * vf.x;
            acc[g][1] += p * vf.y;
          } else {
            acc[g][0] += p * bf2f(vbase[t * D + lane]);
          }
        }
      }
    }
    // rotate the pipeline
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

int paged_attn_nsplit(int bsz, int hq, int hkv) {
  // flash-decoding split factor: fill >=1024 workgroups (256 CUs x 4
  // resident WGs) when the (B, Hkv, group-chunk) grid alone is too small
  // (llama2-70b at TP=8: B x 1 x 1). Shape-only -> hipGraph-stable.
  const int group = hq / hkv;
  const int gc = (group % 8 == 0) ? 8 : 1;
  const int gchunks = (group + gc - 1) / gc;
  const long base = (long)bsz * hkv * gchunks;
  if (base >= 1024) return 1;
  int ns = (int)((1024 + base - 1) / base);
  if (ns > 16) ns = 16;
  return ns;
}

void launch_paged_attn_decode(bf16_t* out, const bf16_t* q,
                              const bf16_t* k_cache, const bf16_t* v_cache,
                              const int* block_table, const int* seq_lens,
                              int bsz, int hq, int hkv, int d, int max_blocks,
                              int block_size, float scale, int nsplit,
                              float* part_o, float* part_ml,
                              hipStream_t stream) {
  const int group = hq / hkv;
  // GC=8 when the GQA group tiles evenly by 8 (llama2-70b, llama3-style);
  // GC=1 otherwise (MHA group=1; falcon's prime group=71 — its MQA KV is
  // tiny and L3-resident, so per-head K re-reads are cheap).
  const int gc = (group % 8 == 0) ? 8 : 1;
  const int gchunks = (group + gc - 1) / gc;
  dim3 grid(bsz, hkv, gchunks * nsplit);
  dim3 block(NW * WAVE);
  #define LAUNCH(DD, GG) \
   hipLaunchKernelGGL(( paged_attn_decode_kernel<DD, GG>), dim3(grid), dim3(block), 0, stream,  \
        out, q, k_cache, v_cache, block_table, seq_lens, hq, hkv, \
        max_blocks, scale, nsplit, part_o, part_ml)
  if (d == 128 && gc == 8) LAUNCH(128, 8);
  else if (d == 128) LAUNCH(128, 1);
  else if (d == 64 && gc == 8) LAUNCH(64, 8);
  else if (d == 64) LAUNCH(64, 1);
  #undef LAUNCH
  if (nsplit > 1) {
    dim3 rgrid(bsz, hq);
    if (d == 128) {
     hipLaunchKernelGGL(( paged_attn_reduce_kernel<128>), dim3(rgrid), dim3(dim3(WAVE)), 0, stream, 
          out, part_o, part_ml, hq, nsplit);
    } else {
     hipLaunchKernelGGL(( paged_attn_reduce_kernel<64>), dim3(rgrid), dim3(dim3(WAVE)), 0, stream, 
          out, part_o, part_ml, hq, nsplit);
    }
  }
}

bool paged_attn_decode_supported(int d, int block_size) {
  return (d == 128 || d == 64) && block_size == BS;
}
