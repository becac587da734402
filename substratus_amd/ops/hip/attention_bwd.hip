// Flash-attention training BACKWARD (first-party trainer hot path —
// torch SDPA fell back to aotriton; the north star is
// hand-written CDNA4 kernels on every hot op).
//
// Standard FA2 decomposition, MI355X-first data movement (the same probed
// MFMA fragment maps, XOR-swizzled LDS tiles and ds_read_b64_tr_b16
// transpose images as attention_prefill.hip):
//
//   delta kernel:  Delta[t,h] = sum_d dO ∘ O            (rowwise, memory-bound)
//   dq kernel:     per 64-row q tile, stream KV chunks:
//                    S  = Q·K^T          (MFMA, C layout [q][k])
//                    P  = exp2(S·scale2 − LSE2[q])       (recomputed, no S kept)
//                    dP = dO·V^T         (MFMA)
//                    dS = P ∘ (dP − Delta[q]) · scale    → bf16 tr-image
//                    dQ += dS·K          (MFMA; K^T via tr16 image)
//   dkv kernel:    per 64-row kv tile, loop the GQA group's q heads,
//                  stream q chunks (q ≥ kv tile start — causal):
//                    S^T  = K·Q^T        (C layout [k][q])
//                    P^T  = exp2(S^T·scale2 − LSE2[q col])
//                    dP^T = V·dO^T
//                    dS^T = P^T ∘ (dP^T − Delta[q col]) · scale
//                    dV += P^T·dO        (dO^T via tr16 image)
//                    dK += dS^T·Q        (Q^T via tr16 image)
//
// GQA is native: dK/dV accumulate over the group's q heads inside the
// kernel — the trainer never materializes repeated K/V heads
// (models/model.py forward_train previously repeat_interleave'd them).
//
// LSE2 is the log2-domain row logsumexp the forward saves
// (attention_prefill.hip epilogue): lse2 = m + log2(l) with
// scale·log2(e) folded into the score domain, so P = exp2(s·scale2 − lse2)
// bit-matches the forward's normalized P up to fp32 rounding.
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;

namespace attnbwd {

DEV_INLINE short4v tr16_read(const short* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) short4v*)p);
}

#define BW_WAVES 4
#define BW_TILE 64   // rows (q or kv) owned per workgroup, 16 per wave
#define BW_CH 64     // streamed chunk length

template <int ROW_BYTES>
DEV_INLINE int swz_chunk(int row, int chunk) {
  constexpr int NCH = ROW_BYTES / 16;
  return row * NCH + (chunk ^ (row & (NCH - 1)));
}

// pack 2 f32 -> bf16x2 via the hardware cvt (round-to-nearest-even)
DEV_INLINE unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

constexpr float LOG2E = 1.44269504088896340736f;

// ---------------------------------------------------------------- delta
// Delta[t, h] = sum_d dO[t,h,d] * O[t,h,d], fp32. One wave per (t, h) row
// pair group: block = 4 waves, grid-strided over T*H rows; lane reads
// D/WAVE*2 bf16 as uint (wave64 × 2 elem = 128 = D).
template <int D>
__global__ void attn_bwd_delta_kernel(float* __restrict__ delta,
                                      const bf16_t* __restrict__ dout,
                                      const bf16_t* __restrict__ out,
                                      long rows) {
  const long wid = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int lane = threadIdx.x % WAVE;
  const long nw = ((long)gridDim.x * blockDim.x) / WAVE;
  constexpr int DU = D / 2;           // uints (2 bf16) per row
  for (long r = wid; r < rows; r += nw) {
    const unsigned* dp = reinterpret_cast<const unsigned*>(dout + r * D);
    const unsigned* op = reinterpret_cast<const unsigned*>(out + r * D);
    float acc = 0.f;
    #pragma unroll
    for (int i = lane; i < DU; i += WAVE) {
      const float2 dv = unpack_bf16x2(dp[i]);
      const float2 ov = unpack_bf16x2(op[i]);
      acc += dv.x * ov.x + dv.y * ov.y;
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      acc += __shfl_xor(acc, off, WAVE);
    if (lane == 0) delta[r] = acc;
  }
}

// ------------------------------------------------------------------- dq
// Grid (num_seqs, HQ, q-tiles); wave w owns q rows [tile*64 + 16w, +16).
template <int D>
__global__ __launch_bounds__(BW_WAVES * WAVE)
void attn_bwd_dq_kernel(bf16_t* __restrict__ dq,        // [T, HQ, D]
                        const bf16_t* __restrict__ q,   // [T, HQ, D]
                        const bf16_t* __restrict__ k,   // [T, HKV, D]
                        const bf16_t* __restrict__ v,   // [T, HKV, D]
                        const bf16_t* __restrict__ dout,
                        const float* __restrict__ lse,   // [T, HQ]
                        const float* __restrict__ delta, // [T, HQ]
                        const int* __restrict__ seq_start,
                        int hq, int hkv, float scale) {
  constexpr int DC32 = D / 32;
  constexpr int DC16 = D / 16;
  const int s = blockIdx.x;
  const int head = blockIdx.y;
  const int tile = blockIdx.z;
  const int kvh = head / (hq / hkv);
  const int base = seq_start[s];
  const int n = seq_start[s + 1] - base;
  const int q0_tile = tile * BW_TILE;
  if (q0_tile >= n) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lhi = lane >> 4;
  const int llo = lane & 15;
  const int q0 = q0_tile + wave * 16;
  const bool wave_live = q0 < n;

  const long qrow_stride = (long)hq * D;
  const long krow_stride = (long)hkv * D;

  // per-wave preloads: Q and dO A-fragments, per-row lse/delta
  short8 qa[DC32], doa[DC32];
  float lse2[4], dlt[4];
  {
    const int qrow = min(q0 + llo, n - 1);
    const long roff = (long)(base + qrow) * qrow_stride + (long)head * D;
    #pragma unroll
    for (int c = 0; c < DC32; ++c) {
      qa[c] = *reinterpret_cast<const short8*>(q + roff + c * 32 + lhi * 8);
      doa[c] = *reinterpret_cast<const short8*>(dout + roff + c * 32 + lhi * 8);
    }
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int rr = min(q0 + lhi * 4 + i, n - 1);
      lse2[i] = lse[(long)(base + rr) * hq + head];
      dlt[i] = delta[(long)(base + rr) * hq + head];
    }
  }

  float4v dq_acc[DC16];
  #pragma unroll
  for (int c = 0; c < DC16; ++c) dq_acc[c] = (float4v){0.f, 0.f, 0.f, 0.f};

  // LDS: K rows (QK^T B-frags), V rows (dO·V^T B-frags), K tr-image
  // (dS·K B-frags), per-wave dS tr-image (dS·K A-frags)
  __shared__ uint4 k_tile[BW_CH * D * 2 / 16];
  __shared__ uint4 v_tile[BW_CH * D * 2 / 16];
  __shared__ short k_img[DC16][BW_CH / 32][2][16][16];
  __shared__ short ds_img[BW_WAVES][BW_CH / 32][2][16][16];

  const int kmax_tile = min(q0_tile + BW_TILE, n);
  const int nchunks = (kmax_tile + BW_CH - 1) / BW_CH;
  constexpr int NSC = BW_CH / 16;
  const float scale2 = scale * LOG2E;

  for (int kc = 0; kc < nchunks; ++kc) {
    const int kbase = kc * BW_CH;
    // ---- stage K (two layouts) + V (row layout) ------------------------
    {
      constexpr int LOADS = BW_CH * D / 8 / (BW_WAVES * WAVE);
      #pragma unroll
      for (int i = 0; i < LOADS; ++i) {
        const int flat = (i * BW_WAVES * WAVE + threadIdx.x) * 8;
        const int row = flat / D, col = flat % D;
        const bool ok = kbase + row < n;
        const long goff = (long)(base + kbase + row) * krow_stride +
                          (long)kvh * D + col;
        uint4 kv4 = ok ? *reinterpret_cast<const uint4*>(k + goff)
                       : make_uint4(0, 0, 0, 0);
        k_tile[swz_chunk<D * 2>(row, col * 2 / 16)] = kv4;
        {
          const int c = col / 16, off = col % 16;
          const int ks = row / 32, kk = row % 32;
          const int sel = (kk % 8) >= 4;
          const int r = (kk / 8) * 4 + (kk % 4);
          *reinterpret_cast<uint4*>(&k_img[c][ks][sel][r][off]) = kv4;
        }
        uint4 vv4 = ok ? *reinterpret_cast<const uint4*>(v + goff)
                       : make_uint4(0, 0, 0, 0);
        v_tile[swz_chunk<D * 2>(row, col * 2 / 16)] = vv4;
      }
    }
    __syncthreads();

    if (wave_live && kbase < min(q0 + 16, n)) {
      #pragma unroll
      for (int sc = 0; sc < NSC; ++sc) {
        // S = Q·K^T and dP = dO·V^T for this 16-key subchunk
        float4v sf = {0.f, 0.f, 0.f, 0.f};
        float4v dpf = {0.f, 0.f, 0.f, 0.f};
        const int key = sc * 16 + llo;
        #pragma unroll
        for (int c = 0; c < DC32; ++c) {
          const uint4 kb4 = k_tile[swz_chunk<D * 2>(key, c * 4 + lhi)];
          sf = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qa[c], *reinterpret_cast<const short8*>(&kb4), sf, 0, 0, 0);
          const uint4 vb4 = v_tile[swz_chunk<D * 2>(key, c * 4 + lhi)];
          dpf = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              doa[c], *reinterpret_cast<const short8*>(&vb4), dpf, 0, 0, 0);
        }
        // dS = P ∘ (dP − Delta) · scale, causal-masked, C layout
        float dsv[4];
        const int kpos = kbase + sc * 16 + llo;
        #pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int qpos = q0 + lhi * 4 + i;
          const bool live = (qpos < n) && (kpos <= qpos);
          const float p =
              live ? __builtin_amdgcn_exp2f(sf[i] * scale2 - lse2[i]) : 0.f;
          dsv[i] = p * (dpf[i] - dlt[i]) * scale;
        }
        // store transposed into the per-wave tr-image (A-frag source)
        const int kk = key & 31;
        *reinterpret_cast<uint2*>(
            &ds_img[wave][key >> 5][(kk % 8) >= 4]
                   [(kk / 8) * 4 + (kk % 4)][lhi * 4]) =
            make_uint2(cvt_pk_bf16(dsv[0], dsv[1]),
                       cvt_pk_bf16(dsv[2], dsv[3]));
      }
      // ---- dQ += dS·K over the chunk's keys in 32-k steps --------------
      #pragma unroll
      for (int ks = 0; ks < BW_CH / 32; ++ks) {
        const short4v alo = tr16_read(&ds_img[wave][ks][0][0][0] + lane * 4);
        const short4v ahi = tr16_read(&ds_img[wave][ks][1][0][0] + lane * 4);
        short8 dsa;
        dsa[0] = alo[0]; dsa[1] = alo[1]; dsa[2] = alo[2]; dsa[3] = alo[3];
        dsa[4] = ahi[0]; dsa[5] = ahi[1]; dsa[6] = ahi[2]; dsa[7] = ahi[3];
        #pragma unroll
        for (int c = 0; c < DC16; ++c) {
          const short4v lo = tr16_read(&k_img[c][ks][0][0][0] + lane * 4);
          const short4v hi = tr16_read(&k_img[c][ks][1][0][0] + lane * 4);
          short8 kb;
          kb[0] = lo[0]; kb[1] = lo[1]; kb[2] = lo[2]; kb[3] = lo[3];
          kb[4] = hi[0]; kb[5] = hi[1]; kb[6] = hi[2]; kb[7] = hi[3];
          dq_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              dsa, kb, dq_acc[c], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  if (wave_live) {
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int qpos = q0 + lhi * 4 + i;
      if (qpos >= n) continue;
      bf16_t* op = dq + (long)(base + qpos) * qrow_stride + (long)head * D;
      #pragma unroll
      for (int c = 0; c < DC16; ++c)
        op[c * 16 + llo] = f2bf(dq_acc[c][i]);
    }
  }
}

// ------------------------------------------------------------------ dkv
// Grid (num_seqs, HQ, kv-tiles) — one block per Q HEAD so MQA/GQA keep the
// full grid parallelism of the dq kernel (a per-kv-head grid with an
// in-kernel group loop was 13x slower on falcon's 71:1 MQA: 71x fewer
// blocks AND Q/dO re-staged serially per head). dK/dV accumulate across
// the group's heads with fp32 unsafeAtomicAdd (hardware
// global_atomic_add_f32); grp==1 skips the atomic.
template <int D>
__global__ __launch_bounds__(BW_WAVES * WAVE)
void attn_bwd_dkv_kernel(float* __restrict__ dk,        // [T, HKV, D] f32
                         float* __restrict__ dv,        // [T, HKV, D] f32
                         const bf16_t* __restrict__ q,  // [T, HQ, D]
                         const bf16_t* __restrict__ kk_, // [T, HKV, D]
                         const bf16_t* __restrict__ v,  // [T, HKV, D]
                         const bf16_t* __restrict__ dout,
                         const float* __restrict__ lse,
                         const float* __restrict__ delta,
                         const int* __restrict__ seq_start,
                         int hq, int hkv, float scale) {
  constexpr int DC32 = D / 32;
  constexpr int DC16 = D / 16;
  const int s = blockIdx.x;
  const int head = blockIdx.y;
  const int tile = blockIdx.z;
  const int grp = hq / hkv;
  const int kvh = head / grp;
  const int base = seq_start[s];
  const int n = seq_start[s + 1] - base;
  const int kv0_tile = tile * BW_TILE;
  if (kv0_tile >= n) return;

  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int lhi = lane >> 4;
  const int llo = lane & 15;
  const int kv0 = kv0_tile + wave * 16;    // this wave's first kv row
  const bool wave_live = kv0 < n;

  const long qrow_stride = (long)hq * D;
  const long krow_stride = (long)hkv * D;

  // per-wave preloads: K and V A-fragments of the wave's 16 kv rows
  short8 ka[DC32], va[DC32];
  {
    const int krow = min(kv0 + llo, n - 1);
    const long roff = (long)(base + krow) * krow_stride + (long)kvh * D;
    #pragma unroll
    for (int c = 0; c < DC32; ++c) {
      ka[c] = *reinterpret_cast<const short8*>(kk_ + roff + c * 32 + lhi * 8);
      va[c] = *reinterpret_cast<const short8*>(v + roff + c * 32 + lhi * 8);
    }
  }

  float4v dk_acc[DC16], dv_acc[DC16];
  #pragma unroll
  for (int c = 0; c < DC16; ++c) {
    dk_acc[c] = (float4v){0.f, 0.f, 0.f, 0.f};
    dv_acc[c] = (float4v){0.f, 0.f, 0.f, 0.f};
  }

  // LDS: Q rows + dO rows (S^T / dP^T B-frags), Q tr-image + dO tr-image
  // (dK·Q / dV·dO B-frags), per-wave P^T and dS^T tr-images (A-frags)
  __shared__ uint4 q_tile[BW_CH * D * 2 / 16];
  __shared__ uint4 do_tile[BW_CH * D * 2 / 16];
  __shared__ short q_img[DC16][BW_CH / 32][2][16][16];
  __shared__ short do_img[DC16][BW_CH / 32][2][16][16];
  __shared__ short pt_img[BW_WAVES][BW_CH / 32][2][16][16];
  __shared__ short dst_img[BW_WAVES][BW_CH / 32][2][16][16];

  constexpr int NSC = BW_CH / 16;
  const float scale2 = scale * LOG2E;
  const int qc_first = kv0_tile / BW_CH;   // BW_TILE == BW_CH: = tile
  const int nqchunks = (n + BW_CH - 1) / BW_CH;

  {
    for (int qc = qc_first; qc < nqchunks; ++qc) {
      const int qbase = qc * BW_CH;
      // ---- stage Q and dO, both layouts --------------------------------
      {
        constexpr int LOADS = BW_CH * D / 8 / (BW_WAVES * WAVE);
        #pragma unroll
        for (int i = 0; i < LOADS; ++i) {
          const int flat = (i * BW_WAVES * WAVE + threadIdx.x) * 8;
          const int row = flat / D, col = flat % D;
          const bool ok = qbase + row < n;
          const long goff = (long)(base + qbase + row) * qrow_stride +
                            (long)head * D + col;
          const int c = col / 16, off = col % 16;
          const int ks = row / 32, kkk = row % 32;
          const int sel = (kkk % 8) >= 4;
          const int r = (kkk / 8) * 4 + (kkk % 4);
          uint4 q4 = ok ? *reinterpret_cast<const uint4*>(q + goff)
                        : make_uint4(0, 0, 0, 0);
          q_tile[swz_chunk<D * 2>(row, col * 2 / 16)] = q4;
          *reinterpret_cast<uint4*>(&q_img[c][ks][sel][r][off]) = q4;
          uint4 d4 = ok ? *reinterpret_cast<const uint4*>(dout + goff)
                        : make_uint4(0, 0, 0, 0);
          do_tile[swz_chunk<D * 2>(row, col * 2 / 16)] = d4;
          *reinterpret_cast<uint4*>(&do_img[c][ks][sel][r][off]) = d4;
        }
      }
      __syncthreads();

      if (wave_live) {
        #pragma unroll
        for (int sc = 0; sc < NSC; ++sc) {
          // S^T = K·Q^T and dP^T = V·dO^T for this 16-q subchunk
          float4v stf = {0.f, 0.f, 0.f, 0.f};
          float4v dptf = {0.f, 0.f, 0.f, 0.f};
          const int qrow = sc * 16 + llo;
          #pragma unroll
          for (int c = 0; c < DC32; ++c) {
            const uint4 qb4 = q_tile[swz_chunk<D * 2>(qrow, c * 4 + lhi)];
            stf = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                ka[c], *reinterpret_cast<const short8*>(&qb4), stf, 0, 0, 0);
            const uint4 db4 = do_tile[swz_chunk<D * 2>(qrow, c * 4 + lhi)];
            dptf = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                va[c], *reinterpret_cast<const short8*>(&db4), dptf, 0, 0, 0);
          }
          // P^T and dS^T, causal-masked, C layout [kv row i][q col llo]
          const int qpos = qbase + sc * 16 + llo;
          const int qcl = min(qpos, n - 1);
          const float lse_q = lse[(long)(base + qcl) * hq + head];
          const float dlt_q = delta[(long)(base + qcl) * hq + head];
          float ptv[4], dstv[4];
          #pragma unroll
          for (int i = 0; i < 4; ++i) {
            const int kpos = kv0 + lhi * 4 + i;
            const bool live = (qpos < n) && (kpos < n) && (qpos >= kpos);
            const float p =
                live ? __builtin_amdgcn_exp2f(stf[i] * scale2 - lse_q) : 0.f;
            ptv[i] = p;
            dstv[i] = p * (dptf[i] - dlt_q) * scale;
          }
          const int qk = qrow & 31;
          const int sel = (qk % 8) >= 4, r = (qk / 8) * 4 + (qk % 4);
          *reinterpret_cast<uint2*>(
              &pt_img[wave][qrow >> 5][sel][r][lhi * 4]) =
              make_uint2(cvt_pk_bf16(ptv[0], ptv[1]),
                         cvt_pk_bf16(ptv[2], ptv[3]));
          *reinterpret_cast<uint2*>(
              &dst_img[wave][qrow >> 5][sel][r][lhi * 4]) =
              make_uint2(cvt_pk_bf16(dstv[0], dstv[1]),
                         cvt_pk_bf16(dstv[2], dstv[3]));
        }
        // ---- dV += P^T·dO, dK += dS^T·Q over the q chunk ---------------
        #pragma unroll
        for (int ks = 0; ks < BW_CH / 32; ++ks) {
          short8 pa, da;
          {
            const short4v lo = tr16_read(&pt_img[wave][ks][0][0][0] + lane * 4);
            const short4v hi = tr16_read(&pt_img[wave][ks][1][0][0] + lane * 4);
            pa[0] = lo[0]; pa[1] = lo[1]; pa[2] = lo[2]; pa[3] = lo[3];
            pa[4] = hi[0]; pa[5] = hi[1]; pa[6] = hi[2]; pa[7] = hi[3];
          }
          {
            const short4v lo = tr16_read(&dst_img[wave][ks][0][0][0] + lane * 4);
            const short4v hi = tr16_read(&dst_img[wave][ks][1][0][0] + lane * 4);
            da[0] = lo[0]; da[1] = lo[1]; da[2] = lo[2]; da[3] = lo[3];
            da[4] = hi[0]; da[5] = hi[1]; da[6] = hi[2]; da[7] = hi[3];
          }
          #pragma unroll
          for (int c = 0; c < DC16; ++c) {
            {
              const short4v lo = tr16_read(&do_img[c][ks][0][0][0] + lane * 4);
              const short4v hi = tr16_read(&do_img[c][ks][1][0][0] + lane * 4);
              short8 db;
              db[0] = lo[0]; db[1] = lo[1]; db[2] = lo[2]; db[3] = lo[3];
              db[4] = hi[0]; db[5] = hi[1]; db[6] = hi[2]; db[7] = hi[3];
              dv_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  pa, db, dv_acc[c], 0, 0, 0);
            }
            {
              const short4v lo = tr16_read(&q_img[c][ks][0][0][0] + lane * 4);
              const short4v hi = tr16_read(&q_img[c][ks][1][0][0] + lane * 4);
              short8 qb;
              qb[0] = lo[0]; qb[1] = lo[1]; qb[2] = lo[2]; qb[3] = lo[3];
              qb[4] = hi[0]; qb[5] = hi[1]; qb[6] = hi[2]; qb[7] = hi[3];
              dk_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  da, qb, dk_acc[c], 0, 0, 0);
            }
          }
        }
      }
      __syncthreads();
    }
  }

  if (wave_live) {
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int kpos = kv0 + lhi * 4 + i;
      if (kpos >= n) continue;
      const long roff = (long)(base + kpos) * krow_stride + (long)kvh * D;
      if (grp == 1) {
        #pragma unroll
        for (int c = 0; c < DC16; ++c) {
          dk[roff + c * 16 + llo] = dk_acc[c][i];
          dv[roff + c * 16 + llo] = dv_acc[c][i];
        }
      } else {
        #pragma unroll
        for (int c = 0; c < DC16; ++c) {
          unsafeAtomicAdd(&dk[roff + c * 16 + llo], dk_acc[c][i]);
          unsafeAtomicAdd(&dv[roff + c * 16 + llo], dv_acc[c][i]);
        }
      }
    }
  }
}

}  // namespace attnbwd

// ------------------------------------------------------------- launchers
void launch_attn_bwd_delta(float* delta, const bf16_t* dout,
                           const bf16_t* out, long rows, int d,
                           hipStream_t stream) {
  const int threads = 256;
  const long waves = (rows + 3) / 4;
  const int blocks = (int)min(waves, (long)2048);
  if (d == 128)
    attnbwd::attn_bwd_delta_kernel<128><<<blocks, threads, 0, stream>>>(
        delta, dout, out, rows);
  else
    attnbwd::attn_bwd_delta_kernel<64><<<blocks, threads, 0, stream>>>(
        delta, dout, out, rows);
}

void launch_attn_bwd(bf16_t* dq, float* dk, float* dv, const bf16_t* q,
                     const bf16_t* k, const bf16_t* v, const bf16_t* dout,
                     const float* lse, const float* delta,
                     const int* seq_start, int num_seqs, int max_len,
                     int hq, int hkv, int d, float scale,
                     hipStream_t stream) {
  const int tiles = (max_len + BW_TILE - 1) / BW_TILE;
  dim3 block(BW_WAVES * WAVE);
  dim3 grid_q(num_seqs, hq, tiles);
  dim3 grid_kv(num_seqs, hq, tiles);
  if (d == 128) {
    attnbwd::attn_bwd_dq_kernel<128><<<grid_q, block, 0, stream>>>(
        dq, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
    attnbwd::attn_bwd_dkv_kernel<128><<<grid_kv, block, 0, stream>>>(
        dk, dv, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
  } else if (d == 64) {
    attnbwd::attn_bwd_dq_kernel<64><<<grid_q, block, 0, stream>>>(
        dq, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
    attnbwd::attn_bwd_dkv_kernel<64><<<grid_kv, block, 0, stream>>>(
        dk, dv, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
  }
}

bool attn_bwd_supported(int d) { return d == 128 || d == 64; }

// dkv alone (the mixed-dispatch backward pairs the V5 dq kernel with THIS
// dkv: the V5-structure dkv measured 2.3x SLOWER — 2104 vs 918 us/call on
// falcon MQA — despite winning on dq/fwd; see attention_v5.hip STATUS)
void launch_attn_bwd_dkv_only(float* dk, float* dv, const bf16_t* q,
                              const bf16_t* k, const bf16_t* v,
                              const bf16_t* dout, const float* lse,
                              const float* delta, const int* seq_start,
                              int num_seqs, int max_len, int hq, int hkv,
                              int d, float scale, hipStream_t stream) {
  const int tiles = (max_len + BW_TILE - 1) / BW_TILE;
  dim3 block(BW_WAVES * WAVE);
  dim3 grid_kv(num_seqs, hq, tiles);
  if (d == 128)
    attnbwd::attn_bwd_dkv_kernel<128><<<grid_kv, block, 0, stream>>>(
        dk, dv, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
  else
    attnbwd::attn_bwd_dkv_kernel<64><<<grid_kv, block, 0, stream>>>(
        dk, dv, q, k, v, dout, lse, delta, seq_start, hq, hkv, scale);
}
