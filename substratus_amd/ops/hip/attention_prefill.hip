// Varlen causal prefill attention (SURVEY.md §2.5 ledger item 2) —
// flash-style tiled attention on MFMA (mfma_f32_16x16x32_bf16) for prompt
// ingestion.
//
// Geometry: one workgroup (4 waves) per (sequence, q-head, 64-row q tile);
// wave w owns q rows [tile*64 + 16w, +16). Keys stream in chunks of 32.
//
// V2 data movement (V1 measured 141 TF — dominated by scalar LDS reads and
// unswizzled layouts):
//   * K and V chunks are staged cooperatively into LDS with coalesced uint4
//     global loads; V is TRANSPOSED at staging (vT[d][k]) so every MFMA
//     B-fragment — K^T for QK^T, V for P·V — is ONE contiguous 16-byte
//     ds_read_b128 per lane instead of 8 scalar reads.
//   * All LDS tiles are XOR-swizzled at 16-byte-chunk granularity
//     (chunk ^= row & (row_chunks-1), guide T2): fragment reads put 16 lanes
//     on 16 DIFFERENT rows at the same column range, which is the textbook
//     16-way ds_read_b128 bank conflict on a linear layout.
//   * P transposes C-layout → A-layout through a swizzled LDS tile the same
//     way (scalar writes, one uint4 read per fragment).
//
// MFMA fragment maps (gfx950, 16x16x32 bf16; verified by the mfma_probe GPU
// test against torch.matmul on asymmetric data):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j]  (j = 0..7, short8)
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C/D: lane l holds C[row = (l>>4)*4 + i][col = l&15]  (i = 0..3, float4)
#include "common.h"

typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(4))) float float4v;

// ds_read_b64_tr_b16 (guide T10): given per-lane LDS address
// base + ((l>>4)*64 + (l&15)) elems into a [16][16] bf16 image, lane l
// receives column (l&15) of rows (l>>4)*4 + j — a free 4x16 transpose.
DEV_INLINE short4v tr16_read(const short* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) short4v*)p);
}

#define PF_WAVES 4
#define QTILE 64   // q rows per workgroup (16 per wave)
#define KCHUNK 64  // keys per iteration (64: half the barriers of 32,
                   // double the MFMA per phase — guide: BK 32->64 +7-16%)

// XOR-swizzled 16-B-chunk address inside a row-major LDS tile.
// row_bytes must be a power-of-two multiple of 16.
template <int ROW_BYTES>
DEV_INLINE int swz_chunk(int row, int chunk) {
  constexpr int NCH = ROW_BYTES / 16;
  return row * NCH + (chunk ^ (row & (NCH - 1)));
}

// VARIANT (ablation probe, guide §5.4 rules 8/17 — ablate before
// optimizing; non-zero variants are TIMING-ONLY, numerics are wrong):
//   0 full   1 no-softmax-VALU   2 no-PV-mfma   3 no-V-staging
//   4 no-QK-mfma (QK loads kept alive)
template <int D, int VARIANT = 0>
__global__ __launch_bounds__(PF_WAVES * WAVE)
void attn_prefill_kernel(bf16_t* __restrict__ out,      // [T, HQ, D]
                         const bf16_t* __restrict__ q,  // [T, HQ, D]
                         const bf16_t* __restrict__ k,  // [T, HKV, D]
                         const bf16_t* __restrict__ v,  // [T, HKV, D]
                         const int* __restrict__ seq_start,  // [S+1]
                         int hq, int hkv, float scale,
                         float* __restrict__ lse) {  // [T, HQ] or null:
                         // log2-domain row logsumexp (m + log2 l, with
                         // scale*log2e folded in) saved for the training
                         // backward (attention_bwd.hip recomputes
                         // P = exp2(s*scale2 - lse))
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
      qa[c] = *reinterpret_cast<const short8*>(qp + c * 32 + lhi * 8);
    }
  }

  float m[4], l[4];             // per held q row (i = 0..3 → q = lhi*4+i)
  float4v o_acc[DC16];
  #pragma unroll
  for (int i = 0; i < 4; ++i) { m[i] = -INFINITY; l[i] = 0.f; }
  #pragma unroll
  for (int c = 0; c < DC16; ++c) o_acc[c] = (float4v){0.f, 0.f, 0.f, 0.f};

  // LDS tiles:
  //   k_tile: [KCHUNK rows][D cols], XOR-swizzled — QK^T B-fragments
  //   v_img:  row-major [16][16] tr-read images, rows k-interleaved
  //           ((k/8)*4 + k%4, k%8>=4 selects the sibling image) so TWO
  //           ds_read_b64_tr_b16 deliver a P·V B-fragment — staging stays
  //           ONE coalesced uint4 write per loaded uint4 (the 8-scalar-write
  //           transpose was 35% of kernel time, profiles/
  //           prefill_ablation_r01.txt)
  //   p_img:  per wave, the SAME k-interleaved tr-read image layout as
  //           v_img but storing P TRANSPOSED (img[r(k)][qrow]): a lane's 4
  //           C-fragment rows (qrow = lhi*4+i) are CONSECUTIVE shorts, so
  //           the whole P store is ONE packed b64 write per 16-key subchunk
  //           (was 16 scalar ds_write_b16), and the P·V A-fragment is two
  //           ds_read_b64_tr_b16 — same probed mapping as the V B-frags
  //           (tr16 out[l][j] = img[(l>>4)*4+j][l&15] = P^T[k][qrow] gives
  //           exactly A[row=llo][k=lhi*8+sel*4+j])
  __shared__ uint4 k_tile[2][KCHUNK * D * 2 / 16];
  __shared__ short v_img[2][DC16][KCHUNK / 32][2][16][16];
  __shared__ short p_img[PF_WAVES][KCHUNK / 32][2][16][16];

  const int kmax_tile = min(q0_tile + QTILE, n);       // causal bound of tile
  const int nchunks = (kmax_tile + KCHUNK - 1) / KCHUNK;
  const int kmax_wave = min(q0 + 16, n);               // this wave's bound
  constexpr int NSC = KCHUNK / 16;

  // ---- software-pipelined chunk loop ------------------------------------
  // The online softmax is a SERIAL VALU chain (rowmax reduce -> exp2 ->
  // rowsum reduce) between the two MFMA phases — the DCE ablation put it at
  // ~35% of kernel time with neither pipe saturated. Skewing the pipeline
  // one chunk (QK^T of chunk kc+1 issues right before softmax of kc, from
  // the other LDS buffer) lets the scheduler interleave those independent
  // chains. Chunk j is staged at iteration j-2 (double-buffered k/v tiles),
  // QK'd at iteration j-1, softmax+PV'd at iteration j; two barriers per
  // iteration, same as the unpipelined loop.

  auto stage_chunk = [&](int kc2, int buf) {
    if (kc2 >= nchunks) return;
    const int kbase = kc2 * KCHUNK;
    constexpr int LOADS = KCHUNK * D / 8 / (PF_WAVES * WAVE);  // per thread
    #pragma unroll
    for (int i = 0; i < LOADS; ++i) {
      const int flat = (i * PF_WAVES * WAVE + threadIdx.x) * 8;  // elem idx
      const int row = flat / D, col = flat % D;
      const bool ok = kbase + row < n;
      const long goff = (long)(base + kbase + row) * krow_stride +
                        (long)kvh * D + col;
      // K: row-major, swizzled by 16-B chunk
      uint4 kv4 = ok ? *reinterpret_cast<const uint4*>(k + goff)
                     : make_uint4(0, 0, 0, 0);
      k_tile[buf][swz_chunk<D * 2>(row, col * 2 / 16)] = kv4;
      uint4 vv4 = ok ? *reinterpret_cast<const uint4*>(v + goff)
                     : make_uint4(0, 0, 0, 0);
      if (VARIANT == 3) {  // keep the loads, skip the image writes
        asm volatile("" :: "v"(vv4.x), "v"(vv4.y));
        continue;
      }
      // one uint4 into the tr-read image: c = col/16, half = col%16
      {
        const int c = col / 16, off = col % 16;   // off in {0, 8}
        const int ks = row / 32, kk = row % 32;
        const int sel = (kk % 8) >= 4;
        const int r = (kk / 8) * 4 + (kk % 4);
        *reinterpret_cast<uint4*>(&v_img[buf][c][ks][sel][r][off]) = vv4;
      }
    }
  };

  // does this wave compute chunk kc2?
  auto wave_has = [&](int kc2) {
    return wave_live && kc2 < nchunks && kc2 * KCHUNK < kmax_wave;
  };

  // QK^T of one chunk: KCHUNK/16 16-key subchunks (B-frags from k_tile)
  auto qk_chunk = [&](int buf, float4v (&sf)[NSC]) {
    #pragma unroll
    for (int sc = 0; sc < NSC; ++sc) {
      float4v acc = {0.f, 0.f, 0.f, 0.f};
      const int key = sc * 16 + llo;   // B col = this lane's key row
      #pragma unroll
      for (int c = 0; c < DC32; ++c) {
        // kdim byte offset = (c*32 + lhi*8)*2 -> chunk c*4 + lhi
        const uint4 kb4 = k_tile[buf][swz_chunk<D * 2>(key, c * 4 + lhi)];
        const short8 kb = *reinterpret_cast<const short8*>(&kb4);
        if (VARIANT == 4) {
          asm volatile("" :: "v"(kb4.x));  // keep the ds_read alive
        } else {
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa[c], kb, acc,
                                                        0, 0, 0);
        }
      }
      sf[sc] = acc;
    }
  };

  // prologue: chunks 0 and 1 staged, QK of chunk 0 in flight
  stage_chunk(0, 0);
  stage_chunk(1, 1);
  __syncthreads();
  float4v sfrag[NSC], sf_nxt[NSC];
  if (wave_has(0)) qk_chunk(0, sfrag);

  for (int kc = 0; kc < nchunks; ++kc) {
    const int cur = kc & 1;
    const int kbase = kc * KCHUNK;
    // ---- QK^T of the NEXT chunk (independent of this chunk's softmax) --
    if (wave_has(kc + 1)) qk_chunk(cur ^ 1, sf_nxt);

    if (wave_has(kc)) {
      bool do_pv = true;
      if (VARIANT == 1) {
        // skip the softmax VALU: p := raw scores (timing-only), l := +16
        #pragma unroll
        for (int sc = 0; sc < NSC; ++sc) {
          unsigned plo_, phi_;
          asm("v_cvt_pk_bf16_f32 %0, %1, %2"
              : "=v"(plo_) : "v"(sfrag[sc][0]), "v"(sfrag[sc][1]));
          asm("v_cvt_pk_bf16_f32 %0, %1, %2"
              : "=v"(phi_) : "v"(sfrag[sc][2]), "v"(sfrag[sc][3]));
          const int kf = sc * 16 + llo, kk = kf & 31;
          *reinterpret_cast<uint2*>(
              &p_img[wave][kf >> 5][(kk % 8) >= 4]
                    [(kk / 8) * 4 + (kk % 4)][lhi * 4]) =
              make_uint2(plo_, phi_);
        }
        #pragma unroll
        for (int i = 0; i < 4; ++i) l[i] += 16.f;
      } else {
        // ---- mask + online softmax (C layout), log2 domain --------------
        // All exponentials run as exp2 with scale*log2(e) folded into the
        // score once (saves a VALU mul per element before every v_exp_f32);
        // chunks entirely below the causal diagonal (kbase+KCHUNK-1 <= q0,
        // in-bounds) skip the per-element mask compare+select.
        const float scale2 = scale * 1.44269504088896340736f;  // log2(e)
        float pm[NSC][4];
        float rowmax[4];
        #pragma unroll
        for (int i = 0; i < 4; ++i) rowmax[i] = -INFINITY;
        if (kbase + KCHUNK - 1 <= q0 && kbase + KCHUNK <= n) {
          #pragma unroll
          for (int sc = 0; sc < NSC; ++sc) {
            #pragma unroll
            for (int i = 0; i < 4; ++i) {
              const float sv = sfrag[sc][i] * scale2;
              pm[sc][i] = sv;
              rowmax[i] = fmaxf(rowmax[i], sv);
            }
          }
        } else {
          int klim[4];
          #pragma unroll
          for (int i = 0; i < 4; ++i)
            klim[i] = min(q0 + lhi * 4 + i, n - 1);
          #pragma unroll
          for (int sc = 0; sc < NSC; ++sc) {
            const int kpos = kbase + sc * 16 + llo;
            #pragma unroll
            for (int i = 0; i < 4; ++i) {
              float sv =
                  (kpos <= klim[i]) ? sfrag[sc][i] * scale2 : -INFINITY;
              pm[sc][i] = sv;
              rowmax[i] = fmaxf(rowmax[i], sv);
            }
          }
        }
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
          alpha[i] =
              (m[i] == -INFINITY) ? 0.f : __builtin_amdgcn_exp2f(m[i] - mn);
          m[i] = mn;
        }
        // P = exp2(S - m); row-sum; one packed b64 write per subchunk into
        // the transposed tr-read image (lane's 4 rows: consecutive shorts)
        float rowsum[4] = {0.f, 0.f, 0.f, 0.f};
        #pragma unroll
        for (int sc = 0; sc < NSC; ++sc) {
          float pv4[4];
          #pragma unroll
          for (int i = 0; i < 4; ++i) {
            pv4[i] = (pm[sc][i] == -INFINITY || m[i] == -INFINITY)
                         ? 0.f : __builtin_amdgcn_exp2f(pm[sc][i] - m[i]);
            rowsum[i] += pv4[i];
          }
          unsigned plo_, phi_;
          asm("v_cvt_pk_bf16_f32 %0, %1, %2"
              : "=v"(plo_) : "v"(pv4[0]), "v"(pv4[1]));
          asm("v_cvt_pk_bf16_f32 %0, %1, %2"
              : "=v"(phi_) : "v"(pv4[2]), "v"(pv4[3]));
          const int kf = sc * 16 + llo, kk = kf & 31;
          *reinterpret_cast<uint2*>(
              &p_img[wave][kf >> 5][(kk % 8) >= 4]
                    [(kk / 8) * 4 + (kk % 4)][lhi * 4]) =
              make_uint2(plo_, phi_);
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
      }
      if (VARIANT == 2) {
        asm volatile("" :: "v"(sfrag[0][0]));  // keep QK^T alive
        do_pv = false;
      }
      if (do_pv) {
        // ---- P·V over KCHUNK keys in 32-k MFMA steps; A and B frags both
        // come from tr-read images (two ds_read_b64_tr_b16 each)
        #pragma unroll
        for (int ks = 0; ks < KCHUNK / 32; ++ks) {
          // A-frag: P^T image -> out[l][j] = P[qrow=llo][k=lhi*8+sel*4+j]
          const short4v alo =
              tr16_read(&p_img[wave][ks][0][0][0] + lane * 4);
          const short4v ahi =
              tr16_read(&p_img[wave][ks][1][0][0] + lane * 4);
          short8 pa;
          pa[0] = alo[0]; pa[1] = alo[1]; pa[2] = alo[2]; pa[3] = alo[3];
          pa[4] = ahi[0]; pa[5] = ahi[1]; pa[6] = ahi[2]; pa[7] = ahi[3];
          #pragma unroll
          for (int c = 0; c < DC16; ++c) {
            // two hardware-transpose reads assemble the B-frag
            // (k = lhi*8+j). ds_read_b64_tr_b16 lane mapping (probed on
            // gfx950, profiles/tr16_probe_r01.txt): with per-lane address
            // base+lane*4 shorts into a row-major [16][16] image,
            // out[l][j] = img[(l>>4)*4 + j][l&15] — the address is lane*4
            // FLAT (a [lhi*4][llo] address has 2-byte low bits the
            // 8B-aligned b64 load silently masks off).
            const short4v lo =
                tr16_read(&v_img[cur][c][ks][0][0][0] + lane * 4);
            const short4v hi =
                tr16_read(&v_img[cur][c][ks][1][0][0] + lane * 4);
            short8 vb;
            vb[0] = lo[0]; vb[1] = lo[1]; vb[2] = lo[2]; vb[3] = lo[3];
            vb[4] = hi[0]; vb[5] = hi[1]; vb[6] = hi[2]; vb[7] = hi[3];
            o_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                pa, vb, o_acc[c], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();                 // every wave done reading buffer cur
    stage_chunk(kc + 2, cur);        // overwrite it with chunk kc+2
    __syncthreads();                 // staged before next iter's QK reads
    #pragma unroll
    for (int sc = 0; sc < NSC; ++sc) sfrag[sc] = sf_nxt[sc];
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
      if (lse != nullptr && llo == 0)
        // v_log_f32 is log BASE 2 — exactly the domain m lives in
        lse[(long)(base + qpos) * hq + head] =
            m[i] + __builtin_amdgcn_logf(l[i]);
    }
  }
}

void launch_attn_prefill(bf16_t* out, const bf16_t* q, const bf16_t* k,
                         const bf16_t* v, const int* seq_start, int num_seqs,
                         int max_len, int hq, int hkv, int d, float scale,
                         hipStream_t stream, float* lse) {
  const int tiles = (max_len + QTILE - 1) / QTILE;
  dim3 grid(num_seqs, hq, tiles);
  dim3 block(PF_WAVES * WAVE);
  if (d == 128) {
    attn_prefill_kernel<128><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, lse);
  } else if (d == 64) {
    attn_prefill_kernel<64><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, lse);
  }
}

bool attn_prefill_supported(int d) { return d == 128 || d == 64; }

// timing-only ablation probe (D=128): variant semantics in the kernel header
void launch_attn_prefill_probe(int variant, bf16_t* out, const bf16_t* q,
                               const bf16_t* k, const bf16_t* v,
                               const int* seq_start, int num_seqs,
                               int max_len, int hq, int hkv, float scale,
                               hipStream_t stream) {
  const int tiles = (max_len + QTILE - 1) / QTILE;
  dim3 grid(num_seqs, hq, tiles);
  dim3 block(PF_WAVES * WAVE);
  switch (variant) {
    case 1: attn_prefill_kernel<128, 1><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, nullptr); break;
    case 2: attn_prefill_kernel<128, 2><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, nullptr); break;
    case 3: attn_prefill_kernel<128, 3><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, nullptr); break;
    case 4: attn_prefill_kernel<128, 4><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, nullptr); break;
    default: attn_prefill_kernel<128, 0><<<grid, block, 0, stream>>>(
        out, q, k, v, seq_start, hq, hkv, scale, nullptr); break;
  }
}

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
