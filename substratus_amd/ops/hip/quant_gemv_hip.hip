#include "hip/hip_runtime.h"
// Fused block-dequant GEMV + dequant kernels for GGUF-quantized weights
// (SURVEY.md §2.5 ledger item 8: "block-dequant fused into GEMV for
// single-token decode" — the first-party replacement for the reference's
// llama.cpp serving path).
//
// Weight storage (repacked at load from GGUF for aligned GPU access):
//   scales: fp16 [N, K/32]
//   qs:     q4_0 -> uint8 [N, K/2] (nibble i of byte b packs weights b and
//           b+16 of the 32-block: llama.cpp layout), q8_0 -> int8 [N, K]
//
// quant_gemv (M <= 8, latency-bound decode): one wave per output row; each
// lane owns blocks lane, lane+64, ... (scale 2 B + 16 B nibbles via uint4);
// x read through L2 (it is re-read by every row — L2/L3 resident);
// wave-reduced fp32 accumulation.
//
// dequant (prefill / large M): block-parallel expansion to bf16, then the
// caller runs hipBLASLt.
#include "common.h"

typedef _Float16 f16_t;

template <int M, bool Q8>
__global__ __launch_bounds__(256)
void quant_gemv_kernel(bf16_t* __restrict__ y,         // [M, N]
                       const bf16_t* __restrict__ x,   // [M, K]
                       const f16_t* __restrict__ scales,  // [N, K/32]
                       const unsigned char* __restrict__ qs,
                       int n, int k) {
  const int row = blockIdx.x * 4 + threadIdx.x / WAVE;  // 4 waves = 4 rows
  if (row >= n) return;
  const int lane = threadIdx.x % WAVE;
  const int nblocks = k / 32;
  const f16_t* srow = scales + (long)row * nblocks;
  float acc[M];
  #pragma unroll
  for (int m = 0; m < M; ++m) acc[m] = 0.f;

  if (!Q8) {
    // 2-deep software pipeline: block b+WAVE's qs/scale loads are issued
    // before block b's dequant-dot. Tail handling is BRANCHLESS: indices
    // clamp to the last block and a zero scale masks the redundant work —
    // a per-lane `if (b < nblocks)` guard around the loads makes hipcc
    // branch and drain vmcnt per iteration (guide §5 trap 4(c)), which
    // measured SLOWER than no pipeline at all.
    const uint4* qrow = reinterpret_cast<const uint4*>(qs + (long)row * k / 2);
    const int last = nblocks - 1;
    int b = lane;
    uint4 nib_c = qrow[min(b, last)];
    float sv0 = (float)srow[min(b, last)];
    float s_c = (b <= last) ? sv0 : 0.f;
    const int iters = (nblocks + WAVE - 1) / WAVE;
    for (int it = 0; it < iters; ++it, b += WAVE) {
      const int bn = min(b + WAVE, last);
      const uint4 nib_n = qrow[bn];
      const float svn = (float)srow[bn];  // load always, select after
      const float s_n = (b + WAVE <= last) ? svn : 0.f;
      const int bcl = min(b, last);       // clamp x address for masked lanes
      const unsigned int w[4] = {nib_c.x, nib_c.y, nib_c.z, nib_c.w};
      // issue every row's x loads before any compute (per-m inline loads
      // serialize the M>1 case: 0.94 TB/s vs 2.34 at M=1)
      uint4 xall[M][4];
      #pragma unroll
      for (int m = 0; m < M; ++m) {
        const uint4* xb =
            reinterpret_cast<const uint4*>(x + (long)m * k + bcl * 32);
        #pragma unroll
        for (int c = 0; c < 4; ++c) xall[m][c] = xb[c];
      }
      #pragma unroll
      for (int m = 0; m < M; ++m) {
        float partial = 0.f;
        #pragma unroll
        for (int c = 0; c < 4; ++c) {   // 4 bytes -> weights i..i+3, i+16..19
          unsigned int word = w[c];
          const uint4 xlo4 = xall[m][c >= 2 ? 1 : 0];
          const uint4 xhi4 = xall[m][c >= 2 ? 3 : 2];
          const unsigned int xlo[2] = {c % 2 == 0 ? xlo4.x : xlo4.z,
                                       c % 2 == 0 ? xlo4.y : xlo4.w};
          const unsigned int xhi[2] = {c % 2 == 0 ? xhi4.x : xhi4.z,
                                       c % 2 == 0 ? xhi4.y : xhi4.w};
          #pragma unroll
          for (int j = 0; j < 4; ++j) {
            const float lo = (float)((word >> (8 * j)) & 0xF) - 8.f;
            const float hi = (float)((word >> (8 * j + 4)) & 0xF) - 8.f;
            const float2 xl = unpack_bf16x2(xlo[j / 2]);
            const float2 xh = unpack_bf16x2(xhi[j / 2]);
            const float xv_lo = (j % 2 == 0) ? xl.x : xl.y;
            const float xv_hi = (j % 2 == 0) ? xh.x : xh.y;
            partial += lo * xv_lo + hi * xv_hi;
          }
        }
        acc[m] += partial * s_c;
      }
      nib_c = nib_n;
      s_c = s_n;
    }
  } else {
    const uint4* qrow = reinterpret_cast<const uint4*>(qs + (long)row * k);
    for (int b = lane; b < nblocks; b += WAVE) {
      const float s = (float)srow[b];
      // 32 int8 = 2 × uint4
      const uint4 q0 = qrow[b * 2], q1 = qrow[b * 2 + 1];
      const unsigned int w[8] = {q0.x, q0.y, q0.z, q0.w,
                                 q1.x, q1.y, q1.z, q1.w};
      uint4 xall8[M][4];
      #pragma unroll
      for (int m = 0; m < M; ++m) {
        const uint4* xb =
            reinterpret_cast<const uint4*>(x + (long)m * k + b * 32);
        #pragma unroll
        for (int c = 0; c < 4; ++c) xall8[m][c] = xb[c];
      }
      #pragma unroll
      for (int m = 0; m < M; ++m) {
        float partial = 0.f;
        #pragma unroll
        for (int c = 0; c < 4; ++c) {
          const uint4 xv = xall8[m][c];
          const unsigned int xs[4] = {xv.x, xv.y, xv.z, xv.w};
          #pragma unroll
          for (int j = 0; j < 2; ++j) {  // 2 words of 4 int8 per uint4-pair
            const unsigned int word = w[c * 2 + j];
            #pragma unroll
            for (int t = 0; t < 4; ++t) {
              const float qv = (float)(signed char)((word >> (8 * t)) & 0xFF);
              const float2 xp = unpack_bf16x2(xs[j * 2 + t / 2]);
              partial += qv * ((t % 2 == 0) ? xp.x : xp.y);
            }
          }
        }
        acc[m] += partial * s;
      }
    }
  }
  #pragma unroll
  for (int m = 0; m < M; ++m) {
    float v = wave_sum(acc[m]);
    if (lane == 0) y[(long)m * n + row] = f2bf(v);
  }
}

// expand quantized rows to bf16 (prefill path feeds hipBLASLt)
template <bool Q8>
__global__ void dequant_kernel(bf16_t* __restrict__ out,   // [N, K]
                               const f16_t* __restrict__ scales,
                               const unsigned char* __restrict__ qs,
                               long nblocks_total, int k) {
  // one 32-weight block per thread
  const long blk = blockIdx.x * (long)blockDim.x + threadIdx.x;
  if (blk >= nblocks_total) return;
  const int nb_per_row = k / 32;
  const long row = blk / nb_per_row;
  const int brow = (int)(blk % nb_per_row);
  const float s = (float)scales[blk];
  bf16_t* o = out + row * k + brow * 32;
  if (!Q8) {
    const unsigned char* q = qs + row * (long)(k / 2) + brow * 16;
    #pragma unroll
    for (int i = 0; i < 16; ++i) {
      const unsigned char byte = q[i];
      o[i] = f2bf(((float)(byte & 0xF) - 8.f) * s);
      o[i + 16] = f2bf(((float)(byte >> 4) - 8.f) * s);
    }
  } else {
    const signed char* q =
        reinterpret_cast<const signed char*>(qs) + row * (long)k + brow * 32;
    #pragma unroll
    for (int i = 0; i < 32; ++i) o[i] = f2bf((float)q[i] * s);
  }
}

void launch_quant_gemv(bf16_t* y, const bf16_t* x, const void* scales,
                       const unsigned char* qs, int m, int n, int k,
                       bool q8, hipStream_t stream) {
  dim3 grid((n + 3) / 4);
  dim3 block(256);
  const f16_t* sc = reinterpret_cast<const f16_t*>(scales);
  // braces matter: torch's hipify rewrites <<<>>> and glues a bare `else`
  // onto the launch otherwise
  #define GEMV_CASE(MM)                                                   \
    do {                                                                  \
      if (q8) {                                                           \
       hipLaunchKernelGGL(( quant_gemv_kernel<MM, true>), dim3(grid), dim3(block), 0, stream,           \
            y, x, sc, qs, n, k);                                          \
      } else {                                                            \
       hipLaunchKernelGGL(( quant_gemv_kernel<MM, false>), dim3(grid), dim3(block), 0, stream,          \
            y, x, sc, qs, n, k);                                          \
      }                                                                   \
    } while (0)
  switch (m) {
    case 1: GEMV_CASE(1); break;
    case 2: GEMV_CASE(2); break;
    case 3: GEMV_CASE(3); break;
    case 4: GEMV_CASE(4); break;
    case 5: GEMV_CASE(5); break;
    case 6: GEMV_CASE(6); break;
    case 7: GEMV_CASE(7); break;
    case 8: GEMV_CASE(8); break;
  }
  #undef GEMV_CASE
}

void launch_dequant(bf16_t* out, const void* scales, const unsigned char* qs,
                    long n, int k, bool q8, hipStream_t stream) {
  const long nblocks = n * (long)(k / 32);
  const int bsz = 256;
  const long grid = (nblocks + bsz - 1) / bsz;
  const f16_t* sc = reinterpret_cast<const f16_t*>(scales);
  if (q8)
   hipLaunchKernelGGL(( dequant_kernel<true>), dim3(dim3((unsigned)grid)), dim3(dim3(bsz)), 0, stream, 
        out, sc, qs, nblocks, k);
  else
   hipLaunchKernelGGL(( dequant_kernel<false>), dim3(dim3((unsigned)grid)), dim3(dim3(bsz)), 0, stream, 
        out, sc, qs, nblocks, k);
}
