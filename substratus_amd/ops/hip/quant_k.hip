// K-quant (GGUF Q4_K / Q6_K) fused GEMV + dequant kernels — the
// super-block formats real ecosystem *.Q4_K_M.gguf files use (the
// reference served them through llama.cpp; models/gguf.py has the codec
// spec and the torch reference these kernels are parity-tested against).
//
// Storage (repacked at load by models/quant.py repack_k for aligned GPU
// access; llama.cpp's packed blocks are 144/210 bytes — 210 is not even
// 4-aligned):
//   q4_K: scales fp16 [N, 2*SB] (d, dmin interleaved per super-block);
//         qs uint8 [N, SB*8 (6-bit sc) | SB*8 (6-bit mn) | K/2 nibbles]
//   q6_K: scales fp16 [N, SB] (d); qs uint8 [N, K/2 ql | K/4 qh | K/16 sc]
//   SB = K/256 super-blocks per row.
//
// GEMV (M <= 8 decode): one wave per output row (4 waves per block);
// q4_K lanes each own a 64-element chunk (2 sub-scales), q6_K lanes a
// 128-element half; x is L2-resident (re-read by every row).
#include "common.h"

typedef _Float16 f16_t;

namespace quantk {

DEV_INLINE float bf_at(const uint4v& v, int i) {
  const unsigned u = ((const unsigned*)&v)[i >> 1];
  return (i & 1) ? __uint_as_float(u & 0xffff0000u)
                 : __uint_as_float((u & 0xffffu) << 16);
}

template <int M, bool Q6>
__global__ __launch_bounds__(256)
void quant_gemv_k_kernel(bf16_t* __restrict__ y,        // [M, N]
                         const bf16_t* __restrict__ x,  // [M, K]
                         const f16_t* __restrict__ scales,
                         const unsigned char* __restrict__ qs,
                         int n, int k) {
  const int row = blockIdx.x * 4 + threadIdx.x / WAVE;
  if (row >= n) return;
  const int lane = threadIdx.x % WAVE;
  const int sb = k / 256;
  float acc[M];
  #pragma unroll
  for (int m = 0; m < M; ++m) acc[m] = 0.f;

  if (!Q6) {
    // ---- q4_K: 64-element chunks, 2 sub-scales each -------------------
    const long row_q = (long)row * (sb * 16 + k / 2);
    const unsigned char* scb = qs + row_q;          // 6-bit sc, SB*8
    const unsigned char* mnb = scb + sb * 8;        // 6-bit mn, SB*8
    const uint4* nib = reinterpret_cast<const uint4*>(mnb + sb * 8);
    const f16_t* sr = scales + (long)row * 2 * sb;
    const int nchunk = k / 64;
    for (int c = lane; c < nchunk; c += WAVE) {
      const int s = c >> 2, qt = c & 3;
      const float d = (float)sr[2 * s];
      const float dm = (float)sr[2 * s + 1];
      const float sc0 = d * scb[s * 8 + 2 * qt];
      const float sc1 = d * scb[s * 8 + 2 * qt + 1];
      const float m0 = dm * mnb[s * 8 + 2 * qt];
      const float m1 = dm * mnb[s * 8 + 2 * qt + 1];
      const uint4 q0 = nib[c * 2];      // 32 bytes = 64 nibbles
      const uint4 q1 = nib[c * 2 + 1];
      const unsigned w[8] = {q0.x, q0.y, q0.z, q0.w, q1.x, q1.y, q1.z, q1.w};
      #pragma unroll
      for (int m = 0; m < M; ++m) {
        const uint4v* xr = reinterpret_cast<const uint4v*>(
            x + (long)m * k + c * 64);
        float dot0 = 0.f, dot1 = 0.f, sx0 = 0.f, sx1 = 0.f;
        #pragma unroll
        for (int i = 0; i < 8; ++i) {         // 8 x-vectors of 8 bf16
          const uint4v xv = xr[i];
          #pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int e = i * 8 + j;          // element in [0, 64)
            const float xvf = bf_at(xv, j);
            // byte e%32 of the chunk holds elem e (lo) / e+32 (hi)
            const unsigned byte = (w[(e & 31) >> 2] >> (((e & 31) & 3) * 8))
                                  & 0xFF;
            const float q = (e < 32) ? (float)(byte & 0xF)
                                     : (float)(byte >> 4);
            if (e < 32) { dot0 += q * xvf; sx0 += xvf; }
            else        { dot1 += q * xvf; sx1 += xvf; }
          }
        }
        acc[m] += sc0 * dot0 - m0 * sx0 + sc1 * dot1 - m1 * sx1;
      }
    }
  } else {
    // ---- q6_K: 128-element halves, 8 sub-scales each -------------------
    const long row_q = (long)row * (k / 2 + k / 4 + k / 16);
    const unsigned char* qlb = qs + row_q;
    const unsigned char* qhb = qlb + k / 2;
    const signed char* sgb = reinterpret_cast<const signed char*>(
        qhb + k / 4);
    const f16_t* sr = scales + (long)row * sb;
    const int nhalf = k / 128;
    for (int c = lane; c < nhalf; c += WAVE) {
      const int s = c >> 1, hf = c & 1;
      const float d = (float)sr[s];
      const uint4* qlv = reinterpret_cast<const uint4*>(
          qlb + s * 128 + hf * 64);
      const uint4* qhv = reinterpret_cast<const uint4*>(
          qhb + s * 64 + hf * 32);
      const signed char* scl = sgb + s * 16 + hf * 8;
      unsigned ql[16], qh[8];
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const uint4 v = qlv[i];
        ql[i * 4] = v.x; ql[i * 4 + 1] = v.y;
        ql[i * 4 + 2] = v.z; ql[i * 4 + 3] = v.w;
      }
      #pragma unroll
      for (int i = 0; i < 2; ++i) {
        const uint4 v = qhv[i];
        qh[i * 4] = v.x; qh[i * 4 + 1] = v.y;
        qh[i * 4 + 2] = v.z; qh[i * 4 + 3] = v.w;
      }
      float sub[8];
      #pragma unroll
      for (int g = 0; g < 8; ++g) sub[g] = (float)scl[g];
      #pragma unroll
      for (int m = 0; m < M; ++m) {
        const uint4v* xr = reinterpret_cast<const uint4v*>(
            x + (long)m * k + s * 256 + hf * 128);
        float gacc[8];
        #pragma unroll
        for (int g = 0; g < 8; ++g) gacc[g] = 0.f;
        #pragma unroll
        for (int qt = 0; qt < 4; ++qt) {
          #pragma unroll
          for (int l = 0; l < 32; ++l) {
            const unsigned qlb_ = (ql[((qt & 1) * 32 + l) >> 2]
                                   >> ((l & 3) * 8)) & 0xFF;
            const unsigned qhb_ = (qh[l >> 2] >> ((l & 3) * 8)) & 0xFF;
            const unsigned lo = (qt < 2) ? (qlb_ & 0xF) : (qlb_ >> 4);
            const int q = (int)(lo | (((qhb_ >> (qt * 2)) & 3) << 4)) - 32;
            const int e = qt * 32 + l;    // element within the half
            const float xvf = bf_at(xr[e >> 3], e & 7);
            gacc[qt * 2 + (l >> 4)] += (float)q * xvf;
          }
        }
        float hsum = 0.f;
        #pragma unroll
        for (int g = 0; g < 8; ++g) hsum += sub[g] * gacc[g];
        acc[m] += d * hsum;
      }
    }
  }
  #pragma unroll
  for (int m = 0; m < M; ++m) {
    float a = acc[m];
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      a += __shfl_xor(a, off, WAVE);
    if (lane == 0) y[(long)m * n + row] = f2bf(a);
  }
}

// ---- dequant (prefill / large-M path: expand to bf16, hipBLASLt after)
template <bool Q6>
__global__ void dequant_k_kernel(bf16_t* __restrict__ out,  // [N, K]
                                 const f16_t* __restrict__ scales,
                                 const unsigned char* __restrict__ qs,
                                 int n, int k) {
  const int sb = k / 256;
  const long total = (long)n * (Q6 ? k / 128 : k / 64);
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    if (!Q6) {
      const int row = (int)(t / (k / 64));
      const int c = (int)(t % (k / 64));
      const int s = c >> 2, qt = c & 3;
      const long row_q = (long)row * (sb * 16 + k / 2);
      const unsigned char* scb = qs + row_q;
      const unsigned char* mnb = scb + sb * 8;
      const unsigned char* nib = mnb + sb * 8 + c * 32;
      const f16_t* sr = scales + (long)row * 2 * sb;
      const float d = (float)sr[2 * s], dm = (float)sr[2 * s + 1];
      const float sc0 = d * scb[s * 8 + 2 * qt];
      const float sc1 = d * scb[s * 8 + 2 * qt + 1];
      const float m0 = dm * mnb[s * 8 + 2 * qt];
      const float m1 = dm * mnb[s * 8 + 2 * qt + 1];
      bf16_t* op = out + (long)row * k + c * 64;
      for (int l = 0; l < 32; ++l) {
        const unsigned b = nib[l];
        op[l] = f2bf(sc0 * (float)(b & 0xF) - m0);
        op[l + 32] = f2bf(sc1 * (float)(b >> 4) - m1);
      }
    } else {
      const int row = (int)(t / (k / 128));
      const int c = (int)(t % (k / 128));
      const int s = c >> 1, hf = c & 1;
      const long row_q = (long)row * (k / 2 + k / 4 + k / 16);
      const unsigned char* qlb = qs + row_q + s * 128 + hf * 64;
      const unsigned char* qhb = qs + row_q + k / 2 + s * 64 + hf * 32;
      const signed char* scl = reinterpret_cast<const signed char*>(
          qs + row_q + k / 2 + k / 4) + s * 16 + hf * 8;
      const float d = (float)scales[(long)row * sb + s];
      bf16_t* op = out + (long)row * k + s * 256 + hf * 128;
      for (int l = 0; l < 32; ++l) {
        const unsigned lo1 = qlb[l], lo2 = qlb[l + 32], h = qhb[l];
        const int e0 = (int)((lo1 & 0xF) | ((h & 3) << 4)) - 32;
        const int e1 = (int)((lo2 & 0xF) | (((h >> 2) & 3) << 4)) - 32;
        const int e2 = (int)((lo1 >> 4) | (((h >> 4) & 3) << 4)) - 32;
        const int e3 = (int)((lo2 >> 4) | (((h >> 6) & 3) << 4)) - 32;
        op[l] = f2bf(d * (float)scl[0 + (l >> 4)] * e0);
        op[l + 32] = f2bf(d * (float)scl[2 + (l >> 4)] * e1);
        op[l + 64] = f2bf(d * (float)scl[4 + (l >> 4)] * e2);
        op[l + 96] = f2bf(d * (float)scl[6 + (l >> 4)] * e3);
      }
    }
  }
}

}  // namespace quantk

void launch_quant_gemv_k(bf16_t* y, const bf16_t* x, const void* scales,
                         const unsigned char* qs, int m, int n, int k,
                         bool q6, hipStream_t stream) {
  dim3 grid((n + 3) / 4), block(256);
  const f16_t* s = (const f16_t*)scales;
  // no macro around <<<>>> — torch's hipify mangles kernel launches
  // inside macro bodies
  switch (m) {
    case 1:
      if (q6) {
        quantk::quant_gemv_k_kernel<1, true><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      } else {
        quantk::quant_gemv_k_kernel<1, false><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      }
      break;
    case 2:
      if (q6) {
        quantk::quant_gemv_k_kernel<2, true><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      } else {
        quantk::quant_gemv_k_kernel<2, false><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      }
      break;
    case 3:
      if (q6) {
        quantk::quant_gemv_k_kernel<3, true><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      } else {
        quantk::quant_gemv_k_kernel<3, false><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      }
      break;
    case 4:
      if (q6) {
        quantk::quant_gemv_k_kernel<4, true><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      } else {
        quantk::quant_gemv_k_kernel<4, false><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      }
      break;
    case 5:
      if (q6) {
        quantk::quant_gemv_k_kernel<5, true><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      } else {
        quantk::quant_gemv_k_kernel<5, false><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      }
      break;
    case 6:
      if (q6) {
        quantk::quant_gemv_k_kernel<6, true><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      } else {
        quantk::quant_gemv_k_kernel<6, false><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      }
      break;
    case 7:
      if (q6) {
        quantk::quant_gemv_k_kernel<7, true><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      } else {
        quantk::quant_gemv_k_kernel<7, false><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      }
      break;
    case 8:
      if (q6) {
        quantk::quant_gemv_k_kernel<8, true><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      } else {
        quantk::quant_gemv_k_kernel<8, false><<<grid, block, 0, stream>>>(
            y, x, s, qs, n, k);
      }
      break;
    default:
      break;
  }
}

void launch_dequant_k(bf16_t* out, const void* scales,
                      const unsigned char* qs, int n, int k, bool q6,
                      hipStream_t stream) {
  const long chunks = (long)n * (q6 ? k / 128 : k / 64);
  const int blocks = (int)min((chunks + 255) / 256, (long)4096);
  if (q6)
    quantk::dequant_k_kernel<true><<<blocks, 256, 0, stream>>>(
        out, (const f16_t*)scales, qs, n, k);
  else
    quantk::dequant_k_kernel<false><<<blocks, 256, 0, stream>>>(
        out, (const f16_t*)scales, qs, n, k);
}
