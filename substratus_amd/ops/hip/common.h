// Common CDNA4 (gfx950) kernel helpers.
//
// Wave width is 64 on CDNA4 (guide §1) — hard-coded; every warp-style idiom
// below is 64-wide. bf16 data moves as packed uint/uint4 (8×bf16 = 16 B per
// lane per load) for full coalescing (guide §5 common-mistake #2).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;

// clang ext_vector (not HIP_vector_type) — required by
// __builtin_nontemporal_load, which rejects class types.
typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;

DEV_INLINE uint4v nt_load_u4(const void* p) {
  return __builtin_nontemporal_load(reinterpret_cast<const uint4v*>(p));
}

typedef __attribute__((ext_vector_type(2))) unsigned int uint2v;

DEV_INLINE uint2v nt_load_u2(const void* p) {
  return __builtin_nontemporal_load(reinterpret_cast<const uint2v*>(p));
}

// OCP fp8 e4m3 <-> f32 via the gfx950 hardware converts (OCP semantics on
// gfx950 — matches torch.float8_e4m3fn).
template <int SEL>
DEV_INLINE float fp8_to_f32(unsigned int word) {
  return __builtin_amdgcn_cvt_f32_fp8(word, SEL);  // sel must be constant
}

DEV_INLINE unsigned char f32_to_fp8(float x) {
  return (unsigned char)(__builtin_amdgcn_cvt_pk_fp8_f32(x, x, 0, false) &
                         0xFF);
}

DEV_INLINE float bf2f(bf16_t x) { return __bfloat162float(x); }
DEV_INLINE bf16_t f2bf(float x) { return __float2bfloat16(x); }

// Unpack a uint holding 2 bf16 into 2 floats.
DEV_INLINE float2 unpack_bf16x2(unsigned int u) {
  float2 r;
  r.x = __uint_as_float((u & 0xffffu) << 16);
  r.y = __uint_as_float(u & 0xffff0000u);
  return r;
}

DEV_INLINE unsigned int pack_bf16x2(float lo, float hi) {
  bf16_t a = f2bf(lo), b = f2bf(hi);
  unsigned short ua = *reinterpret_cast<unsigned short*>(&a);
  unsigned short ub = *reinterpret_cast<unsigned short*>(&b);
  return (unsigned int)ua | ((unsigned int)ub << 16);
}

// --- wave-level reductions (64 lanes) -------------------------------------
DEV_INLINE float wave_sum(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

DEV_INLINE float wave_max(float v) {
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Reduce within 16-lane groups (used by 16-lanes-per-token layouts).
DEV_INLINE float group16_sum(float v) {
  #pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

// --- block-level reductions over NW waves via LDS --------------------------
// Caller provides __shared__ float scratch[NW]; result valid in all threads.
template <int NW>
DEV_INLINE float block_sum(float v, float* scratch) {
  int wave = threadIdx.x / WAVE;
  v = wave_sum(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wave] = v;
  __syncthreads();
  float total = 0.f;
  #pragma unroll
  for (int i = 0; i < NW; ++i) total += scratch[i];
  return total;
}

template <int NW>
DEV_INLINE float block_max(float v, float* scratch) {
  int wave = threadIdx.x / WAVE;
  v = wave_max(v);
  if ((threadIdx.x & (WAVE - 1)) == 0) scratch[wave] = v;
  __syncthreads();
  float m = -INFINITY;
  #pragma unroll
  for (int i = 0; i < NW; ++i) m = fmaxf(m, scratch[i]);
  return m;
}

#define HIP_CHECK_KERNEL()                                            \
  do {                                                                \
    hipError_t e_ = hipGetLastError();                                \
    if (e_ != hipSuccess)                                             \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e_)); \
  } while (0)
