#include "hip/hip_runtime.h"
// Fused RoPE + paged-KV append (SURVEY.md §2.5 ledger item 4).
//
// One 4-wave workgroup per token; waves stride over the Hq + 2·Hkv head
// roles (rotate q in place; rotate k — writing BOTH back in place, prefill's
// dense attention consumes it, AND into the paged K cache; copy v into the
// paged V cache). The cache write is fused into the rotation so k/v never
// round-trip through HBM unrotated. Neox-style rotation (pair (d, d+D/2));
// theta <= 0 disables rotation (opt-family models).
//
// Per-lane trig is computed once per token and reused across ALL heads
// (inv_freq/cos/sin depend only on dim and position). Role chunking keeps
// >= 2048 workgroups in flight at any batch size.
//
// KV8: the cache is fp8 e4m3 (gfx950 v_cvt_pk_fp8_f32 on the write path) —
// the opt-in half-traffic KV mode consumed by paged_attn_decode<KV8>.
//
// Caches: [num_blocks, n_kv_heads, block_size, head_dim] bf16 | fp8.
// slots[n] = block_id * block_size + offset (flat slot index).
#include "common.h"

#define RW 4  // waves per workgroup

template <bool KV8>
__global__ __launch_bounds__(RW * WAVE)
void rope_kv_append_kernel(
    bf16_t* __restrict__ q,              // [ntok, hq, d] rows qs apart
    bf16_t* __restrict__ k,              // [ntok, hkv, d] (rotated in place)
    const bf16_t* __restrict__ v,        // [ntok, hkv, d]
    const int* __restrict__ positions,   // [ntok]
    void* __restrict__ k_cache,          // [nb, hkv, bs, d]
    void* __restrict__ v_cache,
    const int* __restrict__ slots,       // [ntok]
    int hq, int hkv, int d, int block_size, float ln_theta, bool do_rope,
    long qs, long ks, long vs) {         // token-row strides in elements —
                                         // decode passes qkv-split VIEWS so
                                         // the fused projection never copies
  const int tok = blockIdx.x;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int half = d / 2;

  const int slot = slots[tok];
  const long cache_row = ((long)(slot / block_size) * hkv * block_size +
                          (long)(slot % block_size)) * d;  // in elements

  float c = 1.f, s = 0.f;
  if (do_rope && lane < half) {
    const float pos = (float)positions[tok];
    const float inv_freq = __expf(-(float)lane / (float)half * ln_theta);
    __sincosf(pos * inv_freq, &s, &c);
  }

  const int nroles = hq + 2 * hkv;
  for (int role = blockIdx.y * RW + wave; role < nroles;
       role += gridDim.y * RW) {
    if (role >= hq + hkv) {  // V copy
      const int h = role - hq - hkv;
      const uint4* src = reinterpret_cast<const uint4*>(
          v + (long)tok * vs + (long)h * d);
      const long off = cache_row + (long)h * block_size * d;
      if (!KV8) {
        uint4* dst = reinterpret_cast<uint4*>(
            reinterpret_cast<bf16_t*>(v_cache) + off);
        for (int i = lane; i < d / 8; i += WAVE) dst[i] = src[i];
      } else {
        unsigned char* dst = reinterpret_cast<unsigned char*>(v_cache) + off;
        for (int i = lane; i < d / 8; i += WAVE) {
          const uint4 u = src[i];
          float2 f0 = unpack_bf16x2(u.x), f1 = unpack_bf16x2(u.y);
          float2 f2 = unpack_bf16x2(u.z), f3 = unpack_bf16x2(u.w);
          unsigned int lo = 0, hi = 0;
          lo = __builtin_amdgcn_cvt_pk_fp8_f32(f0.x, f0.y, lo, false);
          lo = __builtin_amdgcn_cvt_pk_fp8_f32(f1.x, f1.y, lo, true);
          hi = __builtin_amdgcn_cvt_pk_fp8_f32(f2.x, f2.y, hi, false);
          hi = __builtin_amdgcn_cvt_pk_fp8_f32(f3.x, f3.y, hi, true);
          reinterpret_cast<uint2*>(dst)[i] = make_uint2(lo, hi);
        }
      }
      continue;
    }
    const bool is_q = role < hq;
    bf16_t* src;
    if (is_q) {
      src = q + (long)tok * qs + (long)role * d;
    } else {
      src = k + (long)tok * ks + (long)(role - hq) * d;
    }
    if (!do_rope) {
      if (!is_q) {
        const int h = role - hq;
        const long off = cache_row + (long)h * block_size * d;
        if (!KV8) {
          uint4* dst = reinterpret_cast<uint4*>(
              reinterpret_cast<bf16_t*>(k_cache) + off);
          const uint4* sv = reinterpret_cast<const uint4*>(src);
          for (int i = lane; i < d / 8; i += WAVE) dst[i] = sv[i];
        } else {
          unsigned char* dst = reinterpret_cast<unsigned char*>(k_cache) + off;
          for (int i = lane; i < d; i += WAVE)
            dst[i] = f32_to_fp8(bf2f(src[i]));
        }
      }
      continue;
    }
    if (lane < half) {
      const float x1 = bf2f(src[lane]);
      const float x2 = bf2f(src[lane + half]);
      const float r1 = x1 * c - x2 * s;
      const float r2 = x2 * c + x1 * s;
      src[lane] = f2bf(r1);           // in-place (q always; k for prefill)
      src[lane + half] = f2bf(r2);
      if (!is_q) {
        const int h = role - hq;
        const long off = cache_row + (long)h * block_size * d;
        if (!KV8) {
          bf16_t* dst = reinterpret_cast<bf16_t*>(k_cache) + off;
          dst[lane] = f2bf(r1);
          dst[lane + half] = f2bf(r2);
        } else {
          unsigned char* dst = reinterpret_cast<unsigned char*>(k_cache) + off;
          dst[lane] = f32_to_fp8(r1);
          dst[lane + half] = f32_to_fp8(r2);
        }
      }
    }
  }
}

void launch_rope_kv_append(bf16_t* q, bf16_t* k, const bf16_t* v,
                           const int* positions, void* k_cache,
                           void* v_cache, const int* slots, int ntok,
                           int hq, int hkv, int d, int block_size,
                           float theta, bool kv8, long qs, long ks, long vs,
                           hipStream_t stream) {
  float ln_theta = theta > 0.f ? logf(theta) : 0.f;
  const int nroles = hq + 2 * hkv;
  int chunks = 1;
  while (ntok * chunks < 2048 && chunks * RW < nroles) chunks *= 2;
  dim3 grid(ntok, chunks);
  if (kv8) {
   hipLaunchKernelGGL(( rope_kv_append_kernel<true>), dim3(grid), dim3(dim3(RW * WAVE)), 0, stream, 
        q, k, v, positions, k_cache, v_cache, slots, hq, hkv, d, block_size,
        ln_theta, theta > 0.f, qs, ks, vs);
  } else {
   hipLaunchKernelGGL(( rope_kv_append_kernel<false>), dim3(grid), dim3(dim3(RW * WAVE)), 0, stream, 
        q, k, v, positions, k_cache, v_cache, slots, hq, hkv, d, block_size,
        ln_theta, theta > 0.f, qs, ks, vs);
  }
}
