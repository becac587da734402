#include "hip/hip_runtime.h"
// Fused RoPE + paged-KV append (SURVEY.md §2.5 ledger item 4).
//
// One wave per (token, head-role): rotates q in place, rotates k and scatters
// it into the paged K cache, copies v into the paged V cache — the cache
// write is fused into the rotation so k/v never round-trip through HBM
// unrotated. Neox-style rotation (pair (d, d+D/2)); theta <= 0 disables
// rotation (opt-family models).
//
// Caches: [num_blocks, n_kv_heads, block_size, head_dim] bf16.
// slots[n] = block_id * block_size + offset (flat slot index).
#include "common.h"

// grid: (ntok, n_q_heads + 2*n_kv_heads); block: 64 (one wave).
// head_dim <= 256 assumed (lane handles head_dim/WAVE rotation pairs... here
// each lane handles pair (lane + j*WAVE) for j in [0, head_dim/2/WAVE)).
__global__ void rope_kv_append_kernel(
    bf16_t* __restrict__ q,              // [ntok, hq, d]
    const bf16_t* __restrict__ k,        // [ntok, hkv, d]
    const bf16_t* __restrict__ v,        // [ntok, hkv, d]
    const int* __restrict__ positions,   // [ntok]
    bf16_t* __restrict__ k_cache,        // [nb, hkv, bs, d]
    bf16_t* __restrict__ v_cache,
    const int* __restrict__ slots,       // [ntok]
    int hq, int hkv, int d, int block_size, float inv_theta_log,
    bool do_rope) {
  const int tok = blockIdx.x;
  const int role = blockIdx.y;
  const int lane = threadIdx.x;
  const int half = d / 2;

  const int slot = slots[tok];
  const long cache_row_base =
      ((long)(slot / block_size) * hkv * block_size +
       (long)(slot % block_size)) * d;  // + head * block_size * d

  if (role >= hq + hkv) {  // V copy path: no rotation, straight scatter
    const int h = role - hq - hkv;
    const bf16_t* src = v + ((long)tok * hkv + h) * d;
    bf16_t* dst = v_cache + cache_row_base + (long)h * block_size * d;
    // d bf16 = d/8 uint4 lanes; one wave covers 64*8=512 elems per pass
    const uint4* sv = reinterpret_cast<const uint4*>(src);
    uint4* dv = reinterpret_cast<uint4*>(dst);
    for (int i = lane; i < d / 8; i += WAVE) dv[i] = sv[i];
    return;
  }

  const bool is_q = role < hq;
  bf16_t* x;
  bf16_t* dst;
  if (is_q) {
    x = q + ((long)tok * hq + role) * d;
    dst = x;  // in place
  } else {
    const int h = role - hq;
    x = const_cast<bf16_t*>(k) + ((long)tok * hkv + h) * d;
    dst = k_cache + cache_row_base + (long)h * block_size * d;
  }

  if (!do_rope) {
    if (!is_q) {
      const uint4* sv = reinterpret_cast<const uint4*>(x);
      uint4* dv = reinterpret_cast<uint4*>(dst);
      for (int i = lane; i < d / 8; i += WAVE) dv[i] = sv[i];
    }
    return;
  }

  const float pos = (float)positions[tok];
  // pairs (i, i + half), i in [0, half); lane strides by WAVE
  for (int i = lane; i < half; i += WAVE) {
    float x1 = bf2f(x[i]);
    float x2 = bf2f(x[i + half]);
    // inv_freq = theta^(-i/half) = exp(-(i/half) * ln(theta))
    float inv_freq = __expf(-(float)i / (float)half * inv_theta_log);
    float ang = pos * inv_freq;
    float c, s;
    __sincosf(ang, &s, &c);
    float o1 = x1 * c - x2 * s;
    float o2 = x2 * c + x1 * s;
    dst[i] = f2bf(o1);
    dst[i + half] = f2bf(o2);
  }
}

void launch_rope_kv_append(bf16_t* q, const bf16_t* k, const bf16_t* v,
                           const int* positions, bf16_t* k_cache,
                           bf16_t* v_cache, const int* slots, int ntok,
                           int hq, int hkv, int d, int block_size,
                           float theta, hipStream_t stream) {
  dim3 grid(ntok, hq + 2 * hkv);
  float log_theta = theta > 0.f ? logf(theta) : 0.f;
 hipLaunchKernelGGL(( rope_kv_append_kernel), dim3(grid), dim3(dim3(WAVE)), 0, stream, 
      q, k, v, positions, k_cache, v_cache, slots, hq, hkv, d, block_size,
      log_theta, theta > 0.f);
}
