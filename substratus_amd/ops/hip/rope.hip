// Fused RoPE + paged-KV append (SURVEY.md §2.5 ledger item 4).
//
// One 4-wave workgroup per token; waves stride over the Hq + 2·Hkv head
// roles (rotate q in place; rotate k and scatter into the paged K cache;
// copy v into the paged V cache). The cache write is fused into the rotation
// so k/v never round-trip through HBM unrotated. Neox-style rotation (pair
// (d, d+D/2)); theta <= 0 disables rotation (opt-family models).
//
// Per-lane trig is computed once per (token, lane-dim) into registers and
// reused across ALL q/k heads of the token (inv_freq/cos/sin depend only on
// the dim and position, not the head) — the original one-wave-per-role
// version recomputed exp+sincos per head and measured 39.7 µs/launch
// (profiles/serve_decode_r01_baseline.txt).
//
// Caches: [num_blocks, n_kv_heads, block_size, head_dim] bf16.
// slots[n] = block_id * block_size + offset (flat slot index).
#include "common.h"

#define RW 4  // waves per workgroup

// grid: (ntok, role_chunks); block: RW*64. head_dim/2 <= 64 per lane pass
// (d=128 -> 1 pair per lane; d=64 -> lanes 0..31 active per pass).
// role chunking keeps >=2048 WGs in flight at any batch size.
__global__ __launch_bounds__(RW * WAVE)
void rope_kv_append_kernel(
    bf16_t* __restrict__ q,              // [ntok, hq, d]
    const bf16_t* __restrict__ k,        // [ntok, hkv, d]
    const bf16_t* __restrict__ v,        // [ntok, hkv, d]
    const int* __restrict__ positions,   // [ntok]
    bf16_t* __restrict__ k_cache,        // [nb, hkv, bs, d]
    bf16_t* __restrict__ v_cache,
    const int* __restrict__ slots,       // [ntok]
    int hq, int hkv, int d, int block_size, float ln_theta, bool do_rope) {
  const int tok = blockIdx.x;
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int half = d / 2;

  const int slot = slots[tok];
  const long cache_row = ((long)(slot / block_size) * hkv * block_size +
                          (long)(slot % block_size)) * d;

  // per-lane trig, shared across every head of this token
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
          v + ((long)tok * hkv + h) * d);
      uint4* dst = reinterpret_cast<uint4*>(
          v_cache + cache_row + (long)h * block_size * d);
      for (int i = lane; i < d / 8; i += WAVE) dst[i] = src[i];
      continue;
    }
    const bool is_q = role < hq;
    const bf16_t* src;
    bf16_t* dst;
    bf16_t* dst2 = nullptr;   // k also writes back in place (prefill reads
                              // the rotated k for its dense attention)
    if (is_q) {
      src = q + ((long)tok * hq + role) * d;
      dst = const_cast<bf16_t*>(src);
    } else {
      const int h = role - hq;
      src = k + ((long)tok * hkv + h) * d;
      dst = k_cache + cache_row + (long)h * block_size * d;
      dst2 = const_cast<bf16_t*>(src);
    }
    if (!do_rope) {
      if (!is_q) {
        const uint4* sv = reinterpret_cast<const uint4*>(src);
        uint4* dv = reinterpret_cast<uint4*>(dst);
        for (int i = lane; i < d / 8; i += WAVE) dv[i] = sv[i];
      }
      continue;
    }
    if (lane < half) {
      const float x1 = bf2f(src[lane]);
      const float x2 = bf2f(src[lane + half]);
      const bf16_t o1 = f2bf(x1 * c - x2 * s);
      const bf16_t o2 = f2bf(x2 * c + x1 * s);
      dst[lane] = o1;
      dst[lane + half] = o2;
      if (dst2 != nullptr) {
        dst2[lane] = o1;
        dst2[lane + half] = o2;
      }
    }
  }
}

void launch_rope_kv_append(bf16_t* q, const bf16_t* k, const bf16_t* v,
                           const int* positions, bf16_t* k_cache,
                           bf16_t* v_cache, const int* slots, int ntok,
                           int hq, int hkv, int d, int block_size,
                           float theta, hipStream_t stream) {
  float ln_theta = theta > 0.f ? logf(theta) : 0.f;
  const int nroles = hq + 2 * hkv;
  int chunks = 1;
  while (ntok * chunks < 2048 && chunks * RW < nroles) chunks *= 2;
  rope_kv_append_kernel<<<dim3(ntok, chunks), dim3(RW * WAVE), 0, stream>>>(
      q, k, v, positions, k_cache, v_cache, slots, hq, hkv, d, block_size,
      ln_theta, theta > 0.f);
}
