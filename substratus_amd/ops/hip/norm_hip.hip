#include "hip/hip_runtime.h"
// RMSNorm / LayerNorm kernels (SURVEY.md §2.5 native-component ledger item 3).
//
// HBM-bound streaming ops: one workgroup per row, bf16×8 vector loads
// (guide §5 common-mistake #2: scalar bf16 2.35 TB/s → ×8 vectorized
// 4.89 TB/s). fp32 accumulation; weights are read through L2 every row
// (resident after the first row of the launch).
//
// rms_norm_residual fuses the residual add so the hidden state makes one HBM
// round trip per layer instead of two (guide: fuse elementwise work into the
// producing kernel).
#include "common.h"

// Each thread handles H/BLOCK elements; vector path requires H % 8 == 0.
// The row is cached in REGISTERS between the reduce and the scale pass when
// it fits (hidden <= BLOCK*8*RCAP — every production config: h <= 8192 at
// BLOCK=256 needs 4 uint4/thread), so each norm makes ONE read pass over
// the hidden state instead of two (the decode chain runs 2 norms/layer at
// 1 WG/CU where the second pass is pure latency).
template <int BLOCK, bool RESIDUAL>
__global__ void rms_norm_kernel(bf16_t* __restrict__ out,
                                bf16_t* __restrict__ residual,
                                const bf16_t* __restrict__ x,
                                const bf16_t* __restrict__ weight,
                                int hidden, float eps) {
  constexpr int RCAP = 8;  // max uint4 per thread held in registers
  __shared__ float scratch[BLOCK / WAVE];
  const long row = blockIdx.x;
  const bf16_t* xr = x + row * hidden;
  bf16_t* rr = RESIDUAL ? residual + row * hidden : nullptr;
  bf16_t* orow = out + row * hidden;

  float ss = 0.f;
  const int nvec = hidden / 8;
  const uint4* xv = reinterpret_cast<const uint4*>(xr);
  uint4* rv = RESIDUAL ? reinterpret_cast<uint4*>(rr) : nullptr;
  // statically-indexed register cache (a runtime-indexed array goes to
  // scratch — guide §5.4 rule 20); launcher guarantees nvec <= BLOCK*RCAP
  uint4 keep[RCAP];
  #pragma unroll
  for (int kk = 0; kk < RCAP; ++kk) {
    const int i = threadIdx.x + kk * BLOCK;
    if (i >= nvec) break;
    uint4 u = xv[i];
    float2 f0 = unpack_bf16x2(u.x), f1 = unpack_bf16x2(u.y);
    float2 f2 = unpack_bf16x2(u.z), f3 = unpack_bf16x2(u.w);
    if (RESIDUAL) {
      uint4 r = rv[i];
      float2 g0 = unpack_bf16x2(r.x), g1 = unpack_bf16x2(r.y);
      float2 g2 = unpack_bf16x2(r.z), g3 = unpack_bf16x2(r.w);
      f0.x += g0.x; f0.y += g0.y; f1.x += g1.x; f1.y += g1.y;
      f2.x += g2.x; f2.y += g2.y; f3.x += g3.x; f3.y += g3.y;
      uint4 w;
      w.x = pack_bf16x2(f0.x, f0.y); w.y = pack_bf16x2(f1.x, f1.y);
      w.z = pack_bf16x2(f2.x, f2.y); w.w = pack_bf16x2(f3.x, f3.y);
      rv[i] = w;  // residual stream updated in place
      u = w;
    }
    keep[kk] = u;
    ss += f0.x * f0.x + f0.y * f0.y + f1.x * f1.x + f1.y * f1.y +
          f2.x * f2.x + f2.y * f2.y + f3.x * f3.x + f3.y * f3.y;
  }
  float total = block_sum<BLOCK / WAVE>(ss, scratch);
  float inv = rsqrtf(total / (float)hidden + eps);

  const uint4* wv = reinterpret_cast<const uint4*>(weight);
  uint4* ov = reinterpret_cast<uint4*>(orow);
  #pragma unroll
  for (int kk = 0; kk < RCAP; ++kk) {
    const int i = threadIdx.x + kk * BLOCK;
    if (i >= nvec) break;
    uint4 u = keep[kk];
    uint4 w = wv[i];
    float2 f0 = unpack_bf16x2(u.x), f1 = unpack_bf16x2(u.y);
    float2 f2 = unpack_bf16x2(u.z), f3 = unpack_bf16x2(u.w);
    float2 w0 = unpack_bf16x2(w.x), w1 = unpack_bf16x2(w.y);
    float2 w2 = unpack_bf16x2(w.z), w3 = unpack_bf16x2(w.w);
    uint4 o;
    o.x = pack_bf16x2(f0.x * inv * w0.x, f0.y * inv * w0.y);
    o.y = pack_bf16x2(f1.x * inv * w1.x, f1.y * inv * w1.y);
    o.z = pack_bf16x2(f2.x * inv * w2.x, f2.y * inv * w2.y);
    o.w = pack_bf16x2(f3.x * inv * w3.x, f3.y * inv * w3.y);
    ov[i] = o;
  }
}

template <int BLOCK>
__global__ void layer_norm_kernel(bf16_t* __restrict__ out,
                                  const bf16_t* __restrict__ x,
                                  const bf16_t* __restrict__ weight,
                                  const bf16_t* __restrict__ bias,
                                  int hidden, float eps) {
  __shared__ float scratch[BLOCK / WAVE];
  const long row = blockIdx.x;
  const bf16_t* xr = x + row * hidden;
  bf16_t* orow = out + row * hidden;
  const int nvec = hidden / 8;
  const uint4* xv = reinterpret_cast<const uint4*>(xr);

  float s = 0.f, ss = 0.f;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    uint4 u = xv[i];
    float2 f0 = unpack_bf16x2(u.x), f1 = unpack_bf16x2(u.y);
    float2 f2 = unpack_bf16x2(u.z), f3 = unpack_bf16x2(u.w);
    s += f0.x + f0.y + f1.x + f1.y + f2.x + f2.y + f3.x + f3.y;
    ss += f0.x * f0.x + f0.y * f0.y + f1.x * f1.x + f1.y * f1.y +
          f2.x * f2.x + f2.y * f2.y + f3.x * f3.x + f3.y * f3.y;
  }
  float mean = block_sum<BLOCK / WAVE>(s, scratch) / (float)hidden;
  __syncthreads();  // scratch reuse
  float msq = block_sum<BLOCK / WAVE>(ss, scratch) / (float)hidden;
  float var = msq - mean * mean;
  float inv = rsqrtf(var + eps);

  const uint4* wv = reinterpret_cast<const uint4*>(weight);
  const uint4* bv = reinterpret_cast<const uint4*>(bias);
  uint4* ov = reinterpret_cast<uint4*>(orow);
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    uint4 u = xv[i];
    uint4 w = wv[i], b = bv[i];
    float2 f[4] = {unpack_bf16x2(u.x), unpack_bf16x2(u.y),
                   unpack_bf16x2(u.z), unpack_bf16x2(u.w)};
    float2 wf[4] = {unpack_bf16x2(w.x), unpack_bf16x2(w.y),
                    unpack_bf16x2(w.z), unpack_bf16x2(w.w)};
    float2 bfv[4] = {unpack_bf16x2(b.x), unpack_bf16x2(b.y),
                     unpack_bf16x2(b.z), unpack_bf16x2(b.w)};
    unsigned int o[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      o[j] = pack_bf16x2((f[j].x - mean) * inv * wf[j].x + bfv[j].x,
                         (f[j].y - mean) * inv * wf[j].y + bfv[j].y);
    }
    ov[i] = make_uint4(o[0], o[1], o[2], o[3]);
  }
}

void launch_rms_norm(bf16_t* out, const bf16_t* x, const bf16_t* w,
                     long rows, int hidden, float eps, hipStream_t stream) {
  constexpr int BLOCK = 256;
  // register cache bound: hidden <= BLOCK*8*RCAP = 16384 (all configs)
 hipLaunchKernelGGL(( rms_norm_kernel<BLOCK, false>), dim3(dim3(rows)), dim3(dim3(BLOCK)), 0, stream, 
      out, nullptr, x, w, hidden, eps);
}

void launch_rms_norm_residual(bf16_t* out, bf16_t* residual, const bf16_t* x,
                              const bf16_t* w, long rows, int hidden,
                              float eps, hipStream_t stream) {
  constexpr int BLOCK = 256;
 hipLaunchKernelGGL(( rms_norm_kernel<BLOCK, true>), dim3(dim3(rows)), dim3(dim3(BLOCK)), 0, stream, 
      out, residual, x, w, hidden, eps);
}

void launch_layer_norm(bf16_t* out, const bf16_t* x, const bf16_t* w,
                       const bf16_t* b, long rows, int hidden, float eps,
                       hipStream_t stream) {
  constexpr int BLOCK = 256;
 hipLaunchKernelGGL(( layer_norm_kernel<BLOCK>), dim3(dim3(rows)), dim3(dim3(BLOCK)), 0, stream, 
      out, x, w, b, hidden, eps);
}
