#include "hip/hip_runtime.h"
// Fused cross-entropy forward+backward (SURVEY.md §2.5 ledger item 7):
// log-softmax + NLL loss + dlogits in one pass structure, one workgroup per
// row. dlogits = (softmax - onehot(target)) / n_valid; loss accumulated into
// a single fp32 scalar with atomicAdd (mean over non-ignored targets).
//
// Streaming over V (llama 32000 / falcon 65024): two passes over the row
// (max+sum fused via online rescale, then dlogits write) with bf16×8 loads —
// the row is L2-resident on the second pass.
#include "common.h"

#define CBLOCK 256
#define CNW (CBLOCK / WAVE)

__global__ __launch_bounds__(CBLOCK)
void cross_entropy_kernel(float* __restrict__ loss,       // [1]
                          bf16_t* __restrict__ dlogits,   // [N, V]
                          const bf16_t* __restrict__ logits,
                          const long* __restrict__ targets,  // [N]
                          int vocab, long ignore_index, float inv_nvalid) {
  __shared__ float scratch[CNW];
  const long row = blockIdx.x;
  const long tgt = targets[row];
  const bf16_t* lrow = logits + row * vocab;
  bf16_t* drow = dlogits + row * vocab;
  const int tid = threadIdx.x;
  const bool valid = tgt != ignore_index;

  // pass 1: online max + exp-sum
  float m = -INFINITY, s = 0.f;
  const int nvec = vocab / 8;
  const uint4* lv = reinterpret_cast<const uint4*>(lrow);
  for (int i = tid; i < nvec; i += CBLOCK) {
    uint4 u = lv[i];
    float2 f[4] = {unpack_bf16x2(u.x), unpack_bf16x2(u.y),
                   unpack_bf16x2(u.z), unpack_bf16x2(u.w)};
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      float a = f[j].x, b2 = f[j].y;
      float mx = fmaxf(fmaxf(a, b2), m);
      if (mx > m) { s *= __expf(m - mx); m = mx; }
      s += __expf(a - m) + __expf(b2 - m);
    }
  }
  for (int i = nvec * 8 + tid; i < vocab; i += CBLOCK) {  // tail
    float a = bf2f(lrow[i]);
    float mx = fmaxf(a, m);
    if (mx > m) { s *= __expf(m - mx); m = mx; }
    s += __expf(a - m);
  }
  float M = block_max<CNW>(m, scratch);
  __syncthreads();
  s *= __expf(m - M);
  float Z = block_sum<CNW>(s, scratch);
  const float lse = M + __logf(Z);
  const float invz = 1.f / Z;

  // pass 2: dlogits = (exp(x-M)/Z - onehot) * inv_nvalid (0 if ignored row)
  const float rowscale = valid ? inv_nvalid : 0.f;
  uint4* dv = reinterpret_cast<uint4*>(drow);
  for (int i = tid; i < nvec; i += CBLOCK) {
    uint4 u = lv[i];
    float2 f[4] = {unpack_bf16x2(u.x), unpack_bf16x2(u.y),
                   unpack_bf16x2(u.z), unpack_bf16x2(u.w)};
    unsigned int o[4];
    #pragma unroll
    for (int j = 0; j < 4; ++j) {
      int idx0 = i * 8 + 2 * j, idx1 = idx0 + 1;
      float p0 = __expf(f[j].x - M) * invz - (idx0 == tgt ? 1.f : 0.f);
      float p1 = __expf(f[j].y - M) * invz - (idx1 == tgt ? 1.f : 0.f);
      o[j] = pack_bf16x2(p0 * rowscale, p1 * rowscale);
    }
    dv[i] = make_uint4(o[0], o[1], o[2], o[3]);
  }
  for (int i = nvec * 8 + tid; i < vocab; i += CBLOCK) {
    float p = __expf(bf2f(lrow[i]) - M) * invz - (i == tgt ? 1.f : 0.f);
    drow[i] = f2bf(p * rowscale);
  }

  if (tid == 0 && valid) {
    const float tlogit = bf2f(lrow[tgt]);
    atomicAdd(loss, (lse - tlogit) * inv_nvalid);
  }
}

void launch_cross_entropy(float* loss, bf16_t* dlogits, const bf16_t* logits,
                          const long* targets, long rows, int vocab,
                          long ignore_index, float inv_nvalid,
                          hipStream_t stream) {
 hipLaunchKernelGGL(( cross_entropy_kernel), dim3(dim3(rows)), dim3(dim3(CBLOCK)), 0, stream, 
      loss, dlogits, logits, targets, vocab, ignore_index, inv_nvalid);
}
