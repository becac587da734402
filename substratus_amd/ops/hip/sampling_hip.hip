#include "hip/hip_runtime.h"
// Fused sampling kernel (SURVEY.md §2.5 ledger item 5): temperature / top-k /
// top-p + inverse-CDF draw, one workgroup per logits row.
//
// The row (V bf16, e.g. 32000 for llama2 = 62.5 KiB) is staged once into LDS
// and every subsequent pass (max, top-k threshold search, top-p mass search,
// CDF scan) runs from LDS — HBM is touched exactly once per row. top-k uses a
// binary search on the value domain counting x >= t (monotone in t); top-p
// finds the smallest threshold whose masked exp-mass reaches p·Z. Both avoid
// any sort (sampling runs once per decode step; the passes are LDS-bound).
#include "common.h"

#define SBLOCK 256
#define SNW (SBLOCK / WAVE)

// LDS budget: V bf16 values (<= 65536 -> 128 KiB) + scratch.
extern __shared__ unsigned short s_vals[];  // [V] bf16 bits (scaled logits)

__global__ __launch_bounds__(SBLOCK)
void softmax_sample_kernel(long* __restrict__ out,          // [B]
                           const bf16_t* __restrict__ logits,  // [B, V]
                           const float* __restrict__ temperature,
                           const int* __restrict__ top_k,
                           const float* __restrict__ top_p,
                           const float* __restrict__ uniform, int vocab) {
  __shared__ float scratch[SNW];
  __shared__ float s_bcast;
  __shared__ int s_tok;
  const int b = blockIdx.x;
  const bf16_t* row = logits + (long)b * vocab;
  const float temp = temperature[b];
  const int tid = threadIdx.x;

  // ---- greedy: argmax without staging --------------------------------
  if (temp <= 0.f) {
    float best = -INFINITY;
    int besti = 0;
    for (int i = tid; i < vocab; i += SBLOCK) {
      float v = bf2f(row[i]);
      if (v > best || (v == best && i < besti)) { best = v; besti = i; }
    }
    // wave reduce (value, index) preferring smaller index on ties
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(best, off, WAVE);
      int oi = __shfl_xor(besti, off, WAVE);
      if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
    }
    __shared__ float wv[SNW];
    __shared__ int wi[SNW];
    if ((tid & (WAVE - 1)) == 0) { wv[tid / WAVE] = best; wi[tid / WAVE] = besti; }
    __syncthreads();
    if (tid == 0) {
      for (int w = 1; w < SNW; ++w)
        if (wv[w] > wv[0] || (wv[w] == wv[0] && wi[w] < wi[0])) {
          wv[0] = wv[w]; wi[0] = wi[w];
        }
      out[b] = wi[0];
    }
    return;
  }

  // ---- stage the ORIGINAL bf16 bits into LDS + row max ----------------
  // (temperature is applied in fp32 at every read: staging pre-scaled bf16
  // would re-quantize x/T and shift CDF boundaries vs the fp32 reference)
  const float inv_t = 1.f / temp;
  float mx = -INFINITY;
  for (int i = tid; i < vocab; i += SBLOCK) {
    bf16_t h = row[i];
    s_vals[i] = *reinterpret_cast<unsigned short*>(&h);
    mx = fmaxf(mx, bf2f(h) * inv_t);
  }
  __syncthreads();
  float m = block_max<SNW>(mx, scratch);
  __syncthreads();

  // ---- top-k: binary search threshold on value domain -----------------
  float thresh = -INFINITY;
  const int k = top_k[b];
  if (k > 0 && k < vocab) {
    float lo = m - 64.f, hi = m;  // exp(-64) underflows anyway
    for (int it = 0; it < 24; ++it) {
      float mid = 0.5f * (lo + hi);
      int cnt = 0;
      for (int i = tid; i < vocab; i += SBLOCK) {
        bf16_t h; *reinterpret_cast<unsigned short*>(&h) = s_vals[i];
        if (bf2f(h) * inv_t >= mid) ++cnt;
      }
      __syncthreads();
      int total = (int)block_sum<SNW>((float)cnt, scratch);
      __syncthreads();
      if (total > k) lo = mid; else hi = mid;
    }
    // lo converges to v_{k+1} (count>k), hi to just above it (count<=k):
    // x >= hi keeps exactly the top-k values, ties at v_k included.
    thresh = hi;
  }

  // ---- exp-sum over kept values ---------------------------------------
  float zs = 0.f;
  for (int i = tid; i < vocab; i += SBLOCK) {
    bf16_t h; *reinterpret_cast<unsigned short*>(&h) = s_vals[i];
    float v = bf2f(h) * inv_t;
    if (v >= thresh) zs += __expf(v - m);
  }
  __syncthreads();
  float z = block_sum<SNW>(zs, scratch);
  __syncthreads();

  // ---- top-p: binary search the prob-mass threshold -------------------
  const float p_target = top_p[b];
  if (p_target > 0.f && p_target < 1.f) {
    float lo = thresh == -INFINITY ? m - 64.f : thresh, hi = m;
    for (int it = 0; it < 24; ++it) {
      float mid = 0.5f * (lo + hi);
      float ms = 0.f;
      for (int i = tid; i < vocab; i += SBLOCK) {
        bf16_t h; *reinterpret_cast<unsigned short*>(&h) = s_vals[i];
        float v = bf2f(h) * inv_t;
        if (v >= mid && v >= thresh) ms += __expf(v - m);
      }
      __syncthreads();
      float mass = block_sum<SNW>(ms, scratch);
      __syncthreads();
      if (mass >= p_target * z) lo = mid; else hi = mid;
    }
    thresh = fmaxf(thresh, lo);
    // recompute z over the kept set
    float zs2 = 0.f;
    for (int i = tid; i < vocab; i += SBLOCK) {
      bf16_t h; *reinterpret_cast<unsigned short*>(&h) = s_vals[i];
      float v = bf2f(h) * inv_t;
      if (v >= thresh) zs2 += __expf(v - m);
    }
    __syncthreads();
    z = block_sum<SNW>(zs2, scratch);
    __syncthreads();
  }

  // ---- inverse-CDF draw ------------------------------------------------
  const float r = uniform[b] * z;
  // per-thread chunk sums in token order: thread t owns [t*chunk, (t+1)*chunk)
  const int chunk = (vocab + SBLOCK - 1) / SBLOCK;
  const int beg = tid * chunk, end = min(beg + chunk, vocab);
  float csum = 0.f;
  for (int i = beg; i < end; ++i) {
    bf16_t h; *reinterpret_cast<unsigned short*>(&h) = s_vals[i];
    float v = bf2f(h) * inv_t;
    if (v >= thresh) csum += __expf(v - m);
  }
  // exclusive prefix over the 256 per-thread sums (via LDS, simple scan)
  __shared__ float pre[SBLOCK + 1];
  pre[tid + 1] = csum;
  if (tid == 0) pre[0] = 0.f;
  __syncthreads();
  if (tid == 0) {
    for (int i = 1; i <= SBLOCK; ++i) pre[i] += pre[i - 1];
    s_tok = -1;
  }
  __syncthreads();
  // the thread whose [pre[tid], pre[tid+1]) straddles r walks its chunk
  if (r >= pre[tid] && r < pre[tid + 1]) {
    float acc = pre[tid];
    int sel = -1;
    for (int i = beg; i < end; ++i) {
      bf16_t h; *reinterpret_cast<unsigned short*>(&h) = s_vals[i];
      float v = bf2f(h) * inv_t;
      if (v >= thresh) {
        acc += __expf(v - m);
        if (acc > r) { sel = i; break; }
      }
    }
    if (sel >= 0) s_tok = sel;
  }
  __syncthreads();
  if (tid == 0) {
    int sel = s_tok;
    if (sel < 0) {
      // numerical edge (r ~ z): fall back to the last kept token
      for (int i = vocab - 1; i >= 0; --i) {
        bf16_t h; *reinterpret_cast<unsigned short*>(&h) = s_vals[i];
        if (bf2f(h) * inv_t >= thresh) { sel = i; break; }
      }
    }
    out[b] = sel < 0 ? 0 : sel;
  }
}

void launch_softmax_sample(long* out, const bf16_t* logits,
                           const float* temperature, const int* top_k,
                           const float* top_p, const float* uniform, int bsz,
                           int vocab, hipStream_t stream) {
  size_t lds = (size_t)vocab * sizeof(unsigned short);
  static size_t max_opted = 0;
  if (lds > 65536 && lds > max_opted) {  // opt in to >64 KiB dynamic LDS
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&softmax_sample_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    max_opted = lds;
  }
 hipLaunchKernelGGL(( softmax_sample_kernel), dim3(dim3(bsz)), dim3(dim3(SBLOCK)), lds, stream, 
      out, logits, temperature, top_k, top_p, uniform, vocab);
}
