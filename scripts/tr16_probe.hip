// Empirical probe of ds_read_b64_tr_b16's lane/element mapping on gfx950.
// LDS image [16][16] shorts, value = row*16+col. For several per-lane
// address formulas, dump what each lane receives.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((ext_vector_type(4))) short short4v;

__global__ void probe(short* out) {  // out[formula][lane][j]
  __shared__ short img[256];
  const int lane = threadIdx.x;
  for (int i = lane; i < 256; i += 64) img[i] = (short)i;
  __syncthreads();
  const int lhi = lane >> 4, llo = lane & 15;
  // formula 0: base + (l&15) + (l>>4)*64
  {
    short4v r = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (__attribute__((address_space(3))) short4v*)&img[llo + lhi * 64]);
    for (int j = 0; j < 4; ++j) out[(0 * 64 + lane) * 4 + j] = r[j];
  }
  // formula 1: uniform base
  {
    short4v r = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (__attribute__((address_space(3))) short4v*)&img[0]);
    for (int j = 0; j < 4; ++j) out[(1 * 64 + lane) * 4 + j] = r[j];
  }
  // formula 2: base + lane*4 (contiguous 4-elem chunks per lane)
  {
    short4v r = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (__attribute__((address_space(3))) short4v*)&img[lane * 4]);
    for (int j = 0; j < 4; ++j) out[(2 * 64 + lane) * 4 + j] = r[j];
  }
}

int main() {
  short* d;
  hipMalloc(&d, 3 * 64 * 4 * sizeof(short));
  probe<<<1, 64>>>(d);
  short h[3 * 64 * 4];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  const char* names[3] = {"llo+lhi*64", "uniform0", "lane*4"};
  for (int f = 0; f < 3; ++f) {
    printf("== formula %s ==\n", names[f]);
    for (int l = 0; l < 64; l += 1) {
      printf("lane %2d: %3d %3d %3d %3d\n", l,
             h[(f * 64 + l) * 4], h[(f * 64 + l) * 4 + 1],
             h[(f * 64 + l) * 4 + 2], h[(f * 64 + l) * 4 + 3]);
      if (l % 16 == 3 && l > 3) l += 11;  // sample a few lanes per group
    }
  }
  return 0;
}
