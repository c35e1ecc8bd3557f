// Empirical probe of ds_read_b64_tr_b16 semantics on gfx950:
// LDS[i] = i (as bf16 bit pattern = raw short i), each lane passes a
// chosen address, dump what each lane's 4 elements read.
#include <hip/hip_runtime.h>
#include <cstdio>

using s16x4 = __attribute__((ext_vector_type(4))) short;
using lds_s16x4 = __attribute__((address_space(3))) s16x4;
using lds_short = __attribute__((address_space(3))) short;

__global__ void probe(short* out, int scheme) {
  __shared__ short s[1024];
  const int t = threadIdx.x;
  for (int i = t; i < 1024; i += 64) s[i] = (short)i;
  __syncthreads();
  int addr = 0;
  const int l = t;
  if (scheme == 0) addr = 0;                          // uniform base
  else if (scheme == 1) addr = (l & 15) * 4 + (l >> 4) * 64;  // contiguous 8B slices
  else if (scheme == 2) addr = (l & 15) + (l >> 4) * 64;      // 1-elem lane stride
  else if (scheme == 3) addr = l * 4;                  // fully linear 8B/lane
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (lds_s16x4*)((lds_short*)s + addr));
  for (int j = 0; j < 4; ++j) out[(scheme * 64 + l) * 4 + j] = v[j];
}

int main() {
  short* d;
  hipMalloc(&d, 4 * 64 * 4 * sizeof(short));
  for (int sc = 0; sc < 4; ++sc) probe<<<1, 64>>>(d, sc);
  hipDeviceSynchronize();
  short h[4 * 64 * 4];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  for (int sc = 0; sc < 4; ++sc) {
    printf("scheme %d:\n", sc);
    for (int l = 0; l < 20; ++l) {
      printf("  lane %2d: %4d %4d %4d %4d\n", l,
             h[(sc * 64 + l) * 4], h[(sc * 64 + l) * 4 + 1],
             h[(sc * 64 + l) * 4 + 2], h[(sc * 64 + l) * 4 + 3]);
    }
    printf("  lane 16: %4d %4d %4d %4d  lane 32: %4d %4d %4d %4d\n",
           h[(sc*64+16)*4], h[(sc*64+16)*4+1], h[(sc*64+16)*4+2], h[(sc*64+16)*4+3],
           h[(sc*64+32)*4], h[(sc*64+32)*4+1], h[(sc*64+32)*4+2], h[(sc*64+32)*4+3]);
  }
  return 0;
}
