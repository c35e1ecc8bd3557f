// MFMA fragment-layout probe for v_mfma_f32_16x16x32_bf16 (gfx950).
//
// Empirically determines the lane/element -> (row, k) mapping of the A
// and B operands: each of the 64x8 (lane, element) slots is set to a
// one-hot in turn against an all-ones other operand; the fired D row
// (C/D map is documented: row = (lane>>4)*4 + reg, col = lane&15) gives
// A's row map; pairing one-hot A against one-hot B gives the k
// equivalence classes. Output: two 64x8 int tables (row/k for A, col/k
// for B) written to global memory, printed by tools/mfma_probe.py.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdio>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ inline bf16x8 make_onehot(int lane, int elem, int my_lane) {
  bf16x8 v;
#pragma unroll
  for (int e = 0; e < 8; ++e)
    v[e] = (__bf16)((my_lane == lane && e == elem) ? 1.0f : 0.0f);
  return v;
}

__device__ inline bf16x8 make_ones() {
  bf16x8 v;
#pragma unroll
  for (int e = 0; e < 8; ++e) v[e] = (__bf16)1.0f;
  return v;
}

// out_a: 64x8 rows of A; out_b: 64x8 cols of B;
// out_ka: 64x8 k-class of A elems (index into lane0..3 x elem of B basis)
__global__ void probe_kernel(int* out_a_row, int* out_b_col, int* out_k) {
  const int lane = threadIdx.x;  // one wave
  const bf16x8 ones = make_ones();

  for (int l = 0; l < 64; ++l) {
    for (int e = 0; e < 8; ++e) {
      // ---- A one-hot vs B ones: D[i][*] = 1 at i = row(l,e)
      bf16x8 a = make_onehot(l, e, lane);
      f32x4 d = {0.f, 0.f, 0.f, 0.f};
      d = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, ones, d, 0, 0, 0);
      // lane holds D rows (lane>>4)*4+r, col lane&15
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (d[r] > 0.5f && (lane & 15) == 0) {
          out_a_row[l * 8 + e] = (lane >> 4) * 4 + r;
        }
      }
      // ---- A ones vs B one-hot: D[*][j] = 1 at j = col(l,e)
      bf16x8 b = make_onehot(l, e, lane);
      f32x4 d2 = {0.f, 0.f, 0.f, 0.f};
      d2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ones, b, d2, 0, 0, 0);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (d2[r] > 0.5f && ((lane >> 4) * 4 + r) == 0) {
          out_b_col[l * 8 + e] = lane & 15;
        }
      }
      // ---- k-class: A one-hot (l,e) vs B one-hot basis (lb in row-group
      // {0,16,32,48}, eb 0..7): fires iff k_A(l,e) == k_B(lb,eb).
      for (int g = 0; g < 4; ++g) {
        const int lb = g * 16;  // B lanes 0,16,32,48 share col j=0
        for (int eb = 0; eb < 8; ++eb) {
          bf16x8 b1 = make_onehot(lb, eb, lane);
          f32x4 d3 = {0.f, 0.f, 0.f, 0.f};
          d3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, d3, 0, 0, 0);
          float any = d3[0] + d3[1] + d3[2] + d3[3];
          // reduce across the wave: if ANY lane saw a hit, classes match
          any = __shfl(any, 0) + __shfl(any, 16) + __shfl(any, 32) +
                __shfl(any, 48);
          // cheap wave-or: use ballot on the local value
          const unsigned long long m =
              __ballot(d3[0] + d3[1] + d3[2] + d3[3] > 0.5f);
          if (m != 0ull && lane == 0) {
            out_k[l * 8 + e] = g * 8 + eb;  // k-class id of basis slot
          }
        }
      }
    }
  }
}

}  // namespace

int main() {
  int *a_row, *b_col, *k_class;
  hipMalloc(&a_row, 64 * 8 * sizeof(int));
  hipMalloc(&b_col, 64 * 8 * sizeof(int));
  hipMalloc(&k_class, 64 * 8 * sizeof(int));
  hipMemset(a_row, 0xff, 64 * 8 * sizeof(int));
  hipMemset(b_col, 0xff, 64 * 8 * sizeof(int));
  hipMemset(k_class, 0xff, 64 * 8 * sizeof(int));
  hipLaunchKernelGGL(probe_kernel, dim3(1), dim3(64), 0, 0, a_row, b_col,
                     k_class);
  hipDeviceSynchronize();
  int ha[512], hb[512], hk[512];
  hipMemcpy(ha, a_row, sizeof(ha), hipMemcpyDeviceToHost);
  hipMemcpy(hb, b_col, sizeof(hb), hipMemcpyDeviceToHost);
  hipMemcpy(hk, k_class, sizeof(hk), hipMemcpyDeviceToHost);
  printf("A row map (lane e0..e7):\n");
  for (int l = 0; l < 64; ++l) {
    printf("lane %2d:", l);
    for (int e = 0; e < 8; ++e) printf(" %2d", ha[l * 8 + e]);
    printf("   k-class:");
    for (int e = 0; e < 8; ++e) printf(" %2d", hk[l * 8 + e]);
    printf("\n");
  }
  printf("B col map (lane e0..e7):\n");
  for (int l = 0; l < 64; ++l) {
    printf("lane %2d:", l);
    for (int e = 0; e < 8; ++e) printf(" %2d", hb[l * 8 + e]);
    printf("\n");
  }
  return 0;
}
