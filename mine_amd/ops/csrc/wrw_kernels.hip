// Split-K MFMA weight-gradient (wrw) for the fused reflect-pad 3x3 conv
// (wired into _Conv3x3ReflFn.backward in round 2; numerics tests
// tests/test_gpu_ops.py::test_wrw_matches_torch):
//   dW[k, c, dy, dx] = sum_{n,y,x} gy[n,y,x,k] * xpad[n, y+dy, x+dx, c]
// as a GEMM with M = K (out-channels), N = 9*C taps, contraction over
// the 25M pixels, on v_mfma_f32_16x16x32_bf16 (fragment maps per
// tools/mfma_probe.hip: A row=lane&15 k=(lane>>4)*8+e consecutive).
//
// Decomposition: grid.x = pixel slabs (few hundred), grid.y = K/16
// k-chunks. Each workgroup walks its slab's rows, stages per row:
//   LDS gy^T  [16][W]      (A operand: 8 consecutive PIXELS per lane)
//   LDS x^T   [C][3][W+2]  (B operand: 8 consecutive x at (c, dy row))
// keeps the (16 x 9C) partial in registers across the whole slab, and
// flushes ONCE with fp32 atomics (512 slabs x K*9C cells — ~500 adds
// per cell, negligible contention).
//
// Supported: C % 8 == 0 (grid.z splits the tap axis in 32-channel groups,
// so C > 32 costs one extra gy stage per group), any K (grid.y chunks of
// 16), W <= 1022, stride 1, pad 1 reflect.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int kBlock = 256;
constexpr int MAX_W = 1022;
constexpr int MAX_C = 32;
constexpr int MAX_NCH = 18;  // 9*C/16 <= 18 for C <= 32

__device__ __forceinline__ int reflect1(int v, int n) {
  if (v < 0) v = -v;
  if (v >= n) v = 2 * (n - 1) - v;
  return v;
}

__global__ void __launch_bounds__(kBlock)
conv3x3_wrw_kernel(const __hip_bfloat16* __restrict__ x,   // (N,H,W,C)
                   const __hip_bfloat16* __restrict__ gy,  // (N,H,W,K)
                   float* __restrict__ dw,                 // (K, 9*C) packed
                   int N, int H, int W, int C, int K, int n_slabs) {
  extern __shared__ __hip_bfloat16 lds[];
  // LDS layout: gyT [16][Wpad] then xT [Cg][3][W+2 pad8] for this
  // block's 32-channel group (grid.z picks the group; C <= 32 -> one)
  const int Wg = (W + 7) & ~7;          // gy row padded to 8
  const int Wx = (W + 2 + 7) & ~7;      // x row (+halo) padded to 8
  __hip_bfloat16* s_gy = lds;                       // 16 * Wg
  __hip_bfloat16* s_x = lds + 16 * Wg;              // Cg * 3 * Wx

  const int kc = blockIdx.y;            // k-chunk (16 out-channels)
  const int k0 = kc * 16;
  const int c0 = blockIdx.z * MAX_C;    // channel-group base
  const int Cg = (C - c0) < MAX_C ? (C - c0) : MAX_C;  // channels here
  const int slab = blockIdx.x;
  const int64_t rows_total = (int64_t)N * H;
  const int64_t r_begin = rows_total * slab / n_slabs;
  const int64_t r_end = rows_total * (slab + 1) / n_slabs;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int Cv = Cg / 8;
  const int nch = (9 * Cv + 1) / 2;     // 16-wide N chunks over 9*Cg

  // accumulators: one (16k x 16taps) tile per n-chunk, 4 f32/lane
  f32x4 acc[MAX_NCH];
#pragma unroll
  for (int i = 0; i < MAX_NCH; ++i) acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int px_chunks = (W + 31) / 32;  // 32-pixel contraction chunks

  for (int64_t r = r_begin; r < r_end; ++r) {
    const int n = (int)(r / H);
    const int y = (int)(r % H);

    // ---- stage gy^T: s_gy[k][xx] = gy[n,y,xx,k0+k] ----
    for (int i = threadIdx.x; i < 16 * W; i += kBlock) {
      const int k = i / W, xx = i - (i / W) * W;
      const int kk = k0 + k;
      s_gy[k * Wg + xx] = (kk < K)
          ? gy[(((int64_t)n * H + y) * W + xx) * K + kk]
          : (__hip_bfloat16)0.0f;
    }
    // ---- stage x^T rows y-1..y+1 reflected: s_x[c][row][xx] ----
    for (int i = threadIdx.x; i < Cg * 3 * (W + 2); i += kBlock) {
      const int xx = i % (W + 2);
      const int rem = i / (W + 2);
      const int row = rem % 3;
      const int c = rem / 3;
      const int yy = reflect1(y + row - 1, H);
      const int xs = reflect1(xx - 1, W);
      s_x[(c * 3 + row) * Wx + xx] =
          x[(((int64_t)n * H + yy) * W + xs) * C + c0 + c];
    }
    __syncthreads();

    // ---- contraction over this row's pixels, chunks of 32 ----
    // waves split the chunks round-robin
    for (int pc = wave; pc < px_chunks; pc += 4) {
      const int p0 = pc * 32;
      const int p_lane = p0 + (lane >> 4) * 8;  // this lane's 8 pixels
      // A: gy^T row (lane&15) at 8 consecutive pixels; OOB pixels -> 0
      bf16x8 afrag;
      if (p_lane + 8 <= W) {
        afrag = *reinterpret_cast<const bf16x8*>(
            s_gy + (lane & 15) * Wg + p_lane);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int p = p_lane + e;
          afrag[e] = (p < W) ? s_gy[(lane & 15) * Wg + p]
                             : (__hip_bfloat16)0.0f;
        }
      }
      for (int nc = 0; nc < nch; ++nc) {
        // B column (lane&15) of n-chunk nc -> tap index t = nc*16+(lane&15)
        // over the (cb, tap, ci) k-ordering of the fwd kernel:
        //   col j = (seg_lo..) ... here columns are (c, dy, dx) triples in
        //   the order col = (c*9 + dy*3 + dx)  [c-major taps]
        const int col = nc * 16 + (lane & 15);
        bf16x8 bfrag;
        if (col < 9 * Cg) {
          const int c = col / 9;
          const int tap = col - c * 9;
          const int dy = tap / 3, dx = tap - dy * 3;
          // xpad[y+dy-1, p+dx-1+1] = s_x[c][dy][p + dx]
          const int base = (c * 3 + dy) * Wx + p_lane + dx;
          if (p_lane + 8 <= W) {
            // unaligned by dx: element loads (8x b16)
#pragma unroll
            for (int e = 0; e < 8; ++e) bfrag[e] = s_x[base + e];
          } else {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int p = p_lane + e;
              bfrag[e] = (p < W) ? s_x[base + e] : (__hip_bfloat16)0.0f;
            }
          }
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e) bfrag[e] = (__hip_bfloat16)0.0f;
        }
        acc[nc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                          acc[nc], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- flush: reduce the 4 wave-partials via LDS, one atomic per cell --
  // (reuse the staging LDS as a fp32 scratch of 16 x 16 per n-chunk)
  float* red = reinterpret_cast<float*>(lds);
  for (int nc = 0; nc < nch; ++nc) {
    __syncthreads();
    if (wave == 0) {
      for (int i = threadIdx.x; i < 256; i += 64) red[i] = 0.0f;
    }
    __syncthreads();
    {
      const int jcol = lane & 15;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int krow = (lane >> 4) * 4 + rr;
        atomicAdd(&red[krow * 16 + jcol], acc[nc][rr]);
      }
    }
    __syncthreads();
    // wave 0 writes the workgroup partial to global
    if (wave == 0) {
      for (int i = lane; i < 256; i += 64) {
        const int krow = i / 16, jcol = i - (i / 16) * 16;
        const int kk = k0 + krow;
        const int col = nc * 16 + jcol;       // (c_local, tap) packed
        if (kk < K && col < 9 * Cg) {
          const int cl = col / 9, tap = col - cl * 9;
          atomicAdd(&dw[((int64_t)kk * C + c0 + cl) * 9 + tap], red[i]);
        }
      }
    }
  }
}

}  // namespace

extern "C" void mine_conv3x3_wrw(const void* x, const void* gy, float* dw,
                                 int N, int H, int W, int C, int K,
                                 hipStream_t stream) {
  const int n_slabs = 512 < (int64_t)N * H ? 512 : (int)((int64_t)N * H);
  const int Cg = C < MAX_C ? C : MAX_C;
  const int Wg = (W + 7) & ~7;
  const int Wx = (W + 2 + 7) & ~7;
  const size_t lds = (16 * Wg + (size_t)Cg * 3 * Wx) * sizeof(__hip_bfloat16);
  const dim3 grid(n_slabs, (K + 15) / 16, (C + MAX_C - 1) / MAX_C);
  hipLaunchKernelGGL(conv3x3_wrw_kernel, grid, dim3(kBlock), lds, stream,
                     reinterpret_cast<const __hip_bfloat16*>(x),
                     reinterpret_cast<const __hip_bfloat16*>(gy), dw,
                     N, H, W, C, K, n_slabs);
}
