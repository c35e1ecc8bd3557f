// CDNA4 reflection-pad kernels (fwd gather + bwd gather, no atomics).
//
// The decoder's ConvBlock is ReflectionPad2d(1) + 3x3 conv
// (ref network/monodepth2/layers.py:106-138); at the flagship config the
// eager PyTorch pad pair costs ~24% of the whole train step (profile
// profiles/r01_flagship_kernel_stats.md), with the backward dominated by
// its atomic scatter. Here the backward is a pure GATHER: for input pixel
// (yi, xi) the contributing output coords per axis are {yi + p} plus the
// mirrors {p - yi} (left/top edge band) and {2(H-1) - yi + p} (right/
// bottom band) — at most 3 per axis, 9 combinations, summed in fp32.
// Deterministic (bitwise-reproducible) and atomic-free.
//
// Layout: one kernel over a logically contiguous (N, H, W, C) array.
//   channels_last (B,C,H,W)@NHWC  -> N=B,   C=C (lane-adjacent channels)
//   contiguous    (B,C,H,W)@NCHW  -> N=B*C, C=1
// Both dtypes (float, bf16) with fp32 accumulation for the backward.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

namespace {

constexpr int kBlock = 256;

template <typename T>
__global__ void __launch_bounds__(kBlock)
reflect_pad_fwd_kernel(const T* __restrict__ in, T* __restrict__ out,
                       int N, int H, int W, int C, int pad) {
  const int Ho = H + 2 * pad, Wo = W + 2 * pad;
  const int64_t total = (int64_t)N * Ho * Wo * C;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * kBlock) {
    const int c = (int)(i % C);
    int64_t r = i / C;
    const int xo = (int)(r % Wo);
    r /= Wo;
    const int yo = (int)(r % Ho);
    const int n = (int)(r / Ho);
    int yi = yo - pad;
    if (yi < 0) yi = -yi;
    if (yi >= H) yi = 2 * (H - 1) - yi;
    int xi = xo - pad;
    if (xi < 0) xi = -xi;
    if (xi >= W) xi = 2 * (W - 1) - xi;
    out[i] = in[(((int64_t)n * H + yi) * W + xi) * C + c];
  }
}

template <typename T>
__global__ void __launch_bounds__(kBlock)
reflect_pad_bwd_kernel(const T* __restrict__ gout, T* __restrict__ gin,
                       int N, int H, int W, int C, int pad) {
  const int Ho = H + 2 * pad, Wo = W + 2 * pad;
  const int64_t total = (int64_t)N * H * W * C;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * kBlock) {
    const int c = (int)(i % C);
    int64_t r = i / C;
    const int xi = (int)(r % W);
    r /= W;
    const int yi = (int)(r % H);
    const int n = (int)(r / H);

    int ys[3], xs[3];
    int ny = 0, nx = 0;
    ys[ny++] = yi + pad;
    if (yi >= 1 && yi <= pad) ys[ny++] = pad - yi;
    if (yi >= H - 1 - pad && yi <= H - 2) ys[ny++] = 2 * (H - 1) - yi + pad;
    xs[nx++] = xi + pad;
    if (xi >= 1 && xi <= pad) xs[nx++] = pad - xi;
    if (xi >= W - 1 - pad && xi <= W - 2) xs[nx++] = 2 * (W - 1) - xi + pad;

    float acc = 0.0f;
    const T* gb = gout + (int64_t)n * Ho * Wo * C + c;
    for (int a = 0; a < ny; ++a)
      for (int b = 0; b < nx; ++b)
        acc += (float)gb[((int64_t)ys[a] * Wo + xs[b]) * C];
    gin[i] = (T)acc;
  }
}

inline int grid_for(int64_t total) {
  int64_t g = (total + kBlock - 1) / kBlock;
  return (int)(g < 65535 ? g : 65535);
}

}  // namespace

extern "C" {

void mine_reflect_pad_fwd_f32(const float* in, float* out, int N, int H,
                              int W, int C, int pad, hipStream_t stream) {
  hipLaunchKernelGGL(reflect_pad_fwd_kernel<float>,
                     dim3(grid_for((int64_t)N * (H + 2 * pad) * (W + 2 * pad) * C)),
                     dim3(kBlock), 0, stream, in, out, N, H, W, C, pad);
}

void mine_reflect_pad_fwd_bf16(const void* in, void* out, int N, int H,
                               int W, int C, int pad, hipStream_t stream) {
  hipLaunchKernelGGL(reflect_pad_fwd_kernel<__hip_bfloat16>,
                     dim3(grid_for((int64_t)N * (H + 2 * pad) * (W + 2 * pad) * C)),
                     dim3(kBlock), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(in),
                     reinterpret_cast<__hip_bfloat16*>(out), N, H, W, C, pad);
}

void mine_reflect_pad_bwd_f32(const float* gout, float* gin, int N, int H,
                              int W, int C, int pad, hipStream_t stream) {
  hipLaunchKernelGGL(reflect_pad_bwd_kernel<float>,
                     dim3(grid_for((int64_t)N * H * W * C)),
                     dim3(kBlock), 0, stream, gout, gin, N, H, W, C, pad);
}

void mine_reflect_pad_bwd_bf16(const void* gout, void* gin, int N, int H,
                               int W, int C, int pad, hipStream_t stream) {
  hipLaunchKernelGGL(reflect_pad_bwd_kernel<__hip_bfloat16>,
                     dim3(grid_for((int64_t)N * H * W * C)),
                     dim3(kBlock), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(gout),
                     reinterpret_cast<__hip_bfloat16*>(gin), N, H, W, C, pad);
}

}  // extern "C"
