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

// V consecutive channels per thread (one 16-byte load/store when
// V*sizeof(T) == 16 and C % V == 0)
template <typename T, int V>
struct alignas(sizeof(T) * V) VecT {
  T v[V];
};

template <typename T, int V>
__global__ void __launch_bounds__(kBlock)
reflect_pad_fwd_kernel(const T* __restrict__ in, T* __restrict__ out,
                       int N, int H, int W, int C, int pad) {
  using Vec = VecT<T, V>;
  const int Cv = C / V;
  const int Ho = H + 2 * pad, Wo = W + 2 * pad;
  const int64_t total = (int64_t)N * Ho * Wo * Cv;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * kBlock) {
    const int c = (int)(i % Cv);
    int64_t r = i / Cv;
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
    reinterpret_cast<Vec*>(out)[i] = reinterpret_cast<const Vec*>(
        in)[(((int64_t)n * H + yi) * W + xi) * Cv + c];
  }
}

template <typename T, int V>
__global__ void __launch_bounds__(kBlock)
reflect_pad_bwd_kernel(const T* __restrict__ gout, T* __restrict__ gin,
                       int N, int H, int W, int C, int pad) {
  using Vec = VecT<T, V>;
  const int Cv = C / V;
  const int Ho = H + 2 * pad, Wo = W + 2 * pad;
  const int64_t total = (int64_t)N * H * W * Cv;
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * kBlock) {
    const int c = (int)(i % Cv);
    int64_t r = i / Cv;
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

    float acc[V];
#pragma unroll
    for (int j = 0; j < V; ++j) acc[j] = 0.0f;
    const Vec* gb = reinterpret_cast<const Vec*>(gout) +
                    (int64_t)n * Ho * Wo * Cv + c;
    for (int a = 0; a < ny; ++a)
      for (int b = 0; b < nx; ++b) {
        const Vec g = gb[((int64_t)ys[a] * Wo + xs[b]) * Cv];
#pragma unroll
        for (int j = 0; j < V; ++j) acc[j] += (float)g.v[j];
      }
    Vec o;
#pragma unroll
    for (int j = 0; j < V; ++j) o.v[j] = (T)acc[j];
    reinterpret_cast<Vec*>(gin)[i] = o;
  }
}

inline int grid_for(int64_t total) {
  int64_t g = (total + kBlock - 1) / kBlock;
  return (int)(g < 65535 ? g : 65535);
}

// widest V with V*sizeof(T)<=16 dividing C
template <typename T>
inline int pick_vec(int C) {
  const int vmax = 16 / (int)sizeof(T);
  for (int v = vmax; v > 1; v >>= 1)
    if (C % v == 0) return v;
  return 1;
}

}  // namespace

#define PAD_LAUNCH(KERN, T, TOTAL, ...)                                       \
  do {                                                                        \
    const int v = pick_vec<T>(C);                                             \
    const int64_t tv = (TOTAL) / v;                                           \
    if (v == 8)                                                               \
      hipLaunchKernelGGL((KERN<T, 8>), dim3(grid_for(tv)), dim3(kBlock), 0,   \
                         stream, __VA_ARGS__);                                \
    else if (v == 4)                                                          \
      hipLaunchKernelGGL((KERN<T, 4>), dim3(grid_for(tv)), dim3(kBlock), 0,   \
                         stream, __VA_ARGS__);                                \
    else if (v == 2)                                                          \
      hipLaunchKernelGGL((KERN<T, 2>), dim3(grid_for(tv)), dim3(kBlock), 0,   \
                         stream, __VA_ARGS__);                                \
    else                                                                      \
      hipLaunchKernelGGL((KERN<T, 1>), dim3(grid_for(tv)), dim3(kBlock), 0,   \
                         stream, __VA_ARGS__);                                \
  } while (0)

extern "C" {

void mine_reflect_pad_fwd_f32(const float* in, float* out, int N, int H,
                              int W, int C, int pad, hipStream_t stream) {
  PAD_LAUNCH(reflect_pad_fwd_kernel, float,
             (int64_t)N * (H + 2 * pad) * (W + 2 * pad) * C,
             in, out, N, H, W, C, pad);
}

void mine_reflect_pad_fwd_bf16(const void* in, void* out, int N, int H,
                               int W, int C, int pad, hipStream_t stream) {
  PAD_LAUNCH(reflect_pad_fwd_kernel, __hip_bfloat16,
             (int64_t)N * (H + 2 * pad) * (W + 2 * pad) * C,
             reinterpret_cast<const __hip_bfloat16*>(in),
             reinterpret_cast<__hip_bfloat16*>(out), N, H, W, C, pad);
}

void mine_reflect_pad_bwd_f32(const float* gout, float* gin, int N, int H,
                              int W, int C, int pad, hipStream_t stream) {
  PAD_LAUNCH(reflect_pad_bwd_kernel, float, (int64_t)N * H * W * C,
             gout, gin, N, H, W, C, pad);
}

void mine_reflect_pad_bwd_bf16(const void* gout, void* gin, int N, int H,
                               int W, int C, int pad, hipStream_t stream) {
  PAD_LAUNCH(reflect_pad_bwd_kernel, __hip_bfloat16, (int64_t)N * H * W * C,
             reinterpret_cast<const __hip_bfloat16*>(gout),
             reinterpret_cast<__hip_bfloat16*>(gin), N, H, W, C, pad);
}

}  // extern "C"
