// Nearest x2 upsample, NHWC, bf16/f32 — fwd + bwd.
//
// The decoder's five up-stages (ref network/monodepth2/layers.py:198-201,
// depth_decoder.py:127-133) move ~1.4 GB/step at the flagship config;
// torch's channels_last upsample kernel measured ~15x off the HBM
// roofline there (profiles/r01_*). These kernels are plain widened
// streams: one thread moves an 8-channel vector (16 B bf16) per input
// pixel, writing its four children (fwd) / summing its four children in
// fp32 (bwd).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

namespace {

constexpr int kBlock = 256;

template <typename T, int VEC>
struct VecT;
template <> struct VecT<__hip_bfloat16, 8> {
  using type = __attribute__((ext_vector_type(8))) short;
};
template <> struct VecT<float, 4> {
  using type = __attribute__((ext_vector_type(4))) float;
};

// fwd: out[n, 2y+dy, 2x+dx, c] = in[n, y, x, c]
template <typename T, int VEC>
__global__ void __launch_bounds__(kBlock)
upsample2x_fwd_kernel(const T* __restrict__ in, T* __restrict__ out,
                      int64_t total_v, int W, int C) {
  using V = typename VecT<T, VEC>::type;
  const int Cv = C / VEC;
  const int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x;
  if (i >= total_v) return;
  const int cv = (int)(i % Cv);
  const int64_t rem = i / Cv;
  const int x = (int)(rem % W);
  const int64_t row = rem / W;        // n * H + y
  const V v = *reinterpret_cast<const V*>(in + i * VEC);
  // output row base: rows 2*row and 2*row+1 of the 2W-wide image
  T* o0 = out + ((row * 2) * (int64_t)(2 * W) + 2 * x) * C + cv * VEC;
  T* o1 = o0 + (int64_t)(2 * W) * C;
  *reinterpret_cast<V*>(o0) = v;
  *reinterpret_cast<V*>(o0 + C) = v;
  *reinterpret_cast<V*>(o1) = v;
  *reinterpret_cast<V*>(o1 + C) = v;
}

// bwd: gin[n, y, x, c] = sum over the four children (fp32 accumulate)
template <typename T, int VEC>
__global__ void __launch_bounds__(kBlock)
upsample2x_bwd_kernel(const T* __restrict__ gout, T* __restrict__ gin,
                      int64_t total_v, int W, int C) {
  using V = typename VecT<T, VEC>::type;
  const int Cv = C / VEC;
  const int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x;
  if (i >= total_v) return;
  const int cv = (int)(i % Cv);
  const int64_t rem = i / Cv;
  const int x = (int)(rem % W);
  const int64_t row = rem / W;
  const T* g0 = gout + ((row * 2) * (int64_t)(2 * W) + 2 * x) * C + cv * VEC;
  const T* g1 = g0 + (int64_t)(2 * W) * C;
  const V a = *reinterpret_cast<const V*>(g0);
  const V b = *reinterpret_cast<const V*>(g0 + C);
  const V c = *reinterpret_cast<const V*>(g1);
  const V d = *reinterpret_cast<const V*>(g1 + C);
  V o;
#pragma unroll
  for (int e = 0; e < VEC; ++e) {
    if constexpr (sizeof(T) == 2) {
      // raw-bit short <-> bf16 via the __bf16 builtin type
      const float s = (float)__builtin_bit_cast(__bf16, (short)a[e]) +
                      (float)__builtin_bit_cast(__bf16, (short)b[e]) +
                      (float)__builtin_bit_cast(__bf16, (short)c[e]) +
                      (float)__builtin_bit_cast(__bf16, (short)d[e]);
      o[e] = __builtin_bit_cast(short, (__bf16)s);
    } else {
      o[e] = a[e] + b[e] + c[e] + d[e];
    }
  }
  *reinterpret_cast<V*>(gin + i * VEC) = o;
}

}  // namespace

extern "C" {

void mine_upsample2x_fwd(const void* in, void* out, int64_t N, int64_t H,
                         int64_t W, int64_t C, int is_bf16,
                         hipStream_t stream) {
  if (is_bf16) {
    const int64_t total_v = N * H * W * (C / 8);
    const int64_t grid = (total_v + kBlock - 1) / kBlock;
    hipLaunchKernelGGL((upsample2x_fwd_kernel<__hip_bfloat16, 8>),
                       dim3((uint32_t)grid), dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(in),
                       reinterpret_cast<__hip_bfloat16*>(out),
                       total_v, (int)W, (int)C);
  } else {
    const int64_t total_v = N * H * W * (C / 4);
    const int64_t grid = (total_v + kBlock - 1) / kBlock;
    hipLaunchKernelGGL((upsample2x_fwd_kernel<float, 4>),
                       dim3((uint32_t)grid), dim3(kBlock), 0, stream,
                       reinterpret_cast<const float*>(in),
                       reinterpret_cast<float*>(out),
                       total_v, (int)W, (int)C);
  }
}

void mine_upsample2x_bwd(const void* gout, void* gin, int64_t N, int64_t H,
                         int64_t W, int64_t C, int is_bf16,
                         hipStream_t stream) {
  if (is_bf16) {
    const int64_t total_v = N * H * W * (C / 8);
    const int64_t grid = (total_v + kBlock - 1) / kBlock;
    hipLaunchKernelGGL((upsample2x_bwd_kernel<__hip_bfloat16, 8>),
                       dim3((uint32_t)grid), dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gout),
                       reinterpret_cast<__hip_bfloat16*>(gin),
                       total_v, (int)W, (int)C);
  } else {
    const int64_t total_v = N * H * W * (C / 4);
    const int64_t grid = (total_v + kBlock - 1) / kBlock;
    hipLaunchKernelGGL((upsample2x_bwd_kernel<float, 4>),
                       dim3((uint32_t)grid), dim3(kBlock), 0, stream,
                       reinterpret_cast<const float*>(gout),
                       reinterpret_cast<float*>(gin),
                       total_v, (int)W, (int)C);
  }
}

}  // extern "C"
