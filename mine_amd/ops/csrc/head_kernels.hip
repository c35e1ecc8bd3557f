// Fused MPI head: dispconv output -> packed fp32 MPI, one pass.
//
// The decoder head splits its 4-channel conv output into sigmoid RGB and
// |x|+1e-4 sigma (or sigmoid alpha; ref depth_decoder.py:134-146), and
// the engine packs it (B,S,H,W,4) fp32 for the fused renderer. Eagerly
// that is view + 2 slices + sigmoid + abs + cat + permute + contiguous +
// cast — many full passes over ~200 MB per scale. In channels_last the
// conv output (B*S,4,H,W) is ALREADY laid out (B*S,H,W,4), so the pack
// is a view and this kernel is ONE elementwise pass (and its backward
// one more).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

namespace {

constexpr int kBlock = 256;

__device__ __forceinline__ float sigmoidf(float z) {
  return 1.0f / (1.0f + __expf(-z));
}

template <typename T, bool ALPHA>
__global__ void __launch_bounds__(kBlock)
mpi_head_fwd_kernel(const T* __restrict__ z, float* __restrict__ out,
                    int64_t N) {
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * kBlock) {
    const T* zi = z + i * 4;
    float4 o;
    o.x = sigmoidf((float)zi[0]);
    o.y = sigmoidf((float)zi[1]);
    o.z = sigmoidf((float)zi[2]);
    const float s = (float)zi[3];
    o.w = ALPHA ? sigmoidf(s) : fabsf(s) + 1e-4f;
    *reinterpret_cast<float4*>(out + i * 4) = o;
  }
}

template <typename T, bool ALPHA>
__global__ void __launch_bounds__(kBlock)
mpi_head_bwd_kernel(const T* __restrict__ z, const float* __restrict__ gout,
                    T* __restrict__ gz, int64_t N) {
  for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < N;
       i += (int64_t)gridDim.x * kBlock) {
    const T* zi = z + i * 4;
    const float4 g = *reinterpret_cast<const float4*>(gout + i * 4);
    T* go = gz + i * 4;
    {
      const float s = sigmoidf((float)zi[0]);
      go[0] = (T)(g.x * s * (1.0f - s));
    }
    {
      const float s = sigmoidf((float)zi[1]);
      go[1] = (T)(g.y * s * (1.0f - s));
    }
    {
      const float s = sigmoidf((float)zi[2]);
      go[2] = (T)(g.z * s * (1.0f - s));
    }
    const float v = (float)zi[3];
    if (ALPHA) {
      const float s = sigmoidf(v);
      go[3] = (T)(g.w * s * (1.0f - s));
    } else {
      // d|v|/dv: 0 at v == 0 (torch convention)
      go[3] = (T)(g.w * (v > 0.0f ? 1.0f : (v < 0.0f ? -1.0f : 0.0f)));
    }
  }
}

inline int grid_for(int64_t n) {
  int64_t g = (n + kBlock - 1) / kBlock;
  return (int)(g < 65535 ? g : 65535);
}

}  // namespace

#define EXPORT_HEAD(SUF, T)                                                    \
  extern "C" void mine_mpi_head_fwd_##SUF(const void* z, float* out,           \
                                          int64_t N, int alpha,                \
                                          hipStream_t s) {                     \
    if (alpha)                                                                 \
      hipLaunchKernelGGL((mpi_head_fwd_kernel<T, true>), dim3(grid_for(N)),    \
                         dim3(kBlock), 0, s,                                   \
                         reinterpret_cast<const T*>(z), out, N);               \
    else                                                                       \
      hipLaunchKernelGGL((mpi_head_fwd_kernel<T, false>), dim3(grid_for(N)),   \
                         dim3(kBlock), 0, s,                                   \
                         reinterpret_cast<const T*>(z), out, N);               \
  }                                                                            \
  extern "C" void mine_mpi_head_bwd_##SUF(const void* z, const float* gout,    \
                                          void* gz, int64_t N, int alpha,      \
                                          hipStream_t s) {                     \
    if (alpha)                                                                 \
      hipLaunchKernelGGL((mpi_head_bwd_kernel<T, true>), dim3(grid_for(N)),    \
                         dim3(kBlock), 0, s, reinterpret_cast<const T*>(z),    \
                         gout, reinterpret_cast<T*>(gz), N);                   \
    else                                                                       \
      hipLaunchKernelGGL((mpi_head_bwd_kernel<T, false>), dim3(grid_for(N)),   \
                         dim3(kBlock), 0, s, reinterpret_cast<const T*>(z),    \
                         gout, reinterpret_cast<T*>(gz), N);                   \
  }

EXPORT_HEAD(f32, float)
EXPORT_HEAD(bf16, __hip_bfloat16)
