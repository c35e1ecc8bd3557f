// Fused SSIM (11x11 sigma=1.5 Gaussian window) forward + backward.
//
// The reference computes SSIM with five full-image depthwise convolutions
// plus elementwise maps (ref network/ssim.py:19-39) — ten tensor
// materializations per call, and autograd replays them backward. Here:
//   forward  = ONE kernel: stage the tile + halo in LDS, separable x-blur
//              of the five moment streams (img1, img2, img1^2, img2^2,
//              img1*img2) into LDS, y-blur in registers, SSIM map, block
//              reduction, one atomicAdd per block -> mean.
//   backward = TWO kernels: (1) recompute the blurred moments and emit the
//              three per-pixel adjoint fields
//                F1 = d map/d mu1, F2 = d map/d S11, F3 = d map/d S12
//              (S11 = blur(img1^2), S12 = blur(img1*img2));
//              (2) blur the three fields (the window is symmetric, so the
//              correlation in the chain rule is the same blur) and combine:
//                d img1 = g/N * (blur(F1) + 2*img1*blur(F2) + img2*blur(F3)).
//
// Zero padding (the reference's conv2d padding=5), per-(b,c) plane blocks.

#include <hip/hip_runtime.h>
#include <cstdint>

#define DEV __device__ __forceinline__

namespace {

constexpr int TW = 32;    // tile width
constexpr int TH = 8;     // tile height
constexpr int HALO = 5;   // 11x11 window
constexpr int SW = TW + 2 * HALO;  // 42
constexpr int SH = TH + 2 * HALO;  // 18
constexpr int NT = TW * TH;        // 256 threads
constexpr float C1 = 0.01f * 0.01f;
constexpr float C2 = 0.03f * 0.03f;

__constant__ float c_win[11];

struct Smem {
  float im1[SH][SW];
  float im2[SH][SW];
  // x-blurred moment streams (all staged rows x tile columns)
  float m1[SH][TW];
  float m2[SH][TW];
  float m11[SH][TW];
  float m22[SH][TW];
  float m12[SH][TW];
};

DEV void stage_pair(Smem& sm, const float* __restrict__ a,
                    const float* __restrict__ b, int H, int W,
                    int gx0, int gy0, int tid) {
  for (int i = tid; i < SH * SW; i += NT) {
    const int ry = i / SW, rx = i - ry * SW;
    const int gy = gy0 + ry - HALO, gx = gx0 + rx - HALO;
    const bool ok = (gy >= 0 && gy < H && gx >= 0 && gx < W);
    const int64_t off = (int64_t)gy * W + gx;
    sm.im1[ry][rx] = ok ? a[off] : 0.0f;
    sm.im2[ry][rx] = ok ? b[off] : 0.0f;
  }
}

DEV void xblur_moments(Smem& sm, int tid) {
  for (int i = tid; i < SH * TW; i += NT) {
    const int ry = i / TW, x = i - ry * TW;
    float s1 = 0, s2 = 0, s11 = 0, s22 = 0, s12 = 0;
#pragma unroll
    for (int k = 0; k < 11; ++k) {
      const float w = c_win[k];
      const float v1 = sm.im1[ry][x + k];
      const float v2 = sm.im2[ry][x + k];
      s1 += w * v1;
      s2 += w * v2;
      s11 += w * v1 * v1;
      s22 += w * v2 * v2;
      s12 += w * v1 * v2;
    }
    sm.m1[ry][x] = s1;
    sm.m2[ry][x] = s2;
    sm.m11[ry][x] = s11;
    sm.m22[ry][x] = s22;
    sm.m12[ry][x] = s12;
  }
}

struct Moments {
  float mu1, mu2, S11, S22, S12;
};

DEV Moments yblur(const Smem& sm, int ty, int tx) {
  Moments m{0, 0, 0, 0, 0};
#pragma unroll
  for (int k = 0; k < 11; ++k) {
    const float w = c_win[k];
    m.mu1 += w * sm.m1[ty + k][tx];
    m.mu2 += w * sm.m2[ty + k][tx];
    m.S11 += w * sm.m11[ty + k][tx];
    m.S22 += w * sm.m22[ty + k][tx];
    m.S12 += w * sm.m12[ty + k][tx];
  }
  return m;
}

__global__ void __launch_bounds__(NT)
ssim_fwd_kernel(const float* __restrict__ img1, const float* __restrict__ img2,
                float* __restrict__ out_sum, int BC, int H, int W) {
  __shared__ Smem sm;
  __shared__ float red[NT];

  const int tiles_x = (W + TW - 1) / TW;
  const int tiles_y = (H + TH - 1) / TH;
  const int tile = blockIdx.x;
  const int bc = blockIdx.y;
  const int tile_y = tile / tiles_x;
  const int tile_x = tile - tile_y * tiles_x;
  if (tile_y >= tiles_y) return;

  const int gx0 = tile_x * TW, gy0 = tile_y * TH;
  const int tid = threadIdx.x;
  const int64_t plane = (int64_t)bc * H * W;

  stage_pair(sm, img1 + plane, img2 + plane, H, W, gx0, gy0, tid);
  __syncthreads();
  xblur_moments(sm, tid);
  __syncthreads();

  const int ty = tid / TW, tx = tid - (tid / TW) * TW;
  const int gx = gx0 + tx, gy = gy0 + ty;
  float val = 0.0f;
  if (gx < W && gy < H) {
    const Moments m = yblur(sm, ty, tx);
    const float A1 = 2.0f * m.mu1 * m.mu2 + C1;
    const float A2 = 2.0f * (m.S12 - m.mu1 * m.mu2) + C2;
    const float B1 = m.mu1 * m.mu1 + m.mu2 * m.mu2 + C1;
    const float B2 = (m.S11 - m.mu1 * m.mu1) + (m.S22 - m.mu2 * m.mu2) + C2;
    val = (A1 * A2) / (B1 * B2);
  }
  red[tid] = val;
  __syncthreads();
  for (int step = NT / 2; step > 0; step >>= 1) {
    if (tid < step) red[tid] += red[tid + step];
    __syncthreads();
  }
  if (tid == 0) atomicAdd(out_sum, red[0]);
}

__global__ void __launch_bounds__(NT)
ssim_bwd_fields_kernel(const float* __restrict__ img1,
                       const float* __restrict__ img2,
                       float* __restrict__ F1, float* __restrict__ F2,
                       float* __restrict__ F3, int BC, int H, int W) {
  __shared__ Smem sm;
  const int tiles_x = (W + TW - 1) / TW;
  const int tiles_y = (H + TH - 1) / TH;
  const int tile = blockIdx.x;
  const int bc = blockIdx.y;
  const int tile_y = tile / tiles_x;
  const int tile_x = tile - tile_y * tiles_x;
  if (tile_y >= tiles_y) return;

  const int gx0 = tile_x * TW, gy0 = tile_y * TH;
  const int tid = threadIdx.x;
  const int64_t plane = (int64_t)bc * H * W;

  stage_pair(sm, img1 + plane, img2 + plane, H, W, gx0, gy0, tid);
  __syncthreads();
  xblur_moments(sm, tid);
  __syncthreads();

  const int ty = tid / TW, tx = tid - (tid / TW) * TW;
  const int gx = gx0 + tx, gy = gy0 + ty;
  if (gx < W && gy < H) {
    const Moments m = yblur(sm, ty, tx);
    const float A1 = 2.0f * m.mu1 * m.mu2 + C1;
    const float A2 = 2.0f * (m.S12 - m.mu1 * m.mu2) + C2;
    const float B1 = m.mu1 * m.mu1 + m.mu2 * m.mu2 + C1;
    const float B2 = (m.S11 - m.mu1 * m.mu1) + (m.S22 - m.mu2 * m.mu2) + C2;
    const float iB = 1.0f / (B1 * B2);
    // d map/d mu1 = 2mu2(A2-A1)/(B1B2) + 2mu1 A1A2 (B1-B2)/(B1B2)^2
    const float f1 = 2.0f * m.mu2 * (A2 - A1) * iB +
                     2.0f * m.mu1 * A1 * A2 * (B1 - B2) * iB * iB;
    const float f2 = -A1 * A2 * B1 * iB * iB;  // d map / d S11 = -A1A2/(B1 B2^2)
    const float f3 = 2.0f * A1 * iB;           // d map / d S12
    const int64_t off = plane + (int64_t)gy * W + gx;
    F1[off] = f1;
    F2[off] = f2;
    F3[off] = f3;
  }
}

// combine: grad1 = scale * (blur(F1) + 2*img1*blur(F2) + img2*blur(F3))
__global__ void __launch_bounds__(NT)
ssim_bwd_combine_kernel(const float* __restrict__ F1,
                        const float* __restrict__ F2,
                        const float* __restrict__ F3,
                        const float* __restrict__ img1,
                        const float* __restrict__ img2,
                        const float* __restrict__ gscale,  // device scalar
                        float* __restrict__ grad1,
                        int BC, int H, int W, float inv_numel) {
  __shared__ Smem sm;  // reuse: im1/im2 unused channels stage F3
  __shared__ float f3s[SH][SW];

  const int tiles_x = (W + TW - 1) / TW;
  const int tiles_y = (H + TH - 1) / TH;
  const int tile = blockIdx.x;
  const int bc = blockIdx.y;
  const int tile_y = tile / tiles_x;
  const int tile_x = tile - tile_y * tiles_x;
  if (tile_y >= tiles_y) return;

  const int gx0 = tile_x * TW, gy0 = tile_y * TH;
  const int tid = threadIdx.x;
  const int64_t plane = (int64_t)bc * H * W;

  stage_pair(sm, F1 + plane, F2 + plane, H, W, gx0, gy0, tid);
  for (int i = tid; i < SH * SW; i += NT) {
    const int ry = i / SW, rx = i - (i / SW) * SW;
    const int gy = gy0 + ry - HALO, gx = gx0 + rx - HALO;
    const bool ok = (gy >= 0 && gy < H && gx >= 0 && gx < W);
    f3s[ry][rx] = ok ? F3[plane + (int64_t)gy * W + gx] : 0.0f;
  }
  __syncthreads();

  // x-blur the three fields into the moment arrays (reuse m1/m2/m11)
  for (int i = tid; i < SH * TW; i += NT) {
    const int ry = i / TW, x = i - (i / TW) * TW;
    float s1 = 0, s2 = 0, s3 = 0;
#pragma unroll
    for (int k = 0; k < 11; ++k) {
      const float w = c_win[k];
      s1 += w * sm.im1[ry][x + k];
      s2 += w * sm.im2[ry][x + k];
      s3 += w * f3s[ry][x + k];
    }
    sm.m1[ry][x] = s1;
    sm.m2[ry][x] = s2;
    sm.m11[ry][x] = s3;
  }
  __syncthreads();

  const int ty = tid / TW, tx = tid - (tid / TW) * TW;
  const int gx = gx0 + tx, gy = gy0 + ty;
  if (gx < W && gy < H) {
    float b1 = 0, b2 = 0, b3 = 0;
#pragma unroll
    for (int k = 0; k < 11; ++k) {
      const float w = c_win[k];
      b1 += w * sm.m1[ty + k][tx];
      b2 += w * sm.m2[ty + k][tx];
      b3 += w * sm.m11[ty + k][tx];
    }
    const int64_t off = plane + (int64_t)gy * W + gx;
    const float scale = gscale[0] * inv_numel;
    grad1[off] = scale * (b1 + 2.0f * img1[off] * b2 + img2[off] * b3);
  }
}

}  // namespace

extern "C" {

void mine_ssim_set_window(const float* host_win11) {
  hipMemcpyToSymbol(HIP_SYMBOL(c_win), host_win11, 11 * sizeof(float));
}

void mine_ssim_fwd(const float* img1, const float* img2, float* out_sum,
                   int BC, int H, int W, hipStream_t stream) {
  const int tiles = ((W + TW - 1) / TW) * ((H + TH - 1) / TH);
  hipLaunchKernelGGL(ssim_fwd_kernel, dim3(tiles, BC), dim3(NT), 0, stream,
                     img1, img2, out_sum, BC, H, W);
}

void mine_ssim_bwd(const float* img1, const float* img2, const float* gscale,
                   float* F1, float* F2, float* F3, float* grad1,
                   int BC, int H, int W, hipStream_t stream) {
  const int tiles = ((W + TW - 1) / TW) * ((H + TH - 1) / TH);
  hipLaunchKernelGGL(ssim_bwd_fields_kernel, dim3(tiles, BC), dim3(NT), 0,
                     stream, img1, img2, F1, F2, F3, BC, H, W);
  const float inv_numel = 1.0f / ((float)BC * H * W);
  hipLaunchKernelGGL(ssim_bwd_combine_kernel, dim3(tiles, BC), dim3(NT), 0,
                     stream, F1, F2, F3, img1, img2, gscale, grad1,
                     BC, H, W, inv_numel);
}

}  // extern "C"
