// General MFMA implicit-GEMM convolution family for the encoder / neck /
// base-conv shapes (ResNet-50 stack: 1x1 s1/s2, 3x3 s1/s2, 7x7 s2 stem;
// the decoder's reflect-padded batch-B base convs) — forward, data-grad
// and weight-grad, NHWC bf16, fp32 accumulate.
//
// Design (MI355X): every one of these tensors is small (batch 4,
// <=13 MB — L2/L3 resident) and the library path was LAUNCH/LATENCY
// bound (~7 ms/step over ~60 tiny calls, 13 TF/s aggregate —
// tools/conv_shapes.py). So the kernels optimize for low fixed cost,
// not streaming bandwidth: operands are loaded DIRECTLY from global
// (L2-served; no LDS staging pipeline), one bf16x8 load per MFMA A
// fragment slice, weights pre-packed in exact fragment order
// (mine_amd/ops/conv_general.py).
//
// GEMM view (v_mfma_f32_16x16x32_bf16, fragment maps proven in
// conv_kernels.hip / tools/mfma_probe.hip):
//   M = N*P*Q output pixels (lane&15 = A row), N-dim = K out-channels,
//   contraction k = tap*(C/8)*8 + c (8 consecutive channels of one tap
//   per A slice = one bf16x8 load).
// Forward and data-grad are ONE kernel: the source coordinate of
// output pixel p at tap r is (p*SA + r*SB + SD), divided by SE with a
// remainder check — fwd {SA=stride, SB=1, SD=-pad, SE=1}, data-grad
// (transposed conv) {SA=1, SB=-1, SD=pad, SE=stride} over flipped
// transposed weights. PadMode reflect covers the decoder base convs.
//
// Weight-grad: one workgroup per (pixel-slab, 16 out-ch, 16 in-ch, tap);
// each 32-pixel chunk is transposed through the 1 KiB LDS block images
// of wrw_kernels.hip (single b128 stores in, two ds_read_b64_tr_b16
// out), one MFMA per chunk, fp32 atomics at flush.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using s16x4 = __attribute__((ext_vector_type(4))) short;
using s16x8 = __attribute__((ext_vector_type(8))) short;
using lds_short = __attribute__((address_space(3))) short;
using lds_s16x4 = __attribute__((address_space(3))) s16x4;

constexpr int kBlock = 256;

enum PadMode { PM_ZERO = 0, PM_REFLECT = 1 };

__device__ __forceinline__ int reflect1g(int v, int n) {
  if (v < 0) v = -v;
  if (v >= n) v = 2 * (n - 1) - v;
  return v;
}

// source coordinate of output coord p at tap offset r.
// Returns -1 for "zero contribution".
template <int PAD>
__device__ __forceinline__ int src_coord(int p, int r, int SA, int SB,
                                         int SD, int SE, int n_src) {
  int num = p * SA + r * SB + SD;
  if (SE > 1) {
    // transposed-conv phase check (SE = fwd stride)
    if (num % SE != 0) return -1;
    num /= SE;
  }
  if (PAD == PM_REFLECT) return reflect1g(num, n_src);
  return (num >= 0 && num < n_src) ? num : -1;
}

// ---------------------------------------------------------------------------
// fwd / data-grad igemm: 64 output pixels x 64 out-channels per block
// ---------------------------------------------------------------------------
// RS = R*S (compile-time: 1, 9, 49) so tap decomposition has constant
// divisors; Cv = C/8 runtime.

// SPLIT: gridDim.z slices the contraction (deep-K, tiny-M shapes — the
// decoder's 2048-channel base convs run at M=384 pixels, 24 blocks
// without it); slices atomically accumulate into an fp32 workspace
// that igemm_epilogue_kernel then bias-adds and casts.
template <int Sdim, int RS, int PAD, bool SPLIT>
__global__ void __launch_bounds__(kBlock)
conv_igemm_fwd_kernel(const __hip_bfloat16* __restrict__ x,  // (N,Hs,Ws,C)
                      const __hip_bfloat16* __restrict__ wp, // packed frags
                      const float* __restrict__ bias,        // (K) or null
                      __hip_bfloat16* __restrict__ out,      // (M,K) flat
                      float* __restrict__ ws_out,            // (M,K) fp32
                      int64_t M, int P, int Q, int K,
                      int Hs, int Ws, int C,
                      int SA, int SB, int SD, int SE) {
  const int64_t m_lane = (int64_t)blockIdx.x * 64 +
                         (threadIdx.x >> 6) * 16 + (threadIdx.x & 15);
  const int lane = threadIdx.x & 63;
  const int k0 = blockIdx.y * 64;

  // decompose this lane's A-row pixel
  int n = 0, p = 0, q = 0;
  const bool m_ok = m_lane < M;
  if (m_ok) {
    const int64_t pq = (int64_t)P * Q;
    n = (int)(m_lane / pq);
    const int rem = (int)(m_lane - (int64_t)n * pq);
    p = rem / Q;
    q = rem - p * Q;
  }

  const int Cv = C / 8;
  const int nseg = RS * Cv;
  const int nchunks = (nseg + 3) / 4;
  const int nK = (K + 15) / 16;
  const int nk_here = (k0 + 64 <= K) ? 4 : (nK - blockIdx.y * 4);

  int kc_begin = 0, kc_end = nchunks;
  if (SPLIT) {
    kc_begin = (int)((int64_t)nchunks * blockIdx.z / gridDim.z);
    kc_end = (int)((int64_t)nchunks * (blockIdx.z + 1) / gridDim.z);
  }

  f32x4 acc[4];
#pragma unroll
  for (int a = 0; a < 4; ++a) acc[a] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int kc = kc_begin; kc < kc_end; ++kc) {
    const int seg = kc * 4 + (lane >> 4);
    bf16x8 afrag;
    bool loaded = false;
    if (m_ok && seg < nseg) {
      const int tap = seg / Cv;
      const int coct = seg - tap * Cv;
      const int r = tap / Sdim;
      const int s = tap - r * Sdim;
      const int ys = src_coord<PAD>(p, r, SA, SB, SD, SE, Hs);
      const int xs = src_coord<PAD>(q, s, SA, SB, SD, SE, Ws);
      if (ys >= 0 && xs >= 0) {
        afrag = *reinterpret_cast<const bf16x8*>(
            x + (((int64_t)n * Hs + ys) * Ws + xs) * C + coct * 8);
        loaded = true;
      }
    }
    if (!loaded) {
#pragma unroll
      for (int e = 0; e < 8; ++e) afrag[e] = (__bf16)0.0f;
    }
#pragma unroll
    for (int a = 0; a < 4; ++a) {
      if (a < nk_here) {
        const bf16x8 bfrag = *reinterpret_cast<const bf16x8*>(
            wp + (((int64_t)(blockIdx.y * 4 + a) * nchunks + kc) * 64 +
                  lane) * 8);
        acc[a] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         acc[a], 0, 0, 0);
      }
    }
  }

  // epilogue: C/D row = (lane>>4)*4 + rr -> pixel; col = lane&15 -> k
  const int64_t m_out = (int64_t)blockIdx.x * 64 + (threadIdx.x >> 6) * 16 +
                        (lane >> 4) * 4;
  const int j = lane & 15;
#pragma unroll
  for (int a = 0; a < 4; ++a) {
    if (a >= nk_here) continue;
    const int kout = k0 + a * 16 + j;
    if (kout >= K) continue;
    const float b = bias ? bias[kout] : 0.0f;
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int64_t mm = m_out + rr;
      if (mm < M) {
        if (SPLIT) {
          atomicAdd(&ws_out[mm * K + kout], acc[a][rr]);
        } else {
          out[mm * K + kout] = (__hip_bfloat16)(acc[a][rr] + b);
        }
      }
    }
  }
}

__global__ void __launch_bounds__(kBlock)
igemm_epilogue_kernel(const float* __restrict__ ws,
                      const float* __restrict__ bias,
                      __hip_bfloat16* __restrict__ out, int64_t total,
                      int K) {
  const int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x;
  if (i >= total) return;
  const float b = bias ? bias[(int)(i % K)] : 0.0f;
  out[i] = (__hip_bfloat16)(ws[i] + b);
}

// ---------------------------------------------------------------------------
// weight-grad: (slab, k16, c16 x tap) blocks, 32-pixel LDS-transposed
// chunks, one MFMA per chunk
// ---------------------------------------------------------------------------

// 32-element px-block permuted image (NB = 8 blocks of 4 px):
// fragment pair reads at elem 0 and +32*8.
__device__ __forceinline__ int img_elem32(int col, int j) {
  const int pblk = j >> 2;
  const int bi = (pblk & 1) * 4 + (pblk >> 1);
  return bi * 64 + (j & 3) * 16 + col;
}

__device__ __forceinline__ bf16x8 frag32(const __hip_bfloat16* base) {
  const int l = threadIdx.x & 63;
  const lds_short* pp = (const lds_short*)(base) +
                        (l & 15) * 4 + (l >> 4) * 64;
  s16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_s16x4*)(pp));
  s16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (lds_s16x4*)(pp + 32 * 8));
  s16x8 s = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
  return __builtin_bit_cast(bf16x8, s);
}

template <int Sdim, int PAD>
__global__ void __launch_bounds__(64)
conv_igemm_wrw_kernel(const __hip_bfloat16* __restrict__ x,   // (N,Hs,Ws,C)
                      const __hip_bfloat16* __restrict__ gy,  // (M,K) flat
                      float* __restrict__ dw,                 // (K,C,R,S)
                      int64_t M, int P, int Q, int K,
                      int Hs, int Ws, int C, int RS,
                      int SA, int SB, int SD, int SE, int n_slabs) {
  // one wave per block: gy image + x image, 1 KiB each
  __shared__ __attribute__((aligned(16))) __hip_bfloat16 s_gy[2][512];
  __shared__ __attribute__((aligned(16))) __hip_bfloat16 s_x[2][512];

  const int k0 = blockIdx.y * 16;
  const int tap = blockIdx.z / ((C + 15) / 16);
  const int c0 = (blockIdx.z % ((C + 15) / 16)) * 16;
  const int r = tap / Sdim;
  const int s = tap - r * Sdim;
  const int slab = blockIdx.x;
  const int64_t m_begin = M * slab / n_slabs;
  const int64_t m_end = M * (slab + 1) / n_slabs;
  const int lane = threadIdx.x;
  const int64_t pq = (int64_t)P * Q;

  f32x4 acc = f32x4{0.f, 0.f, 0.f, 0.f};

  int buf = 0;
  for (int64_t ch0 = m_begin; ch0 < m_end; ch0 += 32, buf ^= 1) {
    // ---- stage 32 pixels: 64 lanes, one (px, oct) pair each ----
    // gy: oct in {0,1} covers k0..k0+16; x: oct covers c0..c0+16
    {
      const int px = lane >> 1;          // 0..31
      const int oct = lane & 1;
      const int64_t m = ch0 + px;
      s16x8 vg{0, 0, 0, 0, 0, 0, 0, 0};
      s16x8 vx{0, 0, 0, 0, 0, 0, 0, 0};
      if (m < m_end) {
        const int kb = k0 + oct * 8;
        if (kb + 8 <= K) {
          vg = *reinterpret_cast<const s16x8*>(
              reinterpret_cast<const short*>(gy) + m * K + kb);
        } else if (kb < K) {
          const short* src = reinterpret_cast<const short*>(gy) + m * K;
#pragma unroll
          for (int e = 0; e < 8; ++e)
            vg[e] = (kb + e < K) ? src[kb + e] : (short)0;
        }
        // decompose m -> (n, p, q), map through the tap
        const int n = (int)(m / pq);
        const int rem = (int)(m - (int64_t)n * pq);
        const int p = rem / Q;
        const int q = rem - p * Q;
        const int ys = src_coord<PAD>(p, r, SA, SB, SD, SE, Hs);
        const int xs = src_coord<PAD>(q, s, SA, SB, SD, SE, Ws);
        if (ys >= 0 && xs >= 0) {
          const int cb = c0 + oct * 8;
          const short* xsrc = reinterpret_cast<const short*>(x) +
                              (((int64_t)n * Hs + ys) * Ws + xs) * C;
          if (cb + 8 <= C) {
            vx = *reinterpret_cast<const s16x8*>(xsrc + cb);
          } else if (cb < C) {
#pragma unroll
            for (int e = 0; e < 8; ++e)
              vx[e] = (cb + e < C) ? xsrc[cb + e] : (short)0;
          }
        }
      }
      const int off = img_elem32(oct * 8, px);
      *reinterpret_cast<s16x8*>(
          reinterpret_cast<short*>(s_gy[buf]) + off) = vg;
      *reinterpret_cast<s16x8*>(
          reinterpret_cast<short*>(s_x[buf]) + off) = vx;
    }
    // single wave: ds ops are in-order per wave; wait for the writes
    __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt(0) | vmcnt(0)
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
        frag32(s_gy[buf]), frag32(s_x[buf]), acc, 0, 0, 0);
  }

  // flush: row = k (lane>>4)*4+rr, col = c (lane&15)
  const int jcol = lane & 15;
  const int cc = c0 + jcol;
  if (cc < C) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int kk = k0 + (lane >> 4) * 4 + rr;
      if (kk < K) {
        atomicAdd(&dw[((int64_t)kk * C + cc) * RS + tap], acc[rr]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// fragment pack: one fused gather+cast (index < 0 -> zero). The LUT
// already encodes transposes / channel offsets / padding, so a pack is
// ONE kernel launch regardless of the weight's logical layout.
// ---------------------------------------------------------------------------

template <typename TIN>
__global__ void __launch_bounds__(kBlock)
pack_gather_kernel(const TIN* __restrict__ w, const int* __restrict__ lut,
                   __hip_bfloat16* __restrict__ out, int64_t n) {
  const int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x;
  if (i >= n) return;
  const int idx = lut[i];
  out[i] = (idx >= 0) ? (__hip_bfloat16)(float)w[idx] : (__hip_bfloat16)0.0f;
}

}  // namespace

extern "C" void mine_pack_gather(const void* w, const int* lut, void* out,
                                 int64_t n, int is_fp32,
                                 hipStream_t stream) {
  const dim3 grid((unsigned)((n + kBlock - 1) / kBlock));
  if (is_fp32)
    hipLaunchKernelGGL(pack_gather_kernel<float>, grid, dim3(kBlock), 0,
                       stream, reinterpret_cast<const float*>(w), lut,
                       reinterpret_cast<__hip_bfloat16*>(out), n);
  else
    hipLaunchKernelGGL(pack_gather_kernel<__hip_bfloat16>, grid,
                       dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(w), lut,
                       reinterpret_cast<__hip_bfloat16*>(out), n);
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

#define IGEMM_DISPATCH_RS(SD_, RS_, PADV, KERNEL_CALL)                    \
  if (PADV == PM_REFLECT) {                                               \
    constexpr int Sdim = SD_;  constexpr int RS = RS_;                    \
    constexpr int PAD = PM_REFLECT;                                       \
    KERNEL_CALL;                                                          \
  } else {                                                                \
    constexpr int Sdim = SD_;  constexpr int RS = RS_;                    \
    constexpr int PAD = PM_ZERO;                                          \
    KERNEL_CALL;                                                          \
  }

// splitz > 1 requires ws (pre-zeroed fp32 (M,K)); the epilogue launch
// bias-adds and casts into out.
extern "C" void mine_conv_igemm_fwd(
    const void* x, const void* wp, const float* bias, void* out, float* ws,
    int64_t M, int P, int Q, int K, int Hs, int Ws, int C,
    int R, int S, int SA, int SB, int SD, int SE, int pad_mode, int splitz,
    hipStream_t stream) {
  const dim3 grid((unsigned)((M + 63) / 64), (unsigned)((K + 63) / 64),
                  (unsigned)(splitz > 1 ? splitz : 1));
#define CALL_FWD(SPLIT)                                                  \
  hipLaunchKernelGGL((conv_igemm_fwd_kernel<Sdim, RS, PAD, SPLIT>),      \
                     grid, dim3(kBlock), 0, stream,                      \
                     reinterpret_cast<const __hip_bfloat16*>(x),         \
                     reinterpret_cast<const __hip_bfloat16*>(wp), bias,  \
                     reinterpret_cast<__hip_bfloat16*>(out), ws,         \
                     M, P, Q, K, Hs, Ws, C, SA, SB, SD, SE)
#define CALL_FWD_EITHER                                                  \
  do { if (splitz > 1) { CALL_FWD(true); } else { CALL_FWD(false); } }   \
  while (0)
  if (R == 1 && S == 1) {
    IGEMM_DISPATCH_RS(1, 1, pad_mode, CALL_FWD_EITHER)
  } else if (R == 3 && S == 3) {
    IGEMM_DISPATCH_RS(3, 9, pad_mode, CALL_FWD_EITHER)
  } else if (R == 7 && S == 7) {
    IGEMM_DISPATCH_RS(7, 49, pad_mode, CALL_FWD_EITHER)
  }
  if (splitz > 1) {
    const int64_t total = M * K;
    hipLaunchKernelGGL(igemm_epilogue_kernel,
                       dim3((unsigned)((total + kBlock - 1) / kBlock)),
                       dim3(kBlock), 0, stream, ws, bias,
                       reinterpret_cast<__hip_bfloat16*>(out), total, K);
  }
#undef CALL_FWD_EITHER
#undef CALL_FWD
}

extern "C" void mine_conv_igemm_wrw(
    const void* x, const void* gy, float* dw,
    int64_t M, int P, int Q, int K, int Hs, int Ws, int C,
    int R, int S, int SA, int SB, int SD, int SE, int pad_mode,
    hipStream_t stream) {
  const int cg = (C + 15) / 16;
  const int kc = (K + 15) / 16;
  // fill the chip: ~1024 blocks if the work allows
  int slabs = (int)(1024 / ((int64_t)kc * cg * R * S) + 1);
  const int64_t chunks = (M + 31) / 32;
  if (slabs > chunks) slabs = (int)chunks;
  if (slabs < 1) slabs = 1;
  const dim3 grid((unsigned)slabs, (unsigned)kc, (unsigned)(cg * R * S));
#define CALL_WRW                                                         \
  hipLaunchKernelGGL((conv_igemm_wrw_kernel<Sdim, PAD>), grid,           \
                     dim3(64), 0, stream,                                \
                     reinterpret_cast<const __hip_bfloat16*>(x),         \
                     reinterpret_cast<const __hip_bfloat16*>(gy), dw,    \
                     M, P, Q, K, Hs, Ws, C, R * S, SA, SB, SD, SE, slabs)
  if (R == 1 && S == 1) {
    IGEMM_DISPATCH_RS(1, 1, pad_mode, CALL_WRW)
  } else if (R == 3 && S == 3) {
    IGEMM_DISPATCH_RS(3, 9, pad_mode, CALL_WRW)
  } else if (R == 7 && S == 7) {
    IGEMM_DISPATCH_RS(7, 49, pad_mode, CALL_WRW)
  }
#undef CALL_WRW
}
