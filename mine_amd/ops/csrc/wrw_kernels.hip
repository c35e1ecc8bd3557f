// Split-K MFMA weight-gradient (wrw) for the fused reflect-pad 3x3 conv:
//   dW[k, c, dy, dx] = sum_{n,y,x} gy[n,y,x,k] * xpad[n, y+dy, x+dx, c]
// as a GEMM with M = K (out-channels), N = 9*C taps, contraction over
// the pixels, on v_mfma_f32_16x16x32_bf16.
//
// v2 design (the round-2 rewrite; v1 measured 44 ms/step at the
// flagship config — strided 2-byte global staging and per-element LDS
// fragment loads): both operands are staged with COALESCED bf16x8
// loads + single b128 LDS writes into [4px][16ch] transpose-read
// blocks, and every MFMA fragment is read with TWO
// ds_read_b64_tr_b16 hardware transpose reads (lane l, elem j reads
// lds[(l&15) + j*16 + (l>>4)*64] — the conflict-free pattern of the
// CDNA4 guide). The 3 dx taps are handled by staging THREE
// pixel-shifted copies of the gy tile (A operand), so every fragment
// read stays 4-pixel-block aligned; the 3 dy taps are three staged x
// rows (B operand), no duplication.
//
// Per j-chunk each wave issues 6 A transpose-reads + 12 B
// transpose-reads + 18 MFMAs (Cg = 32) — MFMA-bound, vs v1's 144
// scalar LDS reads per 18 MFMAs.
//
// Decomposition: grid = (pixel slabs, K/16 chunks, C/32 groups); a
// block walks its slab's rows in 254-pixel tiles, accumulates the
// (16k x 9*Cg) partial in registers across the whole slab, and
// flushes ONCE with fp32 atomics.
//
// Supported: C % 16 == 0, any K, any W >= 2, stride 1, pad 1 reflect.

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
constexpr int P_TILE = 254;          // output pixels per x-tile
constexpr int JP = 256;              // contraction span (P_TILE + 2)
constexpr int NB = JP / 4;           // 4-pixel blocks per image row span
constexpr int IMG_ELEMS = 16 * JP;   // one [16ch][JP px] image (4096)
constexpr int N_JCHUNK = JP / 32;    // 8 contraction chunks of 32

__device__ __forceinline__ int reflect1(int v, int n) {
  if (v < 0) v = -v;
  if (v >= n) v = 2 * (n - 1) - v;
  return v;
}

// element offset of (col, j) inside one transpose-read image.
// Blocks of [4 px][16 ch]; pixel-block p stored at index
// (p&1)*NB/2 + p/2 so that a fragment's two tr reads (elems 0..3 and
// 4..7 of lane group g = px 8g..8g+7) land at base and base + JP*8.
__device__ __forceinline__ int img_elem(int col, int j) {
  const int p = j >> 2;
  const int bi = (p & 1) * (NB / 2) + (p >> 1);
  return bi * 64 + (j & 3) * 16 + col;
}

__device__ __forceinline__ bf16x8 frag2(const __hip_bfloat16* lds_base,
                                        int elem_off) {
  // ds_read_b64_tr_b16 has NO internal lane offset: each lane passes
  // its own 8-B-aligned address covering its 4-element slice of the
  // group's [4][16] block; the hardware redistributes column (l&15)
  // of the block to lane l (CDNA4 guide §2 / T10).
  const int l = threadIdx.x & 63;
  const lds_short* p = (const lds_short*)(lds_base) + elem_off +
                       (l & 15) * 4 + (l >> 4) * 64;
  s16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_s16x4*)(p));
  s16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (lds_s16x4*)(p + JP * 8));
  s16x8 s = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
  return __builtin_bit_cast(bf16x8, s);
}

template <int CHALVES>   // compile-time so acc[] indexing stays in regs
__global__ void __launch_bounds__(kBlock)
conv3x3_wrw_kernel(const __hip_bfloat16* __restrict__ x,   // (N,H,W,C)
                   const __hip_bfloat16* __restrict__ gy,  // (N,H,W,K)
                   float* __restrict__ dw,                 // (K,C,3,3)
                   int N, int H, int W, int C, int K, int n_slabs) {
  // LDS: 3 A images (gy, dx-shifted) + 3*CHALVES B images (x rows)
  extern __shared__ __attribute__((aligned(16))) __hip_bfloat16 lds[];

  const int k0 = blockIdx.y * 16;
  const int c0 = blockIdx.z * CHALVES * 16;
  constexpr int chalves = CHALVES;
  const int slab = (int)blockIdx.x;
  const int64_t rows_total = (int64_t)N * H;
  const int64_t r_begin = rows_total * slab / n_slabs;
  const int64_t r_end = rows_total * (slab + 1) / n_slabs;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  __hip_bfloat16* s_A = lds;                       // [3][IMG_ELEMS]
  __hip_bfloat16* s_B = lds + 3 * IMG_ELEMS;       // [3][chalves][IMG_ELEMS]

  constexpr int n_acc = 9 * CHALVES;               // (dx, dy, ch)
  f32x4 acc[n_acc];
#pragma unroll
  for (int i = 0; i < n_acc; ++i) acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};

  constexpr int kocts = 2;                          // 16 k = 2 bf16x8
  constexpr int cocts = 2 * CHALVES;                // Cg = cocts * 8

  for (int64_t r = r_begin; r < r_end; ++r) {
    const int n = (int)(r / H);
    const int y = (int)(r % H);
    const int64_t gy_row = (((int64_t)n * H + y) * W) * K;

    for (int x0 = 0; x0 < W; x0 += P_TILE) {
      const int Weff = (W - x0) < P_TILE ? (W - x0) : P_TILE;

      // Only j < Weff + 2 is ever contracted (jchunks bound below), so
      // staging beyond jstage is skipped — at W = 96 this saves the
      // 2.6x full-JP staging waste. Rounded to the 32-wide chunk so
      // every contracted element is initialized.
      const int jstage = ((Weff + 2 + 31) / 32) * 32;

      // ---- zero the A images (covers dx edges + tile truncation) ----
      __syncthreads();
      {
        // blocks of 4 px: zero img blocks for j < jstage. The permuted
        // block order (even blocks first) keeps the zeroed region two
        // contiguous spans per image.
        const int nb = jstage / 4;
        const int span = (nb / 2 + (nb & 1)) * 64 / 8;  // even-half s16x8s
        const int span2 = (nb / 2) * 64 / 8;            // odd half
        s16x8 z{0, 0, 0, 0, 0, 0, 0, 0};
        for (int d = 0; d < 3; ++d) {
          s16x8* za = reinterpret_cast<s16x8*>(s_A + d * IMG_ELEMS);
          for (int i = tid; i < span; i += kBlock) za[i] = z;
          s16x8* zb = reinterpret_cast<s16x8*>(s_A + d * IMG_ELEMS) +
                      (NB / 2) * 64 / 8;
          for (int i = tid; i < span2; i += kBlock) zb[i] = z;
        }
      }
      __syncthreads();

      // ---- stage A: gy row pixels, three dx-shifted copies ----------
      for (int i = tid; i < P_TILE * kocts; i += kBlock) {
        const int oct = i & 1;
        const int pl = i >> 1;                      // pixel in tile
        const int px = x0 + pl;
        if (px >= W) continue;
        const int kbase = k0 + oct * 8;
        if (kbase >= K) continue;
        s16x8 v;
        if (kbase + 8 <= K) {
          v = *reinterpret_cast<const s16x8*>(
              reinterpret_cast<const short*>(gy + gy_row) + px * K + kbase);
        } else {
          const short* src =
              reinterpret_cast<const short*>(gy + gy_row) + px * K;
#pragma unroll
          for (int e = 0; e < 8; ++e)
            v[e] = (kbase + e < K) ? src[kbase + e] : (short)0;
        }
#pragma unroll
        for (int dx = 0; dx < 3; ++dx) {
          const int j = pl + dx;
          *reinterpret_cast<s16x8*>(
              reinterpret_cast<short*>(s_A + dx * IMG_ELEMS) +
              img_elem(oct * 8, j)) = v;
        }
      }

      // ---- stage B: three reflected x rows, clamped halo ------------
      for (int i = tid; i < 3 * JP * cocts; i += kBlock) {
        const int coct = i % cocts;           // cocts is constexpr
        const int rem = i / cocts;
        const int j = rem % JP;
        const int dy = rem / JP;
        if (j >= jstage) continue;
        int u = x0 - 1 + j;
        if (u > W) u = W;                           // clamp before reflect
        const int us = reflect1(u, W);
        const int yy = reflect1(y + dy - 1, H);
        const s16x8 v = *reinterpret_cast<const s16x8*>(
            reinterpret_cast<const short*>(x) +
            ((((int64_t)n * H + yy) * W + us) * C + c0 + coct * 8));
        *reinterpret_cast<s16x8*>(
            reinterpret_cast<short*>(
                s_B + (dy * chalves + (coct >> 1)) * IMG_ELEMS) +
            img_elem((coct & 1) * 8, j)) = v;
      }
      __syncthreads();

      // ---- contraction: j-chunks split across the 4 waves -----------
      // A is nonzero only for j < Weff + 2
      const int jchunks = (Weff + 2 + 31) / 32;
      for (int jc = wave; jc < jchunks; jc += 4) {
        const int ebase = jc * 32 * 8;              // j0 * 8 elems
        bf16x8 afrag[3];
#pragma unroll
        for (int dx = 0; dx < 3; ++dx)
          afrag[dx] = frag2(s_A + dx * IMG_ELEMS, ebase);
#pragma unroll
        for (int dy = 0; dy < 3; ++dy) {
#pragma unroll
          for (int ch = 0; ch < CHALVES; ++ch) {
            const bf16x8 bfrag =
                frag2(s_B + (dy * CHALVES + ch) * IMG_ELEMS, ebase);
#pragma unroll
            for (int dx = 0; dx < 3; ++dx) {
              const int ai = (dx * 3 + dy) * CHALVES + ch;
              acc[ai] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  afrag[dx], bfrag, acc[ai], 0, 0, 0);
            }
          }
        }
      }
    }
  }

  // ---- flush: reduce the 4 wave-partials via LDS, atomics to global --
  float* red = reinterpret_cast<float*>(lds);
#pragma unroll
  for (int ai = 0; ai < n_acc; ++ai) {
    const int dx = ai / (3 * CHALVES);
    const int dy = (ai / CHALVES) % 3;
    const int ch = ai % CHALVES;
    __syncthreads();
    if (wave == 0)
      for (int i = tid; i < 256; i += 64) red[i] = 0.0f;
    __syncthreads();
    {
      const int jcol = lane & 15;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        const int krow = (lane >> 4) * 4 + rr;
        atomicAdd(&red[krow * 16 + jcol], acc[ai][rr]);
      }
    }
    __syncthreads();
    if (wave == 0) {
      for (int i = lane; i < 256; i += 64) {
        const int krow = i >> 4, jcol = i & 15;
        const int kk = k0 + krow;
        const int cc = c0 + ch * 16 + jcol;
        if (kk < K && cc < C) {
          atomicAdd(&dw[(((int64_t)kk * C + cc) * 3 + dy) * 3 + dx],
                    red[i]);
        }
      }
    }
  }
}

}  // namespace

extern "C" void mine_conv3x3_wrw(const void* x, const void* gy, float* dw,
                                 int N, int H, int W, int C, int K,
                                 hipStream_t stream) {
  const int ch = (C % 32 == 0) ? 2 : 1;   // 32- or 16-channel groups
  const int kc = (K + 15) / 16;
  const int zc = C / (16 * ch);
  int slabs = 512 / (kc * zc);
  if (slabs < 1) slabs = 1;
  if ((int64_t)slabs > (int64_t)N * H) slabs = (int)((int64_t)N * H);
  const size_t lds_bytes =
      (size_t)(3 + 3 * ch) * IMG_ELEMS * sizeof(__hip_bfloat16);
  const dim3 grid(slabs, kc, zc);
  if (ch == 2)
    hipLaunchKernelGGL(conv3x3_wrw_kernel<2>, grid, dim3(kBlock), lds_bytes,
                       stream,
                       reinterpret_cast<const __hip_bfloat16*>(x),
                       reinterpret_cast<const __hip_bfloat16*>(gy), dw,
                       N, H, W, C, K, slabs);
  else
    hipLaunchKernelGGL(conv3x3_wrw_kernel<1>, grid, dim3(kBlock), lds_bytes,
                       stream,
                       reinterpret_cast<const __hip_bfloat16*>(x),
                       reinterpret_cast<const __hip_bfloat16*>(gy), dw,
                       N, H, W, C, K, slabs);
}
