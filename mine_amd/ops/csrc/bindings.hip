// Torch bindings for the mine_amd HIP kernels (gfx950 only).
//
// Tensor-level contracts are documented in mine_amd/ops/renderer.py and
// mine_amd/ops/ssim.py. Everything here is thin glue: shape checks,
// output allocation, stream plumbing.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <vector>

extern "C" {
void mine_src_composite_fwd(const float*, const float*, const float*,
                            const float*, float*, float*, float*, int, int,
                            int, int, int, int, hipStream_t);
void mine_src_composite_bwd(const float*, const float*, const float*,
                            const float*, const float*, const float*,
                            const float*, float*, int, int, int, int, int,
                            int, hipStream_t);
void mine_tgt_composite_fwd(const float*, const float*, const float*,
                            const float*, const float*, float*, float*,
                            float*, int, int, int, int, int, int,
                            hipStream_t);
void mine_tgt_composite_bwd(const float*, const float*, const float*,
                            const float*, const float*, const float*,
                            const float*, const float*, float*, float*,
                            int, int, int, int, int, int, int, hipStream_t);
void mine_conv_igemm_fwd(const void*, const void*, const float*, void*,
                         float*, int64_t, int, int, int, int, int, int, int,
                         int, int, int, int, int, int, int, hipStream_t);
void mine_conv_igemm_wrw(const void*, const void*, float*, int64_t, int, int,
                         int, int, int, int, int, int, int, int, int, int,
                         int, hipStream_t);
void mine_eav2_fwd(const float*, const float*, const float*, float*, int,
                   int, int, int64_t, int64_t, int64_t, int64_t,
                   hipStream_t);
void mine_eav2_bwd(const float*, const float*, const float*, float*, float*,
                   const float*, float*, int, int, int, int64_t, int64_t,
                   int64_t, int64_t, hipStream_t);
void mine_pack_gather(const void*, const int*, void*, int64_t, int,
                      hipStream_t);
void mine_upsample2x_fwd(const void*, void*, int64_t, int64_t, int64_t,
                         int64_t, int, hipStream_t);
void mine_upsample2x_bwd(const void*, void*, int64_t, int64_t, int64_t,
                         int64_t, int, hipStream_t);
void mine_reflect_pad_fwd_f32(const float*, float*, int, int, int, int, int,
                              hipStream_t);
void mine_reflect_pad_fwd_bf16(const void*, void*, int, int, int, int, int,
                               hipStream_t);
void mine_reflect_pad_bwd_f32(const float*, float*, int, int, int, int, int,
                              hipStream_t);
void mine_reflect_pad_bwd_bf16(const void*, void*, int, int, int, int, int,
                               hipStream_t);
void mine_conv3x3_fwd(const void*, const void*, const float*, void*, int,
                      int, int, int, int, int, int, int, int, hipStream_t);
void mine_conv3x3_wrw(const void*, const void*, float*, int, int, int, int,
                      int, hipStream_t);
void mine_mpi_head_fwd_f32(const void*, float*, int64_t, int, hipStream_t);
void mine_mpi_head_fwd_bf16(const void*, float*, int64_t, int, hipStream_t);
void mine_mpi_head_bwd_f32(const void*, const float*, void*, int64_t, int,
                           hipStream_t);
void mine_mpi_head_bwd_bf16(const void*, const float*, void*, int64_t, int,
                            hipStream_t);
void mine_bn_stats_f32(const void*, float*, int64_t, int, hipStream_t);
void mine_bn_stats_bf16(const void*, float*, int64_t, int, hipStream_t);
void mine_bn_finalize(const float*, float*, float*, float*, float*, int64_t,
                      int, float, float, hipStream_t);
void mine_bn_act_fwd_f32(const void*, const void*, const float*, const float*,
                         const float*, const float*, void*, int64_t, int, int,
                         hipStream_t);
void mine_bn_act_fwd_bf16(const void*, const void*, const float*, const float*,
                          const float*, const float*, void*, int64_t, int, int,
                          hipStream_t);
void mine_bn_act_bwd_reduce_f32(const void*, const void*, const void*,
                                const float*, const float*, const float*,
                                const float*, float*, int64_t, int, int,
                                hipStream_t);
void mine_bn_act_bwd_reduce_bf16(const void*, const void*, const void*,
                                 const float*, const float*, const float*,
                                 const float*, float*, int64_t, int, int,
                                 hipStream_t);
void mine_bn_act_bwd_dx_f32(const void*, const void*, const void*,
                            const float*, const float*, const float*,
                            const float*, const float*, void*, void*, int64_t,
                            int, int, hipStream_t);
void mine_bn_act_bwd_dx_bf16(const void*, const void*, const void*,
                             const float*, const float*, const float*,
                             const float*, const float*, void*, void*, int64_t,
                             int, int, hipStream_t);
void mine_ssim_set_window(const float*);
void mine_ssim_fwd(const float*, const float*, float*, int, int, int,
                   hipStream_t);
void mine_ssim_bwd(const float*, const float*, const float*, float*, float*,
                   float*, float*, int, int, int, hipStream_t);
}

namespace {

#define CHECK_IN(x)                                               \
  TORCH_CHECK((x).is_cuda(), #x " must be on GPU");               \
  TORCH_CHECK((x).is_contiguous(), #x " must be contiguous");     \
  TORCH_CHECK((x).scalar_type() == at::kFloat, #x " must be f32")

const float* optr(const at::Tensor& t) {
  return t.numel() ? t.data_ptr<float>() : nullptr;
}

hipStream_t stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

std::vector<at::Tensor> src_composite_fwd(at::Tensor mpi, at::Tensor depths,
                                          at::Tensor kinv, at::Tensor img,
                                          bool bg_inf, bool alpha) {
  CHECK_IN(mpi); CHECK_IN(depths); CHECK_IN(kinv);
  TORCH_CHECK(mpi.dim() == 5 && mpi.size(4) == 4, "mpi must be (B,S,H,W,4)");
  const int B = mpi.size(0), S = mpi.size(1), H = mpi.size(2), W = mpi.size(3);
  TORCH_CHECK(S <= 192, "S too large for the LDS geometry stage");
  const bool blend = img.numel() > 0;
  if (blend) {
    CHECK_IN(img);
    TORCH_CHECK(img.sizes() == at::IntArrayRef({B, H, W, 3}),
                "img must be (B,H,W,3)");
  }
  auto rgb = at::empty({B, 3, H, W}, mpi.options());
  auto depth = at::empty({B, 1, H, W}, mpi.options());
  auto mpi_blend = blend ? at::empty_like(mpi) : at::empty({0}, mpi.options());
  mine_src_composite_fwd(mpi.data_ptr<float>(), depths.data_ptr<float>(),
                         kinv.data_ptr<float>(), optr(img),
                         rgb.data_ptr<float>(), depth.data_ptr<float>(),
                         blend ? mpi_blend.data_ptr<float>() : nullptr,
                         B, S, H, W, bg_inf ? 1 : 0, alpha ? 1 : 0,
                         stream());
  return {rgb, depth, mpi_blend};
}

at::Tensor src_composite_bwd(at::Tensor mpi, at::Tensor depths,
                             at::Tensor kinv, at::Tensor img, bool bg_inf,
                             bool alpha, at::Tensor g_rgb,
                             at::Tensor g_depth, at::Tensor g_blend) {
  CHECK_IN(mpi);
  const int B = mpi.size(0), S = mpi.size(1), H = mpi.size(2), W = mpi.size(3);
  auto grad_mpi = at::empty_like(mpi);
  mine_src_composite_bwd(mpi.data_ptr<float>(), depths.data_ptr<float>(),
                         kinv.data_ptr<float>(), optr(img), optr(g_rgb),
                         optr(g_depth), optr(g_blend),
                         grad_mpi.data_ptr<float>(), B, S, H, W,
                         bg_inf ? 1 : 0, alpha ? 1 : 0, stream());
  return grad_mpi;
}

std::vector<at::Tensor> tgt_composite_fwd(at::Tensor mpi, at::Tensor hinv,
                                          at::Tensor m, at::Tensor tvec,
                                          at::Tensor depths, bool bg_inf,
                                          bool alpha) {
  CHECK_IN(mpi); CHECK_IN(hinv); CHECK_IN(m); CHECK_IN(tvec); CHECK_IN(depths);
  TORCH_CHECK(mpi.dim() == 5 && mpi.size(4) == 4, "mpi must be (B,S,H,W,4)");
  const int B = mpi.size(0), S = mpi.size(1), H = mpi.size(2), W = mpi.size(3);
  TORCH_CHECK(S <= 192, "S too large for the LDS geometry stage");
  TORCH_CHECK(hinv.sizes() == at::IntArrayRef({B, S, 3, 3}));
  auto rgb = at::empty({B, 3, H, W}, mpi.options());
  auto depth = at::empty({B, 1, H, W}, mpi.options());
  auto mask = at::empty({B, 1, H, W}, mpi.options());
  mine_tgt_composite_fwd(mpi.data_ptr<float>(), hinv.data_ptr<float>(),
                         m.data_ptr<float>(), tvec.data_ptr<float>(),
                         depths.data_ptr<float>(), rgb.data_ptr<float>(),
                         depth.data_ptr<float>(), mask.data_ptr<float>(),
                         B, S, H, W, bg_inf ? 1 : 0, alpha ? 1 : 0,
                         stream());
  return {rgb, depth, mask};
}

// mode 1 (default): gather redesign — interior pixels write plain
// per-plane payloads, a per-src-tile gather kernel inverts the map
// (hfwd required). mode 0: the round-1 all-atomic scatter (kept for
// A/B tests and as a fallback).
at::Tensor tgt_composite_bwd(at::Tensor mpi, at::Tensor hinv, at::Tensor hfwd,
                             at::Tensor m, at::Tensor tvec, at::Tensor depths,
                             bool bg_inf, bool alpha, at::Tensor g_rgb,
                             at::Tensor g_depth, int64_t mode) {
  CHECK_IN(mpi);
  const int B = mpi.size(0), S = mpi.size(1), H = mpi.size(2), W = mpi.size(3);
  auto grad_mpi = at::zeros_like(mpi);
  at::Tensor payload;
  float* pay_ptr = nullptr;
  if (mode == 1) {
    TORCH_CHECK(hfwd.numel() == (int64_t)B * S * 9, "hfwd required in gather mode");
    payload = at::zeros_like(mpi);
    pay_ptr = payload.data_ptr<float>();
  }
  mine_tgt_composite_bwd(mpi.data_ptr<float>(), hinv.data_ptr<float>(),
                         mode == 1 ? hfwd.data_ptr<float>() : nullptr,
                         m.data_ptr<float>(), tvec.data_ptr<float>(),
                         depths.data_ptr<float>(), optr(g_rgb), optr(g_depth),
                         grad_mpi.data_ptr<float>(), pay_ptr, B, S, H, W,
                         bg_inf ? 1 : 0, (int)mode, alpha ? 1 : 0, stream());
  return grad_mpi;
}

// --------------------------------------------------------------------------
// general igemm conv (encoder/neck/base shapes; igemm_kernels.hip)

// fused edge-aware smoothness v2 (loss_kernels.hip)
at::Tensor eav2_fwd(at::Tensor disp, at::Tensor img, at::Tensor mean_d) {
  TORCH_CHECK(disp.is_cuda() && disp.is_contiguous() &&
              disp.scalar_type() == at::kFloat &&
              img.scalar_type() == at::kFloat);
  const int B = disp.size(0), H = disp.size(2), W = disp.size(3);
  auto out = at::zeros({2}, disp.options());
  mine_eav2_fwd(disp.data_ptr<float>(), img.data_ptr<float>(),
                mean_d.data_ptr<float>(), out.data_ptr<float>(), B, H, W,
                img.stride(0), img.stride(1), img.stride(2), img.stride(3),
                stream());
  return out;
}

at::Tensor eav2_bwd(at::Tensor disp, at::Tensor img, at::Tensor mean_d,
                    at::Tensor gl) {
  const int B = disp.size(0), H = disp.size(2), W = disp.size(3);
  auto g_d = at::empty({B, 1, H, W}, disp.options());
  auto Tb = at::zeros({B}, disp.options());
  auto grad = at::empty_like(disp);
  mine_eav2_bwd(disp.data_ptr<float>(), img.data_ptr<float>(),
                mean_d.data_ptr<float>(), g_d.data_ptr<float>(),
                Tb.data_ptr<float>(), gl.data_ptr<float>(),
                grad.data_ptr<float>(), B, H, W,
                img.stride(0), img.stride(1), img.stride(2), img.stride(3),
                stream());
  return grad;
}

// one-launch fragment pack: bf16 gather through a device LUT (neg -> 0)
at::Tensor pack_gather(at::Tensor w, at::Tensor lut) {
  TORCH_CHECK(w.is_cuda() && w.is_contiguous() && lut.is_contiguous());
  TORCH_CHECK(lut.scalar_type() == at::kInt);
  TORCH_CHECK(w.scalar_type() == at::kFloat ||
              w.scalar_type() == at::kBFloat16);
  auto out = at::empty({lut.numel()}, w.options().dtype(at::kBFloat16));
  mine_pack_gather(w.data_ptr(), lut.data_ptr<int>(), out.data_ptr(),
                   lut.numel(), w.scalar_type() == at::kFloat ? 1 : 0,
                   stream());
  return out;
}

at::Tensor conv_igemm_fwd(at::Tensor x_flat, at::Tensor wp, at::Tensor bias,
                          int64_t M, int64_t P, int64_t Q, int64_t K,
                          int64_t Hs, int64_t Ws, int64_t C, int64_t R,
                          int64_t S, int64_t SA, int64_t SB, int64_t SD,
                          int64_t SE, int64_t pad_mode) {
  TORCH_CHECK(x_flat.is_cuda() && x_flat.is_contiguous() &&
              wp.is_contiguous());
  TORCH_CHECK(x_flat.scalar_type() == at::kBFloat16 &&
              wp.scalar_type() == at::kBFloat16);
  TORCH_CHECK(C % 8 == 0);
  auto out = at::empty({M * K}, x_flat.options());
  // contraction split-K when the (M, K) grid underfills the 256 CUs
  const int64_t blocks = ((M + 63) / 64) * ((K + 63) / 64);
  const int nchunks = (int)((R * S * (C / 8) + 3) / 4);
  int splitz = 1;
  at::Tensor ws;
  float* ws_ptr = nullptr;
  if (blocks < 192 && nchunks >= 16) {
    splitz = (int)(256 / blocks + 1);
    if (splitz > nchunks / 4) splitz = (int)(nchunks / 4);
    if (splitz > 16) splitz = 16;
    if (splitz > 1) {
      ws = at::zeros({M * K}, x_flat.options().dtype(at::kFloat));
      ws_ptr = ws.data_ptr<float>();
    }
  }
  mine_conv_igemm_fwd(x_flat.data_ptr(), wp.data_ptr(),
                      bias.numel() ? bias.data_ptr<float>() : nullptr,
                      out.data_ptr(), ws_ptr, M, (int)P, (int)Q, (int)K,
                      (int)Hs, (int)Ws, (int)C, (int)R, (int)S, (int)SA,
                      (int)SB, (int)SD, (int)SE, (int)pad_mode, splitz,
                      stream());
  return out;
}

at::Tensor conv_igemm_wrw(at::Tensor x_flat, at::Tensor gy_flat, int64_t M,
                          int64_t P, int64_t Q, int64_t K, int64_t Hs,
                          int64_t Ws, int64_t C, int64_t R, int64_t S,
                          int64_t SA, int64_t SB, int64_t SD, int64_t SE,
                          int64_t pad_mode) {
  TORCH_CHECK(x_flat.is_cuda() && x_flat.is_contiguous() &&
              gy_flat.is_contiguous());
  TORCH_CHECK(x_flat.scalar_type() == at::kBFloat16 &&
              gy_flat.scalar_type() == at::kBFloat16);
  auto dw = at::zeros({K, C, R, S}, x_flat.options().dtype(at::kFloat));
  mine_conv_igemm_wrw(x_flat.data_ptr(), gy_flat.data_ptr(),
                      dw.data_ptr<float>(), M, (int)P, (int)Q, (int)K,
                      (int)Hs, (int)Ws, (int)C, (int)R, (int)S, (int)SA,
                      (int)SB, (int)SD, (int)SE, (int)pad_mode, stream());
  return dw;
}

// --------------------------------------------------------------------------
// nearest x2 upsample — flat logical (N,H,W,C); bf16 needs C%8==0,
// f32 C%4==0 (see resample_kernels.hip).

at::Tensor upsample2x_fwd(at::Tensor in, int64_t N, int64_t H, int64_t W,
                          int64_t C) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous() &&
              in.numel() == N * H * W * C);
  const bool bf16 = in.scalar_type() == at::kBFloat16;
  TORCH_CHECK(bf16 ? (C % 8 == 0) : (C % 4 == 0));
  auto out = at::empty({N * H * W * 4 * C}, in.options());
  mine_upsample2x_fwd(in.data_ptr(), out.data_ptr(), N, H, W, C,
                      bf16 ? 1 : 0, stream());
  return out;
}

at::Tensor upsample2x_bwd(at::Tensor gout, int64_t N, int64_t H, int64_t W,
                          int64_t C) {
  TORCH_CHECK(gout.is_cuda() && gout.is_contiguous() &&
              gout.numel() == N * H * W * 4 * C);
  const bool bf16 = gout.scalar_type() == at::kBFloat16;
  TORCH_CHECK(bf16 ? (C % 8 == 0) : (C % 4 == 0));
  auto gin = at::empty({N * H * W * C}, gout.options());
  mine_upsample2x_bwd(gout.data_ptr(), gin.data_ptr(), N, H, W, C,
                      bf16 ? 1 : 0, stream());
  return gin;
}

// --------------------------------------------------------------------------
// reflection pad — logical (N,H,W,C) layout (see pad_kernels.hip).
// The caller (mine_amd/ops/pad.py) maps channels_last / NCHW onto it.

at::Tensor reflect_pad_fwd(at::Tensor in, int64_t N, int64_t H, int64_t W,
                           int64_t C, int64_t pad) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous());
  TORCH_CHECK(in.numel() == N * H * W * C, "bad logical shape");
  TORCH_CHECK(H > pad && W > pad, "pad must be < spatial size");
  auto out = at::empty({N * (H + 2 * pad) * (W + 2 * pad) * C}, in.options());
  if (in.scalar_type() == at::kFloat) {
    mine_reflect_pad_fwd_f32(in.data_ptr<float>(), out.data_ptr<float>(),
                             N, H, W, C, pad, stream());
  } else if (in.scalar_type() == at::kBFloat16) {
    mine_reflect_pad_fwd_bf16(in.data_ptr(), out.data_ptr(),
                              N, H, W, C, pad, stream());
  } else {
    TORCH_CHECK(false, "reflect_pad: dtype must be f32 or bf16");
  }
  return out;
}

at::Tensor reflect_pad_bwd(at::Tensor gout, int64_t N, int64_t H, int64_t W,
                           int64_t C, int64_t pad) {
  TORCH_CHECK(gout.is_cuda() && gout.is_contiguous());
  TORCH_CHECK(gout.numel() == N * (H + 2 * pad) * (W + 2 * pad) * C);
  auto gin = at::empty({N * H * W * C}, gout.options());
  if (gout.scalar_type() == at::kFloat) {
    mine_reflect_pad_bwd_f32(gout.data_ptr<float>(), gin.data_ptr<float>(),
                             N, H, W, C, pad, stream());
  } else if (gout.scalar_type() == at::kBFloat16) {
    mine_reflect_pad_bwd_bf16(gout.data_ptr(), gin.data_ptr(),
                              N, H, W, C, pad, stream());
  } else {
    TORCH_CHECK(false, "reflect_pad: dtype must be f32 or bf16");
  }
  return gin;
}

// --------------------------------------------------------------------------
// fused BatchNorm + activation (see bn_kernels.hip). Tensors arrive as
// flat logical (M, C) channels_last views; f32 or bf16.

#define BN_DISPATCH(FN, T, ...)                                   \
  do {                                                            \
    if ((T) == at::kFloat) FN##_f32(__VA_ARGS__);                 \
    else if ((T) == at::kBFloat16) FN##_bf16(__VA_ARGS__);        \
    else TORCH_CHECK(false, "bn: dtype must be f32 or bf16");     \
  } while (0)

// fused reflect-pad + 3x3 conv (MFMA); flat NHWC views, bf16.
// (src_h, src_w, off) back a zero-embedded logical image for the
// transposed/data-grad use (pad_mode 1, off 1, logical = src + 2);
// plain forward: src == logical, off == 0.
void conv3x3_fwd(at::Tensor x_flat, at::Tensor wp, at::Tensor bias,
                 at::Tensor out, int64_t N, int64_t H, int64_t W, int64_t C,
                 int64_t K, int64_t pad_mode, int64_t src_h, int64_t src_w,
                 int64_t off) {
  TORCH_CHECK(x_flat.is_cuda() && x_flat.is_contiguous());
  TORCH_CHECK(x_flat.scalar_type() == at::kBFloat16 &&
              wp.scalar_type() == at::kBFloat16 &&
              out.scalar_type() == at::kBFloat16);
  TORCH_CHECK(C % 8 == 0 && x_flat.numel() == N * src_h * src_w * C);
  TORCH_CHECK(out.numel() == N * H * W * K);
  mine_conv3x3_fwd(x_flat.data_ptr(), wp.data_ptr(),
                   bias.numel() ? bias.data_ptr<float>() : nullptr,
                   out.data_ptr(), (int)N, (int)H, (int)W, (int)C, (int)K,
                   (int)pad_mode, (int)src_h, (int)src_w, (int)off,
                   stream());
}

// Split-K MFMA weight gradient (see wrw_kernels.hip);
// returns dW fp32 (K, C, 3, 3). Flat NHWC bf16 inputs.
at::Tensor conv3x3_wrw(at::Tensor x_flat, at::Tensor gy_flat, int64_t N,
                       int64_t H, int64_t W, int64_t C, int64_t K) {
  TORCH_CHECK(x_flat.is_cuda() && x_flat.is_contiguous() &&
              gy_flat.is_contiguous());
  TORCH_CHECK(x_flat.scalar_type() == at::kBFloat16 &&
              gy_flat.scalar_type() == at::kBFloat16);
  TORCH_CHECK(C % 16 == 0 && W >= 2);
  auto dw = at::zeros({K, C, 3, 3}, x_flat.options().dtype(at::kFloat));
  mine_conv3x3_wrw(x_flat.data_ptr(), gy_flat.data_ptr(),
                   dw.data_ptr<float>(), (int)N, (int)H, (int)W, (int)C,
                   (int)K, stream());
  return dw;
}

// fused MPI head over a flat (N,4) view; out is fp32 (N,4)
at::Tensor mpi_head_fwd(at::Tensor z, int64_t N, bool alpha) {
  TORCH_CHECK(z.is_cuda() && z.is_contiguous() && z.numel() == N * 4);
  auto out = at::empty({N * 4}, z.options().dtype(at::kFloat));
  if (z.scalar_type() == at::kFloat)
    mine_mpi_head_fwd_f32(z.data_ptr(), out.data_ptr<float>(), N,
                          alpha ? 1 : 0, stream());
  else if (z.scalar_type() == at::kBFloat16)
    mine_mpi_head_fwd_bf16(z.data_ptr(), out.data_ptr<float>(), N,
                           alpha ? 1 : 0, stream());
  else
    TORCH_CHECK(false, "mpi_head: dtype must be f32 or bf16");
  return out;
}

at::Tensor mpi_head_bwd(at::Tensor z, at::Tensor gout, int64_t N, bool alpha) {
  TORCH_CHECK(gout.is_contiguous() && gout.scalar_type() == at::kFloat);
  auto gz = at::empty_like(z);
  if (z.scalar_type() == at::kFloat)
    mine_mpi_head_bwd_f32(z.data_ptr(), gout.data_ptr<float>(), gz.data_ptr(),
                          N, alpha ? 1 : 0, stream());
  else
    mine_mpi_head_bwd_bf16(z.data_ptr(), gout.data_ptr<float>(),
                           gz.data_ptr(), N, alpha ? 1 : 0, stream());
  return gz;
}

at::Tensor bn_sums(at::Tensor x, int64_t M, int64_t C) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.numel() == M * C);
  auto sums = at::zeros({2 * C}, x.options().dtype(at::kFloat));
  BN_DISPATCH(mine_bn_stats, x.scalar_type(), x.data_ptr(),
              sums.data_ptr<float>(), M, (int)C, stream());
  return sums;
}

std::vector<at::Tensor> bn_stats(at::Tensor x, int64_t M, int64_t C,
                                 at::Tensor running_mean,
                                 at::Tensor running_var, double eps,
                                 double momentum) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.numel() == M * C);
  auto f32 = x.options().dtype(at::kFloat);
  auto sums = at::zeros({2 * C}, f32);
  BN_DISPATCH(mine_bn_stats, x.scalar_type(), x.data_ptr(),
              sums.data_ptr<float>(), M, (int)C, stream());
  auto mean = at::empty({C}, f32);
  auto invstd = at::empty({C}, f32);
  const bool track = running_mean.numel() > 0;
  mine_bn_finalize(sums.data_ptr<float>(), mean.data_ptr<float>(),
                   invstd.data_ptr<float>(),
                   track ? running_mean.data_ptr<float>() : nullptr,
                   track ? running_var.data_ptr<float>() : nullptr,
                   M, (int)C, (float)eps, (float)momentum, stream());
  return {mean, invstd};
}

at::Tensor bn_act_fwd(at::Tensor x, at::Tensor res, at::Tensor mean,
                      at::Tensor invstd, at::Tensor gamma, at::Tensor beta,
                      int64_t M, int64_t C, int64_t act) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.numel() == M * C);
  auto y = at::empty_like(x);
  BN_DISPATCH(mine_bn_act_fwd, x.scalar_type(), x.data_ptr(),
              res.numel() ? res.data_ptr() : nullptr,
              mean.data_ptr<float>(), invstd.data_ptr<float>(),
              gamma.data_ptr<float>(), beta.data_ptr<float>(), y.data_ptr(),
              M, (int)C, (int)act, stream());
  return y;
}

at::Tensor bn_act_bwd_reduce(at::Tensor x, at::Tensor res, at::Tensor gy,
                             at::Tensor mean, at::Tensor invstd,
                             at::Tensor gamma, at::Tensor beta, int64_t M,
                             int64_t C, int64_t act) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && gy.is_contiguous());
  auto red = at::zeros({2 * C}, x.options().dtype(at::kFloat));
  const bool add_relu = act == 4;
  BN_DISPATCH(mine_bn_act_bwd_reduce, x.scalar_type(), x.data_ptr(),
              add_relu ? res.data_ptr() : nullptr, gy.data_ptr(),
              mean.data_ptr<float>(), invstd.data_ptr<float>(),
              gamma.data_ptr<float>(), beta.data_ptr<float>(),
              red.data_ptr<float>(), M, (int)C, (int)act, stream());
  return red;  // (dbeta, dgamma)
}

// M iterates the LOCAL elements; M_norm is the (possibly cross-rank
// global) batch count in the d(mean)/d(var) terms — they differ only
// under SyncBN, where `red` has been all-reduced in between.
std::vector<at::Tensor> bn_act_bwd_dx(at::Tensor x, at::Tensor res,
                                      at::Tensor gy, at::Tensor mean,
                                      at::Tensor invstd, at::Tensor gamma,
                                      at::Tensor beta, at::Tensor red,
                                      int64_t M, int64_t C, int64_t act,
                                      int64_t M_norm) {
  const bool add_relu = act == 4;
  auto scaled = red;
  if (M_norm != M) {
    // the dx kernel divides by its M argument; rescale red instead of
    // adding a second kernel parameter
    scaled = red * ((double)M / (double)M_norm);
  }
  auto dx = at::empty_like(x);
  auto dres = add_relu ? at::empty_like(x) : at::empty({0}, x.options());
  BN_DISPATCH(mine_bn_act_bwd_dx, x.scalar_type(), x.data_ptr(),
              add_relu ? res.data_ptr() : nullptr, gy.data_ptr(),
              mean.data_ptr<float>(), invstd.data_ptr<float>(),
              gamma.data_ptr<float>(), beta.data_ptr<float>(),
              scaled.contiguous().data_ptr<float>(), dx.data_ptr(),
              add_relu ? dres.data_ptr() : nullptr, M, (int)C, (int)act,
              stream());
  return {dx, dres};
}

std::vector<at::Tensor> bn_act_bwd(at::Tensor x, at::Tensor res,
                                   at::Tensor gy, at::Tensor mean,
                                   at::Tensor invstd, at::Tensor gamma,
                                   at::Tensor beta, int64_t M, int64_t C,
                                   int64_t act) {
  auto red = bn_act_bwd_reduce(x, res, gy, mean, invstd, gamma, beta,
                               M, C, act);
  auto dxr = bn_act_bwd_dx(x, res, gy, mean, invstd, gamma, beta, red,
                           M, C, act, M);
  // dbeta = red[:C], dgamma = red[C:]
  return {dxr[0], dxr[1], red.narrow(0, C, C).clone(),
          red.narrow(0, 0, C).clone()};
}

// --------------------------------------------------------------------------

bool g_window_set = false;

void ensure_window() {
  if (g_window_set) return;
  // Gaussian(11, sigma=1.5), normalized (ref network/ssim.py:7-9)
  double w[11], sum = 0.0;
  for (int i = 0; i < 11; ++i) {
    const double d = i - 5;
    w[i] = std::exp(-d * d / (2.0 * 1.5 * 1.5));
    sum += w[i];
  }
  float wf[11];
  for (int i = 0; i < 11; ++i) wf[i] = (float)(w[i] / sum);
  mine_ssim_set_window(wf);
  g_window_set = true;
}

std::vector<at::Tensor> ssim_fwd(at::Tensor img1, at::Tensor img2) {
  CHECK_IN(img1); CHECK_IN(img2);
  TORCH_CHECK(img1.dim() == 4 && img1.sizes() == img2.sizes());
  ensure_window();
  const int BC = img1.size(0) * img1.size(1);
  const int H = img1.size(2), W = img1.size(3);
  auto sum = at::zeros({1}, img1.options());
  mine_ssim_fwd(img1.data_ptr<float>(), img2.data_ptr<float>(),
                sum.data_ptr<float>(), BC, H, W, stream());
  return {(sum / (double)(BC * (int64_t)H * W)).squeeze(0)};
}

at::Tensor ssim_bwd(at::Tensor img1, at::Tensor img2, at::Tensor gscale) {
  CHECK_IN(img1); CHECK_IN(img2);
  ensure_window();
  const int BC = img1.size(0) * img1.size(1);
  const int H = img1.size(2), W = img1.size(3);
  auto F1 = at::empty_like(img1);
  auto F2 = at::empty_like(img1);
  auto F3 = at::empty_like(img1);
  auto grad1 = at::empty_like(img1);
  auto g = gscale.to(img1.options()).contiguous();
  mine_ssim_bwd(img1.data_ptr<float>(), img2.data_ptr<float>(),
                g.data_ptr<float>(), F1.data_ptr<float>(),
                F2.data_ptr<float>(), F3.data_ptr<float>(),
                grad1.data_ptr<float>(), BC, H, W, stream());
  return grad1;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("src_composite_fwd", &src_composite_fwd,
          "fused src-view MPI composite + RGB blend, forward");
  mod.def("src_composite_bwd", &src_composite_bwd);
  mod.def("tgt_composite_fwd", &tgt_composite_fwd,
          "fused plane-sweep warp + z-cull + composite, forward");
  mod.def("tgt_composite_bwd", &tgt_composite_bwd);
  mod.def("ssim_fwd", &ssim_fwd);
  mod.def("ssim_bwd", &ssim_bwd);
  mod.def("eav2_fwd", &eav2_fwd, "fused edge-aware smoothness v2 fwd");
  mod.def("eav2_bwd", &eav2_bwd, "fused edge-aware smoothness v2 bwd");
  mod.def("pack_gather", &pack_gather,
          "one-launch fragment pack through a device LUT");
  mod.def("conv_igemm_fwd", &conv_igemm_fwd,
          "general igemm conv fwd / data-grad (coordinate-remapped)");
  mod.def("conv_igemm_wrw", &conv_igemm_wrw,
          "general igemm conv weight-grad");
  mod.def("upsample2x_fwd", &upsample2x_fwd,
          "nearest x2 upsample fwd, flat NHWC");
  mod.def("upsample2x_bwd", &upsample2x_bwd,
          "nearest x2 upsample bwd (4-child sum), flat NHWC");
  mod.def("reflect_pad_fwd", &reflect_pad_fwd,
          "gather reflection pad over logical (N,H,W,C)");
  mod.def("reflect_pad_bwd", &reflect_pad_bwd,
          "gather (atomic-free) reflection pad backward");
  mod.def("bn_stats", &bn_stats,
          "per-channel mean/invstd + running-stat update");
  mod.def("bn_sums", &bn_sums,
          "per-channel (sum, sumsq) only — for cross-rank SyncBN");
  mod.def("bn_act_fwd", &bn_act_fwd, "fused normalize + activation");
  mod.def("bn_act_bwd", &bn_act_bwd,
          "fused BN+act backward -> dx, dres, dgamma, dbeta");
  mod.def("bn_act_bwd_reduce", &bn_act_bwd_reduce);
  mod.def("bn_act_bwd_dx", &bn_act_bwd_dx);
  mod.def("mpi_head_fwd", &mpi_head_fwd,
          "dispconv out -> packed fp32 MPI (sigmoid rgb, |x|+1e-4 sigma)");
  mod.def("mpi_head_bwd", &mpi_head_bwd);
  mod.def("conv3x3_fwd", &conv3x3_fwd,
          "fused reflect/zero-pad + 3x3 conv on MFMA 16x16x32 tiles");
  mod.def("conv3x3_wrw", &conv3x3_wrw,
          "EXPERIMENTAL split-K MFMA weight gradient (unwired)");
}
