// Torch extension entry: dispatch + launch for the CDNA4 kernel library.
// Single translation unit (the kernel files are header-style templates).

#include <torch/extension.h>

#include "common.h"
#include "elementwise.hip"
#include "gemm.hip"
#include "gemm_mfma.hip"
#include "conv.hip"
#include "losses.hip"
#include "fused_mnist.hip"

#include <ATen/hip/HIPContext.h>

namespace {

inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

inline int grid_1d(long total, int block = 256, int cap = 4096) {
  long g = (total + block - 1) / block;
  return (int)std::min<long>(g, cap);
}

#define CHECK_DEV(t) \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t " must be contiguous on GPU")

#define DISPATCH_FT(scalar_t_tensor, ...)                                  \
  AT_DISPATCH_FLOATING_TYPES(scalar_t_tensor.scalar_type(), "ndta_ops",    \
                             [&] { __VA_ARGS__ });

// ---------------------------------------------------------------- ew --
void dinno_dual_threg(torch::Tensor local,
                      c10::optional<torch::Tensor> remote,
                      torch::Tensor offs, torch::Tensor idx,
                      torch::Tensor duals, torch::Tensor s_out,
                      double rho) {
  CHECK_DEV(local); CHECK_DEV(duals); CHECK_DEV(s_out);
  const long L = duals.size(0), n = duals.size(1);
  DISPATCH_FT(local, {
    hipLaunchKernelGGL(ew::dinno_dual_threg_k<scalar_t>,
        dim3(grid_1d(L * n)), dim3(ew::BLOCK), 0, cur_stream(),
        local.data_ptr<scalar_t>(),
        remote.has_value() ? remote->data_ptr<scalar_t>() : nullptr,
        offs.data_ptr<int>(), idx.data_ptr<int>(),
        duals.data_ptr<scalar_t>(), s_out.data_ptr<scalar_t>(),
        (scalar_t)rho, n, L);
  });
  HIP_CHECK_LAST();
}

void mix_rows(torch::Tensor local, c10::optional<torch::Tensor> remote,
              torch::Tensor offs, torch::Tensor idx, torch::Tensor w,
              torch::Tensor out) {
  CHECK_DEV(local); CHECK_DEV(out);
  TORCH_CHECK(local.data_ptr() != out.data_ptr(),
              "mix_rows: out must not alias the local table");
  const long L = out.size(0), n = out.size(1);
  DISPATCH_FT(local, {
    hipLaunchKernelGGL(ew::mix_rows_k<scalar_t>,
        dim3(grid_1d(L * n)), dim3(ew::BLOCK), 0, cur_stream(),
        local.data_ptr<scalar_t>(),
        remote.has_value() ? remote->data_ptr<scalar_t>() : nullptr,
        offs.data_ptr<int>(), idx.data_ptr<int>(),
        w.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(), n, L);
  });
  HIP_CHECK_LAST();
}

void dsgt_mix(torch::Tensor p_loc, torch::Tensor y_loc,
              c10::optional<torch::Tensor> remote, torch::Tensor offs,
              torch::Tensor idx, torch::Tensor w, torch::Tensor p_out,
              torch::Tensor y_mix, double alpha) {
  CHECK_DEV(p_loc); CHECK_DEV(y_loc); CHECK_DEV(p_out); CHECK_DEV(y_mix);
  TORCH_CHECK(p_loc.data_ptr() != p_out.data_ptr(),
              "dsgt_mix: p_out must not alias p_loc");
  const long L = p_out.size(0), n = p_out.size(1);
  DISPATCH_FT(p_loc, {
    hipLaunchKernelGGL(ew::dsgt_mix_k<scalar_t>,
        dim3(grid_1d(L * n)), dim3(ew::BLOCK), 0, cur_stream(),
        p_loc.data_ptr<scalar_t>(), y_loc.data_ptr<scalar_t>(),
        remote.has_value() ? remote->data_ptr<scalar_t>() : nullptr,
        offs.data_ptr<int>(), idx.data_ptr<int>(),
        w.data_ptr<scalar_t>(), p_out.data_ptr<scalar_t>(),
        y_mix.data_ptr<scalar_t>(), (scalar_t)alpha, n, L);
  });
  HIP_CHECK_LAST();
}

void gather_batch(torch::Tensor X_all, torch::Tensor idx,
                  torch::Tensor out, long idx_stride, long idx_off) {
  CHECK_DEV(X_all); CHECK_DEV(out);
  const long L = X_all.size(0), maxlen = X_all.size(1),
             Fdim = X_all.size(2);
  const long B = out.size(0) / L;
  const long total = out.numel();
  DISPATCH_FT(X_all, {
    hipLaunchKernelGGL(ew::gather_batch_k<scalar_t>,
        dim3(grid_1d(total)), dim3(ew::BLOCK), 0, cur_stream(),
        X_all.data_ptr<scalar_t>(), idx.data_ptr<long>(),
        out.data_ptr<scalar_t>(), maxlen, Fdim, B, idx_stride,
        idx_off, total);
  });
  HIP_CHECK_LAST();
}

void gather_targets(torch::Tensor Y_all, torch::Tensor idx,
                    torch::Tensor out, long idx_stride, long idx_off) {
  CHECK_DEV(Y_all); CHECK_DEV(out);
  const long L = Y_all.size(0), maxlen = Y_all.size(1);
  const long B = out.numel() / L;
  const long total = out.numel();
  if (Y_all.scalar_type() == torch::kLong) {
    hipLaunchKernelGGL(ew::gather_targets_k<long>,
        dim3(grid_1d(total)), dim3(ew::BLOCK), 0, cur_stream(),
        Y_all.data_ptr<long>(), idx.data_ptr<long>(),
        out.data_ptr<long>(), maxlen, B, idx_stride, idx_off, total);
  } else {
    DISPATCH_FT(Y_all, {
      hipLaunchKernelGGL(ew::gather_targets_k<scalar_t>,
          dim3(grid_1d(total)), dim3(ew::BLOCK), 0, cur_stream(),
          Y_all.data_ptr<scalar_t>(), idx.data_ptr<long>(),
          out.data_ptr<scalar_t>(), maxlen, B, idx_stride, idx_off,
          total);
    });
  }
  HIP_CHECK_LAST();
}

void dsgt_y_update(torch::Tensor y_mix, torch::Tensor g_new,
                   torch::Tensor g_old, torch::Tensor y) {
  CHECK_DEV(y);
  const long total = y.numel();
  DISPATCH_FT(y, {
    hipLaunchKernelGGL(ew::dsgt_y_update_k<scalar_t>,
        dim3(grid_1d(total)), dim3(ew::BLOCK), 0, cur_stream(),
        y_mix.data_ptr<scalar_t>(), g_new.data_ptr<scalar_t>(),
        g_old.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(), total);
  });
  HIP_CHECK_LAST();
}

void fused_step(torch::Tensor theta, torch::Tensor grad,
                c10::optional<torch::Tensor> dual,
                c10::optional<torch::Tensor> s,
                c10::optional<torch::Tensor> deg,
                c10::optional<torch::Tensor> m,
                c10::optional<torch::Tensor> v,
                double rho, double lr, double beta1, double beta2,
                double eps, double wd, long step_t, long mode,
                bool first_step, long nparts, bool zero_grad) {
  CHECK_DEV(theta); CHECK_DEV(grad);
  const long L = theta.size(0), n = theta.size(1);
  TORCH_CHECK(grad.numel() == L * nparts * n, "grad/nparts mismatch");
  const bool pen = dual.has_value();
  DISPATCH_FT(theta, {
    const scalar_t bc1 =
        (scalar_t)(1.0 - std::pow(beta1, (double)step_t));
    const scalar_t bc2 =
        (scalar_t)(1.0 - std::pow(beta2, (double)step_t));
    auto launch = [&](auto mode_c, auto pen_c) {
      hipLaunchKernelGGL(
          (ew::fused_step_k<scalar_t, decltype(mode_c)::value,
                            decltype(pen_c)::value>),
          dim3(grid_1d(L * n)), dim3(ew::BLOCK), 0, cur_stream(),
          theta.data_ptr<scalar_t>(), grad.data_ptr<scalar_t>(),
          pen ? dual->data_ptr<scalar_t>() : nullptr,
          pen ? s->data_ptr<scalar_t>() : nullptr,
          pen ? deg->data_ptr<int>() : nullptr,
          m.has_value() ? m->data_ptr<scalar_t>() : nullptr,
          v.has_value() ? v->data_ptr<scalar_t>() : nullptr,
          (scalar_t)rho, (scalar_t)lr, (scalar_t)beta1,
          (scalar_t)beta2, (scalar_t)eps, (scalar_t)wd, bc1, bc2,
          first_step ? 1 : 0, (int)nparts, n, L, zero_grad ? 1 : 0);
    };
    using c0 = std::integral_constant<int, 0>;
    using c1 = std::integral_constant<int, 1>;
    using c2 = std::integral_constant<int, 2>;
    using bt = std::integral_constant<bool, true>;
    using bf = std::integral_constant<bool, false>;
    if (mode == 0) { pen ? launch(c0{}, bt{}) : launch(c0{}, bf{}); }
    else if (mode == 1) { pen ? launch(c1{}, bt{}) : launch(c1{}, bf{}); }
    else { pen ? launch(c2{}, bt{}) : launch(c2{}, bf{}); }
  });
  HIP_CHECK_LAST();
}

void dinno_dual_threg_sched(torch::Tensor local,
                            c10::optional<torch::Tensor> remote,
                            torch::Tensor offs, torch::Tensor idx,
                            torch::Tensor duals, torch::Tensor s_out,
                            torch::Tensor sched) {
  CHECK_DEV(local); CHECK_DEV(duals); CHECK_DEV(s_out);
  const long L = duals.size(0), n = duals.size(1);
  DISPATCH_FT(local, {
    hipLaunchKernelGGL(ew::dinno_dual_threg_sched_k<scalar_t>,
        dim3(grid_1d(L * n)), dim3(ew::BLOCK), 0, cur_stream(),
        local.data_ptr<scalar_t>(),
        remote.has_value() ? remote->data_ptr<scalar_t>() : nullptr,
        offs.data_ptr<int>(), idx.data_ptr<int>(),
        duals.data_ptr<scalar_t>(), s_out.data_ptr<scalar_t>(),
        sched.data_ptr<scalar_t>(), n, L);
  });
  HIP_CHECK_LAST();
}

void fused_step_sched(torch::Tensor theta, torch::Tensor grad,
                      c10::optional<torch::Tensor> dual,
                      c10::optional<torch::Tensor> s,
                      c10::optional<torch::Tensor> deg,
                      c10::optional<torch::Tensor> m,
                      c10::optional<torch::Tensor> v,
                      torch::Tensor sched, long pit,
                      double beta1, double beta2, double eps, double wd,
                      long mode, bool first_step, long nparts) {
  CHECK_DEV(theta); CHECK_DEV(grad);
  const long L = theta.size(0), n = theta.size(1);
  TORCH_CHECK(grad.numel() == L * nparts * n, "grad/nparts mismatch");
  const bool pen = dual.has_value();
  DISPATCH_FT(theta, {
    auto launch = [&](auto mode_c, auto pen_c) {
      hipLaunchKernelGGL(
          (ew::fused_step_sched_k<scalar_t, decltype(mode_c)::value,
                                  decltype(pen_c)::value>),
          dim3(grid_1d(L * n)), dim3(ew::BLOCK), 0, cur_stream(),
          theta.data_ptr<scalar_t>(), grad.data_ptr<scalar_t>(),
          pen ? dual->data_ptr<scalar_t>() : nullptr,
          pen ? s->data_ptr<scalar_t>() : nullptr,
          pen ? deg->data_ptr<int>() : nullptr,
          m.has_value() ? m->data_ptr<scalar_t>() : nullptr,
          v.has_value() ? v->data_ptr<scalar_t>() : nullptr,
          sched.data_ptr<scalar_t>(), (int)pit,
          (scalar_t)beta1, (scalar_t)beta2, (scalar_t)eps,
          (scalar_t)wd, first_step ? 1 : 0, (int)nparts, n, L);
    };
    using c0 = std::integral_constant<int, 0>;
    using c1 = std::integral_constant<int, 1>;
    using c2 = std::integral_constant<int, 2>;
    using bt = std::integral_constant<bool, true>;
    using bf = std::integral_constant<bool, false>;
    if (mode == 0) { pen ? launch(c0{}, bt{}) : launch(c0{}, bf{}); }
    else if (mode == 1) { pen ? launch(c1{}, bt{}) : launch(c1{}, bf{}); }
    else { pen ? launch(c2{}, bt{}) : launch(c2{}, bf{}); }
  });
  HIP_CHECK_LAST();
}

void gather_batch_dev(torch::Tensor X_all, torch::Tensor idx,
                      torch::Tensor out, torch::Tensor offs_dev,
                      long pit, long idx_stride) {
  CHECK_DEV(X_all); CHECK_DEV(out);
  const long L = X_all.size(0), maxlen = X_all.size(1),
             Fdim = X_all.size(2);
  const long B = out.size(0) / L;
  const long total = out.numel();
  DISPATCH_FT(X_all, {
    hipLaunchKernelGGL(ew::gather_batch_dev_k<scalar_t>,
        dim3(grid_1d(total)), dim3(ew::BLOCK), 0, cur_stream(),
        X_all.data_ptr<scalar_t>(), idx.data_ptr<long>(),
        out.data_ptr<scalar_t>(), offs_dev.data_ptr<long>(), (int)pit,
        maxlen, Fdim, B, idx_stride, total);
  });
  HIP_CHECK_LAST();
}

void gather_targets_dev(torch::Tensor Y_all, torch::Tensor idx,
                        torch::Tensor out, torch::Tensor offs_dev,
                        long pit, long idx_stride) {
  CHECK_DEV(Y_all); CHECK_DEV(out);
  const long L = Y_all.size(0), maxlen = Y_all.size(1);
  const long B = out.numel() / L;
  const long total = out.numel();
  if (Y_all.scalar_type() == torch::kLong) {
    hipLaunchKernelGGL(ew::gather_targets_dev_k<long>,
        dim3(grid_1d(total)), dim3(ew::BLOCK), 0, cur_stream(),
        Y_all.data_ptr<long>(), idx.data_ptr<long>(),
        out.data_ptr<long>(), offs_dev.data_ptr<long>(), (int)pit,
        maxlen, B, idx_stride, total);
  } else {
    DISPATCH_FT(Y_all, {
      hipLaunchKernelGGL(ew::gather_targets_dev_k<scalar_t>,
          dim3(grid_1d(total)), dim3(ew::BLOCK), 0, cur_stream(),
          Y_all.data_ptr<scalar_t>(), idx.data_ptr<long>(),
          out.data_ptr<scalar_t>(), offs_dev.data_ptr<long>(),
          (int)pit, maxlen, B, idx_stride, total);
    });
  }
  HIP_CHECK_LAST();
}

void reduce_parts(torch::Tensor parts, torch::Tensor out,
                  long nparts) {
  CHECK_DEV(parts); CHECK_DEV(out);
  const long L = out.size(0), n = out.size(1);
  TORCH_CHECK(parts.numel() == L * nparts * n, "parts shape mismatch");
  DISPATCH_FT(out, {
    hipLaunchKernelGGL(ew::reduce_parts_k<scalar_t>,
        dim3(grid_1d(L * n)), dim3(ew::BLOCK), 0, cur_stream(),
        parts.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
        (int)nparts, n, L);
  });
  HIP_CHECK_LAST();
}

void axpy(torch::Tensor x, torch::Tensor g, double alpha,
          bool zero_grad) {
  CHECK_DEV(x);
  DISPATCH_FT(x, {
    hipLaunchKernelGGL(ew::axpy_k<scalar_t>,
        dim3(grid_1d(x.numel())), dim3(ew::BLOCK), 0, cur_stream(),
        x.data_ptr<scalar_t>(), g.data_ptr<scalar_t>(),
        (scalar_t)alpha, x.numel(), zero_grad ? 1 : 0);
  });
  HIP_CHECK_LAST();
}

// -------------------------------------------------------------- gemm --
void linear_fwd(torch::Tensor X, torch::Tensor theta, torch::Tensor Y,
                c10::optional<torch::Tensor> Z, long w_off, long b_off,
                long M, long I, long O, long act, double scale) {
  CHECK_DEV(X); CHECK_DEV(theta); CHECK_DEV(Y);
  const long L = theta.size(0), n = theta.size(1);
  // MFMA pays only when M fills the 64-row tiles across the chip;
  // small-M layers (MNIST fc, per-node B=64) stay on the VALU kernels.
  // I must be even for the 16-byte staging loads (vec2 alignment).
  const bool use_mfma = (M >= 128 && O >= 16 && I >= 8 && I % 2 == 0);
  DISPATCH_FT(X, {
    auto zp = Z.has_value() ? Z->data_ptr<scalar_t>() : nullptr;
    if (use_mfma) {
      dim3 grid((O + 63) / 64, (M + 63) / 64, L);
      hipLaunchKernelGGL(gmfma::mfma_fwd_k<scalar_t>,
          grid, dim3(256), 0, cur_stream(),
          X.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          Y.data_ptr<scalar_t>(), zp, n, w_off, b_off,
          (int)M, (int)I, (int)O, (int)act, (scalar_t)scale);
    } else if (I <= 4 && M >= 1024 && O >= 64 && O <= 1024 &&
               O % 64 == 0) {
      // register-resident encode kernel: one thread per output
      // column (blockDim == O), W row + bias in registers, scalar
      // X-row loads, compile-time-I guard-free dot
      const long blocks = std::min<long>(M, 8192);
      dim3 grid(blocks, 1, L);
      auto launch_enc = [&](auto ik_tag) {
        constexpr int IK = decltype(ik_tag)::value;
        hipLaunchKernelGGL((gemm::encode_fwd_k<scalar_t, IK>), grid,
            dim3((int)O), 0, cur_stream(), X.data_ptr<scalar_t>(),
            theta.data_ptr<scalar_t>(), Y.data_ptr<scalar_t>(), zp, n,
            w_off, b_off, (int)M, (int)O, (int)act, (scalar_t)scale);
      };
      if (I == 1) launch_enc(std::integral_constant<int, 1>{});
      else if (I == 2) launch_enc(std::integral_constant<int, 2>{});
      else if (I == 3) launch_enc(std::integral_constant<int, 3>{});
      else launch_enc(std::integral_constant<int, 4>{});
    } else if (I <= 4 && M >= 1024) {
      const long total = (long)M * O;
      const long blocks =
          std::min<long>((total + 255) / 256, 2048);
      const size_t shmem = (size_t)(O * I + O) * sizeof(scalar_t);
      hipLaunchKernelGGL((gemm::linear_fwd_smallk_k<scalar_t, 4>),
          dim3(blocks, 1, L), dim3(256), shmem, cur_stream(),
          X.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          Y.data_ptr<scalar_t>(), zp, n, w_off, b_off,
          (int)M, (int)I, (int)O, (int)act, (scalar_t)scale);
    } else {
      dim3 grid((O + 15) / 16, (M + 15) / 16, L);
      hipLaunchKernelGGL(gemm::linear_fwd_k<scalar_t>,
          grid, dim3(16, 16), 0, cur_stream(),
          X.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          Y.data_ptr<scalar_t>(), zp, n, w_off, b_off,
          (int)M, (int)I, (int)O, (int)act, (scalar_t)scale);
    }
  });
  HIP_CHECK_LAST();
}

void act_grad(torch::Tensor dY, torch::Tensor Y,
              c10::optional<torch::Tensor> Z, torch::Tensor dZ,
              long act, double scale) {
  CHECK_DEV(dY); CHECK_DEV(dZ);
  DISPATCH_FT(dY, {
    hipLaunchKernelGGL(gemm::act_grad_k<scalar_t>,
        dim3(grid_1d(dY.numel())), dim3(256), 0, cur_stream(),
        dY.data_ptr<scalar_t>(), Y.data_ptr<scalar_t>(),
        Z.has_value() ? Z->data_ptr<scalar_t>() : nullptr,
        dZ.data_ptr<scalar_t>(), dY.numel(), (int)act, (scalar_t)scale);
  });
  HIP_CHECK_LAST();
}

// dX = dZ @ W with the BELOW layer's activation backward fused when
// act_below != 0 (Yb = below activation output, Zb = its pre-activation
// where needed, i.e. sin_relu).
void linear_bwd_dx(torch::Tensor dZ, torch::Tensor theta,
                   torch::Tensor dX,
                   c10::optional<torch::Tensor> Yb,
                   c10::optional<torch::Tensor> Zb,
                   long act_below, double scale_below,
                   long w_off, long M, long I, long O,
                   c10::optional<torch::Tensor> Xb2,
                   long wb_off, long bb_off, long Ib) {
  CHECK_DEV(dZ); CHECK_DEV(dX);
  const long L = theta.size(0), n = theta.size(1);
  const bool use_mfma =
      (M >= 128 && I >= 16 && O >= 8 && I % 2 == 0 && O % 2 == 0);
  DISPATCH_FT(dZ, {
    auto ybp = Yb.has_value() ? Yb->data_ptr<scalar_t>() : nullptr;
    auto zbp = Zb.has_value() ? Zb->data_ptr<scalar_t>() : nullptr;
    auto xb2p = Xb2.has_value() ? Xb2->data_ptr<scalar_t>() : nullptr;
    TORCH_CHECK(act_below == 0 || ybp != nullptr,
                "act_below needs Yb");
    TORCH_CHECK(xb2p == nullptr || Ib <= 4,
                "z-recompute supports below-layer in_dim <= 4");
    static const bool dx_new = []() {
      const char* e = getenv("NDTA_DX_NEW");
      return !(e && e[0] == '0');
    }();
    if (use_mfma && dx_new) {
      dim3 grid((I + 63) / 64, (M + 63) / 64, L);
      hipLaunchKernelGGL(gmfma::mfma_dx_k<scalar_t>,
          grid, dim3(256), 0, cur_stream(),
          dZ.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          dX.data_ptr<scalar_t>(), ybp, zbp, (int)act_below,
          (scalar_t)scale_below, n, w_off, (int)M, (int)I, (int)O,
          xb2p, wb_off, bb_off, (int)Ib);
    } else if (use_mfma) {
      dim3 grid((I + 63) / 64, (M + 63) / 64, L);
      hipLaunchKernelGGL(gmfma::mfma_dx16_k<scalar_t>,
          grid, dim3(256), 0, cur_stream(),
          dZ.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          dX.data_ptr<scalar_t>(), ybp, zbp, (int)act_below,
          (scalar_t)scale_below, n, w_off, (int)M, (int)I, (int)O,
          xb2p, wb_off, bb_off, (int)Ib);
    } else {
      dim3 grid((I + 15) / 16, (M + 15) / 16, L);
      hipLaunchKernelGGL(gemm::linear_bwd_dx_k<scalar_t>,
          grid, dim3(16, 16), 0, cur_stream(),
          dZ.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          dX.data_ptr<scalar_t>(), ybp, zbp, (int)act_below,
          (scalar_t)scale_below, n, w_off, (int)M, (int)I, (int)O,
          xb2p, wb_off, bb_off, (int)Ib);
    }
  });
  HIP_CHECK_LAST();
}

// dW/db. PRECONDITION for the atomic paths (MFMA and small-chunked):
// the layer's slice of gstack is zeroed (the stacked engine zeroes the
// whole grad stack at backward start). The plain VALU path overwrites.
void linear_bwd_dw(torch::Tensor dZ, torch::Tensor X,
                   torch::Tensor gstack, long w_off, long b_off,
                   long M, long I, long O) {
  CHECK_DEV(dZ); CHECK_DEV(X); CHECK_DEV(gstack);
  const long L = gstack.size(0), n = gstack.size(1);
  DISPATCH_FT(dZ, {
    if (M >= 256 && O % 64 == 0 && I % 16 == 0 && I <= 448 && O <= 512) {
      // full-I tiles: dZ and X each fetched from HBM exactly once;
      // direct-global fragment reads (both operands m-row-major), no
      // LDS/barriers — see gemm_mfma.hip mfma_dw_direct_k rationale.
      // Partial tiles go to per-chunk SLABS + a reduce kernel (plain
      // stores; the nchunk-way atomic burst measured ~2.8x roofline);
      // NDTA_DW_ATOMIC=1 selects the atomic epilogue for A/B runs.
      static const bool force_atomic = []() {
        const char* e = getenv("NDTA_DW_ATOMIC");
        return e && e[0] == '1';
      }();
      const long otiles = O / 64;
      long nchunk = std::max<long>(1, 512 / std::max<long>(1, otiles * L));
      nchunk = std::min<long>(nchunk, (M + 63) / 64);
      dim3 grid(1, otiles, L * nchunk);
      torch::Tensor parts;
      scalar_t* pp = nullptr;
      if (!force_atomic) {
        parts = torch::empty({L * nchunk * O * I}, gstack.options());
        pp = parts.data_ptr<scalar_t>();
      }
      if (I <= 64) {
        // fi_per == 1: NIMAX=1 keeps the load loops guard-free
        hipLaunchKernelGGL((gmfma::mfma_dw_direct_k<scalar_t, 1>),
            grid, dim3(512), 0, cur_stream(),
            dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
            gstack.data_ptr<scalar_t>(), pp, n, w_off, b_off,
            (int)M, (int)I, (int)O, (int)nchunk);
      } else if (I <= 256) {
        hipLaunchKernelGGL((gmfma::mfma_dw_direct_k<scalar_t, 4>),
            grid, dim3(512), 0, cur_stream(),
            dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
            gstack.data_ptr<scalar_t>(), pp, n, w_off, b_off,
            (int)M, (int)I, (int)O, (int)nchunk);
      } else {
        hipLaunchKernelGGL((gmfma::mfma_dw_direct_k<scalar_t, 7>),
            grid, dim3(512), 0, cur_stream(),
            dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
            gstack.data_ptr<scalar_t>(), pp, n, w_off, b_off,
            (int)M, (int)I, (int)O, (int)nchunk);
      }
      if (pp != nullptr) {
        const long tile = O * I;
        dim3 rgrid(grid_1d(tile), 1, L);
        hipLaunchKernelGGL(gmfma::dw_reduce_parts_k<scalar_t>,
            rgrid, dim3(256), 0, cur_stream(),
            pp, gstack.data_ptr<scalar_t>(), n, w_off, tile,
            (int)nchunk);
      }
    } else if (M >= 256 && I >= 16 && O >= 16) {
      // fill the chip: tiles * L * nchunk ≈ 2048 blocks
      const long tiles = ((I + 63) / 64) * ((O + 63) / 64);
      long nchunk = std::max<long>(1, 2048 / std::max<long>(1, tiles * L));
      nchunk = std::min<long>(nchunk, (M + 63) / 64);
      dim3 grid((I + 63) / 64, (O + 63) / 64, L * nchunk);
      hipLaunchKernelGGL(gmfma::mfma_dw_k<scalar_t>,
          grid, dim3(256), 0, cur_stream(),
          dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
          gstack.data_ptr<scalar_t>(), n, w_off, b_off,
          (int)M, (int)I, (int)O, (int)nchunk);
    } else if (M > 2048 && I <= 4 && O <= 256) {
      long nchunk = std::max<long>(1, 1024 / std::max<long>(1, L));
      nchunk = std::min<long>(nchunk, (M + 255) / 256);
      if (I == 2 && O == 256) {  // FourierNet encode shape: all
        // bounds compile-time, unconditional loads (trap 4c)
        hipLaunchKernelGGL((gemm::dw_skinny_i_exact_k<scalar_t, 2>),
            dim3(1, 1, L * nchunk), dim3(256), 0, cur_stream(),
            dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
            gstack.data_ptr<scalar_t>(), n, w_off, b_off,
            (int)M, (int)I, (int)O, (int)nchunk);
      } else {
        hipLaunchKernelGGL((gemm::dw_skinny_i_k<scalar_t, 4>),
            dim3(1, 1, L * nchunk), dim3(256), 0, cur_stream(),
            dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
            gstack.data_ptr<scalar_t>(), n, w_off, b_off,
            (int)M, (int)I, (int)O, (int)nchunk);
      }
    } else if (M > 2048 && O <= 4 && I <= 256) {
      long nchunk = std::max<long>(1, 1024 / std::max<long>(1, L));
      nchunk = std::min<long>(nchunk, (M + 255) / 256);
      hipLaunchKernelGGL((gemm::dw_skinny_o_k<scalar_t, 4>),
          dim3(1, 1, L * nchunk), dim3(256), 0, cur_stream(),
          dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
          gstack.data_ptr<scalar_t>(), n, w_off, b_off,
          (int)M, (int)I, (int)O, (int)nchunk);
    } else if (M > 2048) {
      const long total = O * I + O;
      long nchunk = std::max<long>(
          1, 2048 / std::max<long>(1, ((total + 255) / 256) * L));
      nchunk = std::min<long>(nchunk, (M + 255) / 256);
      dim3 grid((total + 255) / 256, 1, L * nchunk);
      hipLaunchKernelGGL(gemm::dw_small_chunked_k<scalar_t>,
          grid, dim3(256), 0, cur_stream(),
          dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
          gstack.data_ptr<scalar_t>(), n, w_off, b_off,
          (int)M, (int)I, (int)O, (int)nchunk);
    } else {
      dim3 grid((I + 15) / 16, (O + 15) / 16, L);
      hipLaunchKernelGGL(gemm::linear_bwd_dw_k<scalar_t>,
          grid, dim3(16, 16), 0, cur_stream(),
          dZ.data_ptr<scalar_t>(), X.data_ptr<scalar_t>(),
          gstack.data_ptr<scalar_t>(), n, w_off, b_off,
          (int)M, (int)I, (int)O);
    }
  });
  HIP_CHECK_LAST();
}

// -------------------------------------------------------------- conv --
void conv_pool_fwd(torch::Tensor X, torch::Tensor theta, torch::Tensor Y,
                   torch::Tensor idx, long w_off, long b_off, long B,
                   long F, long K, long IMG) {
  CHECK_DEV(X); CHECK_DEV(theta); CHECK_DEV(Y);
  const long L = theta.size(0), n = theta.size(1);
  TORCH_CHECK(K <= 7, "conv_pool_fwd supports kernel size <= 7");
  DISPATCH_FT(X, {
    const size_t shmem =
        (IMG * IMG + F * K * K + F) * sizeof(scalar_t);
    if (K <= 5) {
      hipLaunchKernelGGL((conv::conv_pool_fwd_k<scalar_t, 5>),
          dim3(L * B), dim3(256), shmem, cur_stream(),
          X.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          Y.data_ptr<scalar_t>(), idx.data_ptr<unsigned char>(),
          n, w_off, b_off, (int)B, (int)F, (int)K, (int)IMG,
          (const long*)nullptr, 0L, 0L, 0L);
    } else {
      hipLaunchKernelGGL((conv::conv_pool_fwd_k<scalar_t, 7>),
          dim3(L * B), dim3(256), shmem, cur_stream(),
          X.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          Y.data_ptr<scalar_t>(), idx.data_ptr<unsigned char>(),
          n, w_off, b_off, (int)B, (int)F, (int)K, (int)IMG,
          (const long*)nullptr, 0L, 0L, 0L);
    }
  });
  HIP_CHECK_LAST();
}

void conv_pool_bwd(torch::Tensor dY, torch::Tensor idx, torch::Tensor X,
                   torch::Tensor gstack, long w_off, long b_off, long B,
                   long F, long K, long IMG) {
  CHECK_DEV(dY); CHECK_DEV(X); CHECK_DEV(gstack);
  const long L = gstack.size(0), n = gstack.size(1);
  TORCH_CHECK(K <= 7, "conv_pool_bwd supports kernel size <= 7");
  // batch-chunked (l, f, chunk) grid with block-tree reduction +
  // atomics; measured best of three structures on the MNIST bench
  // (LDS-staged image variant was 2x slower — profiles/README.md)
  const int nchunk = (int)std::min<long>(B, 32);
  DISPATCH_FT(dY, {
    // KMAX sizes the per-thread dW register block: the K<=5 variant
    // saves 48 VGPRs over the generic K<=7 one (occupancy)
    if (K <= 5) {
      hipLaunchKernelGGL((conv::conv_pool_bwd_k<scalar_t, 5>),
          dim3(L * F * nchunk), dim3(256), 0, cur_stream(),
          dY.data_ptr<scalar_t>(), idx.data_ptr<unsigned char>(),
          X.data_ptr<scalar_t>(), gstack.data_ptr<scalar_t>(),
          n, w_off, b_off, (int)B, (int)F, (int)K, (int)IMG, nchunk,
          (const long*)nullptr, 0L, 0L, 0L);
    } else {
      hipLaunchKernelGGL((conv::conv_pool_bwd_k<scalar_t, 7>),
          dim3(L * F * nchunk), dim3(256), 0, cur_stream(),
          dY.data_ptr<scalar_t>(), idx.data_ptr<unsigned char>(),
          X.data_ptr<scalar_t>(), gstack.data_ptr<scalar_t>(),
          n, w_off, b_off, (int)B, (int)F, (int)K, (int)IMG, nchunk,
          (const long*)nullptr, 0L, 0L, 0L);
    }
  });
  HIP_CHECK_LAST();
}

// ------------------------------------------------------------ losses --
void mnist_train_step(
    torch::Tensor X_all, torch::Tensor Y_all, torch::Tensor idx,
    c10::optional<torch::Tensor> offs_dev, torch::Tensor theta,
    torch::Tensor grad, c10::optional<torch::Tensor> loss,
    long pit, long idx_off, long idx_stride,
    long wc_off, long bc_off, long w1_off, long b1_off, long w2_off,
    long b2_off, long B, long F, long K, long IMG, long H, long C,
    long TI, double loss_scale) {
  CHECK_DEV(X_all); CHECK_DEV(theta); CHECK_DEV(grad);
  const long L = theta.size(0), n = theta.size(1);
  const long NT = (B + TI - 1) / TI;
  TORCH_CHECK(grad.numel() == L * NT * n,
              "grad must be the [L, NT, n] per-tile slab buffer");
  const long maxlen = Y_all.size(1);
  TORCH_CHECK(bc_off == wc_off + F * K * K,
              "conv weight/bias must be contiguous in the flat layout");
  TORCH_CHECK(K <= 7, "fused mnist step supports kernel size <= 7");
  TORCH_CHECK(TI >= 1 && TI <= 8,
              "fused mnist step register tiling assumes TI <= 8");
  TORCH_CHECK(H <= 64 * TI || TI * H <= 512,
              "fused mnist step per-thread output budget exceeded");
  const long P = (IMG - (K - 1)) / 2;
  const long PF = F * P * P;
  dim3 grid((B + TI - 1) / TI, 1, L);
  DISPATCH_FT(X_all, {
    const size_t shmem =
        (size_t)(TI * (IMG * IMG + PF + 2 * H + C) + F * K * K + F +
                 C * H + C + 16 * 65) *
            sizeof(scalar_t) +
        ((size_t)TI * PF + 15) / 16 * 16 + (size_t)TI * sizeof(long);
    hipLaunchKernelGGL(fmnist::mnist_train_step_k<scalar_t>,
        grid, dim3(256), shmem, cur_stream(),
        X_all.data_ptr<scalar_t>(), Y_all.data_ptr<long>(),
        idx.data_ptr<long>(),
        offs_dev.has_value() ? offs_dev->data_ptr<long>() : nullptr,
        theta.data_ptr<scalar_t>(), grad.data_ptr<scalar_t>(),
        loss.has_value() ? loss->data_ptr<scalar_t>() : nullptr,
        (int)pit, idx_off, idx_stride, maxlen, n,
        wc_off, bc_off, w1_off, b1_off, w2_off, b2_off,
        (int)B, (int)F, (int)K, (int)IMG, (int)H, (int)C, (int)TI,
        (int)NT, (scalar_t)loss_scale);
  });
  HIP_CHECK_LAST();
}

void logsoftmax(torch::Tensor Z, torch::Tensor P, long C) {
  CHECK_DEV(Z); CHECK_DEV(P);
  const long M = Z.numel() / C;
  DISPATCH_FT(Z, {
    hipLaunchKernelGGL(losses::logsoftmax_k<scalar_t>,
        dim3(grid_1d(M)), dim3(256), 0, cur_stream(),
        Z.data_ptr<scalar_t>(), P.data_ptr<scalar_t>(), M, (int)C);
  });
  HIP_CHECK_LAST();
}

void nll_bwd(torch::Tensor logp, torch::Tensor y, torch::Tensor dZ,
             c10::optional<torch::Tensor> loss, long C, long B,
             double loss_scale) {
  CHECK_DEV(logp); CHECK_DEV(dZ);
  const long M = logp.numel() / C;
  DISPATCH_FT(logp, {
    hipLaunchKernelGGL(losses::nll_bwd_k<scalar_t>,
        dim3(grid_1d(M)), dim3(256), 0, cur_stream(),
        logp.data_ptr<scalar_t>(), y.data_ptr<long>(),
        dZ.data_ptr<scalar_t>(),
        loss.has_value() ? loss->data_ptr<scalar_t>() : nullptr,
        M, (int)C, (int)B, (scalar_t)loss_scale);
  });
  HIP_CHECK_LAST();
}

void nll_fused(torch::Tensor Z, torch::Tensor Y_all,
               torch::Tensor idx, torch::Tensor dZ,
               c10::optional<torch::Tensor> loss,
               c10::optional<torch::Tensor> offs_dev, long pit,
               long idx_stride, long idx_off, long C, long B,
               double loss_scale) {
  CHECK_DEV(Z); CHECK_DEV(dZ);
  const long maxlen = Y_all.size(1);
  const long M = Z.numel() / C;
  DISPATCH_FT(Z, {
    hipLaunchKernelGGL(losses::nll_fused_k<scalar_t>,
        dim3(grid_1d(M)), dim3(256), 0, cur_stream(),
        Z.data_ptr<scalar_t>(), Y_all.data_ptr<long>(),
        idx.data_ptr<long>(), dZ.data_ptr<scalar_t>(),
        loss.has_value() ? loss->data_ptr<scalar_t>() : nullptr,
        offs_dev.has_value() ? offs_dev->data_ptr<long>() : nullptr,
        (int)pit, maxlen, idx_stride, idx_off, M, (int)C, (int)B,
        (scalar_t)loss_scale);
  });
  HIP_CHECK_LAST();
}

void bce_bwd(torch::Tensor p, torch::Tensor tgt, torch::Tensor dZ,
             c10::optional<torch::Tensor> loss, long B,
             double loss_scale) {
  CHECK_DEV(p); CHECK_DEV(dZ);
  DISPATCH_FT(p, {
    hipLaunchKernelGGL(losses::bce_bwd_k<scalar_t>,
        dim3(grid_1d(p.numel())), dim3(256), 0, cur_stream(),
        p.data_ptr<scalar_t>(), tgt.data_ptr<scalar_t>(),
        dZ.data_ptr<scalar_t>(),
        loss.has_value() ? loss->data_ptr<scalar_t>() : nullptr,
        p.numel(), (int)B, (scalar_t)loss_scale);
  });
  HIP_CHECK_LAST();
}

void regression_bwd(torch::Tensor yhat, torch::Tensor tgt,
                    torch::Tensor dY, c10::optional<torch::Tensor> loss,
                    long B, double loss_scale, long mode) {
  CHECK_DEV(yhat); CHECK_DEV(dY);
  DISPATCH_FT(yhat, {
    auto lptr =
        loss.has_value() ? loss->data_ptr<scalar_t>() : nullptr;
    if (mode == 0) {
      hipLaunchKernelGGL((losses::regression_bwd_k<scalar_t, 0>),
          dim3(grid_1d(yhat.numel())), dim3(256), 0, cur_stream(),
          yhat.data_ptr<scalar_t>(), tgt.data_ptr<scalar_t>(),
          dY.data_ptr<scalar_t>(), lptr, yhat.numel(), (int)B,
          (scalar_t)loss_scale);
    } else {
      hipLaunchKernelGGL((losses::regression_bwd_k<scalar_t, 1>),
          dim3(grid_1d(yhat.numel())), dim3(256), 0, cur_stream(),
          yhat.data_ptr<scalar_t>(), tgt.data_ptr<scalar_t>(),
          dY.data_ptr<scalar_t>(), lptr, yhat.numel(), (int)B,
          (scalar_t)loss_scale);
    }
  });
  HIP_CHECK_LAST();
}


// gather-fused conv wrappers: image rows read straight from the
// resident dataset via the sampler's index stream (no gather_batch
// launch, no xb buffer round-trip)
void conv_pool_fwd_idx(torch::Tensor X_all, torch::Tensor src_idx,
                       long idx_stride, long idx_off,
                       torch::Tensor theta, torch::Tensor Y,
                       torch::Tensor idx, long w_off, long b_off,
                       long B, long F, long K, long IMG) {
  CHECK_DEV(X_all); CHECK_DEV(theta); CHECK_DEV(Y);
  const long L = theta.size(0), n = theta.size(1);
  const long maxlen = X_all.size(1);
  TORCH_CHECK(K <= 7, "conv_pool_fwd supports kernel size <= 7");
  DISPATCH_FT(X_all, {
    const size_t shmem =
        (IMG * IMG + F * K * K + F) * sizeof(scalar_t);
    if (K <= 5) {
      hipLaunchKernelGGL((conv::conv_pool_fwd_k<scalar_t, 5>),
          dim3(L * B), dim3(256), shmem, cur_stream(),
          X_all.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          Y.data_ptr<scalar_t>(), idx.data_ptr<unsigned char>(),
          n, w_off, b_off, (int)B, (int)F, (int)K, (int)IMG,
          src_idx.data_ptr<long>(), idx_stride, idx_off, maxlen);
    } else {
      hipLaunchKernelGGL((conv::conv_pool_fwd_k<scalar_t, 7>),
          dim3(L * B), dim3(256), shmem, cur_stream(),
          X_all.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
          Y.data_ptr<scalar_t>(), idx.data_ptr<unsigned char>(),
          n, w_off, b_off, (int)B, (int)F, (int)K, (int)IMG,
          src_idx.data_ptr<long>(), idx_stride, idx_off, maxlen);
    }
  });
  HIP_CHECK_LAST();
}

void conv_pool_bwd_idx(torch::Tensor dY, torch::Tensor idx,
                       torch::Tensor X_all, torch::Tensor src_idx,
                       long idx_stride, long idx_off,
                       torch::Tensor gstack, long w_off, long b_off,
                       long B, long F, long K, long IMG) {
  CHECK_DEV(dY); CHECK_DEV(X_all); CHECK_DEV(gstack);
  const long L = gstack.size(0), n = gstack.size(1);
  const long maxlen = X_all.size(1);
  TORCH_CHECK(K <= 7, "conv_pool_bwd supports kernel size <= 7");
  const int nchunk = (int)std::min<long>(B, 32);
  DISPATCH_FT(dY, {
    if (K <= 5) {
      hipLaunchKernelGGL((conv::conv_pool_bwd_k<scalar_t, 5>),
          dim3(L * F * nchunk), dim3(256), 0, cur_stream(),
          dY.data_ptr<scalar_t>(), idx.data_ptr<unsigned char>(),
          X_all.data_ptr<scalar_t>(), gstack.data_ptr<scalar_t>(),
          n, w_off, b_off, (int)B, (int)F, (int)K, (int)IMG, nchunk,
          src_idx.data_ptr<long>(), idx_stride, idx_off, maxlen);
    } else {
      hipLaunchKernelGGL((conv::conv_pool_bwd_k<scalar_t, 7>),
          dim3(L * F * nchunk), dim3(256), 0, cur_stream(),
          dY.data_ptr<scalar_t>(), idx.data_ptr<unsigned char>(),
          X_all.data_ptr<scalar_t>(), gstack.data_ptr<scalar_t>(),
          n, w_off, b_off, (int)B, (int)F, (int)K, (int)IMG, nchunk,
          src_idx.data_ptr<long>(), idx_stride, idx_off, maxlen);
    }
  });
  HIP_CHECK_LAST();
}

// one-launch fc block: fc1+fc2 fwd, NLL, full fc backward (see
// fused_mnist.hip fc_block_k). Caller pre-zeroes the grad stack.
void fc_block(torch::Tensor x0, torch::Tensor theta,
              torch::Tensor Y_all, torch::Tensor idx, long idx_stride,
              long idx_off, torch::Tensor grad, torch::Tensor dx0,
              torch::Tensor dz1g,
              c10::optional<torch::Tensor> loss, long w1_off,
              long b1_off, long w2_off, long b2_off, long M, long I,
              long H, long C, double loss_scale, bool mask_dx0) {
  CHECK_DEV(x0); CHECK_DEV(theta); CHECK_DEV(grad); CHECK_DEV(dx0);
  CHECK_DEV(dz1g);
  const long L = theta.size(0), n = theta.size(1);
  const long maxlen = Y_all.size(1);
  TORCH_CHECK(H <= 64 && C <= 16, "fc_block: H <= 64, C <= 16");
  const long mtiles = (M + fmnist::FC_RT - 1) / fmnist::FC_RT;
  DISPATCH_FT(x0, {
    hipLaunchKernelGGL(fmnist::fc_block_k<scalar_t>,
        dim3(mtiles, 1, L), dim3(512), 0, cur_stream(),
        x0.data_ptr<scalar_t>(), theta.data_ptr<scalar_t>(),
        Y_all.data_ptr<long>(), idx.data_ptr<long>(), idx_stride,
        idx_off, maxlen, grad.data_ptr<scalar_t>(),
        dx0.data_ptr<scalar_t>(), dz1g.data_ptr<scalar_t>(),
        loss.has_value() ? loss->data_ptr<scalar_t>() : nullptr,
        n, w1_off, b1_off, w2_off, b2_off, (int)M, (int)I, (int)H,
        (int)C, (scalar_t)loss_scale, mask_dx0 ? 1 : 0);
  });
  HIP_CHECK_LAST();
  // fc1 dW/db from the exported dz1 (store-path kernel: overwrite,
  // no atomics)
  linear_bwd_dw(dz1g, x0, grad, w1_off, b1_off, M, I, H);
}

// consensus-error metric: normalized pairwise distances + distance to
// the normalized mean (reference dist_mnist_problem.py:152-175) on the
// all-gathered [N, n] stack — returns (D [N,N], Dm [N,1])
std::vector<torch::Tensor> consensus_cdist(torch::Tensor stack) {
  CHECK_DEV(stack);
  const long N = stack.size(0), n = stack.size(1);
  auto opts = stack.options();
  auto norms = torch::empty({N}, opts);
  auto D = torch::empty({N, N}, opts);
  auto mean = torch::empty({n}, opts);
  auto Dm = torch::empty({N, 1}, opts);
  DISPATCH_FT(stack, {
    hipLaunchKernelGGL(ew::row_norms_k<scalar_t>,
        dim3(N), dim3(ew::BLOCK), 0, cur_stream(),
        stack.data_ptr<scalar_t>(), norms.data_ptr<scalar_t>(), n);
    hipLaunchKernelGGL(ew::pairwise_normed_dist_k<scalar_t>,
        dim3(N * N), dim3(ew::BLOCK), 0, cur_stream(),
        stack.data_ptr<scalar_t>(), norms.data_ptr<scalar_t>(),
        D.data_ptr<scalar_t>(), N, n);
    hipLaunchKernelGGL(ew::normed_mean_k<scalar_t>,
        dim3(grid_1d(n)), dim3(ew::BLOCK), 0, cur_stream(),
        stack.data_ptr<scalar_t>(), norms.data_ptr<scalar_t>(),
        mean.data_ptr<scalar_t>(), N, n);
    hipLaunchKernelGGL(ew::dist_to_mean_k<scalar_t>,
        dim3(N), dim3(ew::BLOCK), 0, cur_stream(),
        stack.data_ptr<scalar_t>(), norms.data_ptr<scalar_t>(),
        mean.data_ptr<scalar_t>(), Dm.data_ptr<scalar_t>(), n);
  });
  HIP_CHECK_LAST();
  return {D, Dm};
}

// ----------------------------------------------------------- chains --
// One pybind call per forward / backward pass instead of one per
// layer-op: host profiling (BENCH r2b timing_breakdown) measured
// ~10 us of python+pybind overhead PER ext call, making the host the
// bottleneck of the MNIST round (~23 calls x 10 us vs ~115 us of GPU
// work). These chains launch exactly the kernels the python loops in
// ops/stacked.py forward()/backward() launched, in the same order.
//
// spec: CPU int64 [nl, 7] rows = (kind, w_off, b_off, in_dim,
//       out_dim, act, ksize); kind 0 = linear, 1 = conv_pool;
//       act = ACT_* id (5 = logsoftmax -> linear 'none' + row kernel).
// scales: CPU float64 [nl] activation scales.
void fwd_chain(torch::Tensor spec, torch::Tensor scales,
               c10::optional<torch::Tensor> X_all,
               c10::optional<torch::Tensor> idx, long idx_stride,
               long idx_off,
               torch::Tensor xb, torch::Tensor theta,
               std::vector<torch::Tensor> acts,
               std::vector<c10::optional<torch::Tensor>> zs,
               std::vector<c10::optional<torch::Tensor>> idxs,
               c10::optional<torch::Tensor> logp, long M,
               bool train_skip_logp) {
  TORCH_CHECK(spec.device().is_cpu() && scales.device().is_cpu(),
              "spec/scales must be CPU tensors");
  const auto sp = spec.accessor<long, 2>();
  const auto sc = scales.accessor<double, 1>();
  const long nl = spec.size(0);
  if (X_all.has_value()) {
    gather_batch(*X_all, *idx, xb, idx_stride, idx_off);
  }
  torch::Tensor cur = xb;
  for (long li = 0; li < nl; ++li) {
    const long kind = sp[li][0], w_off = sp[li][1], b_off = sp[li][2];
    const long in_dim = sp[li][3], out_dim = sp[li][4];
    const long act = sp[li][5], ksize = sp[li][6];
    torch::Tensor out = acts[li];
    if (kind == 1) {
      conv_pool_fwd(cur, theta, out, *idxs[li], w_off, b_off, M,
                    out_dim, ksize, in_dim);
      cur = out;
    } else if (act == ACT_LOGSOFTMAX) {
      linear_fwd(cur, theta, out, c10::nullopt, w_off, b_off, M,
                 in_dim, out_dim, ACT_NONE, 1.0);
      if (!train_skip_logp) {
        logsoftmax(out, *logp, out_dim);
        cur = *logp;
      } else {
        cur = out;
      }
    } else {
      linear_fwd(cur, theta, out, zs[li], w_off, b_off, M, in_dim,
                 out_dim, act, sc[li]);
      cur = out;
    }
  }
}

// Backward chain: loss head + layer loop, mirroring
// ops/stacked.py StackedEngine.backward.
// loss_kind: 0 = fused NLL (classification), 1 = BCE+sigmoid,
//            2 = MSE, 3 = L1.
void bwd_chain(torch::Tensor spec, torch::Tensor scales,
               torch::Tensor xb, torch::Tensor theta,
               torch::Tensor grad,
               std::vector<torch::Tensor> acts,
               std::vector<c10::optional<torch::Tensor>> zs,
               std::vector<torch::Tensor> dzs,
               std::vector<c10::optional<torch::Tensor>> idxs,
               long loss_kind,
               c10::optional<torch::Tensor> Y_all,
               c10::optional<torch::Tensor> idx,
               c10::optional<torch::Tensor> graph_offs, long pit,
               long idx_stride, long idx_off,
               c10::optional<torch::Tensor> yb,
               c10::optional<torch::Tensor> loss, double loss_scale,
               long M, long zero_mode) {
  TORCH_CHECK(spec.device().is_cpu() && scales.device().is_cpu(),
              "spec/scales must be CPU tensors");
  const auto sp = spec.accessor<long, 2>();
  const auto sc = scales.accessor<double, 1>();
  const long nl = spec.size(0);
  if (zero_mode == 1) {
    grad.zero_();
  } else if (zero_mode == 2) {
    // conv slices only (kind == 1 rows)
    for (long li = 0; li < nl; ++li) {
      if (sp[li][0] == 1) {
        const long cnt =
            sp[li][4] * sp[li][6] * sp[li][6] + sp[li][4];
        grad.index({torch::indexing::Slice(),
                    torch::indexing::Slice(sp[li][1],
                                           sp[li][1] + cnt)})
            .zero_();
      }
    }
  }
  if (loss.has_value()) loss->zero_();

  const long last = nl - 1;
  torch::Tensor dz = dzs[last];
  if (loss_kind == 0) {
    nll_fused(acts[last], *Y_all, *idx, dz, loss, graph_offs, pit,
              idx_stride, idx_off, sp[last][4], M, loss_scale);
  } else if (loss_kind == 1) {
    bce_bwd(acts[last], *yb, dz, loss, M, loss_scale);
  } else {
    auto dy = torch::empty_like(dz);
    regression_bwd(acts[last], *yb, dy, loss, M, loss_scale,
                   loss_kind == 2 ? 0 : 1);
    act_grad(dy, acts[last], c10::nullopt, dz, sp[last][5], sc[last]);
  }

  for (long li = last; li >= 0; --li) {
    const long kind = sp[li][0], w_off = sp[li][1], b_off = sp[li][2];
    const long in_dim = sp[li][3], out_dim = sp[li][4];
    torch::Tensor below = li > 0 ? acts[li - 1] : xb;
    dz = dzs[li];
    if (kind == 1) {
      conv_pool_bwd(dz, *idxs[li], below, grad, w_off, b_off, M,
                    out_dim, sp[li][6], in_dim);
      continue;  // conv is the first layer: no dX
    }
    linear_bwd_dw(dz, below, grad, w_off, b_off, M, in_dim, out_dim);
    if (li > 0) {
      const long act_b_raw = sp[li - 1][5];
      const long act_b =
          (act_b_raw == ACT_NONE || act_b_raw == ACT_LOGSOFTMAX)
              ? 0 : act_b_raw;
      c10::optional<torch::Tensor> xb2;
      long wb_off = 0, bb_off = 0, ib = 0;
      if (act_b == ACT_SIN_RELU && !zs[li - 1].has_value()) {
        xb2 = li - 1 > 0 ? acts[li - 2] : xb;
        wb_off = sp[li - 1][1];
        bb_off = sp[li - 1][2];
        ib = sp[li - 1][3];
      }
      linear_bwd_dx(
          dz, theta, dzs[li - 1],
          act_b ? c10::optional<torch::Tensor>(acts[li - 1])
                : c10::nullopt,
          zs[li - 1], act_b, sc[li - 1], w_off, M, in_dim, out_dim,
          xb2, wb_off, bb_off, ib);
    }
  }
}

// keyed bijection of [0, n) written into `out` (int64, contiguous):
// the online-density sampler's shuffle (ops/stacked.py
// _OnlineWindowSampler) — one launch instead of a device randperm's
// rocprim sort chain
void feistel_perm(torch::Tensor out, long n, long lb, long key) {
  CHECK_DEV(out);
  TORCH_CHECK(out.scalar_type() == torch::kLong,
              "feistel_perm: out must be int64");
  TORCH_CHECK(out.numel() >= n, "feistel_perm: out too small");
  int bits = 1;
  while ((1L << bits) < n) ++bits;
  const int hb = std::max(1, (bits + 1) / 2);
  hipLaunchKernelGGL(ew::feistel_perm_k,
      dim3(grid_1d(n)), dim3(ew::BLOCK), 0, cur_stream(),
      out.data_ptr<long>(), n, lb, (unsigned long long)key, hb);
  HIP_CHECK_LAST();
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("dinno_dual_threg", &dinno_dual_threg);
  mod.def("mix_rows", &mix_rows);
  mod.def("dsgt_mix", &dsgt_mix);
  mod.def("dsgt_y_update", &dsgt_y_update);
  mod.def("fused_step", &fused_step);
  mod.def("axpy", &axpy);
  mod.def("reduce_parts", &reduce_parts);
  mod.def("gather_batch", &gather_batch);
  mod.def("gather_batch_dev", &gather_batch_dev);
  mod.def("gather_targets_dev", &gather_targets_dev);
  mod.def("dinno_dual_threg_sched", &dinno_dual_threg_sched);
  mod.def("fused_step_sched", &fused_step_sched);
  mod.def("gather_targets", &gather_targets);
  mod.def("linear_fwd", &linear_fwd);
  mod.def("act_grad", &act_grad);
  mod.def("linear_bwd_dx", &linear_bwd_dx);
  mod.def("linear_bwd_dw", &linear_bwd_dw);
  mod.def("conv_pool_fwd", &conv_pool_fwd);
  mod.def("conv_pool_bwd", &conv_pool_bwd);
  mod.def("conv_pool_fwd_idx", &conv_pool_fwd_idx);
  mod.def("conv_pool_bwd_idx", &conv_pool_bwd_idx);
  mod.def("logsoftmax", &logsoftmax);
  mod.def("nll_bwd", &nll_bwd);
  mod.def("nll_fused", &nll_fused);
  mod.def("mnist_train_step", &mnist_train_step);
  mod.def("bce_bwd", &bce_bwd);
  mod.def("regression_bwd", &regression_bwd);
  mod.def("feistel_perm", &feistel_perm);
  mod.def("consensus_cdist", &consensus_cdist);
  mod.def("fc_block", &fc_block);
  mod.def("fwd_chain", &fwd_chain);
  mod.def("bwd_chain", &bwd_chain);
}
