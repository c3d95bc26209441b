// Fused loss kernels. Each produces the gradient w.r.t. the final
// layer's PRE-activation in one pass (log-softmax+NLL and sigmoid+BCE
// use their combined analytic forms), plus an optional per-node loss
// value (for the train-loss EMA metric) — no autograd, no separate
// softmax/log/clamp launches. Loss semantics follow the reference's
// problem layer: NLLLoss on log-softmax outputs (reference
// problems/dist_mnist_problem.py:65-98), BCELoss on sigmoid outputs
// and MSE/L1 regression (dist_dense_problem.py / experiments'
// make_loss choices).

#include "common.h"

namespace losses {

// log-softmax over rows: logp[m, c] = z[m, c] - logsumexp(z[m, :]).
// C is small (10 for MNIST); one thread per row.
template <typename T>
__global__ void logsoftmax_k(const T* __restrict__ Z, T* __restrict__ P,
                             long M, int C) {
  for (long m = blockIdx.x * 256L + threadIdx.x; m < M;
       m += (long)gridDim.x * 256L) {
    const T* z = Z + m * C;
    T* p = P + m * C;
    T mx = z[0];
    for (int c = 1; c < C; ++c) mx = z[c] > mx ? z[c] : mx;
    T s = T(0);
    for (int c = 0; c < C; ++c) s += ::exp(z[c] - mx);
    const T lse = mx + ::log(s);
    for (int c = 0; c < C; ++c) p[c] = z[c] - lse;
  }
}

// NLL backward through log-softmax: dZ = (exp(logp) - onehot(y)) * w,
// w = loss_scale / B (mean reduction per node, loss_scale folds the
// DiNNO pred-loss coefficient). loss[l] accumulates -mean logp[y].
template <typename T>
__global__ void nll_bwd_k(
    const T* __restrict__ logp, const long* __restrict__ y,
    T* __restrict__ dZ, T* __restrict__ loss,  // loss may be null, [L]
    long M, int C, int B, T loss_scale) {
  const T w = loss_scale / T(B);
  for (long m = blockIdx.x * 256L + threadIdx.x; m < M;
       m += (long)gridDim.x * 256L) {
    const T* p = logp + m * C;
    T* g = dZ + m * C;
    const int t = (int)y[m];
    for (int c = 0; c < C; ++c) {
      g[c] = (::exp(p[c]) - (c == t ? T(1) : T(0))) * w;
    }
    if (loss != nullptr) {
      atomicAdd(&loss[m / B], -p[t] / T(B));
    }
  }
}

// sigmoid+BCE combined backward (final density layer, O == 1):
// p = sigmoid(z) is the layer's stored output; dZ = (p - t)/B directly
// (the classic cancellation), loss[l] += BCE/B with clamped logs.
template <typename T>
__global__ void bce_bwd_k(
    const T* __restrict__ p, const T* __restrict__ tgt,
    T* __restrict__ dZ, T* __restrict__ loss, long M, int B,
    T loss_scale) {
  const T w = loss_scale / T(B);
  const T eps = T(1e-12);
  for (long m = blockIdx.x * 256L + threadIdx.x; m < M;
       m += (long)gridDim.x * 256L) {
    const T pm = p[m], t = tgt[m];
    dZ[m] = (pm - t) * w;
    if (loss != nullptr) {
      const T lp = ::log(pm > eps ? pm : eps);
      const T lq = ::log((T(1) - pm) > eps ? (T(1) - pm) : eps);
      atomicAdd(&loss[m / B], -(t * lp + (T(1) - t) * lq) / T(B));
    }
  }
}

// MSE / L1 backward w.r.t. the final ACTIVATION output (caller chains
// act_grad for the sigmoid). mode 0 = MSE, 1 = L1.
template <typename T, int MODE>
__global__ void regression_bwd_k(
    const T* __restrict__ yhat, const T* __restrict__ tgt,
    T* __restrict__ dY, T* __restrict__ loss, long M, int B,
    T loss_scale) {
  const T w = loss_scale / T(B);
  for (long m = blockIdx.x * 256L + threadIdx.x; m < M;
       m += (long)gridDim.x * 256L) {
    const T d = yhat[m] - tgt[m];
    if (MODE == 0) {
      dY[m] = T(2) * d * w;
      if (loss != nullptr) atomicAdd(&loss[m / B], d * d / T(B));
    } else {
      dY[m] = (d > T(0) ? w : (d < T(0) ? -w : T(0)));
      if (loss != nullptr) {
        atomicAdd(&loss[m / B], (d > T(0) ? d : -d) / T(B));
      }
    }
  }
}

// Fully fused classification backward: from the LOGITS, compute the
// row log-sum-exp, gather the target label from the resident dataset
// (via the index stream + offset — no separate target-gather kernel,
// no materialized log-probs in the training path), and write
// dZ = (softmax - onehot) * loss_scale / B. Offset comes from either
// the literal idx_off or slot `pit` of offs_dev (hipGraph mode).
template <typename T>
__global__ void nll_fused_k(
    const T* __restrict__ Z, const long* __restrict__ Y_all,
    const long* __restrict__ idx, T* __restrict__ dZ,
    T* __restrict__ loss,  // may be null, [L]
    const long* __restrict__ offs_dev, int pit,
    long maxlen, long idx_stride, long idx_off,
    long M, int C, int B, T loss_scale) {
  const T w = loss_scale / T(B);
  const long off = offs_dev ? offs_dev[pit] : idx_off;
  for (long m = blockIdx.x * 256L + threadIdx.x; m < M;
       m += (long)gridDim.x * 256L) {
    const long l = m / B;
    const long b = m - l * B;
    const long src = idx[l * idx_stride + off + b];
    const int t = (int)Y_all[l * maxlen + src];

    const T* z = Z + m * C;
    T* g = dZ + m * C;
    T mx = z[0];
    for (int c = 1; c < C; ++c) mx = z[c] > mx ? z[c] : mx;
    T sum = T(0);
    for (int c = 0; c < C; ++c) sum += ::exp(z[c] - mx);
    const T lse = mx + ::log(sum);
    for (int c = 0; c < C; ++c) {
      g[c] = (::exp(z[c] - lse) - (c == t ? T(1) : T(0))) * w;
    }
    if (loss != nullptr) {
      atomicAdd(&loss[l], -(z[t] - lse) / T(B));
    }
  }
}

}  // namespace losses
