// MNISTConvNet head: Conv2d(1, F, k) + ReLU + MaxPool2d(2), fused
// forward and backward, batched over node replicas.
//
// The conv is the first layer, so backward only needs dW/db (no dX) —
// the whole layer costs two kernels per direction. Images are staged in
// LDS (28*28 doubles = 6.3 KB) so each of the k*k taps reads on-chip.

#include "common.h"

namespace conv {

// X: [L*B, 28*28]; theta: [L, n] flat stack, W at w_off ([F,1,k,k]
// row-major = [F, k*k]), b at b_off; Y: [L*B, F*P*P] pooled+ReLU output;
// idx: [L*B, F*P*P] argmax position (0..3) for pool backward.
// Grid: one block per image, 256 threads. KMAX bounds the tap loops at
// compile time (K<=KMAX checked by the caller) so they fully unroll.
// src_idx != null: the image rows GATHER straight from the resident
// dataset (X = X_all [L, maxlen, IMG*IMG], row = src_idx[l*idx_stride
// + idx_off + b]) — the separate gather_batch launch and the xb
// buffer round-trip disappear from the fused-fc train path.
template <typename T, int KMAX>
__global__ void conv_pool_fwd_k(
    const T* __restrict__ X, const T* __restrict__ theta,
    T* __restrict__ Y, unsigned char* __restrict__ idx,
    long n, long w_off, long b_off, int B, int F, int K, int IMG,
    const long* __restrict__ src_idx, long idx_stride, long idx_off,
    long maxlen) {
  extern __shared__ __align__(16) unsigned char smem_raw[];
  T* img = reinterpret_cast<T*>(smem_raw);             // [IMG*IMG]
  T* wgt = img + IMG * IMG;                            // [F*K*K + F]

  const long lb = blockIdx.x;           // image index in [0, L*B)
  const long l = lb / B;
  const int conv_out = IMG - (K - 1);
  const int P = conv_out / 2;

  // stage image + this node's conv weights/bias
  const long src_row =
      src_idx ? l * maxlen + src_idx[l * idx_stride + idx_off + lb % B]
              : lb;
  for (int t = threadIdx.x; t < IMG * IMG; t += blockDim.x) {
    img[t] = X[src_row * IMG * IMG + t];
  }
  const T* Wg = theta + l * n + w_off;
  const T* bg = theta + l * n + b_off;
  for (int t = threadIdx.x; t < F * K * K; t += blockDim.x) {
    wgt[t] = Wg[t];
  }
  for (int t = threadIdx.x; t < F; t += blockDim.x) {
    wgt[F * K * K + t] = bg[t];
  }
  __syncthreads();

  const int npool = F * P * P;
  for (int t = threadIdx.x; t < npool; t += blockDim.x) {
    const int f = t / (P * P);
    const int py = (t / P) % P;
    const int px = t % P;
    const T* wf = wgt + f * K * K;
    const T bias = wgt[F * K * K + f];

    T best = T(0);       // ReLU floor: max(0, .) pooled
    int best_i = 0;
    #pragma unroll
    for (int d = 0; d < 4; ++d) {
      const int cy = 2 * py + (d >> 1);
      const int cx = 2 * px + (d & 1);
      T acc = bias;
#pragma unroll
      for (int ky = 0; ky < KMAX; ++ky) {
        if (ky >= K) break;
        const T* row = img + (cy + ky) * IMG + cx;
        const T* wr = wf + ky * K;
#pragma unroll
        for (int kx = 0; kx < KMAX; ++kx) {
          if (kx >= K) break;
          acc += row[kx] * wr[kx];
        }
      }
      if (acc > best) { best = acc; best_i = d; }
    }
    Y[lb * npool + t] = best;          // relu(max pre-act) == max(relu)
    idx[lb * npool + t] = (unsigned char)best_i;
  }
}

// Backward to weights/bias: route each pooled grad to its argmax conv
// position, multiply by the image window. dY is the grad AFTER the relu
// mask (pooled output > 0), applied by the caller via act_grad.
//
// Grid: (l, f, batch-chunk) so the chip fills (the v1 one-block-per-
// (l,f) version was 24 blocks on 256 CUs and 41% of round time);
// each thread accumulates a private dW[K*K]+db over its strided share
// of the chunk, waves shuffle-reduce, wave leaders combine in LDS and
// lane 0 atomically adds into the grad stack (the caller zeroes the
// conv slice first).
template <typename T, int KMAX>
__global__ void conv_pool_bwd_k(
    const T* __restrict__ dY, const unsigned char* __restrict__ idx,
    const T* __restrict__ X, T* __restrict__ gstack,
    long n, long w_off, long b_off, int B, int F, int K, int IMG,
    int nchunk,
    const long* __restrict__ src_idx, long idx_stride, long idx_off,
    long maxlen) {
  const int chunk = blockIdx.x % nchunk;
  const int f = (blockIdx.x / nchunk) % F;
  const int l = blockIdx.x / (nchunk * F);
  const int conv_out = IMG - (K - 1);
  const int P = conv_out / 2;
  const int npool = F * P * P;
  const int cb = (B + nchunk - 1) / nchunk;       // images per chunk
  const int b0 = chunk * cb;
  const int b1 = min(B, b0 + cb);

  T dw[KMAX * KMAX];
  T db = T(0);
  #pragma unroll
  for (int i = 0; i < KMAX * KMAX; ++i) dw[i] = T(0);

  const int work = (b1 - b0) * P * P;
  if (K == KMAX) {
    // exact-K fast path: no per-element zero-skip branch and no
    // runtime breaks inside the unrolled GLOBAL-load chain — both
    // forced per-element branch + vmcnt waits around the 25 img loads
    // (trap 4c); unconditional taps pipeline (g = 0 contributes 0).
    // The (g, pool-argmax) pair for iteration t+1 prefetches while
    // iteration t's taps run — idx[o] -> img-address is a 2-hop
    // dependency that otherwise heads every iteration (loads use a
    // CLAMPED index so the prefetch needs no bounds branch).
    auto hdr = [&](long t, T& g, int& d, long& b) {
      const long tc = t < work ? t : (work > 0 ? work - 1 : 0);
      b = b0 + tc / (P * P);
      const long o = ((long)l * B + b) * npool + f * P * P
                     + ((tc / P) % P) * P + tc % P;
      g = dY[o];
      d = idx[o];
    };
    long t = threadIdx.x;
    T g = T(0);
    int d = 0;
    long b = 0;
    if (work > 0) hdr(t, g, d, b);
    for (; t < work; t += blockDim.x) {
      T gn;
      int dn;
      long bn;
      hdr(t + blockDim.x, gn, dn, bn);
      const int py = (int)((t / P) % P);
      const int px = (int)(t % P);
      const int cy = 2 * py + (d >> 1);
      const int cx = 2 * px + (d & 1);
      const long src_row =
          src_idx ? l * maxlen + src_idx[l * idx_stride + idx_off + b]
                  : (long)l * B + b;
      const T* img = X + src_row * IMG * IMG;
      db += g;
#pragma unroll
      for (int ky = 0; ky < KMAX; ++ky) {
        const T* row = img + (cy + ky) * IMG + cx;
#pragma unroll
        for (int kx = 0; kx < KMAX; ++kx) {
          dw[ky * KMAX + kx] += g * row[kx];
        }
      }
      g = gn;
      d = dn;
      b = bn;
    }
  } else {
    for (int t = threadIdx.x; t < work; t += blockDim.x) {
      const int b = b0 + t / (P * P);
      const int py = (t / P) % P;
      const int px = t % P;
      const long lb = (long)l * B + b;
      const long o = lb * npool + f * P * P + py * P + px;
      const T g = dY[o];
      if (g == T(0)) continue;
      const int d = idx[o];
      const int cy = 2 * py + (d >> 1);
      const int cx = 2 * px + (d & 1);
      const long src_row =
          src_idx ? l * maxlen + src_idx[l * idx_stride + idx_off + b]
                  : lb;
      const T* img = X + src_row * IMG * IMG;
      db += g;
      // compile-time bounds + KMAX-strided indices: a runtime
      // `ky*K+kx` subscript makes dw[] dynamically indexed and the
      // compiler spills the whole accumulator to scratch (rule 20)
#pragma unroll
      for (int ky = 0; ky < KMAX; ++ky) {
        if (ky >= K) break;
        const T* row = img + (cy + ky) * IMG + cx;
#pragma unroll
        for (int kx = 0; kx < KMAX; ++kx) {
          if (kx >= K) break;
          dw[ky * KMAX + kx] += g * row[kx];
        }
      }
    }
  }

  // reduce the K*K+1 partials: shuffle-reduce every tap within its
  // wave (no barriers), park the per-wave sums in LDS, then ONE
  // barrier and the first KMAX*KMAX+1 threads finish their tap in
  // parallel. The previous per-tap LDS tree spent 2 barriers per tap
  // (52 for K=5) and dominated the kernel.
  __shared__ T red[(KMAX * KMAX + 1) * 4];  // [tap][wave]
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  T* wslice = gstack + (long)l * n + w_off + (long)f * K * K;
#pragma unroll
  for (int i = 0; i < KMAX * KMAX + 1; ++i) {
    const int ky = i / KMAX, kx = i % KMAX;
    const bool is_db = (i == KMAX * KMAX);
    if (!is_db && (ky >= K || kx >= K)) continue;  // uniform
    const T v = wave_reduce_sum(is_db ? db : dw[i]);
    if (lane == 0) red[i * 4 + wid] = v;
  }
  __syncthreads();
  for (int i = threadIdx.x; i < KMAX * KMAX + 1; i += blockDim.x) {
    const int ky = i / KMAX, kx = i % KMAX;
    const bool is_db = (i == KMAX * KMAX);
    if (!is_db && (ky >= K || kx >= K)) continue;
    const T tot =
        red[i * 4] + red[i * 4 + 1] + red[i * 4 + 2] + red[i * 4 + 3];
    if (is_db) {
      atomicAdd(&gstack[(long)l * n + b_off + f], tot);
    } else {
      atomicAdd(&wslice[ky * K + kx], tot);
    }
  }
}

}  // namespace conv
