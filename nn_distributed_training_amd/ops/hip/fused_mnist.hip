// Fully fused MNIST train step (CDNA4).
//
// One kernel performs, per tile of T images of one node replica:
//   gather (index stream) -> conv+ReLU+maxpool -> fc1+ReLU ->
//   fc2 logits -> log-sum-exp + NLL dZ (targets read from the resident
//   label array) -> fc2 dW/db -> dh1 -> fc1 dW/db -> dpool ->
//   conv dW/db
// with every intermediate living in LDS. The layered engine ran this
// as ~12 launches per primal iteration with every activation making an
// HBM round trip; at n=28,440 the work per launch is tiny and the round
// is kernel-boundary-bound (profiles/README.md), so fusing the chain is
// the lever the per-kernel optimizations could not reach.
//
// Gradients accumulate ATOMICALLY into the [L, n] grad stack (caller
// zeroes it); blocks are independent (one per (l, image-tile)), so no
// inter-workgroup ordering is assumed anywhere. fc1's 27,648 weights
// stay in L2 (221 KB fp64 — larger than LDS); each block streams them
// for fc1-forward and once more for dpool.
//
// LDS budget at T=8, fp64: img 50.2KB + pool 27.6KB + h1/dh1 8KB +
// W2 5.2KB + W1 tile 8.3KB + small ~ 104KB -> 1 block/CU; T is a
// launch parameter (dynamic LDS).

#include "common.h"

namespace fmnist {

template <typename T>
__global__ __launch_bounds__(256) void mnist_train_step_k(
    const T* __restrict__ X_all,      // [L, maxlen, IMG*IMG]
    const long* __restrict__ Y_all,   // [L, maxlen]
    const long* __restrict__ idx,     // [L, S] index stream
    const long* __restrict__ offs_dev,  // nullable; slot `pit`
    const T* __restrict__ theta,      // [L, n]
    T* __restrict__ gparts,           // [L, NT, n] per-tile slabs
    T* __restrict__ loss,             // nullable [L]
    int pit, long idx_off, long idx_stride, long maxlen, long n,
    long wc_off, long bc_off, long w1_off, long b1_off, long w2_off,
    long b2_off,
    int B, int F, int K, int IMG, int H, int C, int TI, int NT,
    T loss_scale) {
  extern __shared__ __align__(16) unsigned char smem_raw[];
  const int conv_out = IMG - (K - 1);
  const int P = conv_out / 2;
  const int PF = F * P * P;          // pooled features (432)

  T* img = reinterpret_cast<T*>(smem_raw);        // [TI][IMG*IMG]
  T* pool = img + (long)TI * IMG * IMG;           // [TI][PF]
  T* h1 = pool + (long)TI * PF;                   // [TI][H]
  T* dh1 = h1 + (long)TI * H;                     // [TI][H]
  T* dz2 = dh1 + (long)TI * H;                    // [TI][C]
  T* wconv = dz2 + (long)TI * C;                  // [F*K*K + F]
  T* w2s = wconv + F * K * K + F;                 // [C*H + C]
  T* wt = w2s + (long)C * H + C;                  // [16][65] W1 tiles
  unsigned char* pidx =
      reinterpret_cast<unsigned char*>(wt + 16 * 65);
  long* src = reinterpret_cast<long*>(
      pidx + ((long)TI * PF + 15) / 16 * 16);     // [TI]

  const long l = blockIdx.z;
  const int tile = blockIdx.x;
  const int t0 = tile * TI;                       // first image index
  const int tcnt = min(TI, B - t0);
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);

  const T* th = theta + l * n;
  // this block's private gradient slab: every weight entry is written
  // exactly once by one thread (plain stores — the atomic version
  // serialized ~1.8M f64 atomics per launch); the fused optimizer step
  // reduces the NT slabs on the fly.
  T* gr = gparts + (l * (long)NT + tile) * n;
  const long off = offs_dev ? offs_dev[pit] : idx_off;

  // ---- P0: resolve sources, stage images + conv weights ----
  if (tid < tcnt) {
    src[tid] = idx[l * idx_stride + off + t0 + tid];
  }
  for (int t = tid; t < F * K * K + F; t += 256) {
    wconv[t] = th[wc_off + t];  // wc..|bc.. contiguous in the layout
  }
  for (int t = tid; t < C * H; t += 256) w2s[t] = th[w2_off + t];
  for (int t = tid; t < C; t += 256) w2s[C * H + t] = th[b2_off + t];
  __syncthreads();
  for (int u = tid; u < tcnt * IMG * IMG; u += 256) {
    const int t = u / (IMG * IMG);
    img[u] = X_all[(l * maxlen + src[t]) * IMG * IMG
                   + (u - t * IMG * IMG)];
  }
  __syncthreads();

  // ---- P1: conv + ReLU + maxpool forward ----
  for (int u = tid; u < tcnt * PF; u += 256) {
    const int t = u / PF;
    const int r = u - t * PF;
    const int f = r / (P * P);
    const int py = (r / P) % P;
    const int px = r % P;
    const T* wf = wconv + f * K * K;
    const T bias = wconv[F * K * K + f];
    const T* im = img + (long)t * IMG * IMG;
    T best = T(0);
    int best_i = 0;
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      const int cy = 2 * py + (d >> 1);
      const int cx = 2 * px + (d & 1);
      T acc = bias;
      for (int ky = 0; ky < K; ++ky) {
        const T* row = im + (cy + ky) * IMG + cx;
        const T* wr = wf + ky * K;
        for (int kx = 0; kx < K; ++kx) acc += row[kx] * wr[kx];
      }
      if (acc > best) { best = acc; best_i = d; }
    }
    pool[u] = best;
    pidx[u] = (unsigned char)best_i;
  }
  __syncthreads();

  // ---- P2: fc1 + ReLU as an LDS-tiled mini-GEMM. Per 16-deep
  // K-tile: stage wt[k][o] = W1[o][k0+k] with consecutive threads
  // reading consecutive W1-row elements (the v1 per-thread serial dot
  // walked W1 with a 3.5KB stride across lanes — the dominant cost of
  // the first fused version), then accumulate (t, o) outputs.
  const T* W1 = th + w1_off;
  {
    T acc[2] = {};  // (t, o) outputs per thread: TI*H/256 <= 2 at TI=8
    for (int k0 = 0; k0 < PF; k0 += 16) {
      for (int u = tid; u < H * 16; u += 256) {
        const int o = u / 16, k = u & 15;
        wt[k * 65 + o] =
            (k0 + k < PF) ? W1[(long)o * PF + k0 + k] : T(0);
      }
      __syncthreads();
      const int kmax = min(16, PF - k0);
#pragma unroll
      for (int a = 0; a < 2; ++a) {
        const int u = tid + a * 256;
        if (u < tcnt * H) {
          const int t = u / H;
          const int o = u - t * H;
          const T* x = pool + (long)t * PF + k0;
          T sacc = T(0);
          for (int k = 0; k < kmax; ++k) {
            sacc += x[k] * wt[k * 65 + o];
          }
          acc[a] += sacc;
        }
      }
      __syncthreads();
    }
#pragma unroll
    for (int a = 0; a < 2; ++a) {
      const int u = tid + a * 256;
      if (u < tcnt * H) {
        const T z = acc[a] + th[b1_off + (u % H)];
        h1[u] = z > T(0) ? z : T(0);
      }
    }
  }
  __syncthreads();

  // ---- P3: fc2 logits (W2 is LDS-resident) + LSE + NLL dZ ----
  for (int u = tid; u < tcnt * C; u += 256) {
    const int t = u / C;
    const int o = u - t * C;
    const T* w = w2s + (long)o * H;
    const T* x = h1 + (long)t * H;
    T acc = w2s[C * H + o];
    for (int i = 0; i < H; ++i) acc += x[i] * w[i];
    dz2[u] = acc;  // logits, converted in place below
  }
  __syncthreads();
  const T wmean = loss_scale / T(B);
  if (tid < tcnt) {
    T* z = dz2 + (long)tid * C;
    const int y = (int)Y_all[l * maxlen + src[tid]];
    T mx = z[0];
    for (int c = 1; c < C; ++c) mx = z[c] > mx ? z[c] : mx;
    T sum = T(0);
    for (int c = 0; c < C; ++c) sum += ::exp(z[c] - mx);
    const T lse = mx + ::log(sum);
    if (loss != nullptr) {
      atomicAdd(&loss[l], -(z[y] - lse) / T(B));
    }
    for (int c = 0; c < C; ++c) {
      z[c] = (::exp(z[c] - lse) - (c == y ? T(1) : T(0))) * wmean;
    }
  }
  __syncthreads();

  // ---- P4: fc2 dW/db ----
  for (int u = tid; u < C * H + C; u += 256) {
    T acc = T(0);
    if (u < C * H) {
      const int o = u / H;
      const int i = u - o * H;
      for (int t = 0; t < tcnt; ++t) {
        acc += dz2[(long)t * C + o] * h1[(long)t * H + i];
      }
      gr[w2_off + u] = acc;
    } else {
      const int o = u - C * H;
      for (int t = 0; t < tcnt; ++t) acc += dz2[(long)t * C + o];
      gr[b2_off + o] = acc;
    }
  }

  // ---- P5: dh1 = dz2 @ W2, ReLU' ----
  for (int u = tid; u < tcnt * H; u += 256) {
    const int t = u / H;
    const int i = u - t * H;
    T acc = T(0);
    for (int o = 0; o < C; ++o) {
      acc += dz2[(long)t * C + o] * w2s[(long)o * H + i];
    }
    dh1[u] = (h1[u] > T(0)) ? acc : T(0);
  }
  __syncthreads();

  // ---- P6: fc1 dW/db ----
  for (int u = tid; u < H * PF; u += 256) {
    const int o = u / PF;
    const int i = u - o * PF;
    T acc = T(0);
    for (int t = 0; t < tcnt; ++t) {
      acc += dh1[(long)t * H + o] * pool[(long)t * PF + i];
    }
    gr[w1_off + u] = acc;
  }
  for (int u = tid; u < H; u += 256) {
    T acc = T(0);
    for (int t = 0; t < tcnt; ++t) acc += dh1[(long)t * H + u];
    gr[b1_off + u] = acc;
  }
  __syncthreads();

  // ---- P7: dpool = dh1 @ W1 as a tiled GEMM over 64-wide i-tiles and
  // 16-deep o-tiles: wt[o][i] staged with consecutive threads reading
  // consecutive W1 elements (coalesced), accumulators in registers.
  // pool is reused as dpool (safe: P6's readers hit the barrier above;
  // the ReLU' mask is applied from the saved sign via a fresh read of
  // pool BEFORE overwrite within the same thread's element).
  {
    for (int i0 = 0; i0 < PF; i0 += 64) {
      const int imax = min(64, PF - i0);
      T acc[2] = {};  // (t, ii) outputs: TI*64/256 = 2 at TI = 8
      for (int o0 = 0; o0 < H; o0 += 16) {
        // wt[o][ii] = W1[o0+o][i0+ii]; thread u: o = u/64, ii = u%64
        for (int u = tid; u < 16 * 64; u += 256) {
          const int o = u / 64, ii = u & 63;
          wt[o * 65 + ii] = (ii < imax)
                                ? W1[(long)(o0 + o) * PF + i0 + ii]
                                : T(0);
        }
        __syncthreads();
#pragma unroll
        for (int a = 0; a < 2; ++a) {
          const int u = tid + a * 256;
          const int t = u / 64;
          const int ii = u & 63;
          if (t < tcnt) {
            const T* g = dh1 + (long)t * H + o0;
            T sacc = T(0);
            for (int o = 0; o < 16; ++o) {
              sacc += g[o] * wt[o * 65 + ii];
            }
            acc[a] += sacc;
          }
        }
        __syncthreads();
      }
#pragma unroll
      for (int a = 0; a < 2; ++a) {
        const int u = tid + a * 256;
        const int t = u / 64;
        const int ii = u & 63;
        if (t < tcnt && ii < imax) {
          const long e = (long)t * PF + i0 + ii;
          pool[e] = (pool[e] > T(0)) ? acc[a] : T(0);
        }
      }
      __syncthreads();
    }
  }
  __syncthreads();

  // ---- P8: conv dW/db via argmax routing ----
  for (int f = 0; f < F; ++f) {
    T dw[7 * 7];
    T db = T(0);
#pragma unroll
    for (int i = 0; i < 7 * 7; ++i) dw[i] = T(0);
    for (int u = tid; u < tcnt * P * P; u += 256) {
      const int t = u / (P * P);
      const int py = (u / P) % P;
      const int px = u % P;
      const long e = (long)t * PF + f * P * P + py * P + px;
      const T g = pool[e];
      if (g == T(0)) continue;
      const int d = pidx[e];
      const int cy = 2 * py + (d >> 1);
      const int cx = 2 * px + (d & 1);
      const T* im = img + (long)t * IMG * IMG;
      db += g;
      for (int ky = 0; ky < K; ++ky) {
        const T* row = im + (cy + ky) * IMG + cx;
        for (int kx = 0; kx < K; ++kx) dw[ky * K + kx] += g * row[kx];
      }
    }
    T* red = wt;  // W1-tile buffer is free in P8; [4] per value
    const int wid = tid / WAVE;
    for (int i = 0; i < K * K + 1; ++i) {
      T v = (i < K * K) ? dw[i] : db;
      v = wave_reduce_sum(v);
      if (lane == 0) red[wid] = v;
      __syncthreads();
      if (tid == 0) {
        const T tot = red[0] + red[1] + red[2] + red[3];
        if (i < K * K) {
          gr[wc_off + (long)f * K * K + i] = tot;
        } else {
          gr[bc_off + f] = tot;
        }
      }
      __syncthreads();
    }
  }
}


// ---------------------------------------------------------------------
// Fused fc BLOCK (round 2): fc1+ReLU -> fc2 logits -> log-softmax+NLL
// -> fc2 dW/db -> dz1 -> fc1 dW/db -> dX0, ONE launch per primal
// iteration (replaces 7 small kernels: 2x linear_fwd, nll_fused,
// 2x linear_bwd_dw, 2x linear_bwd_dx — each near the ~5 us kernel
// floor, ~65 us/iteration of GPU time at the MNIST bench shapes,
// profiles/mnist_r2d_topk.txt). The conv layer stays separate (its
// kernels are not floor-bound).
//
// The three big contractions run on the f64/f32 MFMA matrix cores
// (VERDICT r1 #2): fc1 fwd (K = I), fc1 dW (K = RT rows), dX0
// (K = H), with y1 and dz1 LDS-resident — y1 never touches HBM.
// One block = RT(16) rows of one node; grid (ceil(M/RT), 1, L).
// Requires H <= 64, C <= 16; fc grad slices accumulate atomically
// (caller zeroes the whole grad stack).
constexpr int FC_RT = 16;
template <typename T>
__global__ __launch_bounds__(512) void fc_block_k(
    const T* __restrict__ x0,        // [L*M, I] conv/pool output
    const T* __restrict__ theta,     // [L, n]
    const long* __restrict__ Y_all,  // [L, maxlen] targets
    const long* __restrict__ idx,    // index stream
    long idx_stride, long idx_off, long maxlen,
    T* __restrict__ grad,            // [L, n]
    T* __restrict__ dx0,             // [L*M, I]
    T* __restrict__ dz1g,            // [L*M, H] dz1 out (fc1 dW/db
                                     // run in linear_bwd_dw: the
                                     // in-kernel atomic dW1 was ~884K
                                     // f64 atomics/launch)
    T* __restrict__ loss,            // nullable [L]
    long n, long w1_off, long b1_off, long w2_off, long b2_off,
    int M, int I, int H, int C, T loss_scale, int mask_dx0) {
  // v2: 512 threads = 4 CONSUMER waves (MFMA) + 4 PRODUCER waves
  // (HBM -> double-buffered LDS). W1's 221 KB/block stream was the
  // serial chain of the 256-thread version (stage, barrier, MFMA,
  // barrier x14): wave specialization keeps the stream entirely off
  // the compute path.
  using MF = gmfma::mfma_t<T>;
  using acc_t = typename MF::acc_t;
  typedef T vec2 __attribute__((ext_vector_type(2)));
  constexpr int RT = FC_RT;

  // single LDS arena (trap 4a)
  __shared__ T lds[2 * 32 * (RT + 1) + 2 * 32 * 65 + 2 * RT * 65 +
                   RT * 17 + 16 * 64 + 16];
  T* As = lds;                             // [2][32][RT+1]
  T* Bs = As + 2 * 32 * (RT + 1);          // [2][32][65]
  T* y1s = Bs + 2 * 32 * 65;               // [RT][65]
  T* dz1s = y1s + RT * 65;                 // [RT][65]
  T* dz2s = dz1s + RT * 65;                // [RT][17]
  T* w2s = dz2s + RT * 17;                 // [C*H + C]

  const long l = blockIdx.z;
  const int m0 = blockIdx.x * RT;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;                // 0-3 consume, 4-7 produce
  const bool producer = wid >= 4;
  const T* th = theta + l * n;
  const T* W1 = th + w1_off;
  const T* b1 = th + b1_off;
  const int lo = lane & 15;
  const int lk = lane >> 4;
  const bool fullm = (m0 + RT) <= M;
  const int nst = (I + 31) / 32;

  // P0: W2 + b2 into LDS
  for (int t = tid; t < C * H; t += 512) w2s[t] = th[w2_off + t];
  for (int t = tid; t < C; t += 512) w2s[C * H + t] = th[b2_off + t];

  // producer staging: 256 threads cover the [32 k][RT m] x0 image
  // (1 vec2 pair each) and the [32 k][64 o] W1 image (4 pairs)
  const int ptid = tid - 256;
  const int am = producer ? ptid / 16 : 0;
  const int akp = producer ? ptid % 16 : 0;
  const auto stage = [&](int buf, int k0) {
    T* Ab = As + buf * 32 * (RT + 1);
    T* Bb = Bs + buf * 32 * 65;
    if (fullm && (k0 + 32) <= I && H == 64) {  // guard-free (trap 4c)
      const vec2 va = *reinterpret_cast<const vec2*>(
          &x0[(long)(l * (long)M + m0 + am) * I + k0 + 2 * akp]);
      Ab[(2 * akp) * (RT + 1) + am] = va.x;
      Ab[(2 * akp + 1) * (RT + 1) + am] = va.y;
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const vec2 vb = *reinterpret_cast<const vec2*>(
            &W1[(long)(am + q * 16) * I + k0 + 2 * akp]);
        Bb[(2 * akp) * 65 + am + q * 16] = vb.x;
        Bb[(2 * akp + 1) * 65 + am + q * 16] = vb.y;
      }
    } else {
      const int k_ = k0 + 2 * akp;
      vec2 va = vec2{0, 0};
      if (m0 + am < M) {
        if (k_ + 1 < I) {
          va = *reinterpret_cast<const vec2*>(
              &x0[(long)(l * (long)M + m0 + am) * I + k_]);
        } else if (k_ < I) {
          va.x = x0[(long)(l * (long)M + m0 + am) * I + k_];
        }
      }
      Ab[(2 * akp) * (RT + 1) + am] = va.x;
      Ab[(2 * akp + 1) * (RT + 1) + am] = va.y;
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const int o = am + q * 16;
        vec2 vb = vec2{0, 0};
        if (o < H) {
          if (k_ + 1 < I) {
            vb = *reinterpret_cast<const vec2*>(&W1[(long)o * I + k_]);
          } else if (k_ < I) {
            vb.x = W1[(long)o * I + k_];
          }
        }
        Bb[(2 * akp) * 65 + o] = vb.x;
        Bb[(2 * akp + 1) * 65 + o] = vb.y;
      }
    }
  };

  // P1: fc1 forward on MFMA (consumer waves: one 16-wide o-fragment
  // each) while producers fill the next stage's buffers
  acc_t a1 = {};
  const int o0w = wid * 16;  // consumers: 0..48
  if (producer) stage(0, 0);
  __syncthreads();
  for (int st = 0; st < nst; ++st) {
    if (producer) {
      if (st + 1 < nst) stage((st + 1) & 1, (st + 1) * 32);
    } else {
      const T* Ab = As + (st & 1) * 32 * (RT + 1);
      const T* Bb = Bs + (st & 1) * 32 * 65;
#pragma unroll
      for (int kk = 0; kk < 32; kk += 4) {
        const int ka = kk + lk;
        const T a = Ab[ka * (RT + 1) + lo];
        const T b = Bb[ka * 65 + o0w + lo];
        a1 = MF::mma(a, b, a1);
      }
    }
    __syncthreads();
  }
  // y1 = relu(z1 + b1) into LDS (never stored to HBM)
  if (!producer && o0w < H) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = MF::acc_row(lane, r);
      const int o = o0w + lo;
      const T z = a1[r] + b1[o];
      y1s[m * 65 + o] = z > T(0) ? z : T(0);
    }
  }
  __syncthreads();

  // P2: fc2 logits + row log-softmax + NLL dZ2 (H==64 unrolls the
  // dot loop 8-wide so the LDS loads batch)
  if (H == 64) {
    for (int t = tid; t < RT * C; t += 512) {
      const int m = t / C, c = t % C;
      T z = w2s[C * 64 + c];  // b2
      T part[8] = {};
#pragma unroll
      for (int hq = 0; hq < 8; ++hq) {
#pragma unroll
        for (int hh = 0; hh < 8; ++hh) {
          const int h = hq * 8 + hh;
          part[hh] += y1s[m * 65 + h] * w2s[c * 64 + h];
        }
      }
#pragma unroll
      for (int hh = 0; hh < 8; ++hh) z += part[hh];
      dz2s[m * 17 + c] = z;
    }
  } else {
    for (int t = tid; t < RT * C; t += 512) {
      const int m = t / C, c = t % C;
      T z = w2s[C * H + c];
      for (int h = 0; h < H; ++h) {
        z += y1s[m * 65 + h] * w2s[c * H + h];
      }
      dz2s[m * 17 + c] = z;
    }
  }
  __syncthreads();
  if (tid < RT) {
    const int m = tid;
    if (m0 + m < M) {
      const long tgt =
          Y_all[l * maxlen + idx[l * idx_stride + idx_off + m0 + m]];
      T mx = dz2s[m * 17];
      for (int c = 1; c < C; ++c) {
        mx = dz2s[m * 17 + c] > mx ? dz2s[m * 17 + c] : mx;
      }
      T sum = T(0);
      for (int c = 0; c < C; ++c) sum += ::exp(dz2s[m * 17 + c] - mx);
      const T lse = mx + ::log(sum);
      const T w = loss_scale / T(M);
      if (loss != nullptr) {
        atomicAdd(&loss[l], -(dz2s[m * 17 + (int)tgt] - lse) / T(M));
      }
      for (int c = 0; c < C; ++c) {
        dz2s[m * 17 + c] =
            (::exp(dz2s[m * 17 + c] - lse) - (c == (int)tgt ? T(1)
                                                            : T(0)))
            * w;
      }
    } else {
      for (int c = 0; c < C; ++c) dz2s[m * 17 + c] = T(0);
    }
  }
  __syncthreads();

  // P3: dz1 = (dz2 @ W2) * relu'(y1) -> LDS + global (fc1 dW/db run
  // on the store-path dw kernel); fc2 dW/db
  for (int t = tid; t < RT * H; t += 512) {
    const int m = t / H, h = t % H;
    T s = T(0);
    for (int c = 0; c < C; ++c) {
      s += dz2s[m * 17 + c] * w2s[c * H + h];
    }
    const T v = y1s[m * 65 + h] > T(0) ? s : T(0);
    dz1s[m * 65 + h] = v;
    if (m0 + m < M) {
      dz1g[(long)(l * (long)M + m0 + m) * H + h] = v;
    }
  }
  if (H < 64) {  // zero the pad columns the dX0 MFMA sweeps over
    for (int t = tid; t < RT * (64 - H); t += 512) {
      const int m = t / (64 - H), h = H + t % (64 - H);
      dz1s[m * 65 + h] = T(0);
    }
  }
  for (int t = tid; t < C * H; t += 512) {
    const int c = t / H, h = t % H;
    T s = T(0);
    for (int m = 0; m < RT; ++m) {
      s += dz2s[m * 17 + c] * y1s[m * 65 + h];
    }
    atomicAdd(&grad[l * n + w2_off + c * H + h], s);
  }
  if (tid >= 32 && tid < 32 + C) {
    const int c = tid - 32;
    T s = T(0);
    for (int m = 0; m < RT; ++m) s += dz2s[m * 17 + c];
    atomicAdd(&grad[l * n + b2_off + c], s);
  }
  __syncthreads();

  // P4: dX0 = dz1 @ W1 with the conv relu' mask. A[m][k=o] from LDS,
  // B[k=o][i] = W1 direct from global (coalesced); the 64-wide
  // i-chunks split between the two wave HALVES (half h takes chunks
  // h, h+2, ...), each half prefetching its next chunk's W1 column
  // block while the current chunk's MFMAs issue.
  {
    const int half = wid >> 2;   // 0 or 1
    const int wq = wid & 3;      // wave within the half
    T w1v[16], w1n[16];
    const auto ldw1 = [&](int ixv, T* dst) {
      if (H == 64 && ixv + 16 <= I) {  // guard-free (trap 4c)
#pragma unroll
        for (int t = 0; t < 16; ++t) {
          dst[t] = W1[(long)(4 * t + lk) * I + ixv + lo];
        }
      } else {
#pragma unroll
        for (int t = 0; t < 16; ++t) {
          dst[t] = ((4 * t + lk) < H && ixv + lo < I)
                       ? W1[(long)(4 * t + lk) * I + ixv + lo]
                       : T(0);
        }
      }
    };
    const int i0_first = half * 64;
    if (i0_first < I) {
      ldw1(i0_first + wq * 16, w1v);
      for (int i0 = i0_first; i0 < I; i0 += 128) {
        const int ix = i0 + wq * 16;
        if (i0 + 128 < I) ldw1(i0 + 128 + wq * 16, w1n);
        acc_t ax = {};
#pragma unroll
        for (int t = 0; t < 16; ++t) {
          const T a = dz1s[lo * 65 + 4 * t + lk];
          ax = MF::mma(a, w1v[t], ax);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int m = m0 + MF::acc_row(lane, r);
          const int i = ix + lo;
          if (m < M && i < I) {
            const long off = (long)(l * (long)M + m) * I + i;
            // conv relu' mask: x0 IS the conv block's relu output
            dx0[off] = (!mask_dx0 || x0[off] > T(0)) ? ax[r] : T(0);
          }
        }
#pragma unroll
        for (int t = 0; t < 16; ++t) w1v[t] = w1n[t];
      }
    }
  }
}

}  // namespace fmnist
