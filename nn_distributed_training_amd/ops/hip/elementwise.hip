// Decentralized-optimizer elementwise kernels, batched over the L node
// replicas of one rank (the reference's per-node, per-parameter Python
// loops — optimizers/dsgd.py:37-58, dinno.py:119-125, dsgt.py:58-105 —
// each become ONE launch here).
//
// Layout: parameter stacks are [L, n] row-major. Neighbor vectors live in
// a "table" [R, n] (R = L local snapshot rows + received remote rows);
// per-node neighbor lists are CSR int32 (offsets [L+1], indices into the
// table). All kernels are memory-bound grid-stride loops — the design
// goal is one pass over HBM per algorithm step with everything fused.

#include "common.h"

namespace ew {

constexpr int BLOCK = 256;

// Table rows: r < L -> local stack row, else -> received remote row.
// Avoids ever materializing a concatenated table (the local stack is
// read in place; remote rows land where irecv wrote them).
template <typename T>
DEV_INLINE const T* table_row(const T* local, const T* remote, long L,
                              long n, int r) {
  return r < L ? local + (long)r * n : remote + ((long)r - L) * n;
}

// ---------------------------------------------------------------------
// DiNNO round prologue (reference optimizers/dinno.py:119-124, fused):
//   S_i      = sum_{j in N(i)} th_j
//   dual_i  += rho * (deg_i * th_i - S_i)
//   s_i      = (deg_i * th_i + S_i) / 2     (= sum_j th_reg_j, the only
//                                            reduction the penalty
//                                            gradient needs)
template <typename T>
__global__ void dinno_dual_threg_k(
    const T* __restrict__ local,   // [L, n] round-start snapshot
    const T* __restrict__ remote,  // [R-L, n] received rows (may be null)
    const int* __restrict__ offs,  // [L+1]
    const int* __restrict__ idx,   // CSR neighbor rows
    T* __restrict__ duals,         // [L, n] in/out
    T* __restrict__ s_out,         // [L, n] out
    T rho, long n, long L) {
  const long total = L * n;
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / n;
    const long e = t - l * n;
    const int k0 = offs[l], k1 = offs[l + 1];
    T S = T(0);
    for (int k = k0; k < k1; ++k) {
      S += table_row(local, remote, L, n, idx[k])[e];
    }
    const T th = local[l * n + e];
    const T deg = T(k1 - k0);
    duals[t] += rho * (deg * th - S);
    s_out[t] = (deg * th + S) * T(0.5);
  }
}

// ---------------------------------------------------------------------
// Generic weighted row-combine (DSGD mixing, reference dsgd.py:37-46,
// made snapshot-synchronous):  out_l = sum_k w_k * table[idx_k]
// (the caller includes the self row with weight W_ll in the CSR).
template <typename T>
__global__ void mix_rows_k(
    const T* __restrict__ local, const T* __restrict__ remote,
    const int* __restrict__ offs, const int* __restrict__ idx,
    const T* __restrict__ w, T* __restrict__ out, long n, long L) {
  const long total = L * n;
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / n;
    const long e = t - l * n;
    const int k0 = offs[l], k1 = offs[l + 1];
    T acc = T(0);
    for (int k = k0; k < k1; ++k) {
      acc += w[k] * table_row(local, remote, L, n, idx[k])[e];
    }
    out[t] = acc;
  }
}

// ---------------------------------------------------------------------
// DSGT mixing (reference dsgt.py:58-75, synchronous):
//   p_out_l = sum_k w_k * (p_k - alpha * y_k)
//   y_mix_l = sum_k w_k * y_k
// The table rows bundle [p | y] as [R, 2n] so params and tracker ride
// one exchange (2x comm volume).
template <typename T>
__global__ void dsgt_mix_k(
    const T* __restrict__ p_loc,   // [L, n]
    const T* __restrict__ y_loc,   // [L, n]
    const T* __restrict__ remote,  // [R-L, 2n] bundles [p | y]
    const int* __restrict__ offs, const int* __restrict__ idx,
    const T* __restrict__ w, T* __restrict__ p_out,
    T* __restrict__ y_mix, T alpha, long n, long L) {
  const long total = L * n;
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / n;
    const long e = t - l * n;
    const int k0 = offs[l], k1 = offs[l + 1];
    T accp = T(0), accy = T(0);
    for (int k = k0; k < k1; ++k) {
      const int r = idx[k];
      T pj, yj;
      if (r < L) {
        pj = p_loc[(long)r * n + e];
        yj = y_loc[(long)r * n + e];
      } else {
        const T* row = remote + ((long)r - L) * (2 * n);
        pj = row[e];
        yj = row[n + e];
      }
      accp += w[k] * pj;
      accy += w[k] * yj;
    }
    p_out[t] = accp - alpha * accy;
    y_mix[t] = accy;
  }
}

// DSGT tracker update (reference dsgt.py:87-105):
//   y = y_mix + g_new - g_old ;  g_old = g_new
template <typename T>
__global__ void dsgt_y_update_k(
    const T* __restrict__ y_mix, const T* __restrict__ g_new,
    T* __restrict__ g_old, T* __restrict__ y, long total) {
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const T gn = g_new[t];
    y[t] = y_mix[t] + gn - g_old[t];
    g_old[t] = gn;
  }
}

// ---------------------------------------------------------------------
// Reduce per-tile gradient slabs [L, P, n] into [L, n] (consumers that
// are not the fused step: DSGD's axpy, DSGT's tracker update).
template <typename T>
__global__ void reduce_parts_k(const T* __restrict__ parts,
                               T* __restrict__ out, int nparts,
                               long n, long L) {
  const long total = L * n;
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / n;
    const long e = t - l * n;
    T acc = T(0);
    for (int p = 0; p < nparts; ++p) {
      acc += parts[(l * nparts + p) * n + e];
    }
    out[t] = acc;
  }
}

// ---------------------------------------------------------------------
// Fused primal step. The DiNNO penalty gradient is analytic
// (d/dth [ rho * sum_j ||th - th_reg_j||^2 ] = 2 rho (deg th - s), plus
// the dual term from <th, dual>), so it folds into the optimizer update
// and the autograd-visible loss never materializes (reference
// dinno.py:74-91 builds it through torch.cdist + autograd every primal
// iteration).  mode: 0=Adam 1=AdamW 2=SGD (matching dinno.py:55-72);
// with_penalty=false gives the plain local step (DSGD's dsgd.py:49-58).
// first_step: treat the Adam moments as zero without reading them —
// the per-round m/v zero-fill kernels of the non-persistent DiNNO mode
// fold away (reference recreates the Adam optimizer each round,
// dinno.py:57-72).
template <typename T, int MODE, bool WITH_PENALTY>
__global__ void fused_step_k(
    T* __restrict__ theta, T* __restrict__ grad,
    const T* __restrict__ dual,   // null unless WITH_PENALTY
    const T* __restrict__ s,      // null unless WITH_PENALTY
    const int* __restrict__ deg,  // [L], null unless WITH_PENALTY
    T* __restrict__ m, T* __restrict__ v,  // Adam state (null for SGD)
    T rho, T lr, T beta1, T beta2, T eps, T wd, T bc1, T bc2,
    int first_step, int nparts, long n, long L, int zero_grad) {
  const long total = L * n;
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / n;
    const long e = t - l * n;
    T g;
    if (nparts > 1) {  // per-tile slabs from the fused train step
      g = T(0);
      for (int p = 0; p < nparts; ++p) {
        const long gi = (l * nparts + p) * n + e;
        g += grad[gi];
        if (zero_grad) grad[gi] = T(0);  // next iter's atomics land
                                         // on a clean slate for free
      }
    } else {
      g = grad[t];
      if (zero_grad) grad[t] = T(0);
    }
    // isolated node under a dynamic graph: the golden engine skips the
    // whole primal update (the reference crashes on torch.stack of an
    // empty neighbor list, so "frozen" is this framework's defined
    // behavior) — freeze theta AND the Adam moments to match exactly
    if (WITH_PENALTY && deg[l] == 0) continue;
    T th = theta[t];
    if (WITH_PENALTY) {
      g += dual[t] + T(2) * rho * (T(deg[l]) * th - s[t]);
    }
    if (MODE == 2) {  // SGD
      theta[t] = th - lr * g;
      continue;
    }
    if (MODE == 1) {  // AdamW decoupled weight decay
      th -= lr * wd * th;
    } else if (wd != T(0)) {  // Adam L2
      g += wd * th;
    }
    const T m_prev = first_step ? T(0) : m[t];
    const T v_prev = first_step ? T(0) : v[t];
    const T mt = beta1 * m_prev + (T(1) - beta1) * g;
    const T vt = beta2 * v_prev + (T(1) - beta2) * g * g;
    m[t] = mt;
    v[t] = vt;
    // bc1 = 1-beta1^t, bc2 = 1-beta2^t (host-computed per step)
    theta[t] = th - lr * (mt / bc1) / (::sqrt(vt / bc2) + eps);
  }
}

// Graph-capture variants: the round scalars (rho, lr, per-pit Adam
// bias corrections) live in a device buffer `sched` laid out as
// [rho, lr, bc1_pit0, bc2_pit0, bc1_pit1, ...] and batch offsets in an
// int64 buffer — the host updates them with one async pinned copy per
// round and replays a captured hipGraph with CONSTANT kernel args.
template <typename T>
__global__ void dinno_dual_threg_sched_k(
    const T* __restrict__ local, const T* __restrict__ remote,
    const int* __restrict__ offs, const int* __restrict__ idx,
    T* __restrict__ duals, T* __restrict__ s_out,
    const T* __restrict__ sched, long n, long L) {
  const T rho = sched[0];
  const long total = L * n;
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / n;
    const long e = t - l * n;
    const int k0 = offs[l], k1 = offs[l + 1];
    T S = T(0);
    for (int k = k0; k < k1; ++k) {
      S += table_row(local, remote, L, n, idx[k])[e];
    }
    const T th = local[l * n + e];
    const T deg = T(k1 - k0);
    duals[t] += rho * (deg * th - S);
    s_out[t] = (deg * th + S) * T(0.5);
  }
}

template <typename T, int MODE, bool WITH_PENALTY>
__global__ void fused_step_sched_k(
    T* __restrict__ theta, const T* __restrict__ grad,
    const T* __restrict__ dual, const T* __restrict__ s,
    const int* __restrict__ deg, T* __restrict__ m, T* __restrict__ v,
    const T* __restrict__ sched, int pit,
    T beta1, T beta2, T eps, T wd, int first_step, int nparts, long n, long L) {
  const T rho = sched[0];
  const T lr = sched[1];
  const T bc1 = sched[2 + 2 * pit];
  const T bc2 = sched[3 + 2 * pit];
  const long total = L * n;
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / n;
    const long e = t - l * n;
    // freeze isolated nodes (see fused_step_k)
    if (WITH_PENALTY && deg[l] == 0) continue;
    T th = theta[t];
    T g;
    if (nparts > 1) {  // per-tile slabs from the fused train step
      g = T(0);
      for (int p = 0; p < nparts; ++p) {
        g += grad[(l * nparts + p) * n + e];
      }
    } else {
      g = grad[t];
    }
    if (WITH_PENALTY) {
      g += dual[t] + T(2) * rho * (T(deg[l]) * th - s[t]);
    }
    if (MODE == 2) {
      theta[t] = th - lr * g;
      continue;
    }
    if (MODE == 1) {
      th -= lr * wd * th;
    } else if (wd != T(0)) {
      g += wd * th;
    }
    const T m_prev = first_step ? T(0) : m[t];
    const T v_prev = first_step ? T(0) : v[t];
    const T mt = beta1 * m_prev + (T(1) - beta1) * g;
    const T vt = beta2 * v_prev + (T(1) - beta2) * g * g;
    m[t] = mt;
    v[t] = vt;
    theta[t] = th - lr * (mt / bc1) / (::sqrt(vt / bc2) + eps);
  }
}

// Batch assembly: out[l*B + b, :] = X_all[l, idx[l, b], :] in one
// launch (replaces torch arange + advanced indexing + copies). idx rows
// may be strided views into a longer per-node index stream.
template <typename T>
__global__ void gather_batch_k(
    const T* __restrict__ X_all,       // [L, maxlen, F]
    const long* __restrict__ idx,      // [L, S] index stream
    T* __restrict__ out,               // [L*B, F]
    long maxlen, long Fdim, long B, long idx_stride, long idx_off,
    long total) {
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long f = t % Fdim;
    const long lb = t / Fdim;
    const long l = lb / B;
    const long b = lb - l * B;
    const long src = idx[l * idx_stride + idx_off + b];
    out[t] = X_all[(l * maxlen + src) * Fdim + f];
  }
}

// Same for targets (typed independently: long labels or T densities).
template <typename T>
__global__ void gather_targets_k(
    const T* __restrict__ Y_all, const long* __restrict__ idx,
    T* __restrict__ out, long maxlen, long B, long idx_stride,
    long idx_off, long total) {
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / B;
    const long b = t - l * B;
    out[t] = Y_all[l * maxlen + idx[l * idx_stride + idx_off + b]];
  }
}

// Gather with the batch offset read from a device buffer (slot `pit`
// of offs_dev) — hipGraph-replayable.
template <typename T>
__global__ void gather_batch_dev_k(
    const T* __restrict__ X_all, const long* __restrict__ idx,
    T* __restrict__ out, const long* __restrict__ offs_dev, int pit,
    long maxlen, long Fdim, long B, long idx_stride, long total) {
  const long idx_off = offs_dev[pit];
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long f = t % Fdim;
    const long lb = t / Fdim;
    const long l = lb / B;
    const long b = lb - l * B;
    const long src = idx[l * idx_stride + idx_off + b];
    out[t] = X_all[(l * maxlen + src) * Fdim + f];
  }
}

template <typename T>
__global__ void gather_targets_dev_k(
    const T* __restrict__ Y_all, const long* __restrict__ idx,
    T* __restrict__ out, const long* __restrict__ offs_dev, int pit,
    long maxlen, long B, long idx_stride, long total) {
  const long idx_off = offs_dev[pit];
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    const long l = t / B;
    const long b = t - l * B;
    out[t] = Y_all[l * maxlen + idx[l * idx_stride + idx_off + b]];
  }
}

// Plain axpy: theta -= alpha * grad  (DSGD local step); zero_grad
// clears grad after consuming it (next iteration's atomic kernels
// then need no separate fill launch)
template <typename T>
__global__ void axpy_k(T* __restrict__ x, T* __restrict__ g,
                       T alpha, long total, int zero_grad) {
  for (long t = blockIdx.x * (long)BLOCK + threadIdx.x; t < total;
       t += (long)gridDim.x * BLOCK) {
    x[t] += alpha * g[t];
    if (zero_grad) g[t] = T(0);
  }
}

// Keyed Feistel permutation of [0, n): out[i] = lb + pi(i), pi a
// bijection. ONE elementwise launch replaces the device
// torch.randperm chain (rocprim radix sort + merges + arange/fill —
// ~245 us/round amortized in the online density workload, see
// profiles/dens_final_topk.txt). 4 Feistel rounds over the smallest
// 2^(2*hb) >= n, cycle-walking back into [0, n).
DEV_INLINE unsigned feistel_round_f(unsigned r, unsigned key) {
  unsigned f = r * 0x9E3779B9u + key;
  f ^= f >> 13;
  f *= 0x85EBCA6Bu;
  f ^= f >> 16;
  return f;
}

__global__ void feistel_perm_k(long* __restrict__ out, long n, long lb,
                               unsigned long long key, int half_bits) {
  const unsigned mask = (1u << half_bits) - 1u;
  for (long i = blockIdx.x * (long)BLOCK + threadIdx.x; i < n;
       i += (long)gridDim.x * BLOCK) {
    unsigned long long x = (unsigned long long)i;
    do {
      unsigned lhs = (unsigned)(x >> half_bits);
      unsigned rhs = (unsigned)x & mask;
#pragma unroll
      for (int rnd = 0; rnd < 4; ++rnd) {
        const unsigned f =
            feistel_round_f(rhs, (unsigned)(key >> (16 * rnd)) + rnd)
            & mask;
        const unsigned nl = rhs;
        rhs = lhs ^ f;
        lhs = nl;
      }
      x = ((unsigned long long)lhs << half_bits) | rhs;
    } while (x >= (unsigned long long)n);
    out[i] = lb + (long)x;
  }
}


// ---------------------------------------------------------------------
// Consensus-error metric kernels (eval-only; the one op that still ran
// through torch on the gathered stack — reference
// problems/dist_mnist_problem.py:152-175): row norms, pairwise
// normalized L2 distances, normalized mean vector, distances to mean.

template <typename T>
__global__ void row_norms_k(const T* __restrict__ stack,
                            T* __restrict__ norms, long n) {
  __shared__ T scratch[BLOCK / WAVE];
  const long i = blockIdx.x;
  T acc = T(0);
  for (long e = threadIdx.x; e < n; e += BLOCK) {
    const T v = stack[i * n + e];
    acc += v * v;
  }
  acc = block_reduce_sum<T, BLOCK>(acc, scratch);
  if (threadIdx.x == 0) norms[i] = ::sqrt(acc);
}

// D[i, j] = || stack_i/|stack_i| - stack_j/|stack_j| ||_2
template <typename T>
__global__ void pairwise_normed_dist_k(
    const T* __restrict__ stack, const T* __restrict__ norms,
    T* __restrict__ D, long N, long n) {
  __shared__ T scratch[BLOCK / WAVE];
  const long i = blockIdx.x / N;
  const long j = blockIdx.x % N;
  const T ri = T(1) / norms[i];
  const T rj = T(1) / norms[j];
  T acc = T(0);
  for (long e = threadIdx.x; e < n; e += BLOCK) {
    const T d = stack[i * n + e] * ri - stack[j * n + e] * rj;
    acc += d * d;
  }
  acc = block_reduce_sum<T, BLOCK>(acc, scratch);
  if (threadIdx.x == 0) D[i * N + j] = ::sqrt(acc);
}

template <typename T>
__global__ void normed_mean_k(const T* __restrict__ stack,
                              const T* __restrict__ norms,
                              T* __restrict__ mean, long N, long n) {
  for (long e = blockIdx.x * (long)BLOCK + threadIdx.x; e < n;
       e += (long)gridDim.x * BLOCK) {
    T acc = T(0);
    for (long i = 0; i < N; ++i) {
      acc += stack[i * n + e] / norms[i];
    }
    mean[e] = acc / T(N);
  }
}

template <typename T>
__global__ void dist_to_mean_k(const T* __restrict__ stack,
                               const T* __restrict__ norms,
                               const T* __restrict__ mean,
                               T* __restrict__ Dm, long n) {
  __shared__ T scratch[BLOCK / WAVE];
  const long i = blockIdx.x;
  const T ri = T(1) / norms[i];
  T acc = T(0);
  for (long e = threadIdx.x; e < n; e += BLOCK) {
    const T d = stack[i * n + e] * ri - mean[e];
    acc += d * d;
  }
  acc = block_reduce_sum<T, BLOCK>(acc, scratch);
  if (threadIdx.x == 0) Dm[i] = ::sqrt(acc);
}

}  // namespace ew
