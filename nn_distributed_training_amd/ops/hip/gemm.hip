// Stacked linear layers: fwd/bwd GEMMs batched over the L node replicas
// (blockIdx.z = node). Weights live INSIDE the flat [L, n] parameter
// stack at per-layer offsets (models/spec.py), torch Linear layout
// W[O, I] row-major + bias[O] — so the kernels read the training state
// directly, no per-layer repacking ever happens.
//
// v1 kernels are LDS-tiled 16x16 VALU; gemm_mfma.hip provides the MFMA
// fast path for the GEMM-heavy shapes and the dispatcher in bindings.cpp
// routes between them.

#include "common.h"

namespace gemm {

constexpr int TILE = 16;

// Y[M,O] = act(X[M,I] @ W^T + b); optionally store pre-activation Z.
// X rows are node-l's batch: X + l*M*I. W_l = theta + l*n + w_off.
template <typename T>
__global__ void linear_fwd_k(
    const T* __restrict__ X, const T* __restrict__ theta,
    T* __restrict__ Y, T* __restrict__ Z,  // Z may be null
    long n, long w_off, long b_off, int M, int I, int O,
    int act, T scale) {
  __shared__ T xs[TILE][TILE + 1];
  __shared__ T ws[TILE][TILE + 1];
  const long l = blockIdx.z;
  const T* Xl = X + l * (long)M * I;
  const T* W = theta + l * n + w_off;
  const T* b = theta + l * n + b_off;

  const int m0 = blockIdx.y * TILE;
  const int o0 = blockIdx.x * TILE;
  const int tm = threadIdx.y, to = threadIdx.x;

  T acc = T(0);
  for (int k0 = 0; k0 < I; k0 += TILE) {
    // xs[tm][tk] = X[m0+tm][k0+tk] ; ws[to][tk] = W[o0+to][k0+tk]
    {
      const int m = m0 + tm, k = k0 + to;
      xs[tm][to] = (m < M && k < I) ? Xl[(long)m * I + k] : T(0);
      const int o = o0 + tm;
      ws[tm][to] = (o < O && k < I) ? W[(long)o * I + k] : T(0);
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < TILE; ++k) {
      acc += xs[tm][k] * ws[to][k];
    }
    __syncthreads();
  }
  const int m = m0 + tm, o = o0 + to;
  if (m < M && o < O) {
    const T z = acc + b[o];
    if (Z != nullptr) Z[l * (long)M * O + (long)m * O + o] = z;
    Y[l * (long)M * O + (long)m * O + o] = act_fwd(act, z, scale);
  }
}

// Small-K forward (FourierNet encode: I = 2): the tiled kernel's LDS
// K-loop degenerates there. Weights+bias stay in LDS; one thread per
// output element, the block's threads share one X row (broadcast) and
// write coalesced.
template <typename T, int IMAX>
__global__ void linear_fwd_smallk_k(
    const T* __restrict__ X, const T* __restrict__ theta,
    T* __restrict__ Y, T* __restrict__ Z,
    long n, long w_off, long b_off, int M, int I, int O,
    int act, T scale) {
  extern __shared__ __align__(16) unsigned char smem_raw[];
  T* wb = reinterpret_cast<T*>(smem_raw);  // [O*I + O]
  const long l = blockIdx.z;
  const T* W = theta + l * n + w_off;
  const T* bias = theta + l * n + b_off;
  for (int t = threadIdx.x; t < O * I; t += blockDim.x) wb[t] = W[t];
  for (int t = threadIdx.x; t < O; t += blockDim.x) {
    wb[O * I + t] = bias[t];
  }
  __syncthreads();

  const T* Xl = X + l * (long)M * I;
  T* Yl = Y + l * (long)M * O;
  T* Zl = Z ? Z + l * (long)M * O : nullptr;
  const long total = (long)M * O;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long m = t / O;
    const int o = (int)(t - m * O);
    T acc = wb[O * I + o];
#pragma unroll
    for (int i = 0; i < IMAX; ++i) {
      if (i < I) acc += Xl[m * I + i] * wb[o * I + i];
    }
    if (Zl) Zl[t] = acc;
    Yl[t] = act_fwd(act, acc, scale);
  }
}

// Small-K forward, register-resident variant (the FourierNet encode's
// hot shape: I = 2, O = 256, M = 160k). One thread per output column
// with its W row + bias held in REGISTERS for the whole kernel; the
// X row address is block-uniform each iteration so the loads are
// scalar; I is a compile-time template so the dot is guard-free
// (trap 4c: the generic kernel's `i < I` test inside the unrolled
// loop serializes its loads). Requires O == blockDim.x.
template <typename T, int IK>
__global__ void encode_fwd_k(
    const T* __restrict__ X, const T* __restrict__ theta,
    T* __restrict__ Y, T* __restrict__ Z,
    long n, long w_off, long b_off, int M, int O,
    int act, T scale) {
  const long l = blockIdx.z;
  const int o = threadIdx.x;
  const T* W = theta + l * n + w_off;
  T w[IK];
#pragma unroll
  for (int i = 0; i < IK; ++i) w[i] = W[(long)o * IK + i];
  const T bo = theta[l * n + b_off + o];
  const T* Xl = X + l * (long)M * IK;
  T* Yl = Y + l * (long)M * O;
  T* Zl = Z ? Z + l * (long)M * O : nullptr;
  for (long m = blockIdx.x; m < M; m += gridDim.x) {
    T acc = bo;
#pragma unroll
    for (int i = 0; i < IK; ++i) acc += Xl[m * IK + i] * w[i];
    const long t = m * O + o;
    if (Zl) Zl[t] = acc;
    Yl[t] = act_fwd(act, acc, scale);
  }
}

// dZ = dY * act'(z, y) — fused activation backward, one pass.
template <typename T>
__global__ void act_grad_k(
    const T* __restrict__ dY, const T* __restrict__ Y,
    const T* __restrict__ Z,  // may be null (derivative from Y only)
    T* __restrict__ dZ, long total, int act, T scale) {
  for (long t = blockIdx.x * 256L + threadIdx.x; t < total;
       t += (long)gridDim.x * 256L) {
    const T z = Z ? Z[t] : T(0);
    dZ[t] = dY[t] * act_bwd(act, z, Y[t], scale);
  }
}

// dX[M,I] = dZ[M,O] @ W[O,I], below-layer activation bwd fused
// (Xb2 != null: recompute z_below from the below layer's tiny input
// instead of reading Zb — see gemm_mfma.hip mfma_dx_k)
template <typename T>
__global__ void linear_bwd_dx_k(
    const T* __restrict__ dZ, const T* __restrict__ theta,
    T* __restrict__ dX,
    const T* __restrict__ Yb, const T* __restrict__ Zb,
    int act_below, T scale_below,
    long n, long w_off, int M, int I, int O,
    const T* __restrict__ Xb2, long wb_off, long bb_off, int Ib) {
  __shared__ T gs[TILE][TILE + 1];
  __shared__ T ws[TILE][TILE + 1];
  const long l = blockIdx.z;
  const T* dZl = dZ + l * (long)M * O;
  const T* W = theta + l * n + w_off;

  const int m0 = blockIdx.y * TILE;
  const int i0 = blockIdx.x * TILE;
  const int tm = threadIdx.y, ti = threadIdx.x;

  T acc = T(0);
  for (int k0 = 0; k0 < O; k0 += TILE) {
    {
      const int m = m0 + tm, o = k0 + ti;
      gs[tm][ti] = (m < M && o < O) ? dZl[(long)m * O + o] : T(0);
      const int oo = k0 + tm, i = i0 + ti;
      ws[tm][ti] = (oo < O && i < I) ? W[(long)oo * I + i] : T(0);
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < TILE; ++k) {
      acc += gs[tm][k] * ws[k][ti];
    }
    __syncthreads();
  }
  const int m = m0 + tm, i = i0 + ti;
  if (m < M && i < I) {
    const long off = l * (long)M * I + (long)m * I + i;
    if (act_below != ACT_NONE) {
      T z = T(0);
      if (Xb2 != nullptr) {
        const T* Wb = theta + l * n + wb_off;
        z = theta[l * n + bb_off + i];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          if (j < Ib) {
            z += Xb2[(long)(l * (long)M + m) * Ib + j]
                 * Wb[(long)i * Ib + j];
          }
        }
      } else if (Zb != nullptr) {
        z = Zb[off];
      }
      acc *= act_bwd(act_below, z, Yb ? Yb[off] : T(0), scale_below);
    }
    dX[off] = acc;
  }
}

// dW[O,I] = dZ_l[M,O]^T @ X_l[M,I], written into the [L, n] grad stack
// at the layer's offset; db[O] (column sums of dZ) is fused: the i0==0
// block column accumulates bias partials from its staged dZ tiles while
// it loops over M (the separate bias kernel was an underfilled launch).
template <typename T>
__global__ void linear_bwd_dw_k(
    const T* __restrict__ dZ, const T* __restrict__ X,
    T* __restrict__ gstack, long n, long w_off, long b_off,
    int M, int I, int O) {
  __shared__ T gs[TILE][TILE + 1];
  __shared__ T xs[TILE][TILE + 1];
  const long l = blockIdx.z;
  const T* dZl = dZ + l * (long)M * O;
  const T* Xl = X + l * (long)M * I;

  const int o0 = blockIdx.y * TILE;
  const int i0 = blockIdx.x * TILE;
  const int to = threadIdx.y, ti = threadIdx.x;
  const bool bias_block = (i0 == 0);

  T acc = T(0);
  T bacc = T(0);  // thread (to, ti): partial db[o0+ti] over rows m≡to
  for (int k0 = 0; k0 < M; k0 += TILE) {
    {
      const int m = k0 + to;
      const int o = o0 + ti;
      gs[to][ti] = (m < M && o < O) ? dZl[(long)m * O + o] : T(0);
      const int i = i0 + ti;
      xs[to][ti] = (m < M && i < I) ? Xl[(long)m * I + i] : T(0);
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < TILE; ++k) {
      acc += gs[k][to] * xs[k][ti];
    }
    if (bias_block) bacc += gs[to][ti];
    __syncthreads();
  }
  const int o = o0 + to, i = i0 + ti;
  if (o < O && i < I) {
    gstack[l * n + w_off + (long)o * I + i] = acc;
  }
  if (bias_block) {
    // column-sum bacc over to (16 rows) through LDS, thread row 0 writes
    gs[to][ti] = bacc;
    __syncthreads();
    if (to == 0 && o0 + ti < O) {
      T s = T(0);
#pragma unroll
      for (int r = 0; r < TILE; ++r) s += gs[r][ti];
      gstack[l * n + b_off + o0 + ti] = s;
    }
  }
}

// M-chunked dW+db for skinny layers (I or O < 16, e.g. the FourierNet
// encode's dW[256, 2] with M = 20k): each thread owns one (o, i) weight
// entry (or one bias entry), reduces its M chunk, atomically adds.
// Requires the layer's grad slice zeroed by the caller.
template <typename T>
__global__ void dw_small_chunked_k(
    const T* __restrict__ dZ, const T* __restrict__ X,
    T* __restrict__ gstack, long n, long w_off, long b_off,
    int M, int I, int O, int nchunk) {
  const int chunk = blockIdx.z % nchunk;
  const long l = blockIdx.z / nchunk;
  const int mc = (M + nchunk - 1) / nchunk;
  const int mlo = chunk * mc;
  const int mhi = min(M, mlo + mc);
  const T* dZl = dZ + l * (long)M * O;
  const T* Xl = X + l * (long)M * I;

  const int t = blockIdx.x * blockDim.x + threadIdx.x;
  const int total = O * I + O;  // weights then biases
  if (t >= total) return;
  T acc = T(0);
  if (t < O * I) {
    const int o = t / I, i = t % I;
    for (int m = mlo; m < mhi; ++m) {
      acc += dZl[(long)m * O + o] * Xl[(long)m * I + i];
    }
    atomicAdd(&gstack[l * n + w_off + t], acc);
  } else {
    const int o = t - O * I;
    for (int m = mlo; m < mhi; ++m) acc += dZl[(long)m * O + o];
    atomicAdd(&gstack[l * n + b_off + o], acc);
  }
}

// Skinny-layer dW with fully coalesced row reads (the generic
// dw_small_chunked walks columns with stride-O/stride-I gathers and was
// 8% of the density round). Two shapes:
//   I <= IMAX (FourierNet encode, dW[256, 2]): lanes own o-columns,
//     dZ rows read coalesced, the tiny X row broadcast-loaded;
//   O <= OMAX (density head, dW[1, 64]): lanes own i-columns, X rows
//     read coalesced, the tiny dZ row broadcast-loaded.
// M is chunked over blocks; atomic accumulation (grad slice zeroed).
// Exact-shape fast variant (I == IMAX, O == 4*WAVE): all loop bounds
// compile-time, loads unconditional — the runtime `i < I` / `o < O`
// guards in the generic kernel serialized its load pipeline (trap 4c;
// measured 169 us vs a ~52 us traffic roofline on the encode dW).
// The m-loop unrolls 4-wide for ILP against the ~900-cycle latency.
template <typename T, int IMAX>
__global__ void dw_skinny_i_exact_k(
    const T* __restrict__ dZ, const T* __restrict__ X,
    T* __restrict__ gstack, long n, long w_off, long b_off,
    int M, int I, int O, int nchunk) {
  const int chunk = blockIdx.z % nchunk;
  const long l = blockIdx.z / nchunk;
  const int mc = (M + nchunk - 1) / nchunk;
  const int mlo = chunk * mc;
  const int mhi = min(M, mlo + mc);
  const T* dZl = dZ + l * (long)M * O;
  const T* Xl = X + l * (long)M * I;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;

  T acc[4][IMAX + 1] = {};
  int m = mlo + wid;
  for (; m + 12 < mhi; m += 16) {  // 4 rows per wave-iteration
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const long mr = m + 4 * u;
      T xv[IMAX];
#pragma unroll
      for (int i = 0; i < IMAX; ++i) xv[i] = Xl[mr * I + i];
#pragma unroll
      for (int no = 0; no < 4; ++no) {
        const T g = dZl[mr * O + no * WAVE + lane];
        acc[no][IMAX] += g;
#pragma unroll
        for (int i = 0; i < IMAX; ++i) acc[no][i] += g * xv[i];
      }
    }
  }
  for (; m < mhi; m += 4) {  // tail rows
    T xv[IMAX];
#pragma unroll
    for (int i = 0; i < IMAX; ++i) xv[i] = Xl[(long)m * I + i];
#pragma unroll
    for (int no = 0; no < 4; ++no) {
      const T g = dZl[(long)m * O + no * WAVE + lane];
      acc[no][IMAX] += g;
#pragma unroll
      for (int i = 0; i < IMAX; ++i) acc[no][i] += g * xv[i];
    }
  }
#pragma unroll
  for (int no = 0; no < 4; ++no) {
    const int o = no * WAVE + lane;
#pragma unroll
    for (int i = 0; i < IMAX; ++i) {
      atomicAdd(&gstack[l * n + w_off + (long)o * I + i], acc[no][i]);
    }
    atomicAdd(&gstack[l * n + b_off + o], acc[no][IMAX]);
  }
}

template <typename T, int IMAX>
__global__ void dw_skinny_i_k(
    const T* __restrict__ dZ, const T* __restrict__ X,
    T* __restrict__ gstack, long n, long w_off, long b_off,
    int M, int I, int O, int nchunk) {
  const int chunk = blockIdx.z % nchunk;
  const long l = blockIdx.z / nchunk;
  const int mc = (M + nchunk - 1) / nchunk;
  const int mlo = chunk * mc;
  const int mhi = min(M, mlo + mc);
  const T* dZl = dZ + l * (long)M * O;
  const T* Xl = X + l * (long)M * I;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int NO = (O + WAVE - 1) / WAVE;

  // every acc index is compile-time (full unroll + early break): a
  // runtime-indexed accumulator array would live in SCRATCH, not
  // registers (CDNA guide §5.4 rule 20 — measured 8x slower here)
  T acc[4][IMAX + 1] = {};  // [o-slot][i or bias]; NO <= 4 enforced
  for (int m = mlo + wid; m < mhi; m += 4) {
    T xv[IMAX];
#pragma unroll
    for (int i = 0; i < IMAX; ++i) {
      xv[i] = (i < I) ? Xl[(long)m * I + i] : T(0);  // broadcast load
    }
#pragma unroll
    for (int no = 0; no < 4; ++no) {
      if (no >= NO) break;
      const int o = no * WAVE + lane;
      if (o < O) {
        const T g = dZl[(long)m * O + o];  // coalesced
        acc[no][IMAX] += g;
#pragma unroll
        for (int i = 0; i < IMAX; ++i) {
          if (i < I) acc[no][i] += g * xv[i];
        }
      }
    }
  }
#pragma unroll
  for (int no = 0; no < 4; ++no) {
    if (no >= NO) break;
    const int o = no * WAVE + lane;
    if (o < O) {
#pragma unroll
      for (int i = 0; i < IMAX; ++i) {
        if (i < I) {
          atomicAdd(&gstack[l * n + w_off + (long)o * I + i],
                    acc[no][i]);
        }
      }
      atomicAdd(&gstack[l * n + b_off + o], acc[no][IMAX]);
    }
  }
}

template <typename T, int OMAX>
__global__ void dw_skinny_o_k(
    const T* __restrict__ dZ, const T* __restrict__ X,
    T* __restrict__ gstack, long n, long w_off, long b_off,
    int M, int I, int O, int nchunk) {
  const int chunk = blockIdx.z % nchunk;
  const long l = blockIdx.z / nchunk;
  const int mc = (M + nchunk - 1) / nchunk;
  const int mlo = chunk * mc;
  const int mhi = min(M, mlo + mc);
  const T* dZl = dZ + l * (long)M * O;
  const T* Xl = X + l * (long)M * I;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int NI = (I + WAVE - 1) / WAVE;

  T acc[OMAX][4] = {};  // [o][i-slot]; NI <= 4 enforced
  T bacc[OMAX] = {};
  for (int m = mlo + wid; m < mhi; m += 4) {
    T gv[OMAX];
#pragma unroll
    for (int o = 0; o < OMAX; ++o) {
      gv[o] = (o < O) ? dZl[(long)m * O + o] : T(0);  // broadcast
      if (o < O && lane == 0) bacc[o] += gv[o];
    }
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      if (ni >= NI) break;
      const int i = ni * WAVE + lane;
      if (i < I) {
        const T x = Xl[(long)m * I + i];  // coalesced
#pragma unroll
        for (int o = 0; o < OMAX; ++o) {
          if (o < O) acc[o][ni] += gv[o] * x;
        }
      }
    }
  }
#pragma unroll
  for (int ni = 0; ni < 4; ++ni) {
    if (ni >= NI) break;
    const int i = ni * WAVE + lane;
    if (i < I) {
#pragma unroll
      for (int o = 0; o < OMAX; ++o) {
        if (o < O) {
          atomicAdd(&gstack[l * n + w_off + (long)o * I + i],
                    acc[o][ni]);
        }
      }
    }
  }
  if (lane == 0) {
#pragma unroll
    for (int o = 0; o < OMAX; ++o) {
      if (o < O) atomicAdd(&gstack[l * n + b_off + o], bacc[o]);
    }
  }
}

}  // namespace gemm
