// MFMA-tiled stacked linear kernels (CDNA4 gfx950).
//
// The density workload's GEMMs (M up to 20k rows per node, I/O 64-256)
// run on the matrix cores via v_mfma_f64_16x16x4_f64 (fp64 parity path)
// / v_mfma_f32_16x16x4_f32 (fp32 path — exact f32, §3 of the CDNA guide;
// there is no TF32 on gfx950). Geometry: 256-thread blocks = 4 waves in
// a 2x2 grid, each wave owns a 32x32 output sub-tile as 2x2 16x16 MFMA
// fragments; K advances 16 per LDS stage (4 MFMA k-steps).
//
// Fragment maps for the 16x16x4 shapes (measured with
// tools/mfma_probe.hip on gfx950; verified against torch references in
// tests/test_ops_gpu.py):
//   A operand: lane l holds A[i = l&15][k = l>>4]   (one element)
//   B operand: lane l holds B[k = l>>4][j = l&15]
//   C/D:       4 elements, col l&15; row differs BY DTYPE:
//              f32: (l>>4)*4 + r      f64: 4*r + (l>>4)
//
// Three contraction variants cover fwd/bwd:
//   NT (fwd):  Y[M,O] = act(X[M,I] @ W[O,I]^T + b)   contraction I
//   NN (dx):   dX[M,I] = dZ[M,O] @ W[O,I]            contraction O
//   TN (dw):   dW[O,I] = dZ[M,O]^T @ X[M,I]          contraction M,
//              M-chunked across blocks with atomic accumulation
//              (+ fused db), so 20k-row reductions spread over the
//              256-CU chip instead of serializing in one block's K loop.

#include "common.h"

namespace gmfma {

typedef double f64x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

template <typename T>
struct mfma_t;

template <>
struct mfma_t<double> {
  using acc_t = f64x4;
  static DEV_INLINE acc_t mma(double a, double b, acc_t c) {
    return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
  }
  // measured on gfx950 (tools/mfma_probe.hip): f64 acc rows interleave
  // as 4*reg + (lane>>4) — DIFFERENT from the f32 form
  static DEV_INLINE int acc_row(int lane, int r) {
    return 4 * r + (lane >> 4);
  }
};

template <>
struct mfma_t<float> {
  using acc_t = f32x4;
  static DEV_INLINE acc_t mma(float a, float b, acc_t c) {
    return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
  }
  static DEV_INLINE int acc_row(int lane, int r) {
    return (lane >> 4) * 4 + r;
  }
};

constexpr int BM = 64;   // block tile rows
constexpr int BN = 64;   // block tile cols
constexpr int BK = 16;   // K per LDS stage (BK=32 measured neutral)

// ---------------------------------------------------------------------
// NT: Y[M,O] = act(X[M,I] @ W_l[O,I]^T + b_l)  [+ optional Z store]
//
// Round-2 structure (measured 139 us vs ~105 roofline at BK=16
// single-buffered): BK=32, 16-byte (2xf64) staging loads, and the T14
// write-after-barrier register pipeline — ONE register set holds stage
// s+1 while the MFMA loop runs on the LDS image of stage s, so the HBM
// fetch of the next tile overlaps compute instead of serializing at
// the barrier (cdna_hip_programming.md §5.5 T14 / G15).
// Requires I % 2 == 0 for the vectorized path (dispatch falls back to
// the element loop otherwise via the k-bounds checks: pairs are only
// vector-loaded when both elements are in range).
constexpr int FK = 32;  // fwd K per stage
template <typename T>
__global__ __launch_bounds__(256) void mfma_fwd_k(
    const T* __restrict__ X, const T* __restrict__ theta,
    T* __restrict__ Y, T* __restrict__ Z,
    long n, long w_off, long b_off, int M, int I, int O,
    int act, T scale) {
  using MF = mfma_t<T>;
  using acc_t = typename MF::acc_t;
  typedef T vec2 __attribute__((ext_vector_type(2)));
  __shared__ T As[FK][BM + 1];   // A^T image: As[k][m]
  __shared__ T Bs[FK][BN + 1];   // Bs[k][o] = W[o][k]

  const long l = blockIdx.z;
  const T* Xl = X + l * (long)M * I;
  const T* W = theta + l * n + w_off;
  const T* bias = theta + l * n + b_off;

  const int m0 = blockIdx.y * BM;
  const int o0 = blockIdx.x * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;          // 4 waves: (wm, wn) = (wid>>1, wid&1)
  const int wm = (wid >> 1) * 32;
  const int wn = (wid & 1) * 32;

  // staging assignment: pair u = tid + 256*q covers row u/16 of the
  // tile, k-pair u%16 (16 threads stream 256 contiguous bytes per row)
  const int sm = tid / 16;           // A row (m) / B row (o) of pair 0
  const int skp = tid % 16;          // k-pair within the row
  acc_t acc[2][2] = {};
  vec2 ra[4], rb[4];

  const int nstages = (I + FK - 1) / FK;
  // Full tiles/stages load guard-free: runtime bounds-guards on each
  // load of an unrolled chain force per-element branch + vmcnt(0)
  // waits that serialize the prefetch (trap 4c)
  const bool full_mo = (m0 + BM) <= M && (o0 + BN) <= O;
  const auto load_stage = [&](int k0) {
    if (full_mo && k0 + FK <= I) {
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const int m = sm + q * 16;
        const int k = k0 + 2 * skp;
        ra[q] = *reinterpret_cast<const vec2*>(
            &Xl[(long)(m0 + m) * I + k]);
        rb[q] = *reinterpret_cast<const vec2*>(
            &W[(long)(o0 + m) * I + k]);
      }
    } else {
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const int m = sm + q * 16;
        const int k = k0 + 2 * skp;
        ra[q] = vec2{0, 0};
        rb[q] = vec2{0, 0};
        if (m0 + m < M) {
          if (k + 1 < I) {
            ra[q] = *reinterpret_cast<const vec2*>(
                &Xl[(long)(m0 + m) * I + k]);
          } else if (k < I) {
            ra[q].x = Xl[(long)(m0 + m) * I + k];
          }
        }
        if (o0 + m < O) {
          if (k + 1 < I) {
            rb[q] = *reinterpret_cast<const vec2*>(
                &W[(long)(o0 + m) * I + k]);
          } else if (k < I) {
            rb[q].x = W[(long)(o0 + m) * I + k];
          }
        }
      }
    }
  };
  load_stage(0);

  for (int s = 0; s < nstages; ++s) {
    __syncthreads();  // previous compute done: LDS free
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int m = sm + q * 16;
      As[2 * skp][m] = ra[q].x;
      As[2 * skp + 1][m] = ra[q].y;
      Bs[2 * skp][m] = rb[q].x;
      Bs[2 * skp + 1][m] = rb[q].y;
    }
    if (s + 1 < nstages) {  // issue next stage's fetch NOW (T14)
      load_stage((s + 1) * FK);
    }
    __syncthreads();  // LDS image of stage s visible
    const int klim = min(FK, I - s * FK);
#pragma unroll
    for (int kk = 0; kk < FK; kk += 4) {
      if (kk < klim) {
        const int ka = kk + (lane >> 4);
#pragma unroll
        for (int fm = 0; fm < 2; ++fm) {
          const T a = As[ka][wm + fm * 16 + (lane & 15)];
#pragma unroll
          for (int fn = 0; fn < 2; ++fn) {
            const T b = Bs[ka][wn + fn * 16 + (lane & 15)];
            acc[fm][fn] = MF::mma(a, b, acc[fm][fn]);
          }
        }
      }
    }
  }

  // epilogue: bias + activation, coalesced per-fragment store.
  // Full tiles store guard-free (per-element bounds guards serialize
  // the store chain — trap 4c).
  if (full_mo && Z == nullptr) {
    T bv[2];
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      bv[fn] = bias[o0 + wn + fn * 16 + (lane & 15)];
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int m = m0 + wm + fm * 16 + MF::acc_row(lane, r);
          const int o = o0 + wn + fn * 16 + (lane & 15);
          const long off = l * (long)M * O + (long)m * O + o;
          Y[off] = act_fwd(act, acc[fm][fn][r] + bv[fn], scale);
        }
      }
    }
    return;
  }
#pragma unroll
  for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm + fm * 16 + MF::acc_row(lane, r);
        const int o = o0 + wn + fn * 16 + (lane & 15);
        if (m < M && o < O) {
          const T z = acc[fm][fn][r] + bias[o];
          const long off = l * (long)M * O + (long)m * O + o;
          if (Z != nullptr) Z[off] = z;
          Y[off] = act_fwd(act, z, scale);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------
// NN: dX[M,I] = dZ[M,O] @ W_l[O,I]   (contraction over O), with the
// BELOW layer's activation backward fused into the epilogue
// (dX *= act'(z_below, y_below)): the separate memory-bound act_grad
// pass was ~14% of the density round.
//
// Round-2: same T14 write-after-barrier register pipeline + BK=32 +
// vec2 staging as mfma_fwd_k, and an optional RECOMPUTE mode for the
// below layer's pre-activation: when the below layer has a tiny input
// (the FourierNet encode, Ib = 2), z_below is recomputed from
// Xb2 @ W_below^T + b_below in the epilogue (4 flops + L1-resident
// reads) instead of reading a full [M, I] Zb tensor back from HBM
// (328 MB/call in the density config).
template <typename T>
__global__ __launch_bounds__(256) void mfma_dx_k(
    const T* __restrict__ dZ, const T* __restrict__ theta,
    T* __restrict__ dX,
    const T* __restrict__ Yb, const T* __restrict__ Zb,  // may be null
    int act_below, T scale_below,
    long n, long w_off, int M, int I, int O,
    const T* __restrict__ Xb2,  // below layer INPUT (recompute mode)
    long wb_off, long bb_off, int Ib) {
  using MF = mfma_t<T>;
  using acc_t = typename MF::acc_t;
  typedef T vec2 __attribute__((ext_vector_type(2)));
  __shared__ T As[FK][BM + 1];   // As[k=o][m] = dZ[m][o]
  __shared__ T Bs[FK][BN + 1];   // Bs[k=o][i] = W[o][i]

  const long l = blockIdx.z;
  const T* Gl = dZ + l * (long)M * O;
  const T* W = theta + l * n + w_off;

  const int m0 = blockIdx.y * BM;
  const int i0 = blockIdx.x * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 32;
  const int wn = (wid & 1) * 32;

  const int sm = tid / 16;   // staged row (m for A, o-row for B image)
  const int skp = tid % 16;  // k-pair within the row
  acc_t acc[2][2] = {};
  vec2 ra[4];

  // A: As[k][m] = dZ[m0+m][k]  (k = contraction over O, row-major in
  // dZ so pairs are contiguous); B: Bs[k][i] = W[k][i0+i] (W rows
  // contiguous in i).
  const int nstages = (O + FK - 1) / FK;
  // stage A via vec2 over k (pairs contiguous in dZ rows); stage B
  // with vec2 over i (pairs contiguous in W rows): thread covers W
  // rows bo + q*8 (q < 4 -> 32 k-rows) at i-pair bip.
  const int bo = tid / 32;    // W row (o) of pair 0 for B staging
  const int bip = tid % 32;   // i-pair within the row
  vec2 rb2[4];                // 4 rows per thread over the 32-k stage

  // full tiles/stages load guard-free (trap 4c, see mfma_fwd_k)
  const bool full_mi = (m0 + BM) <= M && (i0 + BN) <= I;
  const auto load_stage = [&](int k0) {
    if (full_mi && k0 + FK <= O) {
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        ra[q] = *reinterpret_cast<const vec2*>(
            &Gl[(long)(m0 + sm + q * 16) * O + k0 + 2 * skp]);
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        rb2[q] = *reinterpret_cast<const vec2*>(
            &W[(long)(k0 + bo + q * 8) * I + i0 + 2 * bip]);
      }
    } else {
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const int m = sm + q * 16;
        const int k = k0 + 2 * skp;
        ra[q] = vec2{0, 0};
        if (m0 + m < M) {
          if (k + 1 < O) {
            ra[q] = *reinterpret_cast<const vec2*>(
                &Gl[(long)(m0 + m) * O + k]);
          } else if (k < O) {
            ra[q].x = Gl[(long)(m0 + m) * O + k];
          }
        }
      }
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        const int o = k0 + bo + q * 8;
        const int i = i0 + 2 * bip;
        rb2[q] = vec2{0, 0};
        if (o < O) {
          if (i + 1 < I) {
            rb2[q] = *reinterpret_cast<const vec2*>(
                &W[(long)o * I + i]);
          } else if (i < I) {
            rb2[q].x = W[(long)o * I + i];
          }
        }
      }
    }
  };
  load_stage(0);

  for (int s = 0; s < nstages; ++s) {
    __syncthreads();
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int m = sm + q * 16;
      As[2 * skp][m] = ra[q].x;
      As[2 * skp + 1][m] = ra[q].y;
    }
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int kb = bo + q * 8;
      Bs[kb][2 * bip] = rb2[q].x;
      Bs[kb][2 * bip + 1] = rb2[q].y;
    }
    if (s + 1 < nstages) {
      load_stage((s + 1) * FK);
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < FK; kk += 4) {
      const int ka = kk + (lane >> 4);
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        const T a = As[ka][wm + fm * 16 + (lane & 15)];
#pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
          const T b = Bs[ka][wn + fn * 16 + (lane & 15)];
          acc[fm][fn] = MF::mma(a, b, acc[fm][fn]);
        }
      }
    }
  }

  const T* Wb = theta + l * n + wb_off;
  const T* bb = theta + l * n + bb_off;
  if (full_mi && act_below == ACT_RELU) {
    // guard-free fast path for the relu-below layers: prefetch the 16
    // Yb mask values unconditionally, then compute + store (per-
    // element bounds guards serialized this epilogue — trap 4c)
    T yv[2][2][4];
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int m = m0 + wm + fm * 16 + MF::acc_row(lane, r);
          const int i = i0 + wn + fn * 16 + (lane & 15);
          yv[fm][fn][r] = Yb[l * (long)M * I + (long)m * I + i];
        }
      }
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int m = m0 + wm + fm * 16 + MF::acc_row(lane, r);
          const int i = i0 + wn + fn * 16 + (lane & 15);
          const long off = l * (long)M * I + (long)m * I + i;
          dX[off] = yv[fm][fn][r] > T(0) ? acc[fm][fn][r] : T(0);
        }
      }
    }
    return;
  }
  if (full_mi && act_below == ACT_SIN_RELU && Xb2 != nullptr
      && Ib == 2) {
    // encode-below fast path: z recomputed from the 2-wide input,
    // all loads issued before the sin/cos block
    T x0v[2][4], x1v[2][4];
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm + fm * 16 + MF::acc_row(lane, r);
        const long xoff = (long)(l * (long)M + m) * 2;
        x0v[fm][r] = Xb2[xoff];
        x1v[fm][r] = Xb2[xoff + 1];
      }
    }
    T w0v[2], w1v[2], bv[2];
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int i = i0 + wn + fn * 16 + (lane & 15);
      w0v[fn] = Wb[(long)i * 2];
      w1v[fn] = Wb[(long)i * 2 + 1];
      bv[fn] = bb[i];
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int m = m0 + wm + fm * 16 + MF::acc_row(lane, r);
          const int i = i0 + wn + fn * 16 + (lane & 15);
          const long off = l * (long)M * I + (long)m * I + i;
          const T z = bv[fn] + x0v[fm][r] * w0v[fn]
                      + x1v[fm][r] * w1v[fn];
          dX[off] = acc[fm][fn][r]
                    * act_bwd(ACT_SIN_RELU, z, T(0), scale_below);
        }
      }
    }
    return;
  }
#pragma unroll
  for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm + fm * 16 + MF::acc_row(lane, r);
        const int i = i0 + wn + fn * 16 + (lane & 15);
        if (m < M && i < I) {
          const long off = l * (long)M * I + (long)m * I + i;
          T v = acc[fm][fn][r];
          if (act_below != ACT_NONE) {
            T z = T(0);
            if (Xb2 != nullptr) {  // recompute z_below (tiny Ib)
              z = bb[i];
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
            v *= act_bwd(act_below, z, Yb ? Yb[off] : T(0),
                         scale_below);
          }
          dX[off] = v;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------
// TN: dW[O,I] partial = dZ_chunk[Mc,O]^T @ X_chunk[Mc,I], atomically
// accumulated into the grad stack; db fused (i-tile 0 blocks).
// Grid: (I/BN, O/BM-rows?, L * nchunk). We tile dW as [O rows][I cols]
// with the same 64x64 block: A[k=m][o] = dZ[m][o], B[k=m][i] = X[m][i].
template <typename T>
__global__ __launch_bounds__(256) void mfma_dw_k(
    const T* __restrict__ dZ, const T* __restrict__ X,
    T* __restrict__ gstack, long n, long w_off, long b_off,
    int M, int I, int O, int nchunk) {
  using MF = mfma_t<T>;
  using acc_t = typename MF::acc_t;
  __shared__ T As[BK][BM + 1];   // As[k=m][o]
  __shared__ T Bs[BK][BN + 1];   // Bs[k=m][i]

  const int chunk = blockIdx.z % nchunk;
  const long l = blockIdx.z / nchunk;
  const T* Gl = dZ + l * (long)M * O;
  const T* Xl = X + l * (long)M * I;

  const int mc = (M + nchunk - 1) / nchunk;
  const int mlo = chunk * mc;
  const int mhi = min(M, mlo + mc);

  const int o0 = blockIdx.y * BM;
  const int i0 = blockIdx.x * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 32;   // o-direction
  const int wn = (wid & 1) * 32;    // i-direction

  acc_t acc[2][2] = {};
  // db rides along: the dZ tile is already staged COALESCED in As, so
  // i-tile-0 blocks column-sum it instead of a separate bias kernel
  // re-reading dZ with stride-O gathers (that kernel was 9% of the
  // density round). Thread tid < BM owns column o = tid.
  T db_acc = T(0);
  const bool bias_block = (i0 == 0);

  for (int k0 = mlo; k0 < mhi; k0 += BK) {
    // o-fastest staging order: consecutive threads read consecutive
    // dZ-row elements (k-fastest was a stride-O gather — uncoalesced,
    // and dW was the #1 kernel of the density round)
    for (int t = tid; t < BM * BK; t += 256) {
      const int k = t / BM, o = t % BM;
      As[k][o] = (o0 + o < O && k0 + k < mhi)
                     ? Gl[(long)(k0 + k) * O + (o0 + o)]
                     : T(0);
    }
    for (int t = tid; t < BN * BK; t += 256) {
      const int i = t % BN, k = t / BN;
      Bs[k][i] = (k0 + k < mhi && i0 + i < I)
                     ? Xl[(long)(k0 + k) * I + (i0 + i)]
                     : T(0);
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const int ka = kk + (lane >> 4);
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        const T a = As[ka][wm + fm * 16 + (lane & 15)];
#pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
          const T b = Bs[ka][wn + fn * 16 + (lane & 15)];
          acc[fm][fn] = MF::mma(a, b, acc[fm][fn]);
        }
      }
    }
    if (bias_block && tid < BM) {
#pragma unroll
      for (int k = 0; k < BK; ++k) db_acc += As[k][tid];
    }
    __syncthreads();
  }
  if (bias_block && tid < BM && o0 + tid < O) {
    atomicAdd(&gstack[l * n + b_off + o0 + tid], db_acc);
  }

#pragma unroll
  for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int o = o0 + wm + fm * 16 + MF::acc_row(lane, r);
        const int i = i0 + wn + fn * 16 + (lane & 15);
        if (o < O && i < I) {
          atomicAdd(&gstack[l * n + w_off + (long)o * I + i],
                    acc[fm][fn][r]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------
// TN full-I variant: dW[O, I] = dZ[M,O]^T @ X[M,I] with one block
// covering a 64-o x FULL-I tile (I <= 448, I%16 == 0, O%64 == 0).
//
// Rationale (round-2 rework of mfma_dw_k, which measured 2.5x over its
// traffic roofline): with 64x64 tiles the dZ panel is re-fetched once
// per i-tile (4x for I=256), and the single-buffered LDS stage
// serializes HBM against MFMA.  Full-I tiles fetch dZ and X from HBM
// exactly once.  Both operands of this contraction (k = m) are
// m-row-major in memory, so the MFMA fragment reads (16 consecutive
// o / i within a row) coalesce DIRECTLY from global: no LDS image, no
// barriers; the CU's L1 serves the (x4 i-wave, x2 o-wave) intra-block
// re-reads.  8 independent accumulators per wave cover the f64 MFMA
// dependent latency; 2 waves/SIMD + unrolled k keep ~50 loads in
// flight per SIMD against the ~900-cycle HBM latency.
// db is accumulated by the i0==0 wave pair from its own a-loads
// (each (m, o) element passes through exactly one (lane, k-class)).
// ---------------------------------------------------------------------
// BK=16 single-buffered dx variant (round-1 structure) kept for A/B:
// the T14 rework above REGRESSED dx at O=64 (199 -> 228 us avg,
// profiles/dens_r2d_topk.txt) — only 2 stages leave no pipeline to
// fill. NDTA_DX_NEW=0 selects this kernel.
// NN: dX[M,I] = dZ[M,O] @ W_l[O,I]   (contraction over O), with the
// BELOW layer's activation backward fused into the epilogue
// (dX *= act'(z_below, y_below)): the separate memory-bound act_grad
// pass was ~14% of the density round.
template <typename T>
__global__ __launch_bounds__(256) void mfma_dx16_k(
    const T* __restrict__ dZ, const T* __restrict__ theta,
    T* __restrict__ dX,
    const T* __restrict__ Yb, const T* __restrict__ Zb,  // may be null
    int act_below, T scale_below,
    long n, long w_off, int M, int I, int O,
    const T* __restrict__ Xb2, long wb_off, long bb_off, int Ib) {
  using MF = mfma_t<T>;
  using acc_t = typename MF::acc_t;
  __shared__ T As[BK][BM + 1];   // As[k=o][m] = dZ[m][o]
  __shared__ T Bs[BK][BN + 1];   // Bs[k=o][i] = W[o][i]

  const long l = blockIdx.z;
  const T* Gl = dZ + l * (long)M * O;
  const T* W = theta + l * n + w_off;

  const int m0 = blockIdx.y * BM;
  const int i0 = blockIdx.x * BN;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = (wid >> 1) * 32;
  const int wn = (wid & 1) * 32;

  acc_t acc[2][2] = {};

  for (int k0 = 0; k0 < O; k0 += BK) {
    for (int t = tid; t < BM * BK; t += 256) {
      const int m = t / BK, k = t % BK;
      As[k][m] = (m0 + m < M && k0 + k < O)
                     ? Gl[(long)(m0 + m) * O + (k0 + k)]
                     : T(0);
    }
    for (int t = tid; t < BN * BK; t += 256) {
      const int i = t % BN, k = t / BN;
      Bs[k][i] = (k0 + k < O && i0 + i < I)
                     ? W[(long)(k0 + k) * I + (i0 + i)]
                     : T(0);
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < BK; kk += 4) {
      const int ka = kk + (lane >> 4);
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        const T a = As[ka][wm + fm * 16 + (lane & 15)];
#pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
          const T b = Bs[ka][wn + fn * 16 + (lane & 15)];
          acc[fm][fn] = MF::mma(a, b, acc[fm][fn]);
        }
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int fm = 0; fm < 2; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm + fm * 16 + MF::acc_row(lane, r);
        const int i = i0 + wn + fn * 16 + (lane & 15);
        if (m < M && i < I) {
          const long off = l * (long)M * I + (long)m * I + i;
          T v = acc[fm][fn][r];
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
            v *= act_bwd(act_below, z, Yb ? Yb[off] : T(0),
                         scale_below);
          }
          dX[off] = v;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------
// parts != null: each block stores its [64, I] partial tile to
// parts[(l*nchunk + chunk)*O*I + (o0+o)*I + i] with PLAIN coalesced
// stores instead of nchunk-way atomicAdd contention on gstack (the
// atomic burst measured ~2.8x the kernel's traffic roofline —
// profiles r2b); dw_reduce_parts_k folds the chunks afterwards.
// db stays atomic either way (64 adds/block).
template <typename T, int NIMAX>
__global__ __launch_bounds__(512) void mfma_dw_direct_k(
    const T* __restrict__ dZ, const T* __restrict__ X,
    T* __restrict__ gstack, T* __restrict__ parts,
    long n, long w_off, long b_off,
    int M, int I, int O, int nchunk) {
  using MF = mfma_t<T>;
  using acc_t = typename MF::acc_t;

  const int chunk = blockIdx.z % nchunk;
  const long l = blockIdx.z / nchunk;
  const T* Gl = dZ + l * (long)M * O;
  const T* Xl = X + l * (long)M * I;

  // 4-aligned chunk bounds so only the LAST chunk has a ragged tail
  int mc = (M + nchunk - 1) / nchunk;
  mc = (mc + 3) & ~3;
  const int mlo = chunk * mc;
  const int mhi = min(M, mlo + mc);
  if (mlo >= mhi) return;

  const int o0 = blockIdx.y * 64;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;            // 8 waves: (wo, wi) = (wid&1, wid>>2)? no:
  const int wo = (wid & 1) * 32;       // 2 o-waves x 4 i-waves
  const int wi = wid >> 1;
  const int fi_per = (I / 16 + 3) / 4; // i-fragments per wave (<= 7)
  const int i0 = wi * fi_per * 16;
  const int ni = min(fi_per, (I - i0) / 16);  // frags this wave owns

  const int lo = lane & 15;            // fragment row/col
  const int lk = lane >> 4;            // k sub-step
  acc_t acc[2][NIMAX] = {};
  T db0 = T(0), db1 = T(0);
  const bool bias_wave = (wi == 0) && (b_off >= 0);

  const T* __restrict__ ga0 = Gl + o0 + wo + lo;
  const T* __restrict__ ga1 = ga0 + 16;
  const T* __restrict__ gb = Xl + i0 + lo;

  int k = mlo;
  if (ni > 0) {
    // software-pipelined: the NEXT k-step's operands load while the
    // current step's MFMAs issue (one step of lookahead keeps ~2x the
    // loads in flight against the ~900-cycle HBM latency)
    bool have = k + 4 <= mhi;
    T a0 = T(0), a1 = T(0), b[NIMAX] = {};
    if (have && ni == NIMAX) {
      const long ka = k + lk;
      a0 = ga0[ka * O];
      a1 = ga1[ka * O];
#pragma unroll
      for (int fi = 0; fi < NIMAX; ++fi) {
        b[fi] = gb[ka * I + fi * 16];
      }
    } else if (have) {
      const long ka = k + lk;
      a0 = ga0[ka * O];
      a1 = ga1[ka * O];
#pragma unroll
      for (int fi = 0; fi < NIMAX; ++fi) {
        if (fi < ni) b[fi] = gb[ka * I + fi * 16];
      }
    }
    // guard-free DEPTH-4 pipeline when this wave owns ALL NIMAX
    // fragments (the I%64==0 shapes, i.e. every density layer): one
    // step of lookahead only covers ~512 MFMA-issue cycles against the
    // ~900-cycle HBM latency, so four k-steps of operands stay in
    // flight; the drain loop keeps the refill guard-free (trap 4c).
    if (ni == NIMAX) {
      const int nfull = (mhi - k) / 4;
      // Guard-free software pipeline with COMPILE-TIME slot indices
      // (a runtime-indexed operand array lives in scratch — rule 20;
      // the first modulo-slot version ran 1.7x SLOWER). The main loop
      // processes D steps per iteration so each slot index is the
      // unrolled d; depth covers the ~900-cycle HBM latency against
      // each step's MFMA-issue cover. f64 MFMA issues at 64 cyc/SIMD,
      // f32 at 32: NIMAX=1 steps cover 2 issues, NIMAX>=4 cover 8 —
      // the cheaper the step, the deeper the pipeline.
      constexpr int D =
          (NIMAX == 1) ? (sizeof(T) == 4 ? 16 : 8)
                       : (sizeof(T) == 4 ? 4 : 2);
      int step = 0;
      if (nfull >= D) {
        T a0p[D], a1p[D], bp[D][NIMAX];
#pragma unroll
        for (int d = 0; d < D; ++d) {
          const long kd = k + 4 * d + lk;
          a0p[d] = ga0[kd * O];
          a1p[d] = ga1[kd * O];
#pragma unroll
          for (int fi = 0; fi < NIMAX; ++fi) {
            bp[d][fi] = gb[kd * I + fi * 16];
          }
        }
        const int body = nfull - D;
        for (; step + D <= body; step += D) {
#pragma unroll
          for (int d = 0; d < D; ++d) {
            const T a0c = a0p[d], a1c = a1p[d];
            T bc[NIMAX];
#pragma unroll
            for (int fi = 0; fi < NIMAX; ++fi) bc[fi] = bp[d][fi];
            const long kf = k + 4 * (step + d + D) + lk;
            a0p[d] = ga0[kf * O];
            a1p[d] = ga1[kf * O];
#pragma unroll
            for (int fi = 0; fi < NIMAX; ++fi) {
              bp[d][fi] = gb[kf * I + fi * 16];
            }
            if (bias_wave) { db0 += a0c; db1 += a1c; }
#pragma unroll
            for (int fi = 0; fi < NIMAX; ++fi) {
              acc[0][fi] = MF::mma(a0c, bc[fi], acc[0][fi]);
              acc[1][fi] = MF::mma(a1c, bc[fi], acc[1][fi]);
            }
          }
        }
        // consume the last in-flight slots (steps step..step+D-1 are
        // resident; anything beyond reloads in the plain tail below)
#pragma unroll
        for (int d = 0; d < D; ++d) {
          if (bias_wave) { db0 += a0p[d]; db1 += a1p[d]; }
#pragma unroll
          for (int fi = 0; fi < NIMAX; ++fi) {
            acc[0][fi] = MF::mma(a0p[d], bp[d][fi], acc[0][fi]);
            acc[1][fi] = MF::mma(a1p[d], bp[d][fi], acc[1][fi]);
          }
        }
        step += D;
      }
      for (; step < nfull; ++step) {  // plain tail (< 2D steps)
        const long kd = k + 4 * step + lk;
        const T a0c = ga0[kd * O];
        const T a1c = ga1[kd * O];
        if (bias_wave) { db0 += a0c; db1 += a1c; }
#pragma unroll
        for (int fi = 0; fi < NIMAX; ++fi) {
          const T bc = gb[kd * I + fi * 16];
          acc[0][fi] = MF::mma(a0c, bc, acc[0][fi]);
          acc[1][fi] = MF::mma(a1c, bc, acc[1][fi]);
        }
      }
      k += 4 * nfull;
      have = false;
      (void)a0; (void)a1; (void)b;
    } else {
      while (have) {
        const int kn = k + 4;
        const bool haven = kn + 4 <= mhi;
        T a0n = T(0), a1n = T(0), bn[NIMAX] = {};
        if (haven) {
          const long kan = kn + lk;
          a0n = ga0[kan * O];
          a1n = ga1[kan * O];
#pragma unroll
          for (int fi = 0; fi < NIMAX; ++fi) {
            if (fi < ni) bn[fi] = gb[kan * I + fi * 16];
          }
        }
        if (bias_wave) { db0 += a0; db1 += a1; }
#pragma unroll
        for (int fi = 0; fi < NIMAX; ++fi) {
          if (fi < ni) {
            acc[0][fi] = MF::mma(a0, b[fi], acc[0][fi]);
            acc[1][fi] = MF::mma(a1, b[fi], acc[1][fi]);
          }
        }
        a0 = a0n;
        a1 = a1n;
#pragma unroll
        for (int fi = 0; fi < NIMAX; ++fi) b[fi] = bn[fi];
        k = kn;
        have = haven;
      }
    }
    if (k < mhi) {  // ragged tail (< 4 rows): zero-padded operands
      const long ka = k + lk;
      const bool ok = ka < mhi;
      const T ta0 = ok ? ga0[ka * O] : T(0);
      const T ta1 = ok ? ga1[ka * O] : T(0);
      if (bias_wave) { db0 += ta0; db1 += ta1; }
#pragma unroll
      for (int fi = 0; fi < NIMAX; ++fi) {
        if (fi < ni) {
          const T tb = ok ? gb[ka * I + fi * 16] : T(0);
          acc[0][fi] = MF::mma(ta0, tb, acc[0][fi]);
          acc[1][fi] = MF::mma(ta1, tb, acc[1][fi]);
        }
      }
    }
  } else if (bias_wave) {
    for (; k + 4 <= mhi; k += 4) {
      const long ka = k + lk;
      db0 += ga0[ka * O];
      db1 += ga1[ka * O];
    }
    if (k < mhi && k + lk < mhi) {
      db0 += ga0[(long)(k + lk) * O];
      db1 += ga1[(long)(k + lk) * O];
    }
  }

  if (bias_wave) {
    // fold the 4 k-classes of each column: lanes lo, lo+16, lo+32, lo+48
#pragma unroll
    for (int off = 32; off >= 16; off >>= 1) {
      db0 += __shfl_down(db0, off, WAVE);
      db1 += __shfl_down(db1, off, WAVE);
    }
    if (lane < 16) {
      atomicAdd(&gstack[l * n + b_off + o0 + wo + lo], db0);
      atomicAdd(&gstack[l * n + b_off + o0 + wo + 16 + lo], db1);
    }
  }

  if (parts != nullptr) {
    T* slab = parts + ((long)blockIdx.z) * (long)O * I;
#pragma unroll
    for (int fo = 0; fo < 2; ++fo) {
#pragma unroll
      for (int fi = 0; fi < NIMAX; ++fi) {
        if (fi < ni) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int o = o0 + wo + fo * 16 + MF::acc_row(lane, r);
            const int i = i0 + fi * 16 + lo;
            slab[(long)o * I + i] = acc[fo][fi][r];
          }
        }
      }
    }
  } else {
#pragma unroll
    for (int fo = 0; fo < 2; ++fo) {
#pragma unroll
      for (int fi = 0; fi < NIMAX; ++fi) {
        if (fi < ni) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int o = o0 + wo + fo * 16 + MF::acc_row(lane, r);
            const int i = i0 + fi * 16 + lo;
            atomicAdd(&gstack[l * n + w_off + (long)o * I + i],
                      acc[fo][fi][r]);
          }
        }
      }
    }
  }
}

// folds dw_direct's per-chunk slabs into the grad stack (WRITE —
// the stack's w slice needs no pre-zero on this path)
template <typename T>
__global__ void dw_reduce_parts_k(
    const T* __restrict__ parts, T* __restrict__ gstack, long n,
    long w_off, long tile, int nchunk) {
  const long l = blockIdx.z;
  const T* base = parts + l * (long)nchunk * tile;
  for (long e = blockIdx.x * (long)blockDim.x + threadIdx.x; e < tile;
       e += (long)gridDim.x * blockDim.x) {
    T sum = T(0);
    for (int c = 0; c < nchunk; ++c) sum += base[c * tile + e];
    gstack[l * n + w_off + e] = sum;
  }
}

}  // namespace gmfma
