// Common device helpers for the CDNA4 (gfx950 / MI355X) kernel library.
//
// Conventions:
//  * wave size is 64 (CDNA) — never 32;
//  * all kernels are templated on the element type T in {float, double};
//  * "stacked" tensors batch the L node replicas of one rank:
//    parameter stacks are [L, n] row-major, activations [L*B, F];
//  * activation tags are compile-time template parameters so the
//    activation fuses into the producing GEMM/conv kernel (HBM-bound
//    elementwise passes are never separate launches).
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// activation tags (keep in sync with ops/functional.py ACT_IDS)
enum ActKind : int {
  ACT_NONE = 0,
  ACT_RELU = 1,
  ACT_SIN_RELU = 2,  // relu(sin(scale*z)) — FourierNet encode (see
                     // models/fourier.py; reference fourier_nn.py:44-58)
  ACT_SIGMOID = 3,
  ACT_TANH = 4,
  ACT_LOGSOFTMAX = 5,  // handled by a dedicated row kernel, not act_fwd
};

template <typename T>
DEV_INLINE T act_fwd(int kind, T z, T scale) {
  switch (kind) {
    case ACT_RELU: return z > T(0) ? z : T(0);
    case ACT_SIN_RELU: {
      T s = ::sin(scale * z);
      return s > T(0) ? s : T(0);
    }
    case ACT_SIGMOID: return T(1) / (T(1) + ::exp(-z));
    case ACT_TANH: return ::tanh(z);
    default: return z;
  }
}

// derivative wrt z given the pre-activation z and the activation output y
template <typename T>
DEV_INLINE T act_bwd(int kind, T z, T y, T scale) {
  switch (kind) {
    case ACT_RELU: return y > T(0) ? T(1) : T(0);
    case ACT_SIN_RELU: {
      T s = ::sin(scale * z);
      return s > T(0) ? scale * ::cos(scale * z) : T(0);
    }
    case ACT_SIGMOID: return y * (T(1) - y);
    case ACT_TANH: return T(1) - y * y;
    default: return T(1);
  }
}

template <typename T>
DEV_INLINE T wave_reduce_sum(T v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    v += __shfl_down(v, off, WAVE);
  }
  return v;
}

template <typename T>
DEV_INLINE T wave_reduce_max(T v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    T o = __shfl_down(v, off, WAVE);
    v = o > v ? o : v;
  }
  return v;
}

// block-level reduction helper (block size multiple of 64, <= 1024)
template <typename T, int BLOCK>
DEV_INLINE T block_reduce_sum(T v, T* lds_scratch /* [BLOCK/WAVE] */) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds_scratch[wid] = v;
  __syncthreads();
  if (wid == 0) {
    v = (lane < BLOCK / WAVE) ? lds_scratch[lane] : T(0);
    v = wave_reduce_sum(v);
  }
  return v;  // valid in wave 0 lane 0
}

#define HIP_CHECK_LAST()                                            \
  do {                                                              \
    hipError_t err_ = hipGetLastError();                            \
    if (err_ != hipSuccess) {                                       \
      TORCH_CHECK(false, "HIP kernel launch failed: ",              \
                  hipGetErrorString(err_));                         \
    }                                                               \
  } while (0)
