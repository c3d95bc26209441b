// Empirical MFMA fragment-layout probe for gfx950 f64/f32 16x16x4.
// Computes D = A*B with distinctive A, B loaded per a PARAMETRIC lane
// mapping, then host-side solves which (i,j) each (lane, reg) holds.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>

typedef double f64x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__global__ void probe64(const double* A, const double* B, double* D) {
  int l = threadIdx.x;            // 64 lanes
  // assumed input maps: a = A[i=l&15][k=l>>4], b = B[k=l>>4][j=l&15]
  double a = A[(l & 15) * 4 + (l >> 4)];
  double b = B[(l >> 4) * 16 + (l & 15)];
  f64x4 acc{};
  acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) D[l * 4 + r] = acc[r];
}

__global__ void probe32(const float* A, const float* B, float* D) {
  int l = threadIdx.x;
  float a = A[(l & 15) * 4 + (l >> 4)];
  float b = B[(l >> 4) * 16 + (l & 15)];
  f32x4 acc{};
  acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) D[l * 4 + r] = acc[r];
}

int main() {
  double A[16 * 4], B[4 * 16], Dref[16 * 16];
  for (int i = 0; i < 16; ++i)
    for (int k = 0; k < 4; ++k) A[i * 4 + k] = 1 + i + 40 * k;
  for (int k = 0; k < 4; ++k)
    for (int j = 0; j < 16; ++j) B[k * 16 + j] = 1 + 3 * j + 1000 * k;
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      double s = 0;
      for (int k = 0; k < 4; ++k) s += A[i * 4 + k] * B[k * 16 + j];
      Dref[i * 16 + j] = s;
    }
  double *dA, *dB, *dD, D[256];
  hipMalloc(&dA, sizeof(A)); hipMalloc(&dB, sizeof(B));
  hipMalloc(&dD, sizeof(D));
  hipMemcpy(dA, A, sizeof(A), hipMemcpyHostToDevice);
  hipMemcpy(dB, B, sizeof(B), hipMemcpyHostToDevice);
  probe64<<<1, 64>>>(dA, dB, dD);
  hipMemcpy(D, dD, sizeof(D), hipMemcpyDeviceToHost);
  printf("f64 16x16x4 mapping (lane,reg) -> (i,j):\n");
  for (int l = 0; l < 64; l += 16) {   // representative lanes
    for (int r = 0; r < 4; ++r) {
      int fi = -1, fj = -1;
      for (int i = 0; i < 16 && fi < 0; ++i)
        for (int j = 0; j < 16; ++j)
          if (fabs(Dref[i * 16 + j] - D[l * 4 + r]) < 1e-9) {
            fi = i; fj = j; break;
          }
      printf("  l=%2d r=%d -> i=%2d j=%2d (v=%.0f)\n", l, r, fi, fj,
             D[l * 4 + r]);
    }
  }
  // also lanes 1 and 17 to pin the j mapping
  for (int l : {1, 17, 33}) {
    for (int r = 0; r < 4; ++r) {
      int fi = -1, fj = -1;
      for (int i = 0; i < 16 && fi < 0; ++i)
        for (int j = 0; j < 16; ++j)
          if (fabs(Dref[i * 16 + j] - D[l * 4 + r]) < 1e-9) {
            fi = i; fj = j; break;
          }
      printf("  l=%2d r=%d -> i=%2d j=%2d\n", l, r, fi, fj);
    }
  }

  float Af[64], Bf[64], Df[256];
  for (int t = 0; t < 64; ++t) { Af[t] = A[t]; Bf[t] = B[t]; }
  float *fA, *fB, *fD;
  hipMalloc(&fA, sizeof(Af)); hipMalloc(&fB, sizeof(Bf));
  hipMalloc(&fD, sizeof(Df));
  hipMemcpy(fA, Af, sizeof(Af), hipMemcpyHostToDevice);
  hipMemcpy(fB, Bf, sizeof(Bf), hipMemcpyHostToDevice);
  probe32<<<1, 64>>>(fA, fB, fD);
  hipMemcpy(Df, fD, sizeof(Df), hipMemcpyDeviceToHost);
  printf("f32 16x16x4 mapping:\n");
  for (int l : {0, 1, 16, 17, 32, 48}) {
    for (int r = 0; r < 4; ++r) {
      int fi = -1, fj = -1;
      for (int i = 0; i < 16 && fi < 0; ++i)
        for (int j = 0; j < 16; ++j)
          if (fabs(Dref[i * 16 + j] - Df[l * 4 + r]) < 1e-3) {
            fi = i; fj = j; break;
          }
      printf("  l=%2d r=%d -> i=%2d j=%2d\n", l, r, fi, fj);
    }
  }
  return 0;
}
