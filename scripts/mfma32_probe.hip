// Verify mfma_f32_32x32x16_bf16 A/B/C fragment layouts on gfx950 HW.
//
// Assumed (generalizing the HW-verified 16x16x32 LA0 layout and the
// guide's measured C layout):
//   A[32m x 16k]: lane l holds A[m = l&31][k = (l>>5)*8 + j], j=0..7
//   B[16k x 32n]: lane l holds B[k = (l>>5)*8 + j][n = l&31]
//   C[32m x 32n]: lane l holds C[m = (reg&3) + 8*(reg>>2) + 4*(l>>5)]
//                              [n = l&31],  reg = 0..15
// Also probes __builtin_amdgcn_permlane32_swap semantics.
//
// Build+run (GPU box):
//   hipcc --offload-arch=gfx950 -O2 scripts/mfma32_probe.hip -o /tmp/p32 && /tmp/p32
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

static float br2f(unsigned short u) {
  union { unsigned int i; float f; } c;
  c.i = ((unsigned int)u) << 16;
  return c.f;
}

__global__ void probe_kernel(const unsigned short* A, const unsigned short* B,
                             float* C) {
  const int l = threadIdx.x;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)A[(l & 31) * 16 + ((l >> 5) * 8 + j)];
    b[j] = (short)B[((l >> 5) * 8 + j) * 32 + (l & 31)];
  }
  f32x16 acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int m = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    const int n = l & 31;
    C[m * 32 + n] = acc[r];
  }
}

__global__ void permlane_kernel(int* out) {
  const int l = threadIdx.x;
  int v0 = l;          // "old"
  int v1 = 1000 + l;   // "src"
  // swap rows 0-31 of v0 with rows 32-63 of v1 (per ISA doc)
  auto pair = __builtin_amdgcn_permlane32_swap(v0, v1, false, false);
  out[l * 2 + 0] = pair[0];
  out[l * 2 + 1] = pair[1];
}

int main() {
  unsigned short *A, *B;
  float *C;
  hipMallocManaged(&A, 32 * 16 * 2);
  hipMallocManaged(&B, 16 * 32 * 2);
  hipMallocManaged(&C, 32 * 32 * 4);
  srand(7);
  auto f2br = [](float f) {
    union { float f; unsigned int i; } c;
    c.f = f;
    return (unsigned short)(c.i >> 16);
  };
  for (int i = 0; i < 32 * 16; ++i)
    A[i] = f2br((rand() % 17 - 8) * 0.25f);
  for (int i = 0; i < 16 * 32; ++i)
    B[i] = f2br((rand() % 17 - 8) * 0.25f);
  hipLaunchKernelGGL(probe_kernel, dim3(1), dim3(64), 0, 0, A, B, C);
  hipDeviceSynchronize();
  int bad = 0;
  for (int m = 0; m < 32 && bad < 5; ++m)
    for (int n = 0; n < 32 && bad < 5; ++n) {
      float ref = 0;
      for (int k = 0; k < 16; ++k)
        ref += br2f(A[m * 16 + k]) * br2f(B[k * 32 + n]);
      if (fabsf(C[m * 32 + n] - ref) > 1e-2f) {
        printf("MISMATCH m=%d n=%d got %f want %f\n", m, n, C[m * 32 + n],
               ref);
        ++bad;
      }
    }
  printf(bad ? "MFMA32 LAYOUT: FAIL\n" : "MFMA32 LAYOUT: OK\n");

  int* P;
  hipMallocManaged(&P, 64 * 2 * 4);
  hipLaunchKernelGGL(permlane_kernel, dim3(1), dim3(64), 0, 0, P);
  hipDeviceSynchronize();
  printf("permlane32_swap lane0=(%d,%d) lane1=(%d,%d) lane32=(%d,%d) "
         "lane33=(%d,%d)\n",
         P[0], P[1], P[2], P[3], P[64], P[65], P[66], P[67]);
  return bad ? 1 : 0;
}
