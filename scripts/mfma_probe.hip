// MFMA operand-layout probe for gfx950 (standalone, no torch).
// Tests candidate lane->element mappings for A/B fragments of
// v_mfma_f32_16x16x32_bf16 and v_mfma_f32_32x32x16_bf16 against a CPU
// reference GEMM with random asymmetric matrices (guide G9: transpose-
// detecting check).  C/D layouts are the guide's verified ones.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

using bf16_t = __hip_bfloat16;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// ---- 16x16x32: D[16,16] = A[16,32] @ B[32,16] -----------------------------
// candidate A layouts: lane l, elem j (0..7)
//   LA0: row = l%16, k = (l/16)*8 + j            (contiguous K block)
//   LA1: row = l%16, k = (l/16)*4 + (j%4) + 16*(j/4)  (CDNA3-style split)
// candidate B layouts (B is [32,16] k-major conceptually):
//   LB0: col = l%16, k = (l/16)*8 + j
//   LB1: col = l%16, k = (l/16)*4 + (j%4) + 16*(j/4)
// C/D (verified, guide §3): col = lane&15, row = (lane>>4)*4 + reg
template <int LA, int LB>
__global__ void probe16(const bf16_t* A, const bf16_t* B, float* D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    int ar = l % 16;
    int ak = (LA == 0) ? (l / 16) * 8 + j : (l / 16) * 4 + (j % 4) + 16 * (j / 4);
    a[j] = __hip_bfloat16_raw(A[ar * 32 + ak]).x;
    int bc = l % 16;
    int bk = (LB == 0) ? (l / 16) * 8 + j : (l / 16) * 4 + (j % 4) + 16 * (j / 4);
    b[j] = __hip_bfloat16_raw(B[bk * 16 + bc]).x;
  }
  f32x4 c = {0, 0, 0, 0};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 4; ++r) {
    int row = (l >> 4) * 4 + r;
    int col = l & 15;
    D[row * 16 + col] = c[r];
  }
}

// ---- 32x32x16: D[32,32] = A[32,16] @ B[16,32] -----------------------------
// candidate A layouts: lane l, elem j (0..7)
//   LA0: row = l%32, k = (l/32)*8 + j
//   LA1: row = l%32, k = (l/32)*4 + (j%4) + 8*(j/4)
// C/D (verified): col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
template <int LA, int LB>
__global__ void probe32(const bf16_t* A, const bf16_t* B, float* D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    int ar = l % 32;
    int ak = (LA == 0) ? (l / 32) * 8 + j : (l / 32) * 4 + (j % 4) + 8 * (j / 4);
    a[j] = __hip_bfloat16_raw(A[ar * 16 + ak]).x;
    int bc = l % 32;
    int bk = (LB == 0) ? (l / 32) * 8 + j : (l / 32) * 4 + (j % 4) + 8 * (j / 4);
    b[j] = __hip_bfloat16_raw(B[bk * 32 + bc]).x;
  }
  f32x16 c;
  for (int i = 0; i < 16; ++i) c[i] = 0.f;
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    int col = l & 31;
    D[row * 32 + col] = c[r];
  }
}

static float frand() { return (float)(rand() % 1000 - 500) / 250.f; }

template <typename LaunchFn>
bool check(const char* name, int M, int N, int K, LaunchFn launch) {
  bf16_t *dA, *dB;
  float* dD;
  hipMalloc(&dA, M * K * sizeof(bf16_t));
  hipMalloc(&dB, K * N * sizeof(bf16_t));
  hipMalloc(&dD, M * N * sizeof(float));
  bf16_t* hA = (bf16_t*)malloc(M * K * sizeof(bf16_t));
  bf16_t* hB = (bf16_t*)malloc(K * N * sizeof(bf16_t));
  float* hD = (float*)malloc(M * N * sizeof(float));
  float* ref = (float*)calloc(M * N, sizeof(float));
  for (int i = 0; i < M * K; ++i) hA[i] = __float2bfloat16(frand());
  for (int i = 0; i < K * N; ++i) hB[i] = __float2bfloat16(frand());
  for (int m = 0; m < M; ++m)
    for (int k = 0; k < K; ++k) {
      float av = __bfloat162float(hA[m * K + k]);
      for (int n = 0; n < N; ++n)
        ref[m * N + n] += av * __bfloat162float(hB[k * N + n]);
    }
  hipMemcpy(dA, hA, M * K * sizeof(bf16_t), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, K * N * sizeof(bf16_t), hipMemcpyHostToDevice);
  launch(dA, dB, dD);
  hipDeviceSynchronize();
  hipMemcpy(hD, dD, M * N * sizeof(float), hipMemcpyDeviceToHost);
  float maxd = 0;
  for (int i = 0; i < M * N; ++i)
    maxd = fmaxf(maxd, fabsf(hD[i] - ref[i]));
  printf("%s: maxdiff %f %s\n", name, maxd, maxd < 0.1f ? "MATCH" : "no");
  hipFree(dA); hipFree(dB); hipFree(dD);
  free(hA); free(hB); free(hD); free(ref);
  return maxd < 0.1f;
}

int main() {
  srand(7);
  check("16x16x32 LA0/LB0", 16, 16, 32, [](bf16_t* a, bf16_t* b, float* d) {
    hipLaunchKernelGGL((probe16<0, 0>), 1, 64, 0, 0, a, b, d); });
  check("16x16x32 LA0/LB1", 16, 16, 32, [](bf16_t* a, bf16_t* b, float* d) {
    hipLaunchKernelGGL((probe16<0, 1>), 1, 64, 0, 0, a, b, d); });
  check("16x16x32 LA1/LB0", 16, 16, 32, [](bf16_t* a, bf16_t* b, float* d) {
    hipLaunchKernelGGL((probe16<1, 0>), 1, 64, 0, 0, a, b, d); });
  check("16x16x32 LA1/LB1", 16, 16, 32, [](bf16_t* a, bf16_t* b, float* d) {
    hipLaunchKernelGGL((probe16<1, 1>), 1, 64, 0, 0, a, b, d); });
  check("32x32x16 LA0/LB0", 32, 32, 16, [](bf16_t* a, bf16_t* b, float* d) {
    hipLaunchKernelGGL((probe32<0, 0>), 1, 64, 0, 0, a, b, d); });
  check("32x32x16 LA0/LB1", 32, 32, 16, [](bf16_t* a, bf16_t* b, float* d) {
    hipLaunchKernelGGL((probe32<0, 1>), 1, 64, 0, 0, a, b, d); });
  check("32x32x16 LA1/LB0", 32, 32, 16, [](bf16_t* a, bf16_t* b, float* d) {
    hipLaunchKernelGGL((probe32<1, 0>), 1, 64, 0, 0, a, b, d); });
  check("32x32x16 LA1/LB1", 32, 32, 16, [](bf16_t* a, bf16_t* b, float* d) {
    hipLaunchKernelGGL((probe32<1, 1>), 1, 64, 0, 0, a, b, d); });
  return 0;
}
