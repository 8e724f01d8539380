// Probe: v_mfma_f32_16x16x32_bf16 A/B fragment lane mapping.
// Hypothesis: A[16x32]: lane l holds row = l&15, k = 8*(l>>4)+j (j=0..8)
//             B[32x16]: lane l holds col = l&15, k = 8*(l>>4)+j
//             C/D: col = lane&15, row = (lane>>4)*4 + reg.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void probe(const __bf16* A, const __bf16* B, float* C) {
  const int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(l & 15) * 32 + 8 * (l >> 4) + j];
    b[j] = B[(l & 15) * 32 + 8 * (l >> 4) + j];   // B^T rows = cols of B
  }
  f32x4 c = {};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  for (int r = 0; r < 4; ++r)
    C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

int main() {
  __bf16 *A, *B; float* C;
  hipMallocManaged(&A, 16 * 32 * sizeof(__bf16));
  hipMallocManaged(&B, 16 * 32 * sizeof(__bf16));
  hipMallocManaged(&C, 256 * sizeof(float));
  float fa[16][32], fb[16][32];
  for (int i = 0; i < 16; ++i)
    for (int k = 0; k < 32; ++k) {
      fa[i][k] = (float)((i * 31 + k * 7) % 13) - 6.0f;
      fb[i][k] = (float)((i * 17 + k * 5) % 11) - 5.0f;  // fb[col][k] = B^T
      A[i * 32 + k] = (__bf16)fa[i][k];
      B[i * 32 + k] = (__bf16)fb[i][k];
    }
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, A, B, C);
  hipDeviceSynchronize();
  int bad = 0;
  for (int r = 0; r < 16; ++r)
    for (int c = 0; c < 16; ++c) {
      float want = 0;
      for (int k = 0; k < 32; ++k) want += fa[r][k] * fb[c][k];
      if (fabsf(C[r * 16 + c] - want) > 1e-3f && bad++ < 5)
        printf("MISMATCH r%d c%d got %f want %f\n", r, c, C[r * 16 + c], want);
    }
  printf(bad ? "FAIL %d\n" : "MAPPING OK\n", bad);
  return 0;
}
