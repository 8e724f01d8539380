// Probe v_mfma_scale_f32_32x32x64_f8f6f4 semantics on gfx950:
//  - operand fragment layout (assumed: lane l holds A[row=l&31][k=32*(l>>5)+j],
//    byte j of the 32-byte register block; same for B with col=l&31)
//  - C/D layout (assumed 32x32 crow map like the bf16 MFMAs)
//  - e8m0 scale operand: byte 127 = 2^0; doubling a scale doubles the row.
// Build & run: hipcc --offload-arch=gfx950 probe_mx.hip -o p && ./p
#include <hip/hip_runtime.h>
#include <stdio.h>
#include <stdint.h>

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// fp8 e4m3fn encode for small exact ints (|v| <= 8)
static uint8_t f8(float v) {
  union { float f; uint32_t u; } x{v};
  if (v == 0.0f) return 0;
  uint32_t sgn = (x.u >> 31) << 7;
  int exp = ((x.u >> 23) & 255) - 127;
  uint32_t man = (x.u >> 20) & 7;      // top 3 mantissa bits
  return (uint8_t)(sgn | ((exp + 7) << 3) | man);
}

__global__ void probe(const uint8_t* A, const uint8_t* B, float* D,
                      int sa, int sb) {
  const int lane = threadIdx.x;
  i32x8 a, b;
  uint8_t ab[32], bb[32];
  for (int j = 0; j < 32; ++j) {
    ab[j] = A[(lane & 31) * 64 + 32 * (lane >> 5) + j];
    bb[j] = B[(lane & 31) * 64 + 32 * (lane >> 5) + j];
  }
  __builtin_memcpy(&a, ab, 32);
  __builtin_memcpy(&b, bb, 32);
  f32x16 c = {};
  c = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, c, 0, 0, 0, sa,
                                                      0, sb);
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
    const int col = lane & 31;
    D[row * 32 + col] = c[r];
  }
}

int main() {
  static uint8_t hA[32 * 64], hB[32 * 64];
  static float ref[32][32];
  srand(7);
  for (int i = 0; i < 32; ++i)
    for (int k = 0; k < 64; ++k) {
      hA[i * 64 + k] = f8((float)(rand() % 9 - 4));
      hB[i * 64 + k] = f8((float)(rand() % 9 - 4));
    }
  auto dec = [](uint8_t u) -> float {
    if ((u & 127) == 0) return 0.0f;
    int exp = ((u >> 3) & 15) - 7;
    float m = 1.0f + (u & 7) / 8.0f;
    float v = ldexpf(m, exp);
    return (u & 128) ? -v : v;
  };
  for (int i = 0; i < 32; ++i)
    for (int j = 0; j < 32; ++j) {
      float s = 0;
      for (int k = 0; k < 64; ++k)
        s += dec(hA[i * 64 + k]) * dec(hB[j * 64 + k]);
      ref[i][j] = s;
    }
  uint8_t *dA, *dB;
  float* dD;
  hipMalloc(&dA, sizeof(hA));
  hipMalloc(&dB, sizeof(hB));
  hipMalloc(&dD, 32 * 32 * 4);
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  for (int t = 0; t < 3; ++t) {
    int sa = t == 1 ? 128 : 127;   // 2^1 vs 2^0
    int sb = t == 2 ? 126 : 127;   // 2^-1
    float mult = (t == 0) ? 1.0f : (t == 1 ? 2.0f : 0.5f);
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB, dD, sa, sb);
    static float hD[32 * 32];
    hipMemcpy(hD, dD, sizeof(hD), hipMemcpyDeviceToHost);
    int bad = 0;
    for (int i = 0; i < 32 && bad < 6; ++i)
      for (int j = 0; j < 32; ++j)
        if (hD[i * 32 + j] != ref[i][j] * mult) {
          if (bad < 6)
            printf("t%d [%d][%d] got %f want %f\n", t, i, j, hD[i * 32 + j],
                   ref[i][j] * mult);
          ++bad;
        }
    printf("test %d (mult %.1f): %s (%d bad)\n", t, mult,
           bad ? "FAIL" : "PASS", bad);
  }
  return 0;
}
