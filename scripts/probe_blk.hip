// Unit-probe of the blocked-image staging + tr16 fragment read used in
// attention.hip: stage a known [32][DP] tile, read B-fragments, dump.
// hipcc --offload-arch=gfx950 probe_blk.hip -o p && ./p
#include <hip/hip_runtime.h>
#include <stdio.h>
#include <stdint.h>

#define DP 160
#define TS 32
typedef __attribute__((ext_vector_type(8))) uint16_t ushortx8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((address_space(3))) bf16x4 lds_bf16x4;

__device__ __forceinline__ int blk_off(int idx, int d) {
  return (idx >> 2) * (DP * 4) + ((d >> 4) << 6) + ((idx & 3) << 4)
         + (d & 15);
}

// out[lane][n][s][j] = element j of the fragment (expected V[16s+8hi+j][32n+(l&31)])
__global__ void probe(const uint16_t* V, uint16_t* out) {
  __shared__ uint16_t img[TS * DP];
  const int t = threadIdx.x;   // 256 threads
  #pragma unroll
  for (int i = 0; i < 3; ++i) {
    const int c = t + i * 256;
    if (c < TS * DP / 8) {
      const int row = c / (DP / 8);
      const int d0 = (c - row * (DP / 8)) * 8;
      ushortx8 v = *reinterpret_cast<const ushortx8*>(V + row * DP + d0);
      *reinterpret_cast<ushortx8*>(img + blk_off(row, d0)) = v;
    }
  }
  __syncthreads();
  if (t >= 64) return;
  const int lane = t;
  for (int n = 0; n < DP / 32; ++n)
    for (int s = 0; s < 2; ++s) {
      const int d = n * 32 + (lane & 31), hi = lane >> 5;
      const int idx0 = 16 * s + 8 * hi;
      auto raddr = [&](int i0) {
        return (i0 >> 2) * (DP * 4) + ((d >> 4) << 6)
               + (((d >> 2) & 3) << 4) + ((d & 3) << 2);
      };
      bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
          (lds_bf16x4*)(img + raddr(idx0)));
      bf16x4 h4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
          (lds_bf16x4*)(img + raddr(idx0 + 4)));
      uint16_t u[8];
      __builtin_memcpy(u, &lo, 8);
      __builtin_memcpy(u + 4, &h4, 8);
      for (int j = 0; j < 8; ++j)
        out[((lane * (DP / 32) + n) * 2 + s) * 8 + j] = u[j];
    }
}

int main() {
  uint16_t *dv, *dout;
  uint16_t hv[TS * DP];
  for (int i = 0; i < TS; ++i)
    for (int d = 0; d < DP; ++d) hv[i * DP + d] = (uint16_t)(i * 256 + d);
  hipMalloc(&dv, sizeof(hv));
  hipMalloc(&dout, 64 * (DP / 32) * 2 * 8 * 2);
  hipMemcpy(dv, hv, sizeof(hv), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3(1), dim3(256), 0, 0, dv, dout);
  static uint16_t hout[64 * (DP / 32) * 2 * 8];
  hipMemcpy(hout, dout, sizeof(hout), hipMemcpyDeviceToHost);
  int bad = 0;
  for (int lane = 0; lane < 64; ++lane)
    for (int n = 0; n < DP / 32; ++n)
      for (int s = 0; s < 2; ++s)
        for (int j = 0; j < 8; ++j) {
          const int d = n * 32 + (lane & 31), hi = lane >> 5;
          const int idx = 16 * s + 8 * hi + j;
          const uint16_t want = (uint16_t)(idx * 256 + d);
          const uint16_t got = hout[((lane * (DP / 32) + n) * 2 + s) * 8 + j];
          if (want != got && bad < 20) {
            printf("lane %d n %d s %d j %d: want idx%d d%d got idx%d d%d\n",
                   lane, n, s, j, idx, d, got >> 8, got & 255);
            ++bad;
          } else if (want != got) ++bad;
        }
  printf(bad ? "FAIL %d mismatches\n" : "PASS\n", bad);
  return 0;
}
