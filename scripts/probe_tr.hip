// Probe ds_read_b64_tr_b16 semantics on gfx950: fill LDS with the element
// index, have each lane read one tr16_b64 at addr = base + lane*8B, and
// dump the 4 returned u16 per lane.  A second variant reads at a uniform
// address.  Build & run:  hipcc --offload-arch=gfx950 probe_tr.hip -o p && ./p
#include <hip/hip_runtime.h>
#include <stdio.h>

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

__global__ void probe(unsigned short* out, int mode) {
  __shared__ unsigned short lds[2048];
  for (int i = threadIdx.x; i < 2048; i += blockDim.x)
    lds[i] = (unsigned short)i;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  auto* p = (__attribute__((address_space(3))) bf16x4*)
      ((__attribute__((address_space(3))) char*)lds
       + (mode == 0 ? lane * 8 : 64));
  bf16x4 r = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  unsigned short u[4];
  __builtin_memcpy(u, &r, 8);
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = u[j];
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 64 * 4 * 2);
  for (int mode = 0; mode < 2; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    unsigned short h[256];
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("mode %d (addr = %s):\n", mode, mode ? "uniform 128B" : "lane*8B");
    for (int l = 0; l < 64; ++l) {
      printf("lane %2d: %4d %4d %4d %4d\n", l, h[l * 4], h[l * 4 + 1],
             h[l * 4 + 2], h[l * 4 + 3]);
    }
  }
  return 0;
}
