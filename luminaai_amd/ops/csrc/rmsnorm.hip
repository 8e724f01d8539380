// RMSNorm forward + backward for CDNA4 (gfx950).
//
// MI355X-native design (replaces the semantics of the reference's
// rms_norm_kernel_optimized, /root/reference/Src/Main_Scripts/core/transformer_ops.cu:61-123,
// which was fp32-only and forward-only):
//  - bf16 in/out with fp32 internal math (and an fp32 instantiation),
//  - vector loads picked by ROW ALIGNMENT: 16-byte when every row base is
//    16B-aligned (H*sizeof(T) % 16 == 0), else 8-byte, else scalar — hidden
//    sizes like 1908 give rows aligned only to 8 bytes, and a misaligned
//    dwordx4 load is a memory fault on CDNA4;
//  - one workgroup (256 threads = 4 waves) per token row, wave shuffle + LDS
//    cross-wave reduction;
//  - true fused backward: dx in the same pass as the dw partial accumulation
//    (LDS fp32 dw tile per block, one atomicAdd sweep per block at the end).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) uint16_t ushortx4;
typedef __attribute__((ext_vector_type(2))) float    floatx2;

// ---- vector sum-of-squares / loads, width-dispatched ---------------------
template <typename E, int BLOCK>
DEV_INLINE float row_sumsq(const typename E::storage* __restrict__ xr, int H,
                           int vec) {
  float ss = 0.f;
  if (sizeof(typename E::storage) == 2) {
    if (vec == 8) {
      const ushortx8* xv = reinterpret_cast<const ushortx8*>(xr);
      const int nv = H / 8;
      for (int i = threadIdx.x; i < nv; i += BLOCK) {
        ushortx8 v = xv[i];
        #pragma unroll
        for (int j = 0; j < 8; ++j) { float f = bf16_to_f32(v[j]); ss += f * f; }
      }
      return ss;
    }
    if (vec == 4) {
      const ushortx4* xv = reinterpret_cast<const ushortx4*>(xr);
      const int nv = H / 4;
      for (int i = threadIdx.x; i < nv; i += BLOCK) {
        ushortx4 v = xv[i];
        #pragma unroll
        for (int j = 0; j < 4; ++j) { float f = bf16_to_f32(v[j]); ss += f * f; }
      }
      return ss;
    }
  } else {
    if (vec >= 4) {
      const floatx4* xv = reinterpret_cast<const floatx4*>(xr);
      const int nv = H / 4;
      for (int i = threadIdx.x; i < nv; i += BLOCK) {
        floatx4 v = xv[i];
        #pragma unroll
        for (int j = 0; j < 4; ++j) ss += v[j] * v[j];
      }
      return ss;
    }
  }
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    float f = E::load(xr + i);
    ss += f * f;
  }
  return ss;
}

// ---------------------------------------------------------------- forward
template <typename E, int BLOCK>
__global__ void rmsnorm_fwd_kernel(const typename E::storage* __restrict__ x,
                                   const typename E::storage* __restrict__ w,
                                   typename E::storage* __restrict__ y,
                                   float* __restrict__ invrms,
                                   int H, float eps, int vec) {
  __shared__ float red[16];
  const int64_t row = blockIdx.x;
  const typename E::storage* xr = x + row * (int64_t)H;
  typename E::storage* yr = y + row * (int64_t)H;

  float ss = row_sumsq<E, BLOCK>(xr, H, vec);
  ss = block_reduce_sum(ss, red);
  const float inv = rsqrtf(ss / (float)H + eps);
  if (threadIdx.x == 0 && invrms) invrms[row] = inv;

  for (int i = threadIdx.x; i < H; i += BLOCK) {
    float f = E::load(xr + i) * inv * E::load(w + i);
    E::store(yr + i, f);
  }
}

// ---------------------------------------------------------------- backward
//   dx_i = inv * (g_i*w_i - x_i * inv^2 * mean_j(g_j*w_j*x_j))
//   dw_i += g_i * x_i * inv        (accumulated across rows)
// Each block owns a strided set of rows; dw partial lives in dynamic LDS
// fp32[H] and is flushed with one atomicAdd per element at block end.
template <typename E, int BLOCK, bool USE_LDS_DW>
__global__ void rmsnorm_bwd_kernel(const typename E::storage* __restrict__ gy,
                                   const typename E::storage* __restrict__ x,
                                   const typename E::storage* __restrict__ w,
                                   const float* __restrict__ invrms,
                                   typename E::storage* __restrict__ dx,
                                   float* __restrict__ dw,  // [H] fp32, zeroed
                                   int64_t N, int H) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* dw_tile = reinterpret_cast<float*>(smem_raw);  // [H] when USE_LDS_DW
  __shared__ float red[16];

  if (USE_LDS_DW) {
    for (int i = threadIdx.x; i < H; i += BLOCK) dw_tile[i] = 0.f;
    __syncthreads();
  }

  for (int64_t row = blockIdx.x; row < N; row += gridDim.x) {
    const typename E::storage* xr = x + row * (int64_t)H;
    const typename E::storage* gr = gy + row * (int64_t)H;
    typename E::storage* dxr = dx + row * (int64_t)H;
    const float inv = invrms[row];

    float dot = 0.f;
    for (int i = threadIdx.x; i < H; i += BLOCK)
      dot += E::load(gr + i) * E::load(w + i) * E::load(xr + i);
    dot = block_reduce_sum(dot, red) / (float)H;

    for (int i = threadIdx.x; i < H; i += BLOCK) {
      float xi = E::load(xr + i);
      float gi = E::load(gr + i);
      float wi = E::load(w + i);
      E::store(dxr + i, inv * (gi * wi - xi * inv * inv * dot));
      if (USE_LDS_DW) dw_tile[i] += gi * xi * inv;
      else atomicAdd(dw + i, gi * xi * inv);
    }
    if (USE_LDS_DW) __syncthreads();
  }

  if (USE_LDS_DW) {
    for (int i = threadIdx.x; i < H; i += BLOCK)
      if (dw_tile[i] != 0.f) atomicAdd(dw + i, dw_tile[i]);
  }
}

// ---------------------------------------------------------------- launchers
static inline int pick_vec(int H, int elem_size) {
  // every row base must be aligned to the vector width
  const int row_bytes = H * elem_size;
  if (elem_size == 2) {
    if (H % 8 == 0 && row_bytes % 16 == 0) return 8;
    if (H % 4 == 0 && row_bytes % 8 == 0) return 4;
    return 1;
  }
  if (H % 4 == 0) return 4;  // 16 B rows for fp32 when H%4==0
  return 1;
}

extern "C" {

hipError_t lumina_rmsnorm_fwd_bf16(const void* x, const void* w, void* y,
                                   float* invrms, int64_t N, int H, float eps,
                                   hipStream_t s) {
  constexpr int B = 256;
  rmsnorm_fwd_kernel<BF16Elem, B><<<(uint32_t)N, B, 0, s>>>(
      (const uint16_t*)x, (const uint16_t*)w, (uint16_t*)y, invrms, H, eps,
      pick_vec(H, 2));
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lumina_rmsnorm_fwd_f32(const void* x, const void* w, void* y,
                                  float* invrms, int64_t N, int H, float eps,
                                  hipStream_t s) {
  constexpr int B = 256;
  rmsnorm_fwd_kernel<F32Elem, B><<<(uint32_t)N, B, 0, s>>>(
      (const float*)x, (const float*)w, (float*)y, invrms, H, eps,
      pick_vec(H, 4));
  HIP_CHECK_LAST();
  return hipSuccess;
}

static inline int bwd_grid(int64_t N) {
  int g = (int)(N < 1024 ? N : 1024);
  return g < 1 ? 1 : g;
}

hipError_t lumina_rmsnorm_bwd_bf16(const void* gy, const void* x, const void* w,
                                   const float* invrms, void* dx, float* dw,
                                   int64_t N, int H, hipStream_t s) {
  constexpr int B = 256;
  size_t lds = (size_t)H * sizeof(float);
  if (lds <= 64 * 1024) {
    rmsnorm_bwd_kernel<BF16Elem, B, true><<<bwd_grid(N), B, lds, s>>>(
        (const uint16_t*)gy, (const uint16_t*)x, (const uint16_t*)w, invrms,
        (uint16_t*)dx, dw, N, H);
  } else {
    rmsnorm_bwd_kernel<BF16Elem, B, false><<<bwd_grid(N), B, 0, s>>>(
        (const uint16_t*)gy, (const uint16_t*)x, (const uint16_t*)w, invrms,
        (uint16_t*)dx, dw, N, H);
  }
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lumina_rmsnorm_bwd_f32(const void* gy, const void* x, const void* w,
                                  const float* invrms, void* dx, float* dw,
                                  int64_t N, int H, hipStream_t s) {
  constexpr int B = 256;
  size_t lds = (size_t)H * sizeof(float);
  if (lds <= 64 * 1024) {
    rmsnorm_bwd_kernel<F32Elem, B, true><<<bwd_grid(N), B, lds, s>>>(
        (const float*)gy, (const float*)x, (const float*)w, invrms,
        (float*)dx, dw, N, H);
  } else {
    rmsnorm_bwd_kernel<F32Elem, B, false><<<bwd_grid(N), B, 0, s>>>(
        (const float*)gy, (const float*)x, (const float*)w, invrms,
        (float*)dx, dw, N, H);
  }
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
