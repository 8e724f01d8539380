// Common CDNA4 (gfx950) kernel utilities for the LuminaAI-AMD framework.
// Hand-written HIP, MI355X-first: 64-wide wavefronts, bf16x8 vector loads,
// fp32 internal math. No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// ---- vector types for 16-byte loads -------------------------------------
typedef __attribute__((ext_vector_type(4))) float    floatx4;
typedef __attribute__((ext_vector_type(4))) uint32_t uintx4;
typedef __attribute__((ext_vector_type(8))) uint16_t ushortx8;

// ---- dtype conversion helpers -------------------------------------------
DEV_INLINE float bf16_to_f32(uint16_t u) {
  union { uint32_t i; float f; } v;
  v.i = ((uint32_t)u) << 16;
  return v.f;
}

DEV_INLINE uint16_t f32_to_bf16(float f) {
  union { float f; uint32_t i; } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t lsb = (v.i >> 16) & 1;
  v.i += 0x7fff + lsb;
  return (uint16_t)(v.i >> 16);
}

// generic element accessors so kernels template over T in {float, bf16-as-u16}
struct F32Elem {
  using storage = float;
  static DEV_INLINE float load(const float* p) { return *p; }
  static DEV_INLINE void store(float* p, float v) { *p = v; }
};
struct BF16Elem {
  using storage = uint16_t;
  static DEV_INLINE float load(const uint16_t* p) { return bf16_to_f32(*p); }
  static DEV_INLINE void store(uint16_t* p, float v) { *p = f32_to_bf16(v); }
};

// ---- wave + block reductions --------------------------------------------
DEV_INLINE float wave_reduce_sum(float v) {
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, WAVE);
  return v;  // valid in lane 0
}

DEV_INLINE float wave_reduce_max(float v) {
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  return v;
}

// Block reduction over up to 16 waves; smem must hold >= 16 floats.
// Returns the result broadcast to all threads.
DEV_INLINE float block_reduce_sum(float v, float* smem) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid  = threadIdx.x / WAVE;
  const int nw   = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nw) ? smem[threadIdx.x] : 0.0f;
  if (wid == 0) {
    r = wave_reduce_sum(r);
    if (lane == 0) smem[0] = r;
  }
  __syncthreads();
  r = smem[0];
  __syncthreads();
  return r;
}

DEV_INLINE float block_reduce_max(float v, float* smem) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid  = threadIdx.x / WAVE;
  const int nw   = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nw) ? smem[threadIdx.x] : -INFINITY;
  if (wid == 0) {
    r = wave_reduce_max(r);
    if (lane == 0) smem[0] = r;
  }
  __syncthreads();
  r = smem[0];
  __syncthreads();
  return r;
}

// ---- grid sizing ---------------------------------------------------------
// memory-bound ops: cap the grid and grid-stride the rest (guide G11)
DEV_INLINE int64_t gstride() { return (int64_t)gridDim.x * blockDim.x; }

static inline int elementwise_grid(int64_t n, int block, int per_thread = 8) {
  int64_t want = (n + (int64_t)block * per_thread - 1) / ((int64_t)block * per_thread);
  int64_t cap = 2048;
  return (int)(want < cap ? (want < 1 ? 1 : want) : cap);
}

#define HIP_CHECK_LAST()                                                   \
  do {                                                                     \
    hipError_t e_ = hipGetLastError();                                     \
    if (e_ != hipSuccess) return e_;                                       \
  } while (0)
