// Fused batch-1 decode-layer kernels for gfx950.
//
// Round-1 decode ran ~160 tok/s: the 31-layer chain of ~18 small kernels
// per layer was latency-bound (profiles/r01_SUMMARY.md "Decode path").
// The guide's launches-baseline (MI355X_MICROARCH.md) shows a decode layer
// as FIVE weight-streaming kernels with non-temporal loads and fused
// epilogues is within 13% of a full persistent engine -- this file is that
// shape:
//   K1 dec_gemv<NORM>          rmsnorm fused into the qkv / gate_up GEMV
//                              (the norm is recomputed per block: x is a
//                              few KB, the weight stream is the cost)
//   K2 dec_rope_cache          RoPE(q,k) + KV-cache append (one tiny kernel)
//   K3 dec_attn                single-token GQA attention over the cache
//   K4 dec_gemv<RESID>         o-proj + residual add
//   K5 dec_gemv<NORM|SWIGLU>   post-norm + gate_up + SiLU*up in one pass
//   K6 dec_gemv<RESID>         down-proj + residual add
// All shapes static, cursor on device -> the whole token step hipGraph-
// captures.  Weights are streamed with nt (non-temporal) loads: each CU
// reads its rows exactly once (guide row nt-weights: -5..10%/layer).

#include "common.h"

typedef const __attribute__((address_space(1))) void* dgas;

#define DEC_NORM   1
#define DEC_RESID  2
#define DEC_SWIGLU 4

DEV_INLINE float silu(float v) { return v / (1.0f + __expf(-v)); }

// ---------------------------------------------------------------------------
// y[N] = W[N(or 2N), K] @ xhat[K] (+residual) (SWIGLU: silu(g)*u over row
// pairs). xhat = rmsnorm(x)*wn when NORM, else x. 4 waves, wave per row.
template <int FLAGS>
__global__ __launch_bounds__(256)
void dec_gemv_kernel(const uint16_t* __restrict__ W,
                     const uint16_t* __restrict__ x,
                     const uint16_t* __restrict__ wn,
                     const uint16_t* __restrict__ resid,
                     uint16_t* __restrict__ y,
                     int N, int K, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);           // [16]
  uint16_t* xs = reinterpret_cast<uint16_t*>(smem + 64); // [K]

  // stage x (normed when asked) into LDS once per block
  if constexpr (FLAGS & DEC_NORM) {
    float ss = 0.0f;
    for (int k = threadIdx.x * 2; k < K; k += 512) {
      const float a = bf16_to_f32(x[k]);
      const float b = k + 1 < K ? bf16_to_f32(x[k + 1]) : 0.0f;
      ss = fmaf(a, a, fmaf(b, b, ss));
    }
    const float inv = rsqrtf(block_reduce_sum(ss, red) / K + eps);
    for (int k = threadIdx.x; k < K; k += 256)
      xs[k] = f32_to_bf16(bf16_to_f32(x[k]) * inv * bf16_to_f32(wn[k]));
  } else {
    for (int k = threadIdx.x; k < K; k += 256)
      xs[k] = x[k];
  }
  __syncthreads();

  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= N) return;
  const int lane = threadIdx.x & 63;
  const uint16_t* w0 = W + (int64_t)row * K;
  const uint16_t* w1 = (FLAGS & DEC_SWIGLU)
      ? W + (int64_t)(N + row) * K : nullptr;

  float acc0 = 0.0f, acc1 = 0.0f;
  int k = lane * 8;
  for (; k + 8 <= K; k += 64 * 8) {
    const ushortx8 xv = *reinterpret_cast<const ushortx8*>(xs + k);
    const ushortx8 wv = __builtin_nontemporal_load(
        reinterpret_cast<const ushortx8*>(w0 + k));
    #pragma unroll
    for (int i = 0; i < 8; ++i)
      acc0 = fmaf(bf16_to_f32(wv[i]), bf16_to_f32(xv[i]), acc0);
    if constexpr (FLAGS & DEC_SWIGLU) {
      const ushortx8 uv = __builtin_nontemporal_load(
          reinterpret_cast<const ushortx8*>(w1 + k));
      #pragma unroll
      for (int i = 0; i < 8; ++i)
        acc1 = fmaf(bf16_to_f32(uv[i]), bf16_to_f32(xv[i]), acc1);
    }
  }
  for (int kk = k; kk < K && kk < k + 8; ++kk) {
    acc0 = fmaf(bf16_to_f32(w0[kk]), bf16_to_f32(xs[kk]), acc0);
    if constexpr (FLAGS & DEC_SWIGLU)
      acc1 = fmaf(bf16_to_f32(w1[kk]), bf16_to_f32(xs[kk]), acc1);
  }
  acc0 = wave_reduce_sum(acc0);
  if constexpr (FLAGS & DEC_SWIGLU) acc1 = wave_reduce_sum(acc1);
  if (lane == 0) {
    float v = acc0;
    if constexpr (FLAGS & DEC_SWIGLU) v = silu(acc0) * acc1;
    if constexpr (FLAGS & DEC_RESID) v += bf16_to_f32(resid[row]);
    y[row] = f32_to_bf16(v);
  }
}

// ---------------------------------------------------------------------------
// RoPE(q, k) + cache append at *pos (device cursor; NOT advanced here).
// qkv: [QS + 2*KVS]; q_out [QS]; kc/vc: [cap, HKV, D].
__global__ void dec_rope_cache_kernel(const uint16_t* __restrict__ qkv,
                                      uint16_t* __restrict__ q_out,
                                      uint16_t* __restrict__ kc,
                                      uint16_t* __restrict__ vc,
                                      const float* __restrict__ cost,
                                      const float* __restrict__ sint,
                                      const int* __restrict__ pos_dev,
                                      int H, int HKV, int D) {
  const int pos = *pos_dev;
  const int half = D / 2;
  const float* c = cost + (int64_t)pos * half;
  const float* s = sint + (int64_t)pos * half;
  const int QS = H * D, KVS = HKV * D;
  auto rope1 = [&](const uint16_t* src, int hd) -> float {
    const int d = hd % D;
    const float v = bf16_to_f32(src[hd]);
    if (d < half)
      return v * c[d] - bf16_to_f32(src[hd + half]) * s[d];
    if (d < 2 * half)
      return v * c[d - half] + bf16_to_f32(src[hd - half]) * s[d - half];
    return v;                                 // odd tail dim unrotated
  };
  for (int i = threadIdx.x + blockIdx.x * blockDim.x; i < QS + 2 * KVS;
       i += blockDim.x * gridDim.x) {
    if (i < QS) {
      q_out[i] = f32_to_bf16(rope1(qkv, i));
    } else if (i < QS + KVS) {
      const int hd = i - QS;
      kc[(int64_t)pos * KVS + hd] = f32_to_bf16(rope1(qkv + QS, hd));
    } else {
      const int hd = i - QS - KVS;
      vc[(int64_t)pos * KVS + hd] = qkv[i];
    }
  }
}

__global__ void dec_advance_kernel(int* pos_dev) {
  if (threadIdx.x == 0) ++*pos_dev;
}

// ---------------------------------------------------------------------------
// single-token GQA attention over the cache: out[h][d] =
// softmax(q_h . k_kv(s) * scale) @ v.  Block per q-head, 4 waves split the
// sequence, online softmax per wave, LDS merge.  len = *pos_dev + 1 (the
// new token was appended at *pos_dev).
__global__ __launch_bounds__(256)
void dec_attn_kernel(const uint16_t* __restrict__ q,
                     const uint16_t* __restrict__ kc,
                     const uint16_t* __restrict__ vc,
                     uint16_t* __restrict__ out,
                     const int* __restrict__ pos_dev,
                     int H, int HKV, int D, float scale) {
  __shared__ float sm[4], sl[4];
  __shared__ float sacc[4][192];

  const int h = blockIdx.x;
  const int kvh = h / (H / HKV);
  const int len = *pos_dev + 1;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int KVS = HKV * D;

  // q fragment: this lane's dims (lane, lane+64, lane+128)
  float qv[3] = {};
  #pragma unroll
  for (int i = 0; i < 3; ++i) {
    const int d = lane + i * 64;
    if (d < D) qv[i] = bf16_to_f32(q[h * D + d]);
  }

  const int chunk = (len + 3) / 4;
  const int s0 = wid * chunk;
  const int s1 = s0 + chunk < len ? s0 + chunk : len;

  float m = -1e30f, l = 0.0f, acc[3] = {};
  for (int sp = s0; sp < s1; ++sp) {
    const uint16_t* kr = kc + (int64_t)sp * KVS + kvh * D;
    float d0 = 0.0f;
    #pragma unroll
    for (int i = 0; i < 3; ++i) {
      const int d = lane + i * 64;
      if (d < D) d0 = fmaf(qv[i], bf16_to_f32(kr[d]), d0);
    }
    d0 = wave_reduce_sum(d0);
    d0 = __shfl(d0, 0, 64) * scale;
    const float nm = fmaxf(m, d0);
    const float f = __expf(m - nm);
    const float p = __expf(d0 - nm);
    m = nm;
    l = l * f + p;
    const uint16_t* vr = vc + (int64_t)sp * KVS + kvh * D;
    #pragma unroll
    for (int i = 0; i < 3; ++i) {
      const int d = lane + i * 64;
      if (d < D) acc[i] = acc[i] * f + p * bf16_to_f32(vr[d]);
    }
  }
  // merge the 4 waves
  if (lane == 0) { sm[wid] = s0 < s1 ? m : -1e30f; sl[wid] = l; }
  #pragma unroll
  for (int i = 0; i < 3; ++i) {
    const int d = lane + i * 64;
    if (d < D) sacc[wid][d] = acc[i];
  }
  __syncthreads();
  if (wid == 0) {
    const float gm = fmaxf(fmaxf(sm[0], sm[1]), fmaxf(sm[2], sm[3]));
    float gl = 0.0f;
    float f[4];
    #pragma unroll
    for (int w = 0; w < 4; ++w) {
      f[w] = __expf(sm[w] - gm);
      gl += sl[w] * f[w];
    }
    const float inv = 1.0f / fmaxf(gl, 1e-30f);
    #pragma unroll
    for (int i = 0; i < 3; ++i) {
      const int d = lane + i * 64;
      if (d < D) {
        float o = 0.0f;
        #pragma unroll
        for (int w = 0; w < 4; ++w) o += sacc[w][d] * f[w];
        out[h * D + d] = f32_to_bf16(o * inv);
      }
    }
  }
}

// ---------------------------------------------------------------------------
extern "C" hipError_t lumina_dec_gemv(const void* W, const void* x,
                                      const void* wn, const void* resid,
                                      void* y, int N, int K, float eps,
                                      int flags, hipStream_t stream) {
  const int lds = 64 + ((K * 2 + 15) & ~15);
  dim3 grid((N + 3) / 4), block(256);
  switch (flags) {
    case 0:
      hipLaunchKernelGGL(dec_gemv_kernel<0>, grid, block, lds, stream,
                         (const uint16_t*)W, (const uint16_t*)x,
                         (const uint16_t*)wn, (const uint16_t*)resid,
                         (uint16_t*)y, N, K, eps);
      break;
    case DEC_NORM:
      hipLaunchKernelGGL(dec_gemv_kernel<DEC_NORM>, grid, block, lds, stream,
                         (const uint16_t*)W, (const uint16_t*)x,
                         (const uint16_t*)wn, (const uint16_t*)resid,
                         (uint16_t*)y, N, K, eps);
      break;
    case DEC_RESID:
      hipLaunchKernelGGL(dec_gemv_kernel<DEC_RESID>, grid, block, lds,
                         stream, (const uint16_t*)W, (const uint16_t*)x,
                         (const uint16_t*)wn, (const uint16_t*)resid,
                         (uint16_t*)y, N, K, eps);
      break;
    case DEC_NORM | DEC_SWIGLU:
      hipLaunchKernelGGL((dec_gemv_kernel<DEC_NORM | DEC_SWIGLU>), grid,
                         block, lds, stream, (const uint16_t*)W,
                         (const uint16_t*)x, (const uint16_t*)wn,
                         (const uint16_t*)resid, (uint16_t*)y, N, K, eps);
      break;
    default:
      return hipErrorInvalidValue;
  }
  return hipGetLastError();
}

extern "C" hipError_t lumina_dec_rope_cache(const void* qkv, void* q_out,
                                            void* kc, void* vc,
                                            const float* cost,
                                            const float* sint,
                                            const int* pos_dev,
                                            int H, int HKV, int D,
                                            hipStream_t stream) {
  const int total = (H + 2 * HKV) * D;
  dim3 grid((total + 255) / 256 < 32 ? (total + 255) / 256 : 32);
  hipLaunchKernelGGL(dec_rope_cache_kernel, grid, dim3(256), 0, stream,
                     (const uint16_t*)qkv, (uint16_t*)q_out, (uint16_t*)kc,
                     (uint16_t*)vc, cost, sint, pos_dev, H, HKV, D);
  return hipGetLastError();
}

extern "C" hipError_t lumina_dec_attn(const void* q, const void* kc,
                                      const void* vc, void* out,
                                      const int* pos_dev, int H, int HKV,
                                      int D, float scale,
                                      hipStream_t stream) {
  if (D > 192) return hipErrorInvalidValue;
  hipLaunchKernelGGL(dec_attn_kernel, dim3(H), dim3(256), 0, stream,
                     (const uint16_t*)q, (const uint16_t*)kc,
                     (const uint16_t*)vc, (uint16_t*)out, pos_dev,
                     H, HKV, D, scale);
  return hipGetLastError();
}

extern "C" hipError_t lumina_dec_advance(int* pos_dev, hipStream_t stream) {
  hipLaunchKernelGGL(dec_advance_kernel, dim3(1), dim3(64), 0, stream,
                     pos_dev);
  return hipGetLastError();
}
