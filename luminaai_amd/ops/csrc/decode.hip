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
#define DEC_SCALE  8

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

  // two rows per wave, interleaved in one k-loop (2x memory-level
  // parallelism per wave; measured NEUTRAL vs one row/wave at b1 decode
  // shapes -- the streams are bandwidth-saturated -- kept for the halved
  // launch grid)
  const int row = blockIdx.x * 8 + (threadIdx.x >> 6) * 2;
  if (row >= N) return;
  const int lane = threadIdx.x & 63;
  const bool two = row + 1 < N;
  const uint16_t* w0 = W + (int64_t)row * K;
  const uint16_t* w0b = w0 + (two ? K : 0);
  const uint16_t* w1 = (FLAGS & DEC_SWIGLU)
      ? W + (int64_t)(N + row) * K : nullptr;
  const uint16_t* w1b = (FLAGS & DEC_SWIGLU) ? w1 + (two ? K : 0) : nullptr;

  float acc0 = 0.0f, acc1 = 0.0f, acc0b = 0.0f, acc1b = 0.0f;
  int k = lane * 8;
  for (; k + 8 <= K; k += 64 * 8) {
    const ushortx8 xv = *reinterpret_cast<const ushortx8*>(xs + k);
    const ushortx8 wv = __builtin_nontemporal_load(
        reinterpret_cast<const ushortx8*>(w0 + k));
    const ushortx8 wvb = __builtin_nontemporal_load(
        reinterpret_cast<const ushortx8*>(w0b + k));
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float xf = bf16_to_f32(xv[i]);
      acc0 = fmaf(bf16_to_f32(wv[i]), xf, acc0);
      acc0b = fmaf(bf16_to_f32(wvb[i]), xf, acc0b);
    }
    if constexpr (FLAGS & DEC_SWIGLU) {
      const ushortx8 uv = __builtin_nontemporal_load(
          reinterpret_cast<const ushortx8*>(w1 + k));
      const ushortx8 uvb = __builtin_nontemporal_load(
          reinterpret_cast<const ushortx8*>(w1b + k));
      #pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float xf = bf16_to_f32(xv[i]);
        acc1 = fmaf(bf16_to_f32(uv[i]), xf, acc1);
        acc1b = fmaf(bf16_to_f32(uvb[i]), xf, acc1b);
      }
    }
  }
  for (int kk = k; kk < K && kk < k + 8; ++kk) {
    const float xf = bf16_to_f32(xs[kk]);
    acc0 = fmaf(bf16_to_f32(w0[kk]), xf, acc0);
    acc0b = fmaf(bf16_to_f32(w0b[kk]), xf, acc0b);
    if constexpr (FLAGS & DEC_SWIGLU) {
      acc1 = fmaf(bf16_to_f32(w1[kk]), xf, acc1);
      acc1b = fmaf(bf16_to_f32(w1b[kk]), xf, acc1b);
    }
  }
  acc0 = wave_reduce_sum(acc0);
  acc0b = wave_reduce_sum(acc0b);
  if constexpr (FLAGS & DEC_SWIGLU) {
    acc1 = wave_reduce_sum(acc1);
    acc1b = wave_reduce_sum(acc1b);
  }
  if (lane == 0) {
    float v = acc0;
    if constexpr (FLAGS & DEC_SWIGLU) v = silu(acc0) * acc1;
    if constexpr (FLAGS & DEC_RESID) v += bf16_to_f32(resid[row]);
    y[row] = f32_to_bf16(v);
    if (two) {
      float vb = acc0b;
      if constexpr (FLAGS & DEC_SWIGLU) vb = silu(acc0b) * acc1b;
      if constexpr (FLAGS & DEC_RESID) vb += bf16_to_f32(resid[row + 1]);
      y[row + 1] = f32_to_bf16(vb);
    }
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
// softmax(q_h . k_kv(s) * scale) @ v.  Block per q-head; lane-per-s QK
// scoring (each lane streams one K row), block softmax over an LDS score
// array, then a d-parallel PV pass with coalesced V rows.  len =
// *pos_dev + 1 (the new token was appended at *pos_dev).
__global__ __launch_bounds__(256)
void dec_attn_kernel(const uint16_t* __restrict__ q,
                     const uint16_t* __restrict__ kc,
                     const uint16_t* __restrict__ vc,
                     uint16_t* __restrict__ out,
                     const int* __restrict__ pos_dev,
                     int H, int HKV, int D, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);            // [16]
  float* qs = red + 16;                                   // [D]
  float* sc = qs + ((D + 3) & ~3);                        // [len]

  const int h = blockIdx.x;
  const int kvh = h / (H / HKV);
  const int len = *pos_dev + 1;
  const int t = threadIdx.x;
  const int KVS = HKV * D;

  for (int d = t; d < D; d += 256)
    qs[d] = bf16_to_f32(q[h * D + d]);
  __syncthreads();

  // ---- pass 1: one K row per thread, vectorized along d
  float pmax = -1e30f;
  for (int sp = t; sp < len; sp += 256) {
    const uint16_t* kr = kc + (int64_t)sp * KVS + kvh * D;
    float dot = 0.0f;
    int d = 0;
    for (; d + 8 <= D; d += 8) {
      const ushortx8 kv8 = *reinterpret_cast<const ushortx8*>(kr + d);
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        dot = fmaf(qs[d + j], bf16_to_f32(kv8[j]), dot);
    }
    for (; d < D; ++d)
      dot = fmaf(qs[d], bf16_to_f32(kr[d]), dot);
    dot *= scale;
    sc[sp] = dot;
    pmax = fmaxf(pmax, dot);
  }
  const float m = block_reduce_max(pmax, red);
  float psum = 0.0f;
  for (int sp = t; sp < len; sp += 256) {
    const float p = __expf(sc[sp] - m);
    sc[sp] = p;
    psum += p;
  }
  const float inv = 1.0f / fmaxf(block_reduce_sum(psum, red), 1e-30f);
  __syncthreads();

  // ---- pass 2: d-parallel weighted V accumulation (coalesced rows)
  for (int d = t; d < D; d += 256) {
    float acc = 0.0f;
    for (int sp = 0; sp < len; ++sp)
      acc = fmaf(sc[sp], bf16_to_f32(vc[(int64_t)sp * KVS + kvh * D + d]),
                 acc);
    out[h * D + d] = f32_to_bf16(acc * inv);
  }
}

// ---------------------------------------------------------------------------
// MoE decode support: xhat = rmsnorm(x)*wn once (router and both experts
// read it), a top-k router kernel, and an expert-indirect GEMV whose
// weight pointer is W + eidx[slot]*estride -- the expert index stays on
// device so the whole MoE token step hipGraph-captures.

__global__ __launch_bounds__(256)
void dec_rmsnorm_kernel(const uint16_t* __restrict__ x,
                        const uint16_t* __restrict__ wn,
                        uint16_t* __restrict__ out, int K, float eps) {
  __shared__ float red[16];
  float ss = 0.0f;
  for (int k = threadIdx.x; k < K; k += 256) {
    const float v = bf16_to_f32(x[k]);
    ss = fmaf(v, v, ss);
  }
  const float inv = rsqrtf(block_reduce_sum(ss, red) / K + eps);
  for (int k = threadIdx.x; k < K; k += 256)
    out[k] = f32_to_bf16(bf16_to_f32(x[k]) * inv * bf16_to_f32(wn[k]));
}

// fused MoE router: xhat = rmsnorm(x)*wn (written out for the expert
// GEMVs), gate logits = Wg[E,h] @ xhat, softmax/temp top-k renormalized
// -> eidx/ew.  One launch instead of dec_rmsnorm + dec_gemv + dec_topk.
// Measured NEUTRAL on wall-clock (2.85 ms/token before and after): the
// three single-block kernels' ~5us durations overlap the graph's other
// costs rather than serializing the token.  Kept for the smaller graph
// (2 fewer nodes per MoE layer).
__global__ __launch_bounds__(256)
void dec_router_kernel(const uint16_t* __restrict__ x,
                       const uint16_t* __restrict__ wn,
                       const uint16_t* __restrict__ Wg,
                       uint16_t* __restrict__ xhat,
                       int* __restrict__ eidx, float* __restrict__ ew,
                       int K, int E, int k, float temp, float eps) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* red = reinterpret_cast<float*>(smem);            // [16]
  float* lg = red + 16;                                   // [E]
  uint16_t* xs = reinterpret_cast<uint16_t*>(lg + ((E + 3) & ~3)); // [K]

  const int t = threadIdx.x;
  float ss = 0.0f;
  for (int kk = t; kk < K; kk += 256) {
    const float v = bf16_to_f32(x[kk]);
    ss = fmaf(v, v, ss);
  }
  const float inv = rsqrtf(block_reduce_sum(ss, red) / K + eps);
  for (int kk = t; kk < K; kk += 256) {
    const uint16_t h = f32_to_bf16(bf16_to_f32(x[kk]) * inv
                                   * bf16_to_f32(wn[kk]));
    xs[kk] = h;
    xhat[kk] = h;
  }
  __syncthreads();

  // gate logits: wave per expert row (rounds of 4)
  const int lane = t & 63;
  const int wave = t >> 6;
  for (int e = wave; e < E; e += 4) {
    const uint16_t* w = Wg + (int64_t)e * K;
    float acc = 0.0f;
    for (int kk = lane * 4; kk + 4 <= K; kk += 64 * 4) {
      #pragma unroll
      for (int j = 0; j < 4; ++j)
        acc = fmaf(bf16_to_f32(w[kk + j]), bf16_to_f32(xs[kk + j]), acc);
    }
    for (int kk = (K & ~255) + lane; kk < K; kk += 64)
      acc = fmaf(bf16_to_f32(w[kk]), bf16_to_f32(xs[kk]), acc);
    acc = wave_reduce_sum(acc);
    if (lane == 0) lg[e] = acc;
  }
  __syncthreads();

  // top-k on wave 0 (identical math to dec_topk_kernel)
  if (wave == 0) {
    float v = lane < E ? lg[lane] / temp : -1e30f;
    const float m = wave_reduce_max(v);
    float p = lane < E ? __expf(v - m) : 0.0f;
    const float Z = wave_reduce_sum(p);
    p /= Z;
    float wsum = 0.0f;
    float pk = p;
    for (int j = 0; j < k; ++j) {
      const float mj = wave_reduce_max(pk);
      const uint64_t hit = __ballot(pk == mj);
      const int who = __ffsll((unsigned long long)hit) - 1;
      if (lane == who) {
        eidx[j] = lane;
        pk = -1.0f;
      }
      if (lane == 0) ew[j] = mj;
      wsum += mj;
    }
    if (lane < k) ew[lane] = ew[lane] / fmaxf(wsum, 1e-9f);
  }
}

// softmax(logits/temp) -> top-k (renormalized over the k), one wave.
// Matches ops/reference.py topk_gating inference semantics (no noise).
__global__ void dec_topk_kernel(const uint16_t* __restrict__ logits,
                                int* __restrict__ eidx,
                                float* __restrict__ ew,
                                int E, int k, float temp) {
  const int lane = threadIdx.x;
  float v = lane < E ? bf16_to_f32(logits[lane]) / temp : -1e30f;
  const float m = wave_reduce_max(v);
  float p = lane < E ? __expf(v - m) : 0.0f;
  const float Z = wave_reduce_sum(p);
  p /= Z;
  float wsum = 0.0f;
  float pk = p;
  for (int j = 0; j < k; ++j) {
    const float mj = wave_reduce_max(pk);
    // first lane holding the max claims the slot (stable tie-break: min id)
    const uint64_t hit = __ballot(pk == mj);
    const int who = __ffsll((unsigned long long)hit) - 1;
    if (lane == who) {
      eidx[j] = lane;
      pk = -1.0f;
    }
    if (lane == 0) ew[j] = mj;
    wsum += mj;
  }
  if (lane < k) ew[lane] = ew[lane] / fmaxf(wsum, 1e-9f);
}

// y[N] = W_e[N(or 2N), K] @ x[K] with W_e = W + eidx[slot]*estride.
// SCALE multiplies the dot by ew[slot] before the residual add.
template <int FLAGS>
__global__ __launch_bounds__(256)
void dec_gemv_moe_kernel(const uint16_t* __restrict__ W,
                         const uint16_t* __restrict__ x,
                         const uint16_t* __restrict__ resid,
                         uint16_t* __restrict__ y,
                         const int* __restrict__ eidx,
                         const float* __restrict__ ew,
                         int slot, int64_t estride, int N, int K) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  uint16_t* xs = reinterpret_cast<uint16_t*>(smem);
  for (int kk = threadIdx.x; kk < K; kk += 256)
    xs[kk] = x[kk];
  __syncthreads();

  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= N) return;
  const int lane = threadIdx.x & 63;
  const uint16_t* We = W + eidx[slot] * estride;
  const uint16_t* w0 = We + (int64_t)row * K;
  const uint16_t* w1 = (FLAGS & DEC_SWIGLU)
      ? We + (int64_t)(N + row) * K : nullptr;

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
    if constexpr (FLAGS & DEC_SCALE) v *= ew[slot];
    if constexpr (FLAGS & DEC_RESID) v += bf16_to_f32(resid[row]);
    y[row] = f32_to_bf16(v);
  }
}

// ---------------------------------------------------------------------------
extern "C" hipError_t lumina_dec_gemv(const void* W, const void* x,
                                      const void* wn, const void* resid,
                                      void* y, int N, int K, float eps,
                                      int flags, hipStream_t stream) {
  const int lds = 64 + ((K * 2 + 15) & ~15);
  dim3 grid((N + 7) / 8), block(256);
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
                                      const int* pos_dev, int cap,
                                      int H, int HKV,
                                      int D, float scale,
                                      hipStream_t stream) {
  if (D > 192) return hipErrorInvalidValue;
  const int lds = (16 + ((D + 3) & ~3)) * 4 + cap * 4;
  if (lds > 160 * 1024) return hipErrorInvalidValue;
  hipLaunchKernelGGL(dec_attn_kernel, dim3(H), dim3(256), lds, stream,
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

extern "C" hipError_t lumina_dec_rmsnorm(const void* x, const void* wn,
                                         void* out, int K, float eps,
                                         hipStream_t stream) {
  hipLaunchKernelGGL(dec_rmsnorm_kernel, dim3(1), dim3(256), 0, stream,
                     (const uint16_t*)x, (const uint16_t*)wn,
                     (uint16_t*)out, K, eps);
  return hipGetLastError();
}

extern "C" hipError_t lumina_dec_topk(const void* logits, int* eidx,
                                      float* ew, int E, int k, float temp,
                                      hipStream_t stream) {
  if (E > 64 || k > E) return hipErrorInvalidValue;
  hipLaunchKernelGGL(dec_topk_kernel, dim3(1), dim3(64), 0, stream,
                     (const uint16_t*)logits, eidx, ew, E, k, temp);
  return hipGetLastError();
}

extern "C" hipError_t lumina_dec_gemv_moe(const void* W, const void* x,
                                          const void* resid, void* y,
                                          const int* eidx, const float* ew,
                                          int slot, int64_t estride,
                                          int N, int K, int flags,
                                          hipStream_t stream) {
  const int lds = (K * 2 + 15) & ~15;
  dim3 grid((N + 3) / 4), block(256);
  switch (flags) {
    case DEC_SWIGLU:
      hipLaunchKernelGGL(dec_gemv_moe_kernel<DEC_SWIGLU>, grid, block, lds,
                         stream, (const uint16_t*)W, (const uint16_t*)x,
                         (const uint16_t*)resid, (uint16_t*)y, eidx, ew,
                         slot, estride, N, K);
      break;
    case DEC_RESID | DEC_SCALE:
      hipLaunchKernelGGL((dec_gemv_moe_kernel<DEC_RESID | DEC_SCALE>), grid,
                         block, lds, stream, (const uint16_t*)W,
                         (const uint16_t*)x, (const uint16_t*)resid,
                         (uint16_t*)y, eidx, ew, slot, estride, N, K);
      break;
    default:
      return hipErrorInvalidValue;
  }
  return hipGetLastError();
}

extern "C" hipError_t lumina_dec_router(const void* x, const void* wn,
                                        const void* Wg, void* xhat,
                                        int* eidx, float* ew, int K, int E,
                                        int k, float temp, float eps,
                                        hipStream_t stream) {
  if (E > 64 || k > E) return hipErrorInvalidValue;
  const int lds = 64 + ((E + 3) & ~3) * 4 + ((K * 2 + 15) & ~15);
  hipLaunchKernelGGL(dec_router_kernel, dim3(1), dim3(256), lds, stream,
                     (const uint16_t*)x, (const uint16_t*)wn,
                     (const uint16_t*)Wg, (uint16_t*)xhat, eidx, ew,
                     K, E, k, temp, eps);
  return hipGetLastError();
}
