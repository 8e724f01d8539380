// Fused MoE dispatch/combine gathers for CDNA4 (gfx950).
//
// Replaces the torch index_select + mask-mul + weighted-sum chains of the
// gather-only MoE plan (ops/interface.py MoEDispatchFn/MoECombineFn; design
// rationale there) with single-pass kernels: one block per destination row,
// 16-byte vector traffic, no intermediate [N*k, h] materialisation for the
// combine (the k gathered rows reduce in registers). The reference computed
// dispatch/combine with atomicAdd scatters (moe_cuda_ops.cu:185-310);
// nothing here uses an atomic.
//
// All kernels operate on 2D contiguous [rows, h] tensors; index vectors are
// int64 (torch long). E = elem type (BF16Elem / F32Elem from common.h).

#include "common.h"

// 8-byte vector helpers for the column loops (h % 4 == 0 fast path; rows of
// 2-byte elems at even h are 8-byte aligned). Scalar tail for other h.
typedef __attribute__((ext_vector_type(4))) uint16_t ushortx4;


// grid-stride over rows helper
#define ROW_LOOP(row, nrows) \
  for (int64_t row = blockIdx.x; row < (nrows); row += gridDim.x)

// ---- dispatch forward: buf[s] = fill[s] ? x[src_tok[s]] : 0 --------------
template <typename E>
__global__ void moe_gather_rows_kernel(const typename E::storage* __restrict__ x,
                                       const int64_t* __restrict__ src_tok,
                                       const bool* __restrict__ fill,
                                       typename E::storage* __restrict__ buf,
                                       int64_t n_slots, int h) {
  ROW_LOOP(s, n_slots) {
    const int64_t src = src_tok[s];
    const bool ok = fill[s];
    typename E::storage* out = buf + s * h;
    const typename E::storage* in = x + src * h;
    if constexpr (sizeof(typename E::storage) == 2) {
      if ((h & 3) == 0) {
        const int hv = h >> 2;
        const ushortx4* iv = reinterpret_cast<const ushortx4*>(in);
        ushortx4* ov = reinterpret_cast<ushortx4*>(out);
        const ushortx4 z = {0, 0, 0, 0};
        for (int c = threadIdx.x; c < hv; c += blockDim.x)
          ov[c] = ok ? iv[c] : z;
        continue;
      }
    }
    for (int c = threadIdx.x; c < h; c += blockDim.x)
      E::store(out + c, ok ? E::load(in + c) : 0.0f);
  }
}

// ---- dispatch backward: gx[t] = sum_j keep * gbuf[slot_tm[t*k+j]] --------
template <typename E>
__global__ void moe_dispatch_bwd_kernel(const typename E::storage* __restrict__ gbuf,
                                        const int64_t* __restrict__ slot_tm,
                                        typename E::storage* __restrict__ gx,
                                        int64_t n_tok, int k, int h,
                                        int64_t n_slots) {
  ROW_LOOP(t, n_tok) {
    typename E::storage* out = gx + t * h;
    if constexpr (sizeof(typename E::storage) == 2) {
      if ((h & 3) == 0) {
        const int hv = h >> 2;
        ushortx4* ov = reinterpret_cast<ushortx4*>(out);
        for (int c = threadIdx.x; c < hv; c += blockDim.x) {
          float a0 = 0, a1 = 0, a2 = 0, a3 = 0;
          for (int j = 0; j < k; ++j) {
            const int64_t s = slot_tm[t * k + j];
            if (s < n_slots) {
              ushortx4 v = reinterpret_cast<const ushortx4*>(
                  gbuf + s * h)[c];
              a0 += bf16_to_f32(v[0]); a1 += bf16_to_f32(v[1]);
              a2 += bf16_to_f32(v[2]); a3 += bf16_to_f32(v[3]);
            }
          }
          ushortx4 o4 = {f32_to_bf16(a0), f32_to_bf16(a1),
                         f32_to_bf16(a2), f32_to_bf16(a3)};
          ov[c] = o4;
        }
        continue;
      }
    }
    for (int c = threadIdx.x; c < h; c += blockDim.x) {
      float acc = 0.0f;
      for (int j = 0; j < k; ++j) {
        const int64_t s = slot_tm[t * k + j];
        if (s < n_slots) acc += E::load(gbuf + s * h + c);
      }
      E::store(out + c, acc);
    }
  }
}

// ---- combine forward: out[t] = sum_j w[t*k+j]*keep * y[slot_tm[t*k+j]] ---
template <typename E>
__global__ void moe_combine_fwd_kernel(const typename E::storage* __restrict__ y,
                                       const float* __restrict__ w_tm,
                                       const int64_t* __restrict__ slot_tm,
                                       typename E::storage* __restrict__ out,
                                       int64_t n_tok, int k, int h,
                                       int64_t n_slots) {
  ROW_LOOP(t, n_tok) {
    typename E::storage* o = out + t * h;
    if constexpr (sizeof(typename E::storage) == 2) {
      if ((h & 3) == 0) {
        const int hv = h >> 2;
        ushortx4* ov = reinterpret_cast<ushortx4*>(o);
        for (int c = threadIdx.x; c < hv; c += blockDim.x) {
          float a0 = 0, a1 = 0, a2 = 0, a3 = 0;
          for (int j = 0; j < k; ++j) {
            const int64_t s = slot_tm[t * k + j];
            if (s < n_slots) {
              const float w = w_tm[t * k + j];
              ushortx4 v = reinterpret_cast<const ushortx4*>(y + s * h)[c];
              a0 = fmaf(w, bf16_to_f32(v[0]), a0);
              a1 = fmaf(w, bf16_to_f32(v[1]), a1);
              a2 = fmaf(w, bf16_to_f32(v[2]), a2);
              a3 = fmaf(w, bf16_to_f32(v[3]), a3);
            }
          }
          ushortx4 o4 = {f32_to_bf16(a0), f32_to_bf16(a1),
                         f32_to_bf16(a2), f32_to_bf16(a3)};
          ov[c] = o4;
        }
        continue;
      }
    }
    for (int c = threadIdx.x; c < h; c += blockDim.x) {
      float acc = 0.0f;
      for (int j = 0; j < k; ++j) {
        const int64_t s = slot_tm[t * k + j];
        if (s < n_slots)
          acc += w_tm[t * k + j] * E::load(y + s * h + c);
      }
      E::store(o + c, acc);
    }
  }
}

// ---- combine backward (y): gy[s] = fill[s] * w_tm[inv[s]] * gout[src_tok[s]]
template <typename E>
__global__ void moe_combine_bwd_y_kernel(const typename E::storage* __restrict__ gout,
                                         const float* __restrict__ w_tm,
                                         const int64_t* __restrict__ inv,
                                         const int64_t* __restrict__ src_tok,
                                         const bool* __restrict__ fill,
                                         typename E::storage* __restrict__ gy,
                                         int64_t n_slots, int64_t n_flat, int h) {
  ROW_LOOP(s, n_slots) {
    const bool ok = fill[s];
    const int64_t f = inv[s] < n_flat ? inv[s] : 0;
    const float w = ok ? w_tm[f] : 0.0f;
    const typename E::storage* g = gout + src_tok[s] * h;
    typename E::storage* o = gy + s * h;
    if constexpr (sizeof(typename E::storage) == 2) {
      if ((h & 3) == 0) {
        const int hv = h >> 2;
        const ushortx4* gv = reinterpret_cast<const ushortx4*>(g);
        ushortx4* ov = reinterpret_cast<ushortx4*>(o);
        for (int c = threadIdx.x; c < hv; c += blockDim.x) {
          ushortx4 in4 = gv[c];
          ushortx4 o4;
          #pragma unroll
          for (int i = 0; i < 4; ++i)
            o4[i] = f32_to_bf16(w * bf16_to_f32(in4[i]));
          ov[c] = o4;
        }
        continue;
      }
    }
    for (int c = threadIdx.x; c < h; c += blockDim.x)
      E::store(o + c, w * E::load(g + c));
  }
}

// ---- combine backward (w): gw[t*k+j] = keep * <gout[t], y[slot]> ---------
// one WAVE per (t, j): lanes stride columns, wave-reduce the dot.
template <typename E>
__global__ void moe_combine_bwd_w_kernel(const typename E::storage* __restrict__ gout,
                                         const typename E::storage* __restrict__ y,
                                         const int64_t* __restrict__ slot_tm,
                                         float* __restrict__ gw,
                                         int64_t n_flat, int k, int h,
                                         int64_t n_slots) {
  const int waves_per_block = blockDim.x / WAVE;
  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  for (int64_t f = (int64_t)blockIdx.x * waves_per_block + wave_id;
       f < n_flat; f += (int64_t)gridDim.x * waves_per_block) {
    const int64_t s = slot_tm[f];
    float acc = 0.0f;
    if (s < n_slots) {
      const int64_t t = f / k;
      const typename E::storage* g = gout + t * h;
      const typename E::storage* yy = y + s * h;
      for (int c = lane; c < h; c += WAVE)
        acc += E::load(g + c) * E::load(yy + c);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) gw[f] = (s < n_slots) ? acc : 0.0f;
  }
}

// ---- launchers -----------------------------------------------------------
static inline int row_grid(int64_t rows) {
  return (int)(rows < 32768 ? rows : 32768);
}

#define DEFINE_LAUNCH(name, kern, grid_rows, ...)                           \
  extern "C" hipError_t name##_bf16(__VA_ARGS__);                           \
  extern "C" hipError_t name##_f32(__VA_ARGS__);

extern "C" {

hipError_t lumina_moe_gather_rows(const void* x, const int64_t* src,
                                  const bool* fill, void* buf,
                                  int64_t n_slots, int h, int is_bf16,
                                  hipStream_t s) {
  dim3 g(row_grid(n_slots)), b(256);
  if (is_bf16)
    hipLaunchKernelGGL((moe_gather_rows_kernel<BF16Elem>), g, b, 0, s,
                       (const uint16_t*)x, src, fill, (uint16_t*)buf,
                       n_slots, h);
  else
    hipLaunchKernelGGL((moe_gather_rows_kernel<F32Elem>), g, b, 0, s,
                       (const float*)x, src, fill, (float*)buf, n_slots, h);
  return hipGetLastError();
}

hipError_t lumina_moe_dispatch_bwd(const void* gbuf, const int64_t* slot_tm,
                                   void* gx, int64_t n_tok, int k, int h,
                                   int64_t n_slots, int is_bf16,
                                   hipStream_t s) {
  dim3 g(row_grid(n_tok)), b(256);
  if (is_bf16)
    hipLaunchKernelGGL((moe_dispatch_bwd_kernel<BF16Elem>), g, b, 0, s,
                       (const uint16_t*)gbuf, slot_tm, (uint16_t*)gx,
                       n_tok, k, h, n_slots);
  else
    hipLaunchKernelGGL((moe_dispatch_bwd_kernel<F32Elem>), g, b, 0, s,
                       (const float*)gbuf, slot_tm, (float*)gx,
                       n_tok, k, h, n_slots);
  return hipGetLastError();
}

hipError_t lumina_moe_combine_fwd(const void* y, const float* w_tm,
                                  const int64_t* slot_tm, void* out,
                                  int64_t n_tok, int k, int h,
                                  int64_t n_slots, int is_bf16,
                                  hipStream_t s) {
  dim3 g(row_grid(n_tok)), b(256);
  if (is_bf16)
    hipLaunchKernelGGL((moe_combine_fwd_kernel<BF16Elem>), g, b, 0, s,
                       (const uint16_t*)y, w_tm, slot_tm, (uint16_t*)out,
                       n_tok, k, h, n_slots);
  else
    hipLaunchKernelGGL((moe_combine_fwd_kernel<F32Elem>), g, b, 0, s,
                       (const float*)y, w_tm, slot_tm, (float*)out,
                       n_tok, k, h, n_slots);
  return hipGetLastError();
}

hipError_t lumina_moe_combine_bwd_y(const void* gout, const float* w_tm,
                                    const int64_t* inv, const int64_t* src,
                                    const bool* fill, void* gy,
                                    int64_t n_slots, int64_t n_flat, int h,
                                    int is_bf16, hipStream_t s) {
  dim3 g(row_grid(n_slots)), b(256);
  if (is_bf16)
    hipLaunchKernelGGL((moe_combine_bwd_y_kernel<BF16Elem>), g, b, 0, s,
                       (const uint16_t*)gout, w_tm, inv, src, fill,
                       (uint16_t*)gy, n_slots, n_flat, h);
  else
    hipLaunchKernelGGL((moe_combine_bwd_y_kernel<F32Elem>), g, b, 0, s,
                       (const float*)gout, w_tm, inv, src, fill,
                       (float*)gy, n_slots, n_flat, h);
  return hipGetLastError();
}

hipError_t lumina_moe_combine_bwd_w(const void* gout, const void* y,
                                    const int64_t* slot_tm, float* gw,
                                    int64_t n_flat, int k, int h,
                                    int64_t n_slots, int is_bf16,
                                    hipStream_t s) {
  dim3 g(row_grid((n_flat + 3) / 4)), b(256);
  if (is_bf16)
    hipLaunchKernelGGL((moe_combine_bwd_w_kernel<BF16Elem>), g, b, 0, s,
                       (const uint16_t*)gout, (const uint16_t*)y, slot_tm,
                       gw, n_flat, k, h, n_slots);
  else
    hipLaunchKernelGGL((moe_combine_bwd_w_kernel<F32Elem>), g, b, 0, s,
                       (const float*)gout, (const float*)y, slot_tm,
                       gw, n_flat, k, h, n_slots);
  return hipGetLastError();
}

}  // extern "C"
