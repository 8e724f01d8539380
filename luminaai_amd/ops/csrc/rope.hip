// Rotary position embedding (half-split rotation) for CDNA4 (gfx950).
//
// Replaces the semantics of the reference rope kernels
// (/root/reference/Src/Main_Scripts/core/transformer_ops.cu:130-215) with an
// MI355X-first design:
//  - operates on the [B, S, H, D] layout straight out of the QKV projection
//    (D fastest => coalesced 16-byte lanes), q and k rotated in ONE launch;
//  - cos/sin table [S, D/2] fp32 precomputed once on device (trig in a hot
//    elementwise kernel turns it VALU-bound; guide Appendix B);
//  - backward is the inverse rotation (transpose of a 2x2 rotation = rotation
//    by -theta), exposed as the same kernel with `conj` = true.
//
// out[..., d]       = x[..., d]   * cos - x[..., d+D/2] * sin     (d < D/2)
// out[..., d+D/2]   = x[..., d+D/2] * cos + x[..., d]   * sin
#include "common.h"

// One thread handles VEC consecutive d-positions of one (b, s, h) row's lower
// half (and the matching upper half): 2*VEC*sizeof(T) bytes per thread.
template <typename E, bool CONJ>
__global__ void rope_kernel(const typename E::storage* __restrict__ q,
                            const typename E::storage* __restrict__ k,
                            typename E::storage* __restrict__ oq,
                            typename E::storage* __restrict__ ok,
                            const float* __restrict__ cs,  // [S_cache, D/2] cos
                            const float* __restrict__ sn,  // [S_cache, D/2] sin
                            const int* __restrict__ pos,   // [B, S] positions or null
                            int64_t BQ,  // B * S * Hq rows for q
                            int64_t BK,  // B * S * Hk rows for k
                            int S, int Hq, int Hk, int D, int pos_offset) {
  const int half = D / 2;
  const int64_t total = (BQ + BK) * half;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += gstride()) {
    const bool is_q = idx < BQ * half;
    const int64_t li = is_q ? idx : idx - BQ * half;
    const int64_t row = li / half;       // (b, s, h) flattened
    const int d = (int)(li - row * half);
    const int H = is_q ? Hq : Hk;
    const int64_t bs = row / H;          // (b, s)
    const int s = (int)(bs % S);
    const int p = pos ? pos[bs] : (s + pos_offset);

    const float c = cs[(int64_t)p * half + d];
    const float sv = CONJ ? -sn[(int64_t)p * half + d] : sn[(int64_t)p * half + d];

    const typename E::storage* xin = is_q ? q : k;
    typename E::storage* xout = is_q ? oq : ok;
    const int64_t base = row * (int64_t)D + d;
    float x1 = E::load(xin + base);
    float x2 = E::load(xin + base + half);
    E::store(xout + base, x1 * c - x2 * sv);
    E::store(xout + base + half, x2 * c + x1 * sv);
    // odd head_dim (e.g. 1908/12 = 159): the last element passes through
    if ((D & 1) && d == 0)
      E::store(xout + row * (int64_t)D + (D - 1),
               E::load(xin + row * (int64_t)D + (D - 1)));
  }
}

extern "C" {

hipError_t lumina_rope_bf16(const void* q, const void* k, void* oq, void* ok,
                            const float* cos_t, const float* sin_t,
                            const int* pos, int64_t B, int S, int Hq, int Hk,
                            int D, int pos_offset, int conj, hipStream_t st) {
  const int64_t BQ = B * (int64_t)S * Hq, BK = B * (int64_t)S * Hk;
  const int64_t total = (BQ + BK) * (D / 2);
  const int block = 256;
  const int grid = elementwise_grid(total, block, 4);
  if (conj)
    rope_kernel<BF16Elem, true><<<grid, block, 0, st>>>(
        (const uint16_t*)q, (const uint16_t*)k, (uint16_t*)oq, (uint16_t*)ok,
        cos_t, sin_t, pos, BQ, BK, S, Hq, Hk, D, pos_offset);
  else
    rope_kernel<BF16Elem, false><<<grid, block, 0, st>>>(
        (const uint16_t*)q, (const uint16_t*)k, (uint16_t*)oq, (uint16_t*)ok,
        cos_t, sin_t, pos, BQ, BK, S, Hq, Hk, D, pos_offset);
  HIP_CHECK_LAST();
  return hipSuccess;
}

hipError_t lumina_rope_f32(const void* q, const void* k, void* oq, void* ok,
                           const float* cos_t, const float* sin_t,
                           const int* pos, int64_t B, int S, int Hq, int Hk,
                           int D, int pos_offset, int conj, hipStream_t st) {
  const int64_t BQ = B * (int64_t)S * Hq, BK = B * (int64_t)S * Hk;
  const int64_t total = (BQ + BK) * (D / 2);
  const int block = 256;
  const int grid = elementwise_grid(total, block, 4);
  if (conj)
    rope_kernel<F32Elem, true><<<grid, block, 0, st>>>(
        (const float*)q, (const float*)k, (float*)oq, (float*)ok,
        cos_t, sin_t, pos, BQ, BK, S, Hq, Hk, D, pos_offset);
  else
    rope_kernel<F32Elem, false><<<grid, block, 0, st>>>(
        (const float*)q, (const float*)k, (float*)oq, (float*)ok,
        cos_t, sin_t, pos, BQ, BK, S, Hq, Hk, D, pos_offset);
  HIP_CHECK_LAST();
  return hipSuccess;
}

}  // extern "C"
