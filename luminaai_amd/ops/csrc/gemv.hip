// Batch-1 GEMV for decoding on CDNA4: y[N] = W[N,K] @ x[K], bf16 in/out,
// fp32 accumulate.
//
// hipBLASLt's batch-1 GEMV path streams the weight matrix at ~0.4 TB/s
// effective on gfx950 (measured: a captured b1 decode step is 6.5 ms for
// 2.6 GB of weights). This kernel is the guide's decode recipe
// (cdna_hip_programming.md §5 GEMV row): no LDS round trip, one wave per
// output row, lanes stride the row in 16-byte chunks (perfectly coalesced),
// x rides L1/L2. Four waves per block; grid = rows/4 (b1 qkv: 2544 rows ->
// 636 blocks, fills all 8 XCDs).

#include "common.h"

template <typename E>
__global__ __launch_bounds__(256)
void gemv_kernel(const typename E::storage* __restrict__ W,
                 const typename E::storage* __restrict__ x,
                 typename E::storage* __restrict__ y,
                 int N, int64_t K) {
  const int row = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= N) return;
  const int lane = threadIdx.x & 63;
  const typename E::storage* w = W + (int64_t)row * K;

  float acc = 0.0f;
  int64_t k = (int64_t)lane * 8;
  if constexpr (sizeof(typename E::storage) == 2) {
    for (; k + 8 <= K; k += 64 * 8) {
      ushortx8 wv = *reinterpret_cast<const ushortx8*>(w + k);
      ushortx8 xv = *reinterpret_cast<const ushortx8*>(x + k);
      #pragma unroll
      for (int i = 0; i < 8; ++i)
        acc = fmaf(bf16_to_f32(wv[i]), bf16_to_f32(xv[i]), acc);
    }
  } else {
    for (; k + 8 <= K; k += 64 * 8) {
      #pragma unroll
      for (int i = 0; i < 8; ++i)
        acc = fmaf(E::load(w + k + i), E::load(x + k + i), acc);
    }
  }
  for (int64_t kk = k; kk < K && kk < k + 8; ++kk)
    acc = fmaf(E::load(w + kk), E::load(x + kk), acc);

  acc = wave_reduce_sum(acc);
  if (lane == 0) E::store(y + row, acc);
}

extern "C" hipError_t lumina_gemv(const void* W, const void* x, void* y,
                                  int N, int64_t K, int is_bf16,
                                  hipStream_t stream) {
  dim3 grid((N + 3) / 4), block(256);
  if (is_bf16)
    hipLaunchKernelGGL((gemv_kernel<BF16Elem>), grid, block, 0, stream,
                       (const uint16_t*)W, (const uint16_t*)x, (uint16_t*)y,
                       N, K);
  else
    hipLaunchKernelGGL((gemv_kernel<F32Elem>), grid, block, 0, stream,
                       (const float*)W, (const float*)x, (float*)y, N, K);
  return hipGetLastError();
}
