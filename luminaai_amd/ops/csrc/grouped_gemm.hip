// Grouped (batched-expert) GEMM, NT form, for CDNA4 / gfx950:
//
//   out[e][m][n] = sum_k A[e][m][k] * B[e][n][k]        (bf16 in, fp32 acc)
//
// This is the MoE expert backward "grad_x = grad_out @ W^T" shape (and any
// A·Bᵀ with both operands row-major, contraction over the LAST dim of both).
// It exists because ROCm's hipBLASLt batched transposed-B bf16 GEMM
// memory-faults on this stack, and the torch workaround materialised a
// contiguous transposed copy of every expert weight (hundreds of MB) per
// layer per step. NT is also the MFMA-native pattern: BOTH fragments are
// K-contiguous, so every LDS read is a 16-byte ds_read_b128.
//
// Structure (cdna_hip_programming.md §5): 128x128 tile, BK=64, 256 threads
// (4 waves as 2x2, 64x64 output per wave), v_mfma_f32_16x16x32_bf16,
// single-LDS-image register-staged pipeline (prefetch tile t+1 into VGPRs
// while computing tile t; ds_write after the barrier — T14), st_16x32 XOR
// swizzle on the LDS image so fragment ds_read_b128 is bank-spread.
// Arbitrary M/N/K (predicated edge loads/stores; zero-filled K tail).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float  f32x4;

#define BM 128
#define BN 128
#define BK 64
#define THREADS 256
// per-thread staging chunks: tile bytes (128*64*2) / (256 threads * 16B) = 4
#define CHUNKS 4

// LDS byte offset for element (row, k) of a [128][64] bf16 tile image with
// the st_16x32 swizzle: flip byte-bit-5 with byte-bit-9 (= row bit 2).
DEV_INLINE int swz(int row, int kbyte) {
  return row * (BK * 2) + (kbyte ^ (((row >> 2) & 1) << 5));
}

// predicated 16-byte (8 x bf16) load: zero-fills out-of-range elements
DEV_INLINE ushortx8 load8_guard(const uint16_t* __restrict__ p, bool row_ok,
                                int k0, int K) {
  ushortx8 v;
  if (row_ok && k0 + 8 <= K) {
    v = *reinterpret_cast<const ushortx8*>(p);
  } else {
    #pragma unroll
    for (int i = 0; i < 8; ++i)
      v[i] = (row_ok && k0 + i < K) ? p[i] : (uint16_t)0;
  }
  return v;
}

__global__ __launch_bounds__(THREADS, 2)
void grouped_gemm_nt_kernel(const uint16_t* __restrict__ Aall,
                            const uint16_t* __restrict__ Ball,
                            uint16_t* __restrict__ Oall,
                            int M, int N, int K,
                            int64_t strideA, int64_t strideB,
                            int64_t strideO) {
  __shared__ uint16_t lds[2 * BM * BK];      // [A tile | B tile], swizzled
  uint16_t* As = lds;
  uint16_t* Bs = lds + BM * BK;

  const int e = blockIdx.z;
  const uint16_t* A = Aall + e * strideA;
  const uint16_t* B = Ball + e * strideB;
  uint16_t* O = Oall + e * strideO;

  const int tileM = blockIdx.x * BM;
  const int tileN = blockIdx.y * BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = wave >> 1;                  // wave row (0..1) -> 64 rows
  const int wn = wave & 1;                   // wave col (0..1) -> 64 cols

  // staging map: chunk c (0..1023) -> row = c>>3, 16B k-group = c&7
  int s_row[CHUNKS], s_koff[CHUNKS];
  #pragma unroll
  for (int i = 0; i < CHUNKS; ++i) {
    int c = t + i * THREADS;
    s_row[i] = c >> 3;
    s_koff[i] = (c & 7) * 8;                 // in elements
  }

  const int KT = (K + BK - 1) / BK;
  ushortx8 ra[CHUNKS], rb[CHUNKS];

  // ---- load tile 0 into registers
  #pragma unroll
  for (int i = 0; i < CHUNKS; ++i) {
    int gm = tileM + s_row[i];
    int gn = tileN + s_row[i];
    int gk = s_koff[i];
    ra[i] = load8_guard(A + (int64_t)gm * K + gk, gm < M, gk, K);
    rb[i] = load8_guard(B + (int64_t)gn * K + gk, gn < N, gk, K);
  }
  // ---- write tile 0 to LDS
  #pragma unroll
  for (int i = 0; i < CHUNKS; ++i) {
    *reinterpret_cast<ushortx8*>(
        reinterpret_cast<char*>(As) + swz(s_row[i], s_koff[i] * 2)) = ra[i];
    *reinterpret_cast<ushortx8*>(
        reinterpret_cast<char*>(Bs) + swz(s_row[i], s_koff[i] * 2)) = rb[i];
  }
  __syncthreads();

  f32x4 acc[4][4] = {};

  const int fr = lane & 15;                  // fragment row/col index
  const int fg = lane >> 4;                  // k-subgroup 0..3 (8 elems each)

  for (int kt = 0; kt < KT; ++kt) {
    // prefetch next K-tile into registers (overlaps the MFMA block below)
    if (kt + 1 < KT) {
      const int k0 = (kt + 1) * BK;
      #pragma unroll
      for (int i = 0; i < CHUNKS; ++i) {
        int gm = tileM + s_row[i];
        int gn = tileN + s_row[i];
        int gk = k0 + s_koff[i];
        ra[i] = load8_guard(A + (int64_t)gm * K + gk, gm < M, gk, K);
        rb[i] = load8_guard(B + (int64_t)gn * K + gk, gn < N, gk, K);
      }
    }

    // compute on the staged tile: 2 x (8 ds_read_b128 + 16 MFMA)
    #pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      bf16x8 af[4], bf[4];
      const int kb = (kk + fg * 8) * 2;      // byte offset of this lane's k8
      #pragma unroll
      for (int m = 0; m < 4; ++m)
        af[m] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(As)
            + swz(wm * 64 + m * 16 + fr, kb));
      #pragma unroll
      for (int n = 0; n < 4; ++n)
        bf[n] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(Bs)
            + swz(wn * 64 + n * 16 + fr, kb));
      #pragma unroll
      for (int m = 0; m < 4; ++m)
        #pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[m], bf[n], acc[m][n], 0, 0, 0);
    }

    __syncthreads();
    if (kt + 1 < KT) {
      #pragma unroll
      for (int i = 0; i < CHUNKS; ++i) {
        *reinterpret_cast<ushortx8*>(
            reinterpret_cast<char*>(As) + swz(s_row[i], s_koff[i] * 2)) = ra[i];
        *reinterpret_cast<ushortx8*>(
            reinterpret_cast<char*>(Bs) + swz(s_row[i], s_koff[i] * 2)) = rb[i];
      }
      __syncthreads();
    }
  }

  // ---- epilogue: C/D map (16x16): col = lane&15, row = (lane>>4)*4 + r
  #pragma unroll
  for (int m = 0; m < 4; ++m) {
    #pragma unroll
    for (int n = 0; n < 4; ++n) {
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = tileM + wm * 64 + m * 16 + fg * 4 + r;
        int col = tileN + wn * 64 + n * 16 + fr;
        if (row < M && col < N)
          O[(int64_t)row * N + col] = f32_to_bf16(acc[m][n][r]);
      }
    }
  }
}

extern "C" void launch_grouped_gemm_nt(const void* A, const void* B, void* O,
                                       int E, int M, int N, int K,
                                       int64_t strideA, int64_t strideB,
                                       int64_t strideO, hipStream_t stream) {
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN, E);
  hipLaunchKernelGGL(grouped_gemm_nt_kernel, grid, dim3(THREADS), 0, stream,
                     (const uint16_t*)A, (const uint16_t*)B, (uint16_t*)O,
                     M, N, K, strideA, strideB, strideO);
}
