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
// st_16x32 XOR-swizzled LDS image. Two staging paths:
//  - GLDS=true (K % 64 == 0): global_load_lds dwordx4 into a double-buffered
//    LDS image, next K-tile in flight during the MFMA block. glds writes
//    lane-linear, so the swizzle is applied to the per-lane GLOBAL source
//    address (guide: "pre-swizzled global src + swizzled ds_read addr").
//    Ragged M/N rows clamp to a valid row; the epilogue masks them.
//  - GLDS=false: predicated register staging + ds_write, handles any K
//    (zero-filled tail) — the fallback for K % 64 != 0.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float  f32x4;

#define BM 128
#define BN 128
#define BK 64
#define THREADS 256
// per-thread staging chunks: tile bytes (128*64*2) / (256 threads * 16B) = 4
#define CHUNKS 4
#define TILE_ELEMS (BM * BK)

// LDS byte offset for element (row, k) of a [128][64] bf16 tile image with
// the st_16x32 swizzle: flip byte-bit-5 with row bit 2.
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

typedef const __attribute__((address_space(1))) void* gas_ptr;
typedef __attribute__((address_space(3))) void* las_ptr;

template <bool GLDS>
__global__ __launch_bounds__(THREADS, 2)
void grouped_gemm_nt_kernel(const uint16_t* __restrict__ Aall,
                            const uint16_t* __restrict__ Ball,
                            uint16_t* __restrict__ Oall,
                            int M, int N, int K,
                            int64_t strideA, int64_t strideB,
                            int64_t strideO) {
  // [A | B] per buffer; glds path double-buffers (64 KiB total)
  __shared__ uint16_t lds[(GLDS ? 4 : 2) * TILE_ELEMS];

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
  const int KT = (K + BK - 1) / BK;
  const int fr = lane & 15;                  // fragment row/col index
  const int fg = lane >> 4;                  // k-subgroup 0..3 (8 elems)

  f32x4 acc[4][4] = {};
  int buf = 0;

  if constexpr (GLDS) {
    // ---- glds staging: wave w, piece i covers chunks c = w*64 + i*256 + l
    const int maxA = M - 1, maxB = N - 1;
    auto issue_tile = [&](int which, int kt) {
      const int k0 = kt * BK;
      uint16_t* base = lds + which * 2 * TILE_ELEMS;
      #pragma unroll
      for (int i = 0; i < CHUNKS; ++i) {
        const int c = wave * 64 + i * 256 + lane;
        const int row = c >> 3;
        const int kbB = ((c & 7) * 16) ^ (((row >> 2) & 1) << 5);
        // A piece
        {
          const int r = row > maxA - tileM ? (maxA - tileM < 0 ? 0 : maxA - tileM) : row;
          const char* gp = reinterpret_cast<const char*>(
              A + (int64_t)(tileM + r) * K + k0) + kbB;
          las_ptr lp = (las_ptr)(reinterpret_cast<char*>(base)
                                 + (wave * 64 + i * 256) * 16);
          __builtin_amdgcn_global_load_lds((gas_ptr)gp, lp, 16, 0, 0);
        }
        // B piece
        {
          const int r = row > maxB - tileN ? (maxB - tileN < 0 ? 0 : maxB - tileN) : row;
          const char* gp = reinterpret_cast<const char*>(
              B + (int64_t)(tileN + r) * K + k0) + kbB;
          las_ptr lp = (las_ptr)(reinterpret_cast<char*>(base + TILE_ELEMS)
                                 + (wave * 64 + i * 256) * 16);
          __builtin_amdgcn_global_load_lds((gas_ptr)gp, lp, 16, 0, 0);
        }
      }
    };

    issue_tile(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    for (int kt = 0; kt < KT; ++kt) {
      if (kt + 1 < KT)
        issue_tile(buf ^ 1, kt + 1);        // in flight under the MFMAs
      const uint16_t* As = lds + buf * 2 * TILE_ELEMS;
      const uint16_t* Bs = As + TILE_ELEMS;
      #pragma unroll
      for (int kk = 0; kk < BK; kk += 32) {
        bf16x8 af[4], bf[4];
        const int kb = (kk + fg * 8) * 2;
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
      __syncthreads();                      // drains the in-flight glds too
      buf ^= 1;
    }
  } else {
    // ---- predicated register staging (any K)
    uint16_t* As = lds;
    uint16_t* Bs = lds + TILE_ELEMS;
    int s_row[CHUNKS], s_koff[CHUNKS];
    #pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
      int c = t + i * THREADS;
      s_row[i] = c >> 3;
      s_koff[i] = (c & 7) * 8;               // in elements
    }
    ushortx8 ra[CHUNKS], rb[CHUNKS];
    // block-uniform fast path: interior tiles with a complete K-step need no
    // per-element guards (the guarded path measured ~180 TF vs ~650)
    const bool intA = tileM + BM <= M;
    const bool intB = tileN + BN <= N;

    auto stage_regs = [&](int k0) {
      const bool kfull = k0 + BK <= K;
      #pragma unroll
      for (int i = 0; i < CHUNKS; ++i) {
        const int gm = tileM + s_row[i];
        const int gn = tileN + s_row[i];
        const int gk = k0 + s_koff[i];
        if (intA && kfull)
          ra[i] = *reinterpret_cast<const ushortx8*>(A + (int64_t)gm * K + gk);
        else
          ra[i] = load8_guard(A + (int64_t)gm * K + gk, gm < M, gk, K);
        if (intB && kfull)
          rb[i] = *reinterpret_cast<const ushortx8*>(B + (int64_t)gn * K + gk);
        else
          rb[i] = load8_guard(B + (int64_t)gn * K + gk, gn < N, gk, K);
      }
    };

    stage_regs(0);
    #pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
      *reinterpret_cast<ushortx8*>(
          reinterpret_cast<char*>(As) + swz(s_row[i], s_koff[i] * 2)) = ra[i];
      *reinterpret_cast<ushortx8*>(
          reinterpret_cast<char*>(Bs) + swz(s_row[i], s_koff[i] * 2)) = rb[i];
    }
    __syncthreads();

    for (int kt = 0; kt < KT; ++kt) {
      if (kt + 1 < KT)
        stage_regs((kt + 1) * BK);
      #pragma unroll
      for (int kk = 0; kk < BK; kk += 32) {
        bf16x8 af[4], bf[4];
        const int kb = (kk + fg * 8) * 2;
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
  const bool aligned = (K % BK == 0)
      && ((reinterpret_cast<uintptr_t>(A) & 15) == 0)
      && ((reinterpret_cast<uintptr_t>(B) & 15) == 0);
  if (aligned) {
    hipLaunchKernelGGL(grouped_gemm_nt_kernel<true>, grid, dim3(THREADS), 0,
                       stream, (const uint16_t*)A, (const uint16_t*)B,
                       (uint16_t*)O, M, N, K, strideA, strideB, strideO);
  } else {
    hipLaunchKernelGGL(grouped_gemm_nt_kernel<false>, grid, dim3(THREADS), 0,
                       stream, (const uint16_t*)A, (const uint16_t*)B,
                       (uint16_t*)O, M, N, K, strideA, strideB, strideO);
  }
}

// ---------------------------------------------------------------------------
// v2: same 128x128/BK=64 glds structure on v_mfma_f32_32x32x16_bf16
// (per-wave 64x64 = 2x2 fragments of 32x32; higher MFMA ceiling than the
// 16x16 shape: 2382 vs 2075 TF ubench). Fragment maps (cdna4):
//   A/B: lane l holds 8 contiguous k at row/col = l&31, k8 = (l>>5)
//   C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
// The 32-lane fragment groups hit LDS banks differently from the 16-lane
// pattern, so the swizzle is a template knob measured on hardware:
//   SWZ 0: byte5 ^= row bit2 (st_16x32, as v1)
//   SWZ 1: byte5:4 ^= row bits1:0 (spreads 32-row groups over 4 slots)
typedef __attribute__((ext_vector_type(16))) float f32x16;

template <int SWZ>
DEV_INLINE int swz2(int row, int kbyte) {
  if (SWZ == 0) return row * (BK * 2) + (kbyte ^ (((row >> 2) & 1) << 5));
  return row * (BK * 2) + (kbyte ^ ((row & 3) << 4));
}

template <int SWZ>
__global__ __launch_bounds__(THREADS, 2)
void grouped_gemm_nt32_kernel(const uint16_t* __restrict__ Aall,
                              const uint16_t* __restrict__ Ball,
                              uint16_t* __restrict__ Oall,
                              int M, int N, int K,
                              int64_t strideA, int64_t strideB,
                              int64_t strideO) {
  __shared__ uint16_t lds[4 * TILE_ELEMS];
  const int e = blockIdx.z;
  const uint16_t* A = Aall + e * strideA;
  const uint16_t* B = Ball + e * strideB;
  uint16_t* O = Oall + e * strideO;
  const int tileM = blockIdx.x * BM;
  const int tileN = blockIdx.y * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 1;
  const int wn = wave & 1;
  const int KT = K / BK;                     // aligned-only kernel
  const int fr = lane & 31;
  const int fg = lane >> 5;                  // k8-group 0..1
  const int maxA = M - 1, maxB = N - 1;

  f32x16 acc[2][2] = {};
  int buf = 0;

  auto issue_tile = [&](int which, int kt) {
    const int k0 = kt * BK;
    uint16_t* base = lds + which * 2 * TILE_ELEMS;
    #pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
      const int c = wave * 64 + i * 256 + lane;
      const int row = c >> 3;
      const int kbB = swz2<SWZ>(0, (c & 7) * 16);  // swizzle of kbyte only
      const int kbBr = swz2<SWZ>(row, (c & 7) * 16) - row * (BK * 2);
      {
        const int r = row > maxA - tileM ? (maxA - tileM < 0 ? 0 : maxA - tileM) : row;
        const char* gp = reinterpret_cast<const char*>(
            A + (int64_t)(tileM + r) * K + k0) + kbBr;
        las_ptr lp = (las_ptr)(reinterpret_cast<char*>(base)
                               + (wave * 64 + i * 256) * 16);
        __builtin_amdgcn_global_load_lds((gas_ptr)gp, lp, 16, 0, 0);
      }
      {
        const int r = row > maxB - tileN ? (maxB - tileN < 0 ? 0 : maxB - tileN) : row;
        const char* gp = reinterpret_cast<const char*>(
            B + (int64_t)(tileN + r) * K + k0) + kbBr;
        las_ptr lp = (las_ptr)(reinterpret_cast<char*>(base + TILE_ELEMS)
                               + (wave * 64 + i * 256) * 16);
        __builtin_amdgcn_global_load_lds((gas_ptr)gp, lp, 16, 0, 0);
      }
    }
  };

  issue_tile(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int kt = 0; kt < KT; ++kt) {
    if (kt + 1 < KT) issue_tile(buf ^ 1, kt + 1);
    const uint16_t* As = lds + buf * 2 * TILE_ELEMS;
    const uint16_t* Bs = As + TILE_ELEMS;
    #pragma unroll
    for (int kk = 0; kk < BK; kk += 16) {
      bf16x8 af[2], bf[2];
      const int kb = (kk + fg * 8) * 2;
      #pragma unroll
      for (int m = 0; m < 2; ++m)
        af[m] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(As)
            + swz2<SWZ>(wm * 64 + m * 32 + fr, kb));
      #pragma unroll
      for (int n = 0; n < 2; ++n)
        bf[n] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(Bs)
            + swz2<SWZ>(wn * 64 + n * 32 + fr, kb));
      #pragma unroll
      for (int m = 0; m < 2; ++m)
        #pragma unroll
        for (int n = 0; n < 2; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[m], bf[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();
    buf ^= 1;
  }

  #pragma unroll
  for (int m = 0; m < 2; ++m) {
    #pragma unroll
    for (int n = 0; n < 2; ++n) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = tileM + wm * 64 + m * 32
                  + (r & 3) + 8 * (r >> 2) + 4 * fg;
        int col = tileN + wn * 64 + n * 32 + fr;
        if (row < M && col < N)
          O[(int64_t)row * N + col] = f32_to_bf16(acc[m][n][r]);
      }
    }
  }
}

extern "C" void launch_grouped_gemm_nt_v2(const void* A, const void* B,
                                          void* O, int E, int M, int N, int K,
                                          int64_t sA, int64_t sB, int64_t sO,
                                          int swz_mode, hipStream_t stream) {
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN, E);
  if (swz_mode == 0)
    hipLaunchKernelGGL(grouped_gemm_nt32_kernel<0>, grid, dim3(THREADS), 0,
                       stream, (const uint16_t*)A, (const uint16_t*)B,
                       (uint16_t*)O, M, N, K, sA, sB, sO);
  else
    hipLaunchKernelGGL(grouped_gemm_nt32_kernel<1>, grid, dim3(THREADS), 0,
                       stream, (const uint16_t*)A, (const uint16_t*)B,
                       (uint16_t*)O, M, N, K, sA, sB, sO);
}

// ---------------------------------------------------------------------------
// v3: 16x16x32 with THREE LDS buffers, counted vmcnt and raw s_barrier —
// tiles stay in flight across barriers instead of draining (the 2-buffer
// path's __syncthreads carries a vmcnt(0) while a glds is outstanding).
// Per-wave glds queues are same-aged across waves (uniform staging order),
// so one counted vmcnt at every wave + a barrier guarantees the tile all
// waves are about to read has fully landed. 96 KiB LDS -> 1 block/CU.
template <int DUMMY>
__global__ __launch_bounds__(THREADS, 1)
void grouped_gemm_nt3_kernel(const uint16_t* __restrict__ Aall,
                             const uint16_t* __restrict__ Ball,
                             uint16_t* __restrict__ Oall,
                             int M, int N, int K,
                             int64_t strideA, int64_t strideB,
                             int64_t strideO) {
  __shared__ uint16_t lds[6 * TILE_ELEMS];   // 3 buffers x [A|B]
  const int e = blockIdx.z;
  const uint16_t* A = Aall + e * strideA;
  const uint16_t* B = Ball + e * strideB;
  uint16_t* O = Oall + e * strideO;
  const int tileM = blockIdx.x * BM;
  const int tileN = blockIdx.y * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 1;
  const int wn = wave & 1;
  const int KT = K / BK;
  const int fr = lane & 15;
  const int fg = lane >> 4;
  const int maxA = M - 1, maxB = N - 1;

  f32x4 acc[4][4] = {};

  auto issue_tile = [&](int which, int kt) {
    const int k0 = kt * BK;
    uint16_t* base = lds + which * 2 * TILE_ELEMS;
    #pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
      const int c = wave * 64 + i * 256 + lane;
      const int row = c >> 3;
      const int kbB = ((c & 7) * 16) ^ (((row >> 2) & 1) << 5);
      {
        const int r = row > maxA - tileM ? (maxA - tileM < 0 ? 0 : maxA - tileM) : row;
        const char* gp = reinterpret_cast<const char*>(
            A + (int64_t)(tileM + r) * K + k0) + kbB;
        las_ptr lp = (las_ptr)(reinterpret_cast<char*>(base)
                               + (wave * 64 + i * 256) * 16);
        __builtin_amdgcn_global_load_lds((gas_ptr)gp, lp, 16, 0, 0);
      }
      {
        const int r = row > maxB - tileN ? (maxB - tileN < 0 ? 0 : maxB - tileN) : row;
        const char* gp = reinterpret_cast<const char*>(
            B + (int64_t)(tileN + r) * K + k0) + kbB;
        las_ptr lp = (las_ptr)(reinterpret_cast<char*>(base + TILE_ELEMS)
                               + (wave * 64 + i * 256) * 16);
        __builtin_amdgcn_global_load_lds((gas_ptr)gp, lp, 16, 0, 0);
      }
    }
  };

  // prologue: t0 and t1 in flight; wait for t0 only (t1 keeps flying)
  issue_tile(0, 0);
  if (KT > 1) issue_tile(1, 1);
  if (KT > 1)
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");   // t1's 8 glds may stay outstanding
  else
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int kt = 0; kt < KT; ++kt) {
    if (kt + 2 < KT)
      issue_tile((kt + 2) % 3, kt + 2);
    const uint16_t* As = lds + (kt % 3) * 2 * TILE_ELEMS;
    const uint16_t* Bs = As + TILE_ELEMS;
    #pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      bf16x8 af[4], bf[4];
      const int kb = (kk + fg * 8) * 2;
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
    // next iteration reads tile kt+1: wait until only the newest tile's
    // glds can still be in flight, then rendezvous
    if (kt + 2 < KT)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

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

extern "C" void launch_grouped_gemm_nt_v3(const void* A, const void* B,
                                          void* O, int E, int M, int N, int K,
                                          int64_t sA, int64_t sB, int64_t sO,
                                          hipStream_t stream) {
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN, E);
  hipLaunchKernelGGL(grouped_gemm_nt3_kernel<0>, grid, dim3(THREADS), 0,
                     stream, (const uint16_t*)A, (const uint16_t*)B,
                     (uint16_t*)O, M, N, K, sA, sB, sO);
}

// ---------------------------------------------------------------------------
// v4: v1's 2-buffer glds geometry, but raw s_barrier + issue-then-counted-
// vmcnt ordering: each iteration first issues tile kt+1 (8 glds), then waits
// vmcnt(8) — draining exactly the OLDER tile kt while kt+1 stays in flight —
// computes, and crosses a raw barrier with the prefetch still outstanding
// (same-aged per-wave queues make the counted wait a global guarantee).
// Keeps 64 KiB LDS -> 2 blocks/CU, unlike the 3-buffer v3.
template <int DUMMY>
__global__ __launch_bounds__(THREADS, 2)
void grouped_gemm_nt4_kernel(const uint16_t* __restrict__ Aall,
                             const uint16_t* __restrict__ Ball,
                             uint16_t* __restrict__ Oall,
                             int M, int N, int K,
                             int64_t strideA, int64_t strideB,
                             int64_t strideO) {
  __shared__ uint16_t lds[4 * TILE_ELEMS];
  const int e = blockIdx.z;
  const uint16_t* A = Aall + e * strideA;
  const uint16_t* B = Ball + e * strideB;
  uint16_t* O = Oall + e * strideO;
  const int tileM = blockIdx.x * BM;
  const int tileN = blockIdx.y * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 1;
  const int wn = wave & 1;
  const int KT = K / BK;
  const int fr = lane & 15;
  const int fg = lane >> 4;
  const int maxA = M - 1, maxB = N - 1;
  f32x4 acc[4][4] = {};

  auto issue_tile = [&](int which, int kt) {
    const int k0 = kt * BK;
    uint16_t* base = lds + which * 2 * TILE_ELEMS;
    #pragma unroll
    for (int i = 0; i < CHUNKS; ++i) {
      const int c = wave * 64 + i * 256 + lane;
      const int row = c >> 3;
      const int kbB = ((c & 7) * 16) ^ (((row >> 2) & 1) << 5);
      {
        const int r = row > maxA - tileM ? (maxA - tileM < 0 ? 0 : maxA - tileM) : row;
        const char* gp = reinterpret_cast<const char*>(
            A + (int64_t)(tileM + r) * K + k0) + kbB;
        las_ptr lp = (las_ptr)(reinterpret_cast<char*>(base)
                               + (wave * 64 + i * 256) * 16);
        __builtin_amdgcn_global_load_lds((gas_ptr)gp, lp, 16, 0, 0);
      }
      {
        const int r = row > maxB - tileN ? (maxB - tileN < 0 ? 0 : maxB - tileN) : row;
        const char* gp = reinterpret_cast<const char*>(
            B + (int64_t)(tileN + r) * K + k0) + kbB;
        las_ptr lp = (las_ptr)(reinterpret_cast<char*>(base + TILE_ELEMS)
                               + (wave * 64 + i * 256) * 16);
        __builtin_amdgcn_global_load_lds((gas_ptr)gp, lp, 16, 0, 0);
      }
    }
  };

  issue_tile(0, 0);
  int buf = 0;
  for (int kt = 0; kt < KT; ++kt) {
    if (kt + 1 < KT) {
      issue_tile(buf ^ 1, kt + 1);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");   // tile kt landed
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();      // all waves see tile kt in LDS
    const uint16_t* As = lds + buf * 2 * TILE_ELEMS;
    const uint16_t* Bs = As + TILE_ELEMS;
    #pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      bf16x8 af[4], bf[4];
      const int kb = (kk + fg * 8) * 2;
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
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();      // done reading buf; next iter reuses it
    buf ^= 1;
  }

  #pragma unroll
  for (int m = 0; m < 4; ++m)
    #pragma unroll
    for (int n = 0; n < 4; ++n)
      #pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = tileM + wm * 64 + m * 16 + fg * 4 + r;
        int col = tileN + wn * 64 + n * 16 + fr;
        if (row < M && col < N)
          O[(int64_t)row * N + col] = f32_to_bf16(acc[m][n][r]);
      }
}

extern "C" void launch_grouped_gemm_nt_v4(const void* A, const void* B,
                                          void* O, int E, int M, int N, int K,
                                          int64_t sA, int64_t sB, int64_t sO,
                                          hipStream_t stream) {
  dim3 grid((M + BM - 1) / BM, (N + BN - 1) / BN, E);
  hipLaunchKernelGGL(grouped_gemm_nt4_kernel<0>, grid, dim3(THREADS), 0,
                     stream, (const uint16_t*)A, (const uint16_t*)B,
                     (uint16_t*)O, M, N, K, sA, sB, sO);
}
