// MX-fp8 (OCP e4m3fn + e8m0 row scales) forward GEMM path for gfx950.
//
// Plain (non-scaled) fp8 MFMA runs at the bf16 rate on this chip; only the
// block-scaled v_mfma_scale_* instructions reach the ~5 PF fp8 peak
// (MI355X_MICROARCH.md §Matrix cores: MX K=128 measured 4647 TF).  Round-1
// measured exactly that: _scaled_mm e4m3 was throughput-FLAT vs bf16.
//
// Scheme: per-ROW e8m0 (power-of-two) scales on both operands -- coarser
// than OCP MX's per-32-element blocks but standard "rowwise" fp8 training
// granularity; the scale rides the MFMA scale operand (every 32-element
// block of a row shares the row scale), so the instruction still runs at
// the MX rate and the output needs no epilogue rescale.
//
// Kernels:
//   mx_quant_rows : bf16 [R, K] -> fp8 [R, Kp] (zero-padded) + u8 e8m0 [R]
//   mx_quant_cols : bf16 [K, N] -> fp8 [N, Kp] TRANSPOSED + u8 e8m0 [N]
//                   (weights quantize once per step into NT layout, so the
//                    GEMM has a single operand form)
//   gg_mx_nt      : C[e] = A[e] . B[e]^T, fp8 in / bf16 out, grouped.
//
// The forward-only discipline (bf16 backward) matches the usual fp8
// training recipe; the backward reuses the bf16 NT grouped kernel.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) int i32x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(2))) short s16x2;
typedef const __attribute__((address_space(1))) void* mx_gas;
typedef __attribute__((address_space(3))) void* mx_las;

#define MXQ_WAVES 4

// ---- e8m0 helpers ---------------------------------------------------------
// scale byte b encodes 2^(b-127); we pick e = ceil(log2(amax / 448)) so the
// scaled row fits e4m3 (|x|/2^e <= 448).
DEV_INLINE int e8m0_from_amax(float amax) {
  if (!(amax > 0.0f)) return 127;              // zero/NaN row -> scale 1
  union { float f; uint32_t u; } v{amax / 448.0f};
  int e = (int)((v.u >> 23) & 255) - 127;
  if ((v.u & 0x7fffff) != 0) ++e;              // ceil for non powers of two
  e = e < -127 ? -127 : (e > 127 ? 127 : e);
  return e + 127;
}

// ---------------------------------------------------------------------------
// rowwise quant: one wave per row
__global__ void mx_quant_rows_kernel(const uint16_t* __restrict__ X,
                                     uint8_t* __restrict__ Q,
                                     uint8_t* __restrict__ S,
                                     int64_t R, int K, int Kp) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  for (int64_t r = (int64_t)blockIdx.x * MXQ_WAVES + wid; r < R;
       r += (int64_t)gridDim.x * MXQ_WAVES) {
    const uint16_t* row = X + r * K;
    float amax = 0.0f;
    for (int k0 = lane * 8; k0 < K; k0 += 64 * 8) {
      #pragma unroll
      for (int j = 0; j < 8; ++j)
        if (k0 + j < K) amax = fmaxf(amax, fabsf(bf16_to_f32(row[k0 + j])));
    }
    amax = wave_reduce_max(amax);
    amax = __shfl(amax, 0, 64);
    const int sb = e8m0_from_amax(amax);
    if (lane == 0) S[r] = (uint8_t)sb;
    const float inv = exp2f((float)(127 - sb));   // 1 / 2^e
    uint8_t* qrow = Q + r * Kp;
    for (int k0 = lane * 8; k0 < Kp; k0 += 64 * 8) {
      uint8_t out[8];
      #pragma unroll
      for (int j = 0; j < 8; j += 2) {
        float a = (k0 + j < K) ? bf16_to_f32(row[k0 + j]) * inv : 0.0f;
        float b = (k0 + j + 1 < K) ? bf16_to_f32(row[k0 + j + 1]) * inv : 0.0f;
        union { s16x2 v; uint8_t b[4]; } t{};
        t.v = __builtin_amdgcn_cvt_scalef32_pk_fp8_f32(t.v, a, b, 1.0f,
                                                       false);
        out[j] = t.b[0];
        out[j + 1] = t.b[1];
      }
      if (k0 + 8 <= Kp)
        *reinterpret_cast<uint64_t*>(qrow + k0) =
            *reinterpret_cast<const uint64_t*>(out);
      else
        for (int j = 0; j < 8 && k0 + j < Kp; ++j) qrow[k0 + j] = out[j];
    }
  }
}

// ---------------------------------------------------------------------------
// column quant + transpose: W [K, N] bf16 -> Wq [N, Kp] fp8, S [N].
// Phase kernels: col-amax (coalesced along N), then 32x32 LDS-tiled
// transpose with quantization.
// tiled column-amax: block covers [512 k x 256 n], coalesced along N;
// per-column partials combine via atomicMax on the uint32 view (valid for
// non-negative floats).  grid (N/256, K/512, E).
__global__ void mx_colmax_kernel(const uint16_t* __restrict__ Wall,
                                 float* __restrict__ amaxall,
                                 int K, int N) {
  const uint16_t* W = Wall + (int64_t)blockIdx.z * K * N;
  float* amax = amaxall + (int64_t)blockIdx.z * N;
  const int n = blockIdx.x * blockDim.x + threadIdx.x;
  if (n >= N) return;
  const int k0 = blockIdx.y * 512;
  const int k1 = k0 + 512 < K ? k0 + 512 : K;
  float m = 0.0f;
  for (int k = k0; k < k1; ++k)
    m = fmaxf(m, fabsf(bf16_to_f32(W[(int64_t)k * N + n])));
  atomicMax(reinterpret_cast<unsigned*>(amax + n),
            __float_as_uint(m));
}

__global__ void mx_quant_t_kernel(const uint16_t* __restrict__ Wall,
                                  const float* __restrict__ amaxall,
                                  uint8_t* __restrict__ Qall,
                                  uint8_t* __restrict__ Sall,
                                  int K, int N, int Kp) {
  const uint16_t* W = Wall + (int64_t)blockIdx.z * K * N;
  const float* amax = amaxall + (int64_t)blockIdx.z * N;
  uint8_t* Q = Qall + (int64_t)blockIdx.z * N * Kp;
  uint8_t* S = Sall + (int64_t)blockIdx.z * N;
  __shared__ float tile[32][33];
  const int kb = blockIdx.x * 32;
  const int nb = blockIdx.y * 32;
  const int tx = threadIdx.x & 31;   // fast dim
  const int ty = threadIdx.x >> 5;   // 8 rows per pass
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int k = kb + ty + i * 8;
    const int n = nb + tx;
    tile[ty + i * 8][tx] = (k < K && n < N)
        ? bf16_to_f32(W[(int64_t)k * N + n]) : 0.0f;
  }
  __syncthreads();
  // write transposed: row = n, cols = k
  const int n = nb + ty;
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int nn = n + i * 8;
    if (nn >= N) continue;
    const int sb = e8m0_from_amax(amax[nn]);
    if (kb == 0 && tx == 0) S[nn] = (uint8_t)sb;
    const float inv = exp2f((float)(127 - sb));
    const int k = kb + tx;
    if (k < Kp) {
      float v = tile[tx][ty + i * 8] * inv;
      union { s16x2 s; uint8_t b[4]; } t{};
      t.s = __builtin_amdgcn_cvt_scalef32_pk_fp8_f32(t.s, v, 0.0f, 1.0f,
                                                     false);
      Q[(int64_t)nn * Kp + k] = t.b[0];
    }
  }
}

// zero the K-pad tail of the transposed quant output ([N, Kp], K..Kp-1)
__global__ void mx_zero_tail_kernel(uint8_t* __restrict__ Q, int64_t NE,
                                    int K, int Kp) {
  const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int tail = Kp - K;
  if (i < NE * tail)
    Q[(i / tail) * Kp + K + (i % tail)] = 0;
}

// ---------------------------------------------------------------------------
// grouped MX NT GEMM: C[e][M][N] = A[e][M][Kp] . B[e][N][Kp]^T  (fp8 in,
// bf16 out, row scales SA [e][M], SB [e][N]).  128x128 tile, BK=128
// (2 x mfma_scale_32x32x64), 4 waves as 2x2 of 64x64, glds double buffer,
// ONE __shared__ array (hipcc drains the glds queue otherwise).
#define MX_BM 128
#define MX_BK 128
#define MX_TILE (MX_BM * MX_BK)        // bytes per operand tile (fp8)

// [128][128] fp8 image: 8 slots of 16B per 128B row; XOR the slot with
// (row ^ row>>3)&7 -- for every colliding row distance (2,4,8,12 at the
// 32-dword row stride) the key differs, so b128 fragment reads stay
// conflict-free.
DEV_INLINE int mx_key(int row) { return (row ^ (row >> 3)) & 7; }
DEV_INLINE int mx_img(int row, int kbyte) {
  const int slot = ((kbyte >> 4) & 7) ^ mx_key(row);
  return row * MX_BK + slot * 16 + (kbyte & 15);
}

__global__ __launch_bounds__(256, 2)
void gg_mx_nt_kernel(const uint8_t* __restrict__ Aall,
                     const uint8_t* __restrict__ Ball,
                     const uint8_t* __restrict__ SAall,
                     const uint8_t* __restrict__ SBall,
                     uint16_t* __restrict__ Oall,
                     int M, int N, int K,
                     int64_t sA, int64_t sB, int64_t sO) {
  // 2-deep tile-pair ring (64 KiB -> 2 blocks/CU): the j+1 prefetch issued
  // at the top of step j targets the buffer step j-1 retired; the counted
  // drain at the step end exposes ~300 cycles of HBM latency per step,
  // which the co-resident second block covers.
  __shared__ uint8_t lds[4 * MX_TILE];

  const int e = blockIdx.z;
  const uint8_t* A = Aall + e * sA;
  const uint8_t* B = Ball + e * sB;
  const uint8_t* SA = SAall + e * M;
  const uint8_t* SB = SBall + e * N;
  uint16_t* O = Oall + e * sO;

  const int tileM = blockIdx.x * MX_BM;
  const int tileN = blockIdx.y * MX_BM;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = wave >> 1, wn = wave & 1;
  const int fr = lane & 31, fg = lane >> 5;
  const int NKT = K / MX_BK;
  const int maxA = M - 1, maxB = N - 1;

  // per-lane row scales (e8m0 exponents biased by 127), 2 m/n frags each
  int sa_e[2], sb_e[2];
  #pragma unroll
  for (int m = 0; m < 2; ++m) {
    int r = tileM + wm * 64 + m * 32 + fr;
    sa_e[m] = SA[r > maxA ? maxA : r];
    r = tileN + wn * 64 + m * 32 + fr;
    sb_e[m] = SB[r > maxB ? maxB : r];
  }

  // staging: 16 KiB tile = 1024 chunks of 16B -> 4 chunks per thread
  int c_row[4], c_koff[4];
  #pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int c = t + i * 256;
    c_row[i] = c >> 3;                         // 8 chunks per row
    const int s = c & 7;
    c_koff[i] = (s ^ mx_key(c_row[i])) * 16;   // source swizzle (rule 21)
  }

  auto issue = [&](const uint8_t* P, int maxR, int tileR, int j, int which) {
    uint8_t* base = lds + ((j & 1) * 2 + which) * MX_TILE;
    const int k0 = j * MX_BK;
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
      int r = c_row[i];
      r = r > maxR - tileR ? (maxR - tileR < 0 ? 0 : maxR - tileR) : r;
      const char* gp = reinterpret_cast<const char*>(
          P + (int64_t)(tileR + r) * K + k0 + c_koff[i]);
      mx_las lp = (mx_las)(reinterpret_cast<char*>(base)
                           + (t + i * 256) * 16);
      __builtin_amdgcn_global_load_lds((mx_gas)gp, lp, 16, 0, 0);
    }
  };

  f32x16 acc[2][2] = {};

  issue(A, maxA, tileM, 0, 0);
  issue(B, maxB, tileN, 0, 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int j = 0; j < NKT; ++j) {
    if (j + 1 < NKT) {
      issue(A, maxA, tileM, j + 1, 0);
      issue(B, maxB, tileN, j + 1, 1);
    }
    const uint8_t* As = lds + (j & 1) * 2 * MX_TILE;
    const uint8_t* Bs = As + MX_TILE;
    #pragma unroll
    for (int kk = 0; kk < 2; ++kk) {           // two 64-deep mfma steps
      i32x8 af[2], bf[2];
      #pragma unroll
      for (int m = 0; m < 2; ++m) {
        const int row = wm * 64 + m * 32 + fr;
        const int kb = kk * 64 + fg * 32;
        // 32 fp8 = two 16B slots (consecutive, swizzled independently)
        uintx4 lo = *reinterpret_cast<const uintx4*>(As + mx_img(row, kb));
        uintx4 hi = *reinterpret_cast<const uintx4*>(As + mx_img(row, kb + 16));
        union { struct { uintx4 a, b; } p; i32x8 v; } u;
        u.p.a = lo; u.p.b = hi;
        af[m] = u.v;
      }
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        const int row = wn * 64 + n * 32 + fr;
        const int kb = kk * 64 + fg * 32;
        uintx4 lo = *reinterpret_cast<const uintx4*>(Bs + mx_img(row, kb));
        uintx4 hi = *reinterpret_cast<const uintx4*>(Bs + mx_img(row, kb + 16));
        union { struct { uintx4 a, b; } p; i32x8 v; } u;
        u.p.a = lo; u.p.b = hi;
        bf[n] = u.v;
      }
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int m = 0; m < 2; ++m)
        #pragma unroll
        for (int n = 0; n < 2; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
              af[m], bf[n], acc[m][n], 0, 0, 0, sa_e[m], 0, sb_e[n]);
      __builtin_amdgcn_s_setprio(0);
    }
    if (j + 1 < NKT) {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  #pragma unroll
  for (int m = 0; m < 2; ++m)
    #pragma unroll
    for (int n = 0; n < 2; ++n)
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = tileM + wm * 64 + m * 32
                        + (r & 3) + 8 * (r >> 2) + 4 * fg;
        const int col = tileN + wn * 64 + n * 32 + fr;
        if (row < M && col < N)
          O[(int64_t)row * N + col] = f32_to_bf16(acc[m][n][r]);
      }
}

// ---------------------------------------------------------------------------
extern "C" hipError_t lumina_mx_quant_rows(const void* X, void* Q, void* S,
                                           int64_t R, int K, int Kp,
                                           hipStream_t stream) {
  int grid = (int)((R + MXQ_WAVES - 1) / MXQ_WAVES);
  grid = grid > 8192 ? 8192 : grid;
  hipLaunchKernelGGL(mx_quant_rows_kernel, dim3(grid), dim3(64 * MXQ_WAVES),
                     0, stream, (const uint16_t*)X, (uint8_t*)Q, (uint8_t*)S,
                     R, K, Kp);
  return hipGetLastError();
}

extern "C" hipError_t lumina_mx_quant_cols(const void* W, void* Q, void* S,
                                           void* amax_ws, int E, int K,
                                           int N, int Kp,
                                           hipStream_t stream) {
  hipError_t e = hipMemsetAsync(amax_ws, 0, (int64_t)E * N * 4, stream);
  if (e != hipSuccess) return e;
  hipLaunchKernelGGL(mx_colmax_kernel,
                     dim3((N + 255) / 256, (K + 511) / 512, E), dim3(256), 0,
                     stream, (const uint16_t*)W, (float*)amax_ws, K, N);
  e = hipGetLastError();
  if (e != hipSuccess) return e;
  dim3 g((K + 31) / 32, (N + 31) / 32, E);
  hipLaunchKernelGGL(mx_quant_t_kernel, g, dim3(256), 0, stream,
                     (const uint16_t*)W, (const float*)amax_ws, (uint8_t*)Q,
                     (uint8_t*)S, K, N, Kp);
  e = hipGetLastError();
  if (e != hipSuccess) return e;
  if (Kp > K) {
    int64_t tot = (int64_t)E * N * (Kp - K);
    hipLaunchKernelGGL(mx_zero_tail_kernel,
                       dim3((int)((tot + 255) / 256)), dim3(256), 0, stream,
                       (uint8_t*)Q, (int64_t)E * N, K, Kp);
    e = hipGetLastError();
  }
  return e;
}

extern "C" hipError_t lumina_gg_mx_nt(const void* A, const void* B,
                                      const void* SA, const void* SB,
                                      void* O, int E, int M, int N, int K,
                                      int64_t sA, int64_t sB, int64_t sO,
                                      hipStream_t stream) {
  dim3 grid((M + MX_BM - 1) / MX_BM, (N + MX_BM - 1) / MX_BM, E);
  hipLaunchKernelGGL(gg_mx_nt_kernel, grid, dim3(256), 0, stream,
                     (const uint8_t*)A, (const uint8_t*)B,
                     (const uint8_t*)SA, (const uint8_t*)SB,
                     (uint16_t*)O, M, N, K, sA, sB, sO);
  return hipGetLastError();
}
