// 256^2 / BK=64 / 16x16x32-MFMA grouped NT GEMM for gfx950 -- the guide's
// 8-phase template structure (cdna_hip_programming.md "The 256^2 8-phase
// template"): st_16x32 LDS swizzle, glds half-tile staging, one
// ds_read||glds||MFMA interleaved phase per (K-sub, M-quadrant), raw
// s_barrier pairs, setprio around the MFMA block, counted vmcnt only at
// K-tile boundaries.  Replaces gemm8p's 32x32x16/BK=32 structure on the
// expert grad_x path (gemm8p measured hipBLASLt parity; the template's
// finer phase interleave is the documented lever past it).
//
// Reference capability: grouped expert GEMMs (ColossalAI MLPExperts
// bmm path); this kernel is the NT form out[e] = A[e] @ B[e]^T with
// A [E,M,K], B [E,N,K] row-major, K % 64 == 0 (caller zero-pads).
//
// STATUS (round 2, measured): numerics-correct at every probed shape
// (rel-max err ~2-3e-3 vs fp32 matmul) but 816-885 TF on the b1 grad_x
// shapes / 885 at 4096^3 -- BEHIND gemm8p's 903-1084.  The exposed
// read segments between the per-phase barrier pairs are the gap (the
// guide's 1330 TF figure needs the hand-scheduled fine interleave its
// example file carries; reconstructing that from the prose alone
// plateaued here).  Kept as an experiment; the grad_x path stays on
// gemm8p.  A barrier-free compiler-scheduled variant (pf2=0) memory-
// faulted and is disabled pending a debug pass.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((address_space(3))) void* las_vptr;
typedef const __attribute__((address_space(1))) void* gas_cptr;

#define T8_THREADS 512
#define T8_BM 256
#define T8_BN 256
#define T8_BK 64
#define T8_HALF_ELEMS (128 * 64)          // one half-tile image, 16 KiB

// st_16x32 swizzle: XOR byte-bit-5 with byte-bit-9 within each 1024-byte
// (8-row) subtile -- spreads the ds_read_b128 lane groups of a fragment
// column over 4 bank slots (guide: bank-conflict 37.8M -> 267K).
DEV_INLINE int t8_swz(int byte) { return byte ^ (((byte >> 9) & 1) << 5); }

// ---------------------------------------------------------------------------
template <int PF2>   // PF2: halves prefetched per phase in phases 0..1 (2)
                     // vs one per phase across all 4 (1)
__global__ __launch_bounds__(T8_THREADS, 1)
void gg8t_kernel(const uint16_t* __restrict__ Aall,
                 const uint16_t* __restrict__ Ball,
                 uint16_t* __restrict__ Oall,
                 int M, int N, int K,
                 int64_t sA, int64_t sB, int64_t sO) {
  // one __shared__ array (the two-array glds vmcnt(0) trap, guide Sec.5a):
  // 2 buffers x [A h0 | A h1 | B h0 | B h1] x 16 KiB = 128 KiB
  __shared__ uint16_t lds_all[2 * 4 * T8_HALF_ELEMS];

  const int e = blockIdx.z;
  const uint16_t* A = Aall + e * sA;
  const uint16_t* B = Ball + e * sB;
  uint16_t* O = Oall + e * sO;

  // XCD-aware bijective remap (guide T1)
  const int nwg = gridDim.x * gridDim.y;
  const int orig = blockIdx.x + blockIdx.y * gridDim.x;
  const int q = nwg >> 3, r = nwg & 7, xcd = orig & 7;
  const int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                   + (orig >> 3);
  const int tileM = (wgid % gridDim.x) * T8_BM;
  const int tileN = (wgid / gridDim.x) * T8_BN;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wr = wave >> 2;            // M half (128 rows)
  const int wc = wave & 3;             // N quarter (64 cols)
  const int l15 = lane & 15;
  const int l4 = lane >> 4;            // 0..3: k-subgroup / C row group
  const int NKT = K / T8_BK;

  // ---- staging source coords: dest chunk c sits at swizzled home, so the
  // SOURCE coordinates are swz(16c) (involution; glds dest is lane-linear)
  int st_row[2], st_kb[2];
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int lb = t8_swz(16 * (t + i * T8_THREADS));
    st_row[i] = lb >> 7;               // logical row in the half image
    st_kb[i] = lb & 127;               // byte offset within the 64-elem row
  }

  // issue one half-tile (2 glds/thread): op 0/1 = A/B, h = row half
  auto issue_half = [&](int kt, int op, int h) {
    const uint16_t* base = op ? B : A;
    const int lim = (op ? N : M) - 1;
    const int t0 = op ? tileN : tileM;
    uint16_t* slot = lds_all + ((kt & 1) * 4 + op * 2 + h) * T8_HALF_ELEMS;
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      int row = t0 + h * 128 + st_row[i];
      row = row > lim ? (lim < 0 ? 0 : lim) : row;
      const char* gp = reinterpret_cast<const char*>(
          base + (int64_t)row * K + kt * T8_BK) + st_kb[i];
      las_vptr lp = (las_vptr)(reinterpret_cast<char*>(slot)
                               + (t + i * T8_THREADS) * 16);
      __builtin_amdgcn_global_load_lds((gas_cptr)gp, lp, 16, 0, 0);
    }
  };

  f32x4 acc[8][4] = {};

  // ---- prologue: kt 0 fully staged
  #pragma unroll
  for (int h = 0; h < 4; ++h)
    issue_half(0, h >> 1, h & 1);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const uint16_t* Ah = lds_all + (wr * T8_HALF_ELEMS);
  const uint16_t* Bh = lds_all + ((2 + (wc >> 1)) * T8_HALF_ELEMS);
  const int arow0 = l15;                        // + m*16
  const int brow0 = (wc & 1) * 64 + l15;        // + n*16

  for (int kt = 0; kt < NKT; ++kt) {
    const int bb = (kt & 1) * 4 * T8_HALF_ELEMS;
    const uint16_t* Abase = Ah + bb;
    const uint16_t* Bbase = Bh + bb;
    const bool pf = kt + 1 < NKT;
    // 4 phases: (ks, mh) -- one C M-quadrant x one K=32 sub-step each
    #pragma unroll
    for (int phx = 0; phx < 4; ++phx) {
      const int ks = phx >> 1, mh = phx & 1;
      bf16x8 af[4], bf[4];
      const int kbyte = ks * 64 + 16 * l4;
      #pragma unroll
      for (int m = 0; m < 4; ++m)
        af[m] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(Abase)
            + t8_swz(((mh * 4 + m) * 16 + arow0) * 128 + kbyte));
      #pragma unroll
      for (int n = 0; n < 4; ++n)
        bf[n] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(Bbase)
            + t8_swz((n * 16 + brow0) * 128 + kbyte));
      if (pf) {
        if (PF2 == 4) {                 // all 4 halves up front (phase 0):
          if (phx == 0)                 // 3-4 phases of flight time, the
            #pragma unroll              // boundary drain is then free
            for (int h = 0; h < 4; ++h)
              issue_half(kt + 1, h >> 1, h & 1);
        } else if (PF2 == 2) {          // 2 halves in phases 0-1
          if (phx < 2) {
            issue_half(kt + 1, phx, 0);
            issue_half(kt + 1, phx, 1);
          }
        } else {                        // 1 half per phase (PF2 0 or 1)
          issue_half(kt + 1, phx >> 1, phx & 1);
        }
      }
      if (PF2 != 0) {
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int m = 0; m < 4; ++m)
        #pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[mh * 4 + m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[m], bf[n], acc[mh * 4 + m][n], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // K-tile boundary wait folded into the last phase's closing
      // barrier: next kt's ds_reads hit the buffer the in-flight glds
      // are filling
      if (phx == 3 && pf)
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      if (PF2 != 0 || phx == 3)
        __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue
  const int r0 = tileM + wr * 128;
  const int c0 = tileN + wc * 64;
  #pragma unroll
  for (int m = 0; m < 8; ++m) {
    #pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int row = r0 + m * 16 + l4 * 4 + rr;
      if (row < M) {
        #pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int col = c0 + n * 16 + l15;
          if (col < N)
            O[(int64_t)row * N + col] = f32_to_bf16(acc[m][n][rr]);
        }
      }
    }
  }
}

extern "C" hipError_t lumina_gg8t_nt(const void* A, const void* B, void* O,
                                     int E, int M, int N, int K,
                                     int64_t sA, int64_t sB, int64_t sO,
                                     int pf2, hipStream_t stream) {
  if (K % T8_BK) return hipErrorInvalidValue;
  dim3 grid((M + T8_BM - 1) / T8_BM, (N + T8_BN - 1) / T8_BN, E);
  if (pf2 == 0) return hipErrorInvalidValue;  // barrier-free variant
                                              // faulted; see file header
  if (pf2 == 4)
    hipLaunchKernelGGL(gg8t_kernel<4>, grid, dim3(T8_THREADS), 0, stream,
                       (const uint16_t*)A, (const uint16_t*)B, (uint16_t*)O,
                       M, N, K, sA, sB, sO);
  else if (pf2 == 2)
    hipLaunchKernelGGL(gg8t_kernel<2>, grid, dim3(T8_THREADS), 0, stream,
                       (const uint16_t*)A, (const uint16_t*)B, (uint16_t*)O,
                       M, N, K, sA, sB, sO);
  else
    hipLaunchKernelGGL(gg8t_kernel<1>, grid, dim3(T8_THREADS), 0, stream,
                       (const uint16_t*)A, (const uint16_t*)B, (uint16_t*)O,
                       M, N, K, sA, sB, sO);
  return hipGetLastError();
}
