// Deep-pipelined 256x256 grouped GEMM for CDNA4 / gfx950 (the "8-phase"
// class structure of cdna_hip_programming.md §5: 256² tile, counted-vmcnt
// glds pipeline, raw barriers, setprio).  Round-1's 128² 2-barrier grouped
// kernels ceilinged at ~850-910 TF; this is the documented path past it.
//
// Geometry:
//   tile 256x256, K streamed in 32-deep steps; 8 waves (512 threads) as
//   2(M) x 4(N), per-wave output 128x64 = 8x4 fragments of 16x16;
//   v_mfma_f32_16x16x32_bf16 -> 32 MFMA per wave per K-step.
//   LDS: ring of 8 half-slots of 16 KiB ([256 rows][32 k] bf16) = 128 KiB;
//   K-step j uses slots (2j, 2j+1) mod 8 (A, B).  During step j the two
//   glds pairs for step j+2 are issued into the slots step j-2 retired, so
//   entering step j needs only s_waitcnt vmcnt(6) (3 half-slots = 6 glds
//   per wave in flight) -- loads stay in flight across the single raw
//   barrier per step (guide T3+T4).
//
// Operand forms:
//   NT:  A [E,M,K], B [E,N,K], both row-major K-contiguous (MoE grad_x,
//        and any A.B^T).  Both LDS images row-major with the XOR slot
//        swizzle; fragments via ds_read_b128.
//   NN:  A [E,M,K], B [E,K,N] (expert FORWARD x.W without transposing the
//        weights).  The B image is the [4 k][16 n]-blocked transpose-read
//        layout; fragments via 2x ds_read_b64_tr_b16 (semantics probed in
//        scripts/probe_tr.hip: lane supplies its own 8B-aligned address,
//        hardware redistributes out[l][j] = in[4j + ((l>>2)&3)][l&3]).
//
// K must be a multiple of 32 (callers pad; luminaai pads 1908 -> 1920).
// M and N tails are handled by clamped staging + guarded epilogue.

#include "common.h"
#include <stdlib.h>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float  f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((address_space(3))) bf16x4 lds_bf16x4_t;
typedef const __attribute__((address_space(1))) void* gas_cptr;
typedef __attribute__((address_space(3))) void* las_vptr;

#define G8_THREADS 512
#define G8_BM 256
#define G8_BN 256
#define G8_BK 32
// one half-slot: [256 rows][32 k] bf16 = 16 KiB; ring of 8
#define SLOT_ELEMS (256 * 32)

// ---- row-image addressing (A always; B in NT form) ------------------------
// [256 rows][32 k] bf16: 64-byte rows, 4 slots of 16B; XOR the slot with
// row>>2 for conflict-free ds_read_b128 column... fragment reads.
DEV_INLINE int g8_rimg(int row, int kbyte) {
  const int slot = (kbyte >> 4) ^ ((row >> 2) & 3);
  return row * 64 + slot * 16 + (kbyte & 15);
}

// glds writes lane-linear, so the swizzle lives on the SOURCE k-offset
// (rule 21): LDS 16B chunk c = row*4 + s holds global k-chunk (s ^ ((row>>2)&3)).
DEV_INLINE int g8_src_koff(int row, int s) {     // in elements
  return (s ^ ((row >> 2) & 3)) * 8;
}

// ---- blocked B image for the NN form --------------------------------------
// element (k, n) of the [BK=32 k][256 n] tile at element offset
//   (k>>2)*(256*4 + 8) + ((n>>4)<<6) + ((k&3)<<4) + (n&15)
// (8-element inter-quad pad keeps row reads off 4-way banks; tr reads are
// in-block and unaffected).
DEV_INLINE int g8_blk(int k, int n) {
  return (k >> 2) * (256 * 4 + 8) + ((n >> 4) << 6) + ((k & 3) << 4)
         + (n & 15);
}
DEV_INLINE int g8_blk_raddr(int k, int n) {      // tr16 read address
  return (k >> 2) * (256 * 4 + 8) + ((n >> 4) << 6) + (((n >> 2) & 3) << 4)
         + ((n & 3) << 2);
}
#define BSLOT_ELEMS (8 * (256 * 4 + 8))          // 8 k-quads

DEV_INLINE bf16x4 g8_tr16(const uint16_t* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_bf16x4_t*)(p));
}

// ---------------------------------------------------------------------------
// kernel: one 256x256 output tile per workgroup, z = expert
// NN: B is [K, N] (k-major rows), else [N, K]
template <bool NN>
__global__ __launch_bounds__(G8_THREADS, 1)
void gg8p_kernel(const uint16_t* __restrict__ Aall,
                 const uint16_t* __restrict__ Ball,
                 uint16_t* __restrict__ Oall,
                 int M, int N, int K, int Kb,
                 int64_t sA, int64_t sB, int64_t sO, int noremap) {
  // ONE __shared__ array only: a second __shared__ object makes hipcc emit
  // s_waitcnt vmcnt(0) before the first ds_read of every k-step of a glds
  // pipeline (guide Sec.5 'Three .s-level traps', a) -- that full drain was
  // measured here as a 2x slowdown.  A slots at 0, B slots after.
  __shared__ uint16_t lds_all[4 * SLOT_ELEMS
                              + 4 * (NN ? BSLOT_ELEMS : SLOT_ELEMS)];
  uint16_t (*lsA)[SLOT_ELEMS] =
      reinterpret_cast<uint16_t (*)[SLOT_ELEMS]>(lds_all);
  uint16_t (*lsB)[NN ? BSLOT_ELEMS : SLOT_ELEMS] =
      reinterpret_cast<uint16_t (*)[NN ? BSLOT_ELEMS : SLOT_ELEMS]>(
          lds_all + 4 * SLOT_ELEMS);

  const int e = blockIdx.z;
  const uint16_t* A = Aall + e * sA;
  const uint16_t* B = Ball + e * sB;
  uint16_t* O = Oall + e * sO;

  // XCD-aware bijective remap (guide T1): give each XCD a contiguous chunk
  // of the per-expert tile space so co-resident tiles share A/B panels in
  // their XCD's L2.
  const int nwg = gridDim.x * gridDim.y;
  const int orig = blockIdx.x + blockIdx.y * gridDim.x;
  const int q = nwg >> 3, r = nwg & 7, xcd = orig & 7;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
             + (orig >> 3);
  if (noremap) wgid = orig;        // A/B knob (8192^3 regression diag)
  const int tileM = (wgid % gridDim.x) * G8_BM;
  const int tileN = (wgid / gridDim.x) * G8_BN;
  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wave = t >> 6;
  const int wm = wave >> 2;            // 0..1 -> 128 rows
  const int wn = wave & 3;             // 0..3 -> 64 cols
  const int fr = lane & 31;            // 32x32 fragment row/col
  const int fg = lane >> 5;            // k-half-subgroup (8 elems)
  const int NKS = K / G8_BK;           // 32-deep K-steps

  // ---- staging source coordinates (threadIdx-only, computed once) ----
  // A half: 16 KiB = 1024 chunks; thread owns chunks t and t+512.
  // chunk c -> row = c>>2, slot s = c&3 (swizzled source k-chunk).
  int a_row[2], a_koff[2];
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int c = t + i * G8_THREADS;
    a_row[i] = c >> 2;
    a_koff[i] = g8_src_koff(a_row[i], c & 3);
  }
  // B half: NT mirrors A; NN stages [32 k][256 n] k-major rows: chunk c of
  // the blocked image covers (k = ?, n0 = ?) per g8_blk layout: iterate the
  // blocked image linearly in 16B chunks; padded quads make the mapping
  // irregular, so precompute (k, n0) from the chunk index directly.
  int b_k[2], b_n0[2], b_ldsoff[2];
  #pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int c = t + i * G8_THREADS;
    if (NN) {
      // blocked image: quad kq = k>>2 holds 256*4 elements of data in
      // chunks: within quad: 16 n-blocks x 4 k x (2 halves of 8 n).
      // chunk index within quad: cq = c - kq*128  (128 chunks of data)
      const int kq = c >> 7;           // 128 16B-chunks per k-quad
      const int cq = c & 127;
      // data chunk cq: n-block nb = cq>>3, k-in-quad kk = (cq>>1)&3,
      // half h = cq&1  -> element (k = kq*4+kk, n = nb*16 + h*8)
      const int nb = cq >> 3, kk = (cq >> 1) & 3, hh = cq & 1;
      b_k[i] = kq * 4 + kk;
      b_n0[i] = nb * 16 + hh * 8;
      b_ldsoff[i] = g8_blk(b_k[i], b_n0[i]);
    } else {
      b_k[i] = a_koff[i];              // same layout as A
      b_n0[i] = a_row[i];              // "row" is the n index
      b_ldsoff[i] = 0;
    }
  }

  const int maxA = M - 1, maxB = N - 1;

  // issue one A half (k-step j) into ring slot (j & 3)
  auto issue_A = [&](int j) {
    const int k0 = j * G8_BK;
    uint16_t* base = lsA[j & 3];
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      int r = a_row[i];
      r = r > maxA - tileM ? (maxA - tileM < 0 ? 0 : maxA - tileM) : r;
      const char* gp = reinterpret_cast<const char*>(
          A + (int64_t)(tileM + r) * K + k0 + a_koff[i]);
      las_vptr lp = (las_vptr)(reinterpret_cast<char*>(base)
                               + (t + i * G8_THREADS) * 16);
      __builtin_amdgcn_global_load_lds((gas_cptr)gp, lp, 16, 0, 0);
    }
  };
  auto issue_B = [&](int j) {
    const int k0 = j * G8_BK;
    uint16_t* base = lsB[j & 3];
    #pragma unroll
    for (int i = 0; i < 2; ++i) {
      if (NN) {
        // source: B[k0 + b_k][tileN + b_n0 .. +7]; clamp k to B's valid
        // rows (caller zero-pads A's K instead) and keep the 16B chunk
        // inside the row
        int n = b_n0[i];
        const int nlim = N - 8 - tileN;
        n = n > nlim ? (nlim > 0 ? (nlim & ~7) : 0) : n;
        int k = k0 + b_k[i];
        k = k >= Kb ? Kb - 1 : k;
        const char* gp = reinterpret_cast<const char*>(
            B + (int64_t)k * N + tileN + n);
        las_vptr lp = (las_vptr)(reinterpret_cast<char*>(base)
                                 + b_ldsoff[i] * 2);
        __builtin_amdgcn_global_load_lds((gas_cptr)gp, lp, 16, 0, 0);
      } else {
        int r = b_n0[i];
        r = r > maxB - tileN ? (maxB - tileN < 0 ? 0 : maxB - tileN) : r;
        const char* gp = reinterpret_cast<const char*>(
            B + (int64_t)(tileN + r) * K + k0 + b_k[i]);
        las_vptr lp = (las_vptr)(reinterpret_cast<char*>(base)
                                 + (t + i * G8_THREADS) * 16);
        __builtin_amdgcn_global_load_lds((gas_cptr)gp, lp, 16, 0, 0);
      }
    }
  };

  // 32x32x16 MFMA: per-wave 128x64 output = 4(M) x 2(N) fragments
  f32x16 acc[4][2] = {};

  // ---- prologue: steps 0..2 in flight; wait for step 0 only
  issue_A(0); issue_B(0);
  if (NKS > 1) { issue_A(1); issue_B(1); }
  if (NKS > 2) { issue_A(2); issue_B(2); }
  asm volatile("s_waitcnt vmcnt(8)" ::: "memory");   // step 0 landed
  __builtin_amdgcn_s_barrier();

  for (int j = 0; j < NKS; ++j) {
    const uint16_t* As = lsA[j & 3];
    const uint16_t* Bs = lsB[j & 3];

    // ---- 16 MFMAs (32x32x16, two 16-deep k-halves) over this K-step.
    // No explicit lgkmcnt: hipcc emits counted lgkm waits between each
    // ds_read and its consuming MFMA (near-optimal fine scheduling).
    // two k-half phases per step: {frag reads for this half, one glds
    // issue, 8 MFMAs} -- keeps memory issue interleaved with the matrix
    // pipe instead of a coarse all-reads-then-all-MFMA block
    #pragma unroll
    for (int hh = 0; hh < 2; ++hh) {
      bf16x8 af[4], bf[2];
      #pragma unroll
      for (int m = 0; m < 4; ++m)
        af[m] = *reinterpret_cast<const bf16x8*>(
            reinterpret_cast<const char*>(As)
            + g8_rimg(wm * 128 + m * 32 + fr, hh * 32 + fg * 16));
      #pragma unroll
      for (int n = 0; n < 2; ++n) {
        if (NN) {
          const int nn = wn * 64 + n * 32 + fr;
          const int k0 = hh * 16 + fg * 8;
          bf16x4 lo = g8_tr16(Bs + g8_blk_raddr(k0, nn));
          bf16x4 hi = g8_tr16(Bs + g8_blk_raddr(k0 + 4, nn));
          union { struct { bf16x4 a, b; } p; bf16x8 v; } u;
          u.p.a = lo; u.p.b = hi;
          bf[n] = u.v;
        } else {
          bf[n] = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const char*>(Bs)
              + g8_rimg(wn * 64 + n * 32 + fr, hh * 32 + fg * 16));
        }
      }
      if (j + 3 < NKS) {
        if (hh == 0) issue_A(j + 3);
        else issue_B(j + 3);
      }
      // template phase discipline: rendezvous, drain the phase's LDS
      // reads, then a pure back-to-back MFMA cluster under setprio
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int m = 0; m < 4; ++m)
        #pragma unroll
        for (int n = 0; n < 2; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[m], bf[n], acc[m][n], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }

    // next step's slots must be landed before any wave reads them: allow
    // steps j+2 and j+3 (8 glds) to stay in flight across the barrier.
    if (j + 1 < NKS) {
      if (j + 3 < NKS)
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      else if (j + 2 < NKS)
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue: C map (32x32): col = l&31, row = (r&3)+8*(r>>2)+4*fg
  #pragma unroll
  for (int m = 0; m < 4; ++m) {
    #pragma unroll
    for (int n = 0; n < 2; ++n) {
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int row = tileM + wm * 128 + m * 32
                        + (r & 3) + 8 * (r >> 2) + 4 * fg;
        const int col = tileN + wn * 64 + n * 32 + fr;
        if (row < M && col < N)
          O[(int64_t)row * N + col] = f32_to_bf16(acc[m][n][r]);
      }
    }
  }
}

extern "C" void launch_gg8p(const void* A, const void* B, void* O,
                            int E, int M, int N, int K, int Kb,
                            int64_t sA, int64_t sB, int64_t sO,
                            int nn_form, hipStream_t stream) {
  dim3 grid((M + G8_BM - 1) / G8_BM, (N + G8_BN - 1) / G8_BN, E);
  // round-2 diag: the XCD-contiguous remap LOSES at every probed size
  // (8192^3: 743 vs 1040 TF; b1 grad_x shapes: 971 vs 1012) -- at these
  // grids an XCD's contiguous chunk walks whole M-columns, thrashing its
  // L2 on the A panel.  Linear order is the default; LUMINA_GG8P_REMAP=1
  // re-enables the remap for A/B.
  static const int noremap = getenv("LUMINA_GG8P_REMAP") == nullptr;
  if (nn_form)
    hipLaunchKernelGGL(gg8p_kernel<true>, grid, dim3(G8_THREADS), 0, stream,
                       (const uint16_t*)A, (const uint16_t*)B, (uint16_t*)O,
                       M, N, K, Kb, sA, sB, sO, noremap);
  else
    hipLaunchKernelGGL(gg8p_kernel<false>, grid, dim3(G8_THREADS), 0, stream,
                       (const uint16_t*)A, (const uint16_t*)B, (uint16_t*)O,
                       M, N, K, Kb, sA, sB, sO, noremap);
}
