// bf16 MFMA GEMM, 256^2-tile 8-phase pipelined structure (CDNA4 guide §5
// "The 256² 8-phase template"). The high tier of the GPU health-validation
// burn-in: the 128^2 double-buffered kernel (gemm_bf16.hip) measures ~835 TF;
// this structure's published figures are ~1330 TF @4096^3 on random operands.
//
// Geometry (guide table): BM=BN=256, BK=64, 8 waves as 2(M)x4(N) = 512
// threads; per-wave output 128x64; LDS = 2 buffers x 4 half-tiles x
// [128][64] bf16 = 128 KiB; 2 global_load_lds (16 B/lane) per half-tile per
// thread; st_16x32 LDS XOR swizzle applied on the glds SOURCE address and
// the ds_read address (both-sides-or-neither, guide §5.4 rule 21 — the LDS
// destination stays lane-linear because glds writes base + lane*16).
//
// Schedule derivation (this file's invariants):
//   * half-tile h (4 per K-tile: 0=A-half0, 1=B-half0, 2=A-half1,
//     3=B-half1) lives in LDS slot h%8;
//   * tile t is computed in 4 phases j=0..3 over C-quadrants
//     (qm,qn) = (0,0),(0,1),(1,1),(1,0): A fragments re-read when qm
//     changes (phases 0 and 2, 8 ds_read_b128), B fragments kept in two
//     register sets read at phases 0 and 1 (4 ds_read_b128 each) — so a
//     slot is never ds_read after the phase that may overwrite it;
//   * phase j stages half h = 4t+7+j (slot (4t+7+j)%8): j=0 writes the
//     other buffer; j=1..3 write this buffer's slots whose last read was
//     phase 0 (A0), 0 (B0), 2 (A1) respectively — always behind the
//     barrier that follows those reads;
//   * s_waitcnt vmcnt at the END of phase 3 (guide: "phases 4 and 8"),
//     before the trailing barrier: all of tile t+1's halves complete,
//     <=3 half-tiles (6 loads) of tile t+2 still in flight;
//   * raw s_barrier + inline-asm lgkmcnt(0) + sched_barrier(0) (rule 18:
//     hipcc hoists register-only MFMA past an asm lgkmcnt without it);
//     __syncthreads is never used in the loop (its fence would drain the
//     in-flight glds to vmcnt(0) — the "glds span barrier" trap).
//
// Constraints: M,N multiples of 256; K multiple of 64 with K >= 192
// (prologue stages 7 half-tiles = tile0 + 3/4 of tile1).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace gemm_bf16_8ph {

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

constexpr int BM = 256;
constexpr int BN = 256;
constexpr int BK = 64;
constexpr int THREADS = 512;            // 8 waves, 2(M) x 4(N)
constexpr int HALF_ELEMS = 128 * BK;    // one half-tile: [128][64] bf16
constexpr int SLOTS = 8;                // 2 buffers x 4 half-tiles

__device__ inline void glds16(const __hip_bfloat16* gsrc, __hip_bfloat16* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds, 16, 0, 0);
}

// LDS XOR swizzle (guide T2, full byte_off ^= (row&15)<<4 form): a
// ds_read_b128 16-lane group whose rows are distinct mod 16 becomes
// conflict-free. On a 128-B row the 4-bit xor's top bit crosses the row
// boundary, so at chunk granularity the involution decomposes into
//   lds_row  = row ^ ((row>>3)&1)        (row-parity flip from xor bit 3)
//   lds_k    = k ^ ((row&7)<<3)          (chunk xor from xor bits 0-2)
// Evolution measured on this kernel (PMC SQ_LDS_BANK_CONFLICT per 4096^3
// dispatch): st_16x32 single-bit 12.6M conflicts -> (row&7) 8-slot form
// ~1073 TF -> this form. Same involution on the glds SOURCE (stage_half
// inverts it) and the ds_read address (rule 21: both-sides-or-neither).
// row-parity variant (full (row&15)<<4 xor) measured SLOWER: 881-951 TF vs
// 1020-1073 for the chunk-only form — the scattered glds source row costs
// more in fetch than the residual 2-way conflicts. Keep chunk-only.
__device__ inline int swz_row(int row) { return row; }
__device__ inline int swz_k(int row, int k) { return k ^ ((row & 7) << 3); }

// Stage one [128][64] half-tile: 1024 16-B slots, 512 threads x 2 passes.
// LDS destination is lane-linear (slot order == lds address order); the
// swizzle permutes which global 16-B chunk lands in each slot: linear slot
// (row_l, chunk_l) holds global (src_row, chunk_l ^ (src_row&7)) where
// src_row = row_l ^ ((row_l>>3)&1) (self-inverse row map).
__device__ inline void stage_half(const __hip_bfloat16* gbase, int ldk,
                                  __hip_bfloat16* half_base) {
  const int t = threadIdx.x;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int slot = pass * THREADS + t;
    int row_l = slot >> 3;
    int chunk = (slot & 7) * 8;
    int src_row = swz_row(row_l);
    glds16(gbase + src_row * ldk + swz_k(src_row, chunk),
           half_base + row_l * BK + chunk);
  }
}

__device__ inline bf16x8 read_frag(const __hip_bfloat16* half_base, int row,
                                   int kbase) {
  return *reinterpret_cast<const bf16x8*>(half_base + swz_row(row) * BK +
                                          swz_k(row, kbase));
}

__device__ inline void wait_lgkm0_fence() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
}

__device__ inline void wait_vmcnt(int halves_outstanding) {
  switch (halves_outstanding) {
    case 0:
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      break;
    case 1:
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
      break;
    case 2:
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      break;
    default:
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      break;
  }
  __builtin_amdgcn_sched_barrier(0);
}

__global__ __launch_bounds__(THREADS, 2) void gemm_bf16_8phase_kernel(
    const __hip_bfloat16* __restrict__ A,   // [M][K]
    const __hip_bfloat16* __restrict__ Bt,  // [N][K]
    float* __restrict__ C,                  // [M][N]
    int M, int N, int K) {
  __shared__ __hip_bfloat16 lds[SLOTS * HALF_ELEMS];  // 128 KiB

  const int tiles_n = N / BN;
  const int bm = ((int)blockIdx.x / tiles_n) * BM;
  const int bn = ((int)blockIdx.x % tiles_n) * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;        // 0..1: 64-row band within each A half
  const int wn = wave & 3;         // 0..3: 32-col band within each B half

  const int frow = lane & 15;
  const int fk = (lane >> 4) * 8;

  const __hip_bfloat16* Ablk = A + (long)bm * K;
  const __hip_bfloat16* Bblk = Bt + (long)bn * K;
  const int T = K / BK;

  // acc[qm][fm'][qn][fn'] — quadrant-major, all indices compile-time
  f32x4 acc[2][4][2][2] = {};

  // stage half h into slot h%8 (h%4: 0=A0, 1=B0, 2=A1, 3=B1)
  auto stage_h = [&](int h) {
    int c = h & 3;
    int tt = h >> 2;
    const __hip_bfloat16* g;
    if (c == 0)
      g = Ablk + tt * BK;
    else if (c == 1)
      g = Bblk + tt * BK;
    else if (c == 2)
      g = Ablk + (long)128 * K + tt * BK;
    else
      g = Bblk + (long)128 * K + tt * BK;
    stage_half(g, K, lds + (h & 7) * HALF_ELEMS);
  };

  // ---- prologue: tile 0 fully + 3 halves of tile 1 (guide's 4 then +3)
  for (int h = 0; h < 4 && h < 4 * T; ++h) stage_h(h);
  wait_vmcnt(2);  // A0,B0 of tile 0 complete
  for (int h = 4; h < 7 && h < 4 * T; ++h) stage_h(h);
  wait_vmcnt(3);  // all of tile 0 complete (<=3 halves of tile 1 in flight)
  __builtin_amdgcn_s_barrier();

  bf16x8 afrag[4][2];       // qm-current A set: 4 m-frags x 2 k-halves
  bf16x8 bfrag[2][2][2];    // [qn][fn'][kk] — both B sets kept

  for (int t = 0; t < T; ++t) {
    const __hip_bfloat16* As0 = lds + ((4 * t + 0) & 7) * HALF_ELEMS;
    const __hip_bfloat16* Bs0 = lds + ((4 * t + 1) & 7) * HALF_ELEMS;
    const __hip_bfloat16* As1 = lds + ((4 * t + 2) & 7) * HALF_ELEMS;
    const __hip_bfloat16* Bs1 = lds + ((4 * t + 3) & 7) * HALF_ELEMS;
    const int arow = wm * 64 + frow;
    const int brow = wn * 32 + frow;

    // ---- phase 0: quadrant (0,0); read A(qm=0) + B set 0; stage h=4t+7
#pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        afrag[fm][kk] = read_frag(As0, arow + fm * 16, kk * 32 + fk);
    }
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        bfrag[0][fn][kk] = read_frag(Bs0, brow + fn * 16, kk * 32 + fk);
    }
    if (4 * t + 7 < 4 * T) stage_h(4 * t + 7);
    // 12 ds_reads issued this phase: partial wait before the barrier lets
    // the first reads land while the rest fly (guide template's optional
    // lgkmcnt(8) line)
    asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)   // kk OUTER: 8 independent MFMAs between
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)  // accumulator reuse (dependent-latency
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)  // hiding, guide §3 MFMA u-bench)
          acc[0][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[fm][kk], bfrag[0][fn][kk], acc[0][fm][0][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 1: quadrant (0,1); read B set 1; stage h=4t+8
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        bfrag[1][fn][kk] = read_frag(Bs1, brow + fn * 16, kk * 32 + fk);
    }
    if (4 * t + 8 < 4 * T) stage_h(4 * t + 8);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)   // kk OUTER: 8 independent MFMAs between
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)  // accumulator reuse (dependent-latency
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)  // hiding, guide §3 MFMA u-bench)
          acc[0][fm][1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[fm][kk], bfrag[1][fn][kk], acc[0][fm][1][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 2: quadrant (1,1); re-read A(qm=1); stage h=4t+9
#pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        afrag[fm][kk] = read_frag(As1, arow + fm * 16, kk * 32 + fk);
    }
    if (4 * t + 9 < 4 * T) stage_h(4 * t + 9);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)   // kk OUTER: 8 independent MFMAs between
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)  // accumulator reuse (dependent-latency
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)  // hiding, guide §3 MFMA u-bench)
          acc[1][fm][1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[fm][kk], bfrag[1][fn][kk], acc[1][fm][1][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 3: quadrant (1,0); B set 0 still live; stage h=4t+10
    if (4 * t + 10 < 4 * T) stage_h(4 * t + 10);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)   // kk OUTER: 8 independent MFMAs between
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)  // accumulator reuse (dependent-latency
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)  // hiding, guide §3 MFMA u-bench)
          acc[1][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[fm][kk], bfrag[0][fn][kk], acc[1][fm][0][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
    // tile t+1's halves must be complete before its phase-0 ds_reads;
    // <= staged-ahead halves of tile t+2 may remain in flight
    {
      int staged = min(4 * T, 4 * t + 11);
      int allowed = staged - 4 * (t + 2);
      wait_vmcnt(allowed < 0 ? 0 : allowed);
    }
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue: C/D map col = lane&15, row = (lane>>4)*4 + r (guide §3)
  const int ccol = lane & 15;
  const int crow = (lane >> 4) * 4;
#pragma unroll
  for (int qm = 0; qm < 2; ++qm)
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int qn = 0; qn < 2; ++qn)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn) {
          int row0 = bm + qm * 128 + wm * 64 + fm * 16 + crow;
          int col = bn + qn * 128 + wn * 32 + fn * 16 + ccol;
#pragma unroll
          for (int r = 0; r < 4; ++r)
            C[(long)(row0 + r) * N + col] = acc[qm][fm][qn][fn][r];
        }
}

}  // namespace gemm_bf16_8ph
