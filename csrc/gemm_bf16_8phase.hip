// bf16 MFMA GEMM, 256^2-tile 8-phase pipelined structure (CDNA4 guide §5
// "The 256² 8-phase template"). The high tier of the GPU health-validation
// burn-in: the 128^2 double-buffered kernel (gemm_bf16.hip) measures ~835 TF;
// this structure's published figures are ~1330 TF @4096^3 on random operands.
//
// Geometry (guide table): BM=BN=256, BK=64, 8 waves as 2(M)x4(N) = 512
// threads; per-wave output 128x64; LDS = 2 buffers x 4 half-tiles x
// [128][64] bf16 = 128 KiB; 2 global_load_lds (16 B/lane) per half-tile per
// thread; LDS XOR swizzle applied on the glds SOURCE address and the
// ds_read address (both-sides-or-neither, guide §5.4 rule 21 — the LDS
// destination stays lane-linear because glds writes base + lane*16).
//
// Template axes (round-2 tuning; within-probe A/B via gemm_bf16_8ph_ab):
//   XCD:  0 = linear blockIdx; 1 = bijective XCD-contiguous remap (guide T1:
//         each XCD gets a contiguous tile band -> neighbor tiles share
//         operand panels in that XCD's private L2; +10±3% on bf16 GEMM @8k).
//   PIPE: 0 = round-1 schedule (reads 12/4/8/0 per phase, single A set,
//         one vmcnt(6) per K-tile at phase 3's tail);
//         2 = PIPE=0 schedule with the K-loop unrolled 2 tiles per
//         iteration ("8 phases/iter, 2 K-tiles/iter" as the guide's
//         template states): every LDS slot index becomes a compile-time
//         constant, removing the per-phase (4t+i)&7 address arithmetic;
//         3 = PIPE=2 without the manual lgkmcnt fences: the ds_reads are
//         compiler-generated, so the compiler already inserts the minimal
//         s_waitcnt before each dependent MFMA — the manual
//         lgkmcnt(0)+sched_barrier(0) pair is a full drain plus a hard
//         scheduling wall on top of it (measured A/B to decide);
//         4 = PIPE=0 with phases 2+3 merged (3 barrier-pairs per K-tile
//         instead of 4; 32 MFMAs in the merged cluster). The h=4t+10 stage
//         cannot stay in the merged phase — it overwrites the As1 slot
//         while OTHER waves' A1 ds_reads may be in flight (only a barrier
//         after every wave's lgkm drain makes the slot safe) — so it is
//         deferred to the next tile's phase 0, costing one half-tile of
//         prefetch depth at the tile boundary. Tests whether barrier
//         count, not phase interleave, is the residual cost;
//         5 = PIPE=0 with the static setprio form (guide T5: one
//         s_setprio(1) for the younger wave half before the loop, no
//         per-cluster flips);
//         1 = phase-ahead schedule (reads 4/4/8/8: the A(qm=0) fragments of
//         tile t+1 are read during tile t's phase 3, so phase 0 starts its
//         MFMA with zero A-read latency; needs a second A register set,
//         +32 VGPR, and a counted vmcnt at the tail of phases 0/2/3 placed
//         BEFORE the barrier so the following phase's ds_reads are
//         collectively (cross-wave) covered).
//
// PIPE=1 schedule invariants (half h of tile t: 4t+{0=A0,1=B0,2=A1,3=B1},
// staged at phase j of tile t as h=4t+7+j, i.e. 7 halves ahead):
//   p0: read B0(t)   [cover: p3(t-1) tail vmcnt target 4t+1]; MFMA q(0,0)
//       with A-set0 (read at p3(t-1)) + B-set0;  tail vmcnt -> 4t+3
//   p1: read B1(t)   [cover: p0 tail]; MFMA q(0,1) A-set0 + B-set1
//   p2: read A1(t) -> A-set1 [cover: p0 tail, 4t+2 <= 4t+3]; MFMA q(1,1);
//       tail vmcnt -> 4t+4 (next tile's A0)
//   p3: read A0(t+1) -> A-set0 (last used p1); MFMA q(1,0) A-set1 + B-set0
//       behind lgkmcnt(8) (the 8 fresh reads may stay outstanding — nothing
//       in this phase consumes them); tail vmcnt -> 4t+5 (next B0)
// Every tail vmcnt precedes the phase's closing s_barrier: a per-wave
// s_waitcnt only covers that wave's own glds, so the barrier is what makes
// the guarantee collective before another wave's ds_read consumes the slot.
// Register WAR: A-set0 is rewritten at p3 while q(1,0) reads A-set1; B-set0
// rewritten at p0 after its last read in p3(t-1); slots are never ds_read
// after the phase that may glds-overwrite them (same argument as PIPE=0).
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
constexpr int NXCD = 8;

__device__ inline void glds16(const __hip_bfloat16* gsrc, __hip_bfloat16* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds, 16, 0, 0);
}

// LDS XOR swizzle (chunk-xor form): a ds_read_b128 16-lane group whose rows
// are distinct mod 8 spreads over 8 16-B slots. Measured on this kernel
// (PMC SQ_LDS_BANK_CONFLICT per 4096^3 dispatch): st_16x32 single-bit
// 12.6M conflicts; this form 0. The full (row&15)<<4 conflict-free form
// measured SLOWER (881-951 vs 1020-1073 TF): its row-scattered glds source
// costs more in fetch than the residual conflicts gain. Same involution on
// the glds SOURCE (stage_half inverts it) and the ds_read address
// (rule 21: both-sides-or-neither).
__device__ inline int swz_row(int row) { return row; }
template <int SWZV = 0>
__device__ inline int swz_k(int row, int k) {
  // SWZV=1 folds row bit 3 into the hash (see gemm_fp8_mx.hip: it fixed
  // the fp8 sibling's r/r+8 b128-group collisions, +11% there)
  int g = (row & 7) ^ (SWZV ? ((row >> 3) & 1) : 0);
  return k ^ (g << 3);
}

// Stage one [128][64] half-tile: 1024 16-B slots, 512 threads x 2 passes.
// LDS destination is lane-linear (slot order == lds address order); the
// swizzle permutes which global 16-B chunk lands in each slot.
template <int SWZV = 0>
__device__ inline void stage_half(const __hip_bfloat16* gbase, int ldk,
                                  __hip_bfloat16* half_base) {
  const int t = threadIdx.x;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int slot = pass * THREADS + t;
    int row_l = slot >> 3;
    int chunk = (slot & 7) * 8;
    int src_row = swz_row(row_l);
    glds16(gbase + src_row * ldk + swz_k<SWZV>(src_row, chunk),
           half_base + row_l * BK + chunk);
  }
}

template <int SWZV = 0>
__device__ inline bf16x8 read_frag(const __hip_bfloat16* half_base, int row,
                                   int kbase) {
  return *reinterpret_cast<const bf16x8*>(half_base + swz_row(row) * BK +
                                          swz_k<SWZV>(row, kbase));
}

__device__ inline void wait_lgkm0_fence() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
}

__device__ inline void wait_lgkm8_fence() {
  // allow up to 8 ds_reads to stay outstanding (the phase-ahead A reads)
  asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
}

// wait until at most `halves_outstanding` half-tiles (2 glds each) of this
// wave's global_load_lds traffic remain in flight
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
    case 3:
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      break;
    case 4:
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      break;
    default:
      asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
      break;
  }
  __builtin_amdgcn_sched_barrier(0);
}

// Ensure (collectively with the following barrier) that half `target` has
// landed: allow only the halves staged after it to remain outstanding.
__device__ inline void wait_half_landed(int staged_through, int target) {
  int allowed = staged_through - target;  // halves issued after `target`
  wait_vmcnt(allowed < 0 ? 0 : (allowed > 5 ? 5 : allowed));
}


// One K-tile (4 phases) of the PIPE=0 schedule with compile-time LDS slot
// indices. SBASE = (4t)&7 — 0 for even tiles, 4 for odd ones — makes every
// slot and staging destination a constant. LGKM selects the manual
// lgkmcnt fences (PIPE=2) vs compiler-managed waits (PIPE=3).
template <int SBASE, bool LGKM, int SWZV, typename STAGE>
__device__ inline void tile4(const __hip_bfloat16* lds_c, STAGE&& stage_hs,
                             int t, int T, int arow, int brow, int fk,
                             bf16x8 (&afrag)[4][2], bf16x8 (&bfrag)[2][2][2],
                             f32x4 (&acc)[2][4][2][2]) {
  const __hip_bfloat16* As0 = lds_c + ((SBASE + 0) & 7) * HALF_ELEMS;
  const __hip_bfloat16* Bs0 = lds_c + ((SBASE + 1) & 7) * HALF_ELEMS;
  const __hip_bfloat16* As1 = lds_c + ((SBASE + 2) & 7) * HALF_ELEMS;
  const __hip_bfloat16* Bs1 = lds_c + ((SBASE + 3) & 7) * HALF_ELEMS;

  // ---- phase 0
#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      afrag[fm][kk] = read_frag<SWZV>(As0, arow + fm * 16, kk * 32 + fk);
#pragma unroll
  for (int fn = 0; fn < 2; ++fn)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      bfrag[0][fn][kk] = read_frag<SWZV>(Bs0, brow + fn * 16, kk * 32 + fk);
  if (4 * t + 7 < 4 * T) stage_hs(4 * t + 7, (SBASE + 7) & 7);
  if (LGKM) asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  if (LGKM) wait_lgkm0_fence();
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        acc[0][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm][kk], bfrag[0][fn][kk], acc[0][fm][0][fn], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---- phase 1
#pragma unroll
  for (int fn = 0; fn < 2; ++fn)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      bfrag[1][fn][kk] = read_frag<SWZV>(Bs1, brow + fn * 16, kk * 32 + fk);
  if (4 * t + 8 < 4 * T) stage_hs(4 * t + 8, (SBASE + 8) & 7);
  __builtin_amdgcn_s_barrier();
  if (LGKM) wait_lgkm0_fence();
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        acc[0][fm][1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm][kk], bfrag[1][fn][kk], acc[0][fm][1][fn], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---- phase 2
#pragma unroll
  for (int fm = 0; fm < 4; ++fm)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      afrag[fm][kk] = read_frag<SWZV>(As1, arow + fm * 16, kk * 32 + fk);
  if (4 * t + 9 < 4 * T) stage_hs(4 * t + 9, (SBASE + 9) & 7);
  __builtin_amdgcn_s_barrier();
  if (LGKM) wait_lgkm0_fence();
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        acc[1][fm][1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm][kk], bfrag[1][fn][kk], acc[1][fm][1][fn], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  __builtin_amdgcn_s_barrier();

  // ---- phase 3
  if (4 * t + 10 < 4 * T) stage_hs(4 * t + 10, (SBASE + 10) & 7);
  __builtin_amdgcn_s_barrier();
  if (LGKM) wait_lgkm0_fence();
  __builtin_amdgcn_s_setprio(1);
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        acc[1][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[fm][kk], bfrag[0][fn][kk], acc[1][fm][0][fn], 0, 0, 0);
  __builtin_amdgcn_s_setprio(0);
  {
    int staged = min(4 * T, 4 * t + 11);
    int allowed = staged - 4 * (t + 2);
    wait_vmcnt(allowed < 0 ? 0 : (allowed > 3 ? 3 : allowed));
  }
  __builtin_amdgcn_s_barrier();
}

template <int XCD, int PIPE, int SWZV = 0>
__global__ __launch_bounds__(THREADS, 2) void gemm_bf16_8phase_kernel(
    const __hip_bfloat16* __restrict__ A,   // [M][K]
    const __hip_bfloat16* __restrict__ Bt,  // [N][K]
    float* __restrict__ C,                  // [M][N]
    int M, int N, int K) {
  __shared__ __hip_bfloat16 lds[SLOTS * HALF_ELEMS];  // 128 KiB

  int bid = (int)blockIdx.x;
  if (XCD) {
    // bijective XCD-contiguous remap (guide T1): hardware round-robins
    // consecutive blockIdx across the 8 XCDs; give XCD x the contiguous
    // tile range instead so neighboring tiles (shared A panel) hit its L2
    int nwg = (int)gridDim.x;
    int q = nwg / NXCD, r = nwg % NXCD;
    int xcd = bid % NXCD, seq = bid / NXCD;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + seq;
  }
  const int tiles_n = N / BN;
  const int bm = (bid / tiles_n) * BM;
  const int bn = (bid % tiles_n) * BN;
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
    stage_half<SWZV>(g, K, lds + (h & 7) * HALF_ELEMS);
  };

  const int arow = wm * 64 + frow;
  const int brow = wn * 32 + frow;

  if (PIPE >= 2) {
    // ---- unrolled schedule: 2 K-tiles (8 phases) per iteration ----------
    for (int h = 0; h < 4 && h < 4 * T; ++h) stage_h(h);
    wait_vmcnt(2);
    for (int h = 4; h < 7 && h < 4 * T; ++h) stage_h(h);
    wait_vmcnt(3);
    __builtin_amdgcn_s_barrier();

    bf16x8 afrag[4][2];
    bf16x8 bfrag[2][2][2];
    auto stage_hs = [&](int h, int slot) {
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
      stage_half<SWZV>(g, K, lds + slot * HALF_ELEMS);
    };
    constexpr bool LG = (PIPE == 2);
    int t = 0;
    for (; t + 1 < T; t += 2) {
      tile4<0, LG, SWZV>(lds, stage_hs, t, T, arow, brow, fk, afrag, bfrag, acc);
      tile4<4, LG, SWZV>(lds, stage_hs, t + 1, T, arow, brow, fk, afrag, bfrag, acc);
    }
    if (t < T)  // odd T tail (t even here, slot base 0)
      tile4<0, LG, SWZV>(lds, stage_hs, t, T, arow, brow, fk, afrag, bfrag, acc);
  } else if (PIPE == 0 || PIPE == 4 || PIPE == 5) {
    // ---- round-1 schedule (+PIPE 4/5 micro-variants) ---------------------
    // prologue: tile 0 fully + 3 halves of tile 1 (guide's 4 then +3)
    for (int h = 0; h < 4 && h < 4 * T; ++h) stage_h(h);
    wait_vmcnt(2);  // A0,B0 of tile 0 complete
    for (int h = 4; h < 7 && h < 4 * T; ++h) stage_h(h);
    wait_vmcnt(3);  // all of tile 0 complete (<=3 halves of tile 1 in flight)
    __builtin_amdgcn_s_barrier();

    constexpr bool CLUSTER_PRIO = (PIPE != 5);
    if (PIPE == 5) {
      // static T5 form: the younger dispatch half cedes VALU arbitration to
      // the older half on every segment; one standing priority boost for it
      // removes the start-of-segment penalty (condition is wave-uniform)
      if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
        if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(1);
    }

    bf16x8 afrag[4][2];       // qm-current A set: 4 m-frags x 2 k-halves
    bf16x8 bfrag[2][2][2];    // [qn][fn'][kk] — both B sets kept

    for (int t = 0; t < T; ++t) {
      const __hip_bfloat16* As0 = lds + ((4 * t + 0) & 7) * HALF_ELEMS;
      const __hip_bfloat16* Bs0 = lds + ((4 * t + 1) & 7) * HALF_ELEMS;
      const __hip_bfloat16* As1 = lds + ((4 * t + 2) & 7) * HALF_ELEMS;
      const __hip_bfloat16* Bs1 = lds + ((4 * t + 3) & 7) * HALF_ELEMS;

      // ---- phase 0: quadrant (0,0); read A(qm=0) + B set 0; stage h=4t+7
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          afrag[fm][kk] = read_frag<SWZV>(As0, arow + fm * 16, kk * 32 + fk);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          bfrag[0][fn][kk] = read_frag<SWZV>(Bs0, brow + fn * 16, kk * 32 + fk);
      }
      if (PIPE == 4 && t > 0 && 4 * t + 6 < 4 * T)
        stage_h(4 * t + 6);  // deferred from the previous merged phase
      if (4 * t + 7 < 4 * T) stage_h(4 * t + 7);
      // 12 ds_reads issued this phase: partial wait before the barrier lets
      // the first reads land while the rest fly
      asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      wait_lgkm0_fence();
      if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[0][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[fm][kk], bfrag[0][fn][kk], acc[0][fm][0][fn], 0, 0, 0);
      if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();

      // ---- phase 1: quadrant (0,1); read B set 1; stage h=4t+8
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          bfrag[1][fn][kk] = read_frag<SWZV>(Bs1, brow + fn * 16, kk * 32 + fk);
      }
      if (4 * t + 8 < 4 * T) stage_h(4 * t + 8);
      __builtin_amdgcn_s_barrier();
      wait_lgkm0_fence();
      if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[0][fm][1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[fm][kk], bfrag[1][fn][kk], acc[0][fm][1][fn], 0, 0, 0);
      if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();

      // ---- phase 2: quadrant (1,1); re-read A(qm=1); stage h=4t+9
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          afrag[fm][kk] = read_frag<SWZV>(As1, arow + fm * 16, kk * 32 + fk);
      }
      if (4 * t + 9 < 4 * T) stage_h(4 * t + 9);
      __builtin_amdgcn_s_barrier();
      wait_lgkm0_fence();
      if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[1][fm][1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[fm][kk], bfrag[1][fn][kk], acc[1][fm][1][fn], 0, 0, 0);
      if (PIPE == 4) {
        // merged phase 3: quadrant (1,0), B set 0 still live
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
#pragma unroll
          for (int fm = 0; fm < 4; ++fm)
#pragma unroll
            for (int fn = 0; fn < 2; ++fn)
              acc[1][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  afrag[fm][kk], bfrag[0][fn][kk], acc[1][fm][0][fn], 0, 0, 0);
      }
      if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(0);
      if (PIPE != 4) {
        __builtin_amdgcn_s_barrier();

        // ---- phase 3: quadrant (1,0); B set 0 still live; stage h=4t+10
        if (4 * t + 10 < 4 * T) stage_h(4 * t + 10);
        __builtin_amdgcn_s_barrier();
        wait_lgkm0_fence();
        if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
#pragma unroll
          for (int fm = 0; fm < 4; ++fm)
#pragma unroll
            for (int fn = 0; fn < 2; ++fn)
              acc[1][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  afrag[fm][kk], bfrag[0][fn][kk], acc[1][fm][0][fn], 0, 0, 0);
        if (CLUSTER_PRIO) __builtin_amdgcn_s_setprio(0);
      }
      // tile t+1's halves must be complete before its phase-0 ds_reads
      // (PIPE=4 defers the 4t+10 stage, so one fewer half is in flight)
      {
        int staged = min(4 * T, PIPE == 4 ? 4 * t + 10 : 4 * t + 11);
        int allowed = staged - 4 * (t + 2);
        wait_vmcnt(allowed < 0 ? 0 : (allowed > 3 ? 3 : allowed));
      }
      __builtin_amdgcn_s_barrier();
    }
  } else {
    // ---- PIPE=1: phase-ahead schedule (see header comment) ---------------
    for (int h = 0; h < 7 && h < 4 * T; ++h) stage_h(h);
    // prologue covers: tile0's A0 (for the prologue A reads) AND B0
    // (consumed by phase 0's in-phase reads after the barrier)
    wait_half_landed(min(7, 4 * T) - 1, 1);
    __builtin_amdgcn_s_barrier();

    bf16x8 a0frag[4][2];      // A qm=0 set (read at p3 of the previous tile)
    bf16x8 a1frag[4][2];      // A qm=1 set (read at p2)
    bf16x8 bfrag[2][2][2];    // [set][fn'][kk]

    // prologue A-set0 prime (plays the role of p3(t=-1))
#pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        a0frag[fm][kk] = read_frag<SWZV>(lds + 0 * HALF_ELEMS, arow + fm * 16,
                                   kk * 32 + fk);
    }

    for (int t = 0; t < T; ++t) {
      const __hip_bfloat16* Bs0 = lds + ((4 * t + 1) & 7) * HALF_ELEMS;
      const __hip_bfloat16* As1 = lds + ((4 * t + 2) & 7) * HALF_ELEMS;
      const __hip_bfloat16* Bs1 = lds + ((4 * t + 3) & 7) * HALF_ELEMS;
      const __hip_bfloat16* As0n = lds + ((4 * t + 4) & 7) * HALF_ELEMS;

      // ---- phase 0: read B0(t); MFMA q(0,0) = a0 x b0; stage 4t+7
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          bfrag[0][fn][kk] = read_frag<SWZV>(Bs0, brow + fn * 16, kk * 32 + fk);
      }
      if (4 * t + 7 < 4 * T) stage_h(4 * t + 7);
      __builtin_amdgcn_s_barrier();
      wait_lgkm0_fence();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[0][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a0frag[fm][kk], bfrag[0][fn][kk], acc[0][fm][0][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // cover p1's B1(t) (=half 4t+3) collectively via the barrier
      wait_half_landed(min(4 * T, 4 * t + 8) - 1, 4 * t + 3);
      __builtin_amdgcn_s_barrier();

      // ---- phase 1: read B1(t); MFMA q(0,1) = a0 x b1; stage 4t+8
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          bfrag[1][fn][kk] = read_frag<SWZV>(Bs1, brow + fn * 16, kk * 32 + fk);
      }
      if (4 * t + 8 < 4 * T) stage_h(4 * t + 8);
      __builtin_amdgcn_s_barrier();
      wait_lgkm0_fence();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[0][fm][1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a0frag[fm][kk], bfrag[1][fn][kk], acc[0][fm][1][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();

      // ---- phase 2: read A1(t); MFMA q(1,1) = a1 x b1; stage 4t+9
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          a1frag[fm][kk] = read_frag<SWZV>(As1, arow + fm * 16, kk * 32 + fk);
      }
      if (4 * t + 9 < 4 * T) stage_h(4 * t + 9);
      __builtin_amdgcn_s_barrier();
      wait_lgkm0_fence();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[1][fm][1][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a1frag[fm][kk], bfrag[1][fn][kk], acc[1][fm][1][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // cover p3's A0(t+1) (=half 4t+4)
      if (t + 1 < T) wait_half_landed(min(4 * T, 4 * t + 10) - 1, 4 * t + 4);
      __builtin_amdgcn_s_barrier();

      // ---- phase 3: read A0(t+1) phase-AHEAD; MFMA q(1,0) = a1 x b0;
      //      stage 4t+10
      if (t + 1 < T) {
#pragma unroll
        for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            a0frag[fm][kk] = read_frag<SWZV>(As0n, arow + fm * 16, kk * 32 + fk);
        }
      }
      if (4 * t + 10 < 4 * T) stage_h(4 * t + 10);
      __builtin_amdgcn_s_barrier();
      // the 8 fresh A reads may stay outstanding: q(1,0) uses none of them
      wait_lgkm8_fence();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[1][fm][0][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a1frag[fm][kk], bfrag[0][fn][kk], acc[1][fm][0][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      // cover next p0's B0(t+1) (=half 4t+5)
      if (t + 1 < T) wait_half_landed(min(4 * T, 4 * t + 11) - 1, 4 * t + 5);
      __builtin_amdgcn_s_barrier();
    }
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
