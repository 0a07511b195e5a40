// MX-fp8 (OCP e4m3) MFMA GEMM — 256^2 tile, 8-phase schedule, K-step 128.
//
// gfx950's only high-rate fp8 path is the block-scaled MX instruction
// (guide §3: there is NO non-scaled mfma_f32_*x64_fp8; the MX-scaled
// 16x16x128 runs at ~4.6 PF µbench vs 2.1 PF for non-scaled fp8). This
// kernel runs it as a plain fp8 GEMM: every 32-element block scale is
// e8m0 127 (x1.0), so numerics are exactly "e4m3 inputs, fp32 accumulate".
//
// Operand layout (verified on hardware by csrc-probe, max err 0.0 vs exact
// integer reference — see profiles/gemm_kernel_stats.md round-2 fp8 note):
//   A: lane l holds row (l&15), k bytes (l>>4)*32 .. +32 (8 VGPRs, v8i32)
//   B: lane l holds col (l&15) of B = row of Bt, same k range
//   C/D: col = lane&15, row = (lane>>4)*4 + reg (shape-determined,
//        dtype-independent on gfx950)
//   scales: opsel 0, low byte 127 on both operands; cbsz=0 blgp=0 (e4m3).
//
// Structure: the bf16 8-phase template's schedule verbatim (PIPE=0 of
// gemm_bf16_8phase.hip — measured optimal there across 6 variant families)
// with BK=128 fp8: a half-tile is [128][128] fp8 = 16 KiB — the SAME byte
// shape as the bf16 [128][64] half, so staging (2 glds x 16B per thread),
// the chunk-xor LDS swizzle (byte ^= (row&7)<<4), slot cadence, counted
// vmcnt and barriers are unchanged; only the fragment reads (2x
// ds_read_b128 = 32 B per fragment) and the MFMA (8 independent
// 16x16x128 per phase, no kk loop) differ.
//
// FMT template axis: 0 = fp8 e4m3 (K-step 128), 4 = fp4 e2m1 (K-step 256 —
// nibble-packed, so a [128][256-fp4] half-tile is the SAME 16 KiB byte
// shape; each quadrant phase then runs two 16x16x128 MFMAs per fragment
// pair, one per 128-element k-half, at the 4x fp4 rate). The fp4 operand
// layout (low nibble = even k; 16 bytes in the low 4 dwords of the v8i32
// operand; cbsz=blgp=4) was verified on hardware with the same
// single-MFMA exact-integer probe as fp8 (max abs err 0.0).
//
// Shipped defaults (each measured in both A/B interleave orders; the
// tuning ladder with PMC evidence is profiles/gemm_kernel_stats.md):
//   fp8: SWZV=1 (row-bit-3 swizzle; bank conflicts 44M -> 0, +11%),
//        unmerged phases — 2.22 PF @8192^3 random operands;
//   fp4: SWZV=0, MP23=1 (merged phases 2+3, +1-2%) — 4.02-4.09 PF,
//        bit-exact vs the dequantized fp32 reference.
// Rejected-by-measurement variants stay selectable for A/B: shape 17
// (alternate swizzle), 18 (quad-transpose dwordx4 epilogue), 19/20
// (merge toggles), 32 (the 32x32x64 instruction: only 2 independent
// accumulators per phase against its 64-cycle dependent latency).
//
// Constraints: M,N multiples of 256; K multiple of 128 (fp8) / 256 (fp4),
// K >= 2 K-steps.
#include <hip/hip_runtime.h>

namespace gemm_fp8_mx {

using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using i32x4 = __attribute__((ext_vector_type(4))) int;
using i32x8 = __attribute__((ext_vector_type(8))) int;

constexpr int BM = 256;
constexpr int BN = 256;
constexpr int BKB = 128;                 // K-step in BYTES (fp8: 128 elems; fp4: 256)
constexpr int THREADS = 512;             // 8 waves, 2(M) x 4(N)
constexpr int HALF_BYTES = 128 * BKB;    // [128][128 B] = 16 KiB
constexpr int SLOTS = 8;

__device__ inline void glds16(const unsigned char* gsrc, unsigned char* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds, 16, 0, 0);
}

// chunk-xor swizzle in 16-B granules — SWZV selects the row hash g(row):
//   0: g = row&7 (the bf16 sibling's form; fp8's two-chunk fragment reads
//      measured 44M SQ_LDS_BANK_CONFLICT/6-dispatch with it — rows r and
//      r+8 always collide under a 3-bit hash of row&7)
//   1: g = (row&7) ^ ((row>>3)&1) — folds row bit 3 into the hash, making
//      the 16 rows of a b128 lane group land on 16 disjoint 4-bank spans
//      (derivation in profiles/gemm_kernel_stats.md). Any per-row xor is
//      self-inverse, so staging source and reads share the formula.
template <int SWZV>
__device__ inline int swz_byte_v(int row, int byte_off) {
  int g = (row & 7) ^ (SWZV ? ((row >> 3) & 1) : 0);
  return byte_off ^ (g << 4);
}

__device__ inline int swz_byte(int row, int byte_off) {
  return swz_byte_v<0>(row, byte_off);
}

template <int SWZV>
__device__ inline void stage_half(const unsigned char* gbase, int ldk,
                                  unsigned char* half_base) {
  const int t = threadIdx.x;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int slot = pass * THREADS + t;
    int row_l = slot >> 3;
    int byte_off = (slot & 7) * 16;
    glds16(gbase + row_l * ldk + swz_byte_v<SWZV>(row_l, byte_off),
           half_base + row_l * BKB + byte_off);
  }
}

// one fp8 MFMA fragment (32 B): two independently-swizzled 16-B chunks
template <int SWZV>
__device__ inline i32x8 read_frag(const unsigned char* half_base, int row,
                                  int kb) {
  i32x4 lo = *reinterpret_cast<const i32x4*>(half_base + row * BKB +
                                             swz_byte_v<SWZV>(row, kb));
  i32x4 hi = *reinterpret_cast<const i32x4*>(half_base + row * BKB +
                                             swz_byte_v<SWZV>(row, kb + 16));
  return i32x8{lo[0], lo[1], lo[2], lo[3], hi[0], hi[1], hi[2], hi[3]};
}

// one fp4 MFMA fragment (16 B = 32 nibbles) in the low 4 dwords
template <int SWZV>
__device__ inline i32x8 read_frag4(const unsigned char* half_base, int row,
                                   int kb) {
  i32x4 lo = *reinterpret_cast<const i32x4*>(half_base + row * BKB +
                                             swz_byte_v<SWZV>(row, kb));
  return i32x8{lo[0], lo[1], lo[2], lo[3], 0, 0, 0, 0};
}

__device__ inline void wait_lgkm0_fence() {
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
}

__device__ inline void wait_vmcnt(int halves_outstanding) {
  switch (halves_outstanding) {
    case 0: asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); break;
    case 1: asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); break;
    case 2: asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); break;
    default: asm volatile("s_waitcnt vmcnt(6)" ::: "memory"); break;
  }
  __builtin_amdgcn_sched_barrier(0);
}

#define MX_MFMA(FMT, a, b, c) \
  __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4((a), (b), (c), (FMT), (FMT), 0, 127, 0, 127)

// In-quad 4x4 transpose of an MFMA f32x4 fragment (rows r0..r0+3, col =
// own lane) so lane 4g+j ends holding row r0+j across the quad's 4 cols —
// one global_store_dwordx4 instead of four scattered dword stores. Two
// bit-exchange rounds (element-bit b <-> lane-bit b) via shfl_xor, which
// hipcc lowers to DPP quad_perm (no LDS). Guide T-epilogue: a store tail
// is often ISSUE-bound, so 4x fewer store instructions at equal bytes.
__device__ inline f32x4 quad_transpose(f32x4 v, int lane) {
  const int j = lane & 3;
  f32x4 u, w;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float ex = __shfl_xor(v[r ^ 1], 1);
    u[r] = ((r & 1) == (j & 1)) ? v[r] : ex;
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float ex = __shfl_xor(u[r ^ 2], 2);
    w[r] = ((r & 2) == (j & 2)) ? u[r] : ex;
  }
  return w;
}

// MP23: merge phases 2+3 into one barrier section (3 barrier-pairs per
// K-tile). The h=4t+10 stage is deferred to the next tile's phase 0 (it
// would overwrite the As1 slot while other waves' A1 reads are in flight
// — same derivation as the bf16 sibling's variant 8). bf16 measured -3%;
// re-measured here because the MX clusters retire ~4x faster, making the
// per-phase barrier overhead relatively larger.
template <int FMT, int SWZV = 0, int EPI = 0, int MP23 = 0>
__global__ __launch_bounds__(THREADS, 2) void gemm_fp8_mx_kernel(
    const unsigned char* __restrict__ A,   // [M][K] packed
    const unsigned char* __restrict__ Bt,  // [N][K] packed
    float* __restrict__ C,                 // [M][N]
    int M, int N, int K) {                 // K in ELEMENTS
  __shared__ unsigned char lds[SLOTS * HALF_BYTES];  // 128 KiB

  const int tiles_n = N / BN;
  const int bm = ((int)blockIdx.x / tiles_n) * BM;
  const int bn = ((int)blockIdx.x % tiles_n) * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  const int frow = lane & 15;
  const int fkb = (lane >> 4) * 32;   // fp8: 32-byte k-slice per lane
  const int fkb4 = (lane >> 4) * 16;  // fp4: 16-byte slice per k-half

  const int Kb = (FMT == 4) ? K / 2 : K;   // row stride in bytes
  const unsigned char* Ablk = A + (long)bm * Kb;
  const unsigned char* Bblk = Bt + (long)bn * Kb;
  const int T = Kb / BKB;

  f32x4 acc[2][4][2][2] = {};

  auto stage_h = [&](int h) {
    int c = h & 3;
    int tt = h >> 2;
    const unsigned char* g;
    if (c == 0)
      g = Ablk + tt * BKB;
    else if (c == 1)
      g = Bblk + tt * BKB;
    else if (c == 2)
      g = Ablk + (long)128 * Kb + tt * BKB;
    else
      g = Bblk + (long)128 * Kb + tt * BKB;
    stage_half<SWZV>(g, Kb, lds + (h & 7) * HALF_BYTES);
  };

  for (int h = 0; h < 4 && h < 4 * T; ++h) stage_h(h);
  wait_vmcnt(2);
  for (int h = 4; h < 7 && h < 4 * T; ++h) stage_h(h);
  wait_vmcnt(3);
  __builtin_amdgcn_s_barrier();

  i32x8 afrag[4];
  i32x8 bfrag[2][2];
  i32x8 afrag4[4][2];     // fp4: one 16-B fragment per 128-element k-half
  i32x8 bfrag4[2][2][2];

  const int arow = wm * 64 + frow;
  const int brow = wn * 32 + frow;

  for (int t = 0; t < T; ++t) {
    const unsigned char* As0 = lds + ((4 * t + 0) & 7) * HALF_BYTES;
    const unsigned char* Bs0 = lds + ((4 * t + 1) & 7) * HALF_BYTES;
    const unsigned char* As1 = lds + ((4 * t + 2) & 7) * HALF_BYTES;
    const unsigned char* Bs1 = lds + ((4 * t + 3) & 7) * HALF_BYTES;

    // ---- phase 0: q(0,0); read A(qm0)+B set 0; stage 4t+7
    if (FMT == 4) {
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          afrag4[fm][ks] = read_frag4<SWZV>(As0, arow + fm * 16, ks * 64 + fkb4);
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          bfrag4[0][fn][ks] = read_frag4<SWZV>(Bs0, brow + fn * 16, ks * 64 + fkb4);
    } else {
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
        afrag[fm] = read_frag<SWZV>(As0, arow + fm * 16, fkb);
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        bfrag[0][fn] = read_frag<SWZV>(Bs0, brow + fn * 16, fkb);
    }
    if (MP23 && t > 0 && 4 * t + 6 < 4 * T)
      stage_h(4 * t + 6);  // deferred from the previous merged phase
    if (4 * t + 7 < 4 * T) stage_h(4 * t + 7);
    asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        if (FMT == 4) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[0][fm][0][fn] = MX_MFMA(4, afrag4[fm][ks], bfrag4[0][fn][ks],
                                          acc[0][fm][0][fn]);
        } else {
          acc[0][fm][0][fn] = MX_MFMA(0, afrag[fm], bfrag[0][fn], acc[0][fm][0][fn]);
        }
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 1: q(0,1); read B set 1; stage 4t+8
    if (FMT == 4) {
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          bfrag4[1][fn][ks] = read_frag4<SWZV>(Bs1, brow + fn * 16, ks * 64 + fkb4);
    } else {
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        bfrag[1][fn] = read_frag<SWZV>(Bs1, brow + fn * 16, fkb);
    }
    if (4 * t + 8 < 4 * T) stage_h(4 * t + 8);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        if (FMT == 4) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[0][fm][1][fn] = MX_MFMA(4, afrag4[fm][ks], bfrag4[1][fn][ks],
                                          acc[0][fm][1][fn]);
        } else {
          acc[0][fm][1][fn] = MX_MFMA(0, afrag[fm], bfrag[1][fn], acc[0][fm][1][fn]);
        }
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 2: q(1,1); re-read A(qm1); stage 4t+9
    if (FMT == 4) {
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          afrag4[fm][ks] = read_frag4<SWZV>(As1, arow + fm * 16, ks * 64 + fkb4);
    } else {
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
        afrag[fm] = read_frag<SWZV>(As1, arow + fm * 16, fkb);
    }
    if (4 * t + 9 < 4 * T) stage_h(4 * t + 9);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        if (FMT == 4) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[1][fm][1][fn] = MX_MFMA(4, afrag4[fm][ks], bfrag4[1][fn][ks],
                                          acc[1][fm][1][fn]);
        } else {
          acc[1][fm][1][fn] = MX_MFMA(0, afrag[fm], bfrag[1][fn], acc[1][fm][1][fn]);
        }
    if (MP23) {
      // merged phase 3: q(1,0), B set 0 still live in registers
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          if (FMT == 4) {
#pragma unroll
            for (int ks = 0; ks < 2; ++ks)
              acc[1][fm][0][fn] = MX_MFMA(4, afrag4[fm][ks], bfrag4[0][fn][ks],
                                            acc[1][fm][0][fn]);
          } else {
            acc[1][fm][0][fn] = MX_MFMA(0, afrag[fm], bfrag[0][fn], acc[1][fm][0][fn]);
          }
      __builtin_amdgcn_s_setprio(0);
      {
        int staged = min(4 * T, 4 * t + 10);
        int allowed = staged - 4 * (t + 2);
        wait_vmcnt(allowed < 0 ? 0 : (allowed > 3 ? 3 : allowed));
      }
      __builtin_amdgcn_s_barrier();
      continue;
    }
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 3: q(1,0); B set 0 still live; stage 4t+10
    if (4 * t + 10 < 4 * T) stage_h(4 * t + 10);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        if (FMT == 4) {
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[1][fm][0][fn] = MX_MFMA(4, afrag4[fm][ks], bfrag4[0][fn][ks],
                                          acc[1][fm][0][fn]);
        } else {
          acc[1][fm][0][fn] = MX_MFMA(0, afrag[fm], bfrag[0][fn], acc[1][fm][0][fn]);
        }
    __builtin_amdgcn_s_setprio(0);
    {
      int staged = min(4 * T, 4 * t + 11);
      int allowed = staged - 4 * (t + 2);
      wait_vmcnt(allowed < 0 ? 0 : (allowed > 3 ? 3 : allowed));
    }
    __builtin_amdgcn_s_barrier();
  }

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
          if (EPI == 1) {
            f32x4 w = quad_transpose(acc[qm][fm][qn][fn], lane);
            int colq = bn + qn * 128 + wn * 32 + fn * 16 + (lane & 12);
            *reinterpret_cast<f32x4*>(
                &C[(long)(row0 + (lane & 3)) * N + colq]) = w;
          } else {
            int col = bn + qn * 128 + wn * 32 + fn * 16 + ccol;
#pragma unroll
            for (int r = 0; r < 4; ++r)
              C[(long)(row0 + r) * N + col] = acc[qm][fm][qn][fn][r];
          }
        }
}


#define MX_MFMA32(FMT, a, b, c) \
  __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4((a), (b), (c), (FMT), (FMT), 0, 127, 0, 127)

// 32x32x64 variant: same 256^2 tile / 8-wave / 8-phase schedule and the
// same 16 KiB half-tile staging; fragments are 32-row/32-col, each phase
// runs {fp8: 2 m-frags x 2 k-steps, fp4: 2 x 4} MFMAs of the 32x32x64
// scaled instruction (~25% higher uarch ceiling than 16x16x128 — guide §3
// µbench 9099 vs 7228 TF fp4). Operand/C-D layouts hardware-verified by
// csrc-probe (max abs err 0.0): A lane l = row l&31, k (l>>5)*32 elements;
// C col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5).
template <int FMT, int SWZV = 0>  // FMT: 0 = fp8 e4m3, 4 = fp4 e2m1
__global__ __launch_bounds__(THREADS, 2) void gemm_mx32_kernel(
    const unsigned char* __restrict__ A,   // [M][K] packed
    const unsigned char* __restrict__ Bt,  // [N][K] packed
    float* __restrict__ C,                 // [M][N]
    int M, int N, int K) {                 // K in ELEMENTS
  __shared__ unsigned char lds[SLOTS * HALF_BYTES];  // 128 KiB

  constexpr int KSTEPS = (FMT == 4) ? 4 : 2;     // 32B(fp4)/64B(fp8) per step
  constexpr int STEPB = BKB / KSTEPS;            // bytes per k-step row-slice
  constexpr int LANEB = STEPB / 2;               // per-lane slice (l>>5 half)

  const int tiles_n = N / BN;
  const int bm = ((int)blockIdx.x / tiles_n) * BM;
  const int bn = ((int)blockIdx.x % tiles_n) * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;

  const int frow = lane & 31;
  const int fkb = (lane >> 5) * LANEB;

  const int Kb = (FMT == 4) ? K / 2 : K;
  const unsigned char* Ablk = A + (long)bm * Kb;
  const unsigned char* Bblk = Bt + (long)bn * Kb;
  const int T = Kb / BKB;

  f32x16 acc[2][2][2] = {};  // [qm][fm][qn]; fn==0 (one 32-col frag/wave)

  auto stage_h = [&](int h) {
    int c = h & 3;
    int tt = h >> 2;
    const unsigned char* g;
    if (c == 0)
      g = Ablk + tt * BKB;
    else if (c == 1)
      g = Bblk + tt * BKB;
    else if (c == 2)
      g = Ablk + (long)128 * Kb + tt * BKB;
    else
      g = Bblk + (long)128 * Kb + tt * BKB;
    stage_half<SWZV>(g, Kb, lds + (h & 7) * HALF_BYTES);
  };

  for (int h = 0; h < 4 && h < 4 * T; ++h) stage_h(h);
  wait_vmcnt(2);
  for (int h = 4; h < 7 && h < 4 * T; ++h) stage_h(h);
  wait_vmcnt(3);
  __builtin_amdgcn_s_barrier();

  i32x8 afrag[2][KSTEPS];
  i32x8 bfrag[2][KSTEPS];  // [set][ks]

  const int arow = wm * 64 + frow;
  const int brow = wn * 32 + frow;

  auto read_f = [&](const unsigned char* base, int row, int ks) -> i32x8 {
    if (FMT == 4) return read_frag4<SWZV>(base, row, ks * STEPB + fkb);
    return read_frag<SWZV>(base, row, ks * STEPB + fkb);
  };

  for (int t = 0; t < T; ++t) {
    const unsigned char* As0 = lds + ((4 * t + 0) & 7) * HALF_BYTES;
    const unsigned char* Bs0 = lds + ((4 * t + 1) & 7) * HALF_BYTES;
    const unsigned char* As1 = lds + ((4 * t + 2) & 7) * HALF_BYTES;
    const unsigned char* Bs1 = lds + ((4 * t + 3) & 7) * HALF_BYTES;

    // ---- phase 0: q(0,0)
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks)
        afrag[fm][ks] = read_f(As0, arow + fm * 32, ks);
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
      bfrag[0][ks] = read_f(Bs0, brow, ks);
    if (4 * t + 7 < 4 * T) stage_h(4 * t + 7);
    asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
        acc[0][fm][0] = MX_MFMA32(FMT, afrag[fm][ks], bfrag[0][ks], acc[0][fm][0]);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 1: q(0,1)
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
      bfrag[1][ks] = read_f(Bs1, brow, ks);
    if (4 * t + 8 < 4 * T) stage_h(4 * t + 8);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
        acc[0][fm][1] = MX_MFMA32(FMT, afrag[fm][ks], bfrag[1][ks], acc[0][fm][1]);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 2: q(1,1)
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks)
        afrag[fm][ks] = read_f(As1, arow + fm * 32, ks);
    if (4 * t + 9 < 4 * T) stage_h(4 * t + 9);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
        acc[1][fm][1] = MX_MFMA32(FMT, afrag[fm][ks], bfrag[1][ks], acc[1][fm][1]);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---- phase 3: q(1,0)
    if (4 * t + 10 < 4 * T) stage_h(4 * t + 10);
    __builtin_amdgcn_s_barrier();
    wait_lgkm0_fence();
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks)
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
        acc[1][fm][0] = MX_MFMA32(FMT, afrag[fm][ks], bfrag[0][ks], acc[1][fm][0]);
    __builtin_amdgcn_s_setprio(0);
    {
      int staged = min(4 * T, 4 * t + 11);
      int allowed = staged - 4 * (t + 2);
      wait_vmcnt(allowed < 0 ? 0 : (allowed > 3 ? 3 : allowed));
    }
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: 32x32 C/D map
  const int ccol = lane & 31;
  const int rbase = 4 * (lane >> 5);
#pragma unroll
  for (int qm = 0; qm < 2; ++qm)
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int qn = 0; qn < 2; ++qn) {
        int row0 = bm + qm * 128 + wm * 64 + fm * 32;
        int col = bn + qn * 128 + wn * 32 + ccol;
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
          int r = (reg & 3) + 8 * (reg >> 2) + rbase;
          C[(long)(row0 + r) * N + col] = acc[qm][fm][qn][reg];
        }
      }
}

// pseudorandom VALID e4m3 fill: full sign/mantissa variation, exponents
// bounded so products stay finite (bench-honesty: zero or sign-stuck fills
// inflate TF via DVFS — guide §5.4 rule 25)
__global__ void fill_e4m3_hash_kernel(unsigned char* p, size_t n, unsigned seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    unsigned h = (unsigned)(i * 2654435761u) ^ seed;
    h ^= h >> 13;
    h *= 0x85ebca6bu;
    h ^= h >> 16;
    // sign = bit 0; exponent in [4..11] (values ~2^-3 .. 2^4); mantissa 3 bits
    unsigned char b = (unsigned char)(((h & 1u) << 7) | ((((h >> 1) & 7u) + 4u) << 3) |
                                      ((h >> 4) & 7u));
    p[i] = b;
  }
}

// pseudorandom e2m1 (fp4) nibble fill: all 16 codes are valid values
__global__ void fill_e2m1_hash_kernel(unsigned char* p, size_t n, unsigned seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    unsigned h = (unsigned)(i * 2654435761u) ^ seed;
    h ^= h >> 13;
    h *= 0x85ebca6bu;
    h ^= h >> 16;
    p[i] = (unsigned char)(h & 0xFF);
  }
}

}  // namespace gemm_fp8_mx
