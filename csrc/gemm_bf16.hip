// bf16 MFMA GEMM — the GPU health-validation workload.
//
// Role in the control plane: the node-validation endpoint measures each free
// GPU's dense bf16 throughput before scheduling (a burn-in; complements the
// HBM probe and RCCL smoke). This is the CDNA4 guide's "ladder step 3"
// structure (§5: 128x128 tile, global_load_lds width-16 staging,
// double-buffered LDS, 2 barriers per K-step — measured ~874 TF at 4096^3 on
// gfx950 there; that structure's known ceiling is ~900 TF).
//
// Conventions:
//   * C[M][N] fp32 = A[M][K] @ B^T with B stored TRANSPOSED [N][K] (the
//     ladder's B^T input): both operands stage identically and each MFMA
//     fragment is one contiguous ds_read_b128.
//   * mfma_f32_16x16x32_bf16 per-lane maps (verified on hardware by
//     tests/test_gpu.py): A[l&15][(l>>4)*8+j], B[(l>>4)*8+j][l&15] — with
//     B^T rows being N-columns, the B fragment read is row-shaped too.
//   * acc accumulates in-place across K (guide §3 K-loop recipe); fragment
//     arrays are indexed by compile-time-unrolled constants only (§5.4
//     rule 20: runtime-indexed ext_vector arrays spill to scratch).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace gemm_bf16 {

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

constexpr int BM = 128;
constexpr int BN = 128;
constexpr int WAVES_M = 2;
constexpr int WAVES_N = 2;
constexpr int THREADS = WAVES_M * WAVES_N * 64;  // 256
constexpr int FRAGS_M = BM / WAVES_M / 16;       // 4
constexpr int FRAGS_N = BN / WAVES_N / 16;       // 4

__device__ inline void glds16(const __hip_bfloat16* gsrc, __hip_bfloat16* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)gsrc,
      (__attribute__((address_space(3))) unsigned int*)lds, 16, 0, 0);
}

// Stage a [128][BK] tile from row-major [rows][ld] global memory into linear
// LDS. 256 threads x 16 B per glds pass; BM*BK*2B / 4 KiB passes per tile.
// The LDS image is lane-linear by construction (glds writes base + lane*16).
template <int BK>
__device__ inline void stage_tile(const __hip_bfloat16* g, int ld,
                                  __hip_bfloat16* lds) {
  const int t = threadIdx.x;
  constexpr int kSlotsPerRow = BK * 2 / 16;  // 16B slots per row
  constexpr int kPasses = BM * BK * 2 / (THREADS * 16);
#pragma unroll
  for (int pass = 0; pass < kPasses; ++pass) {
    int slot = pass * THREADS + t;
    int row = slot / kSlotsPerRow;
    int kchunk = (slot % kSlotsPerRow) * 8;  // 8 bf16 per 16B slot
    glds16(g + row * ld + kchunk, lds + row * BK + kchunk);
  }
}

template <int BK>
__global__ __launch_bounds__(THREADS) void gemm_bf16_tile_kernel(
    const __hip_bfloat16* __restrict__ A,   // [M][K]
    const __hip_bfloat16* __restrict__ Bt,  // [N][K] (B transposed)
    float* __restrict__ C,                  // [M][N]
    int M, int N, int K, int use_swizzle) {
  __shared__ __hip_bfloat16 lds[2][2][BM * BK];  // [dbuf][A/B][tile]

  // XCD-aware remap: the dispatcher places block b on XCD b%8, so renumber
  // blocks XCD-major to keep neighboring tiles on one XCD's L2. Bijective
  // form (guide §5: the naive remap is non-bijective when nwg%8 != 0).
  // Pays ~+10% when HBM-bound (large N); gated off for small grids where
  // the working set is L3-resident and the remap costs ~2%.
  int wgid = blockIdx.x;
  const int nwg = gridDim.x;
  if (use_swizzle) {
    const int xcd = wgid & 7;
    const int pos = wgid >> 3;
    const int q = nwg >> 3;
    const int r = nwg & 7;
    wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int tiles_n = N / BN;
  const int bm = (wgid / tiles_n) * BM;
  const int bn = (wgid % tiles_n) * BN;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = (wave / WAVES_N) * (BM / WAVES_M);  // 0 or 64
  const int wn = (wave % WAVES_N) * (BN / WAVES_N);

  const int frow = lane & 15;        // fragment row (A) / col-row (Bt)
  const int fk = (lane >> 4) * 8;    // fragment k base

  f32x4 acc[FRAGS_M][FRAGS_N] = {};

  stage_tile<BK>(A + bm * K, K, &lds[0][0][0]);
  stage_tile<BK>(Bt + bn * K, K, &lds[0][1][0]);

  int buf = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    // Barrier does double duty: drains the in-flight glds for THIS buffer
    // (the fence inside __syncthreads carries vmcnt(0) while an LDS-DMA is
    // outstanding) and separates last iteration's fragment reads from the
    // prefetch below that overwrites their buffer.
    __syncthreads();
    // prefetch the NEXT K-tile after the barrier: it overlaps the MFMA
    // phase below and is drained at the next iteration's barrier.
    if (k0 + BK < K) {
      stage_tile<BK>(A + bm * K + (k0 + BK), K, &lds[buf ^ 1][0][0]);
      stage_tile<BK>(Bt + bn * K + (k0 + BK), K, &lds[buf ^ 1][1][0]);
    }

    const __hip_bfloat16* As = &lds[buf][0][0];
    const __hip_bfloat16* Bs = &lds[buf][1][0];
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 afrag[FRAGS_M];
      bf16x8 bfrag[FRAGS_N];
#pragma unroll
      for (int fm = 0; fm < FRAGS_M; ++fm) {
        afrag[fm] = *reinterpret_cast<const bf16x8*>(
            As + (wm + fm * 16 + frow) * BK + kk * 32 + fk);
      }
#pragma unroll
      for (int fn = 0; fn < FRAGS_N; ++fn) {
        bfrag[fn] = *reinterpret_cast<const bf16x8*>(
            Bs + (wn + fn * 16 + frow) * BK + kk * 32 + fk);
      }
#pragma unroll
      for (int fm = 0; fm < FRAGS_M; ++fm) {
#pragma unroll
        for (int fn = 0; fn < FRAGS_N; ++fn) {
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[fm], bfrag[fn], acc[fm][fn], 0, 0, 0);
        }
      }
    }
    buf ^= 1;
  }

  // epilogue: C/D map col = lane&15, row = (lane>>4)*4 + r (guide §3)
  const int ccol = lane & 15;
  const int crow = (lane >> 4) * 4;
#pragma unroll
  for (int fm = 0; fm < FRAGS_M; ++fm) {
#pragma unroll
    for (int fn = 0; fn < FRAGS_N; ++fn) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        C[(bm + wm + fm * 16 + crow + r) * N + (bn + wn + fn * 16 + ccol)] =
            acc[fm][fn][r];
      }
    }
  }
}

}  // namespace gemm_bf16
