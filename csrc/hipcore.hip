// MI355X (gfx950 / CDNA4) native core for the control plane.
//
// This is the native layer the reference lacks entirely (the
// reference's only GPU interaction is forking nvidia-smi and delegating
// device wiring to nvidia-docker). Components:
//
//   * xGMI / HBM bandwidth probe kernels: vectorized float4 grid-stride
//     copy, local (HBM stream) and peer-to-peer (pull over xGMI). Feeds the
//     scheduler's adjacency matrix (parallel/topology.py).
//   * MFMA warm-up / validation tiles: the exact-f32 16x16x4 MFMA (operand
//     maps documented in the CDNA4 guide) and the bf16 16x16x32 MFMA —
//     clock ramp before measuring, numerics check against torch fp32.
//
// Wave width is 64 on CDNA4; tiles and launch shapes below are sized for
// 64-lane wavefronts and a 256-CU / 8-XCD chip (grids ≫ 256 workgroups).
//
// Build: hipcc --offload-arch=gfx950 (driven by gpu_docker_api_amd/ops/build.py).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <map>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <algorithm>
#include <functional>
#include <string>
#include <vector>

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string(#expr) + " failed: " +              \
                               hipGetErrorString(_e));                         \
    }                                                                          \
  } while (0)

namespace {

// ---------------------------------------------------------------------------
// Copy kernels (bandwidth probe + numerics-testable copy)
// ---------------------------------------------------------------------------

// Grid-stride float4 copy: 16 B per lane per iteration — one coalesced
// 1 KiB transaction per wave64 (CDNA4 guide §2, "Global memory coalescing").
__global__ void copy_f32x4_kernel(const float4* __restrict__ src,
                                  float4* __restrict__ dst, size_t n4) {
  size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) dst[i] = src[i];
}

// 4x-unrolled variant: four independent dwordx4 loads in flight per thread
// before any store — deeper MLP to hide HBM latency on large streams.
__global__ void copy_f32x4_x4_kernel(const float4* __restrict__ src,
                                     float4* __restrict__ dst, size_t n4) {
  size_t stride = (size_t)gridDim.x * blockDim.x;
  size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  for (; i + 3 * stride < n4; i += 4 * stride) {
    float4 a = src[i];
    float4 b = src[i + stride];
    float4 c = src[i + 2 * stride];
    float4 d = src[i + 3 * stride];
    dst[i] = a;
    dst[i + stride] = b;
    dst[i + 2 * stride] = c;
    dst[i + 3 * stride] = d;
  }
  for (; i < n4; i += stride) dst[i] = src[i];
}

// Tail-safe scalar copy for non-multiple-of-4 sizes.
__global__ void copy_f32_tail_kernel(const float* __restrict__ src,
                                     float* __restrict__ dst, size_t n,
                                     size_t offset) {
  size_t i = offset + blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  if (i < n) dst[i] = src[i];
}

void launch_copy_f32(const float* src, float* dst, size_t n, hipStream_t stream) {
  size_t n4 = n / 4;
  constexpr int kBlock = 256;
  if (n4 > 0) {
    // ≫ 256 workgroups to fill 256 CUs across 8 XCDs; cap to keep launch sane
    int grid = (int)std::min<size_t>((n4 + kBlock - 1) / kBlock, 32768);
    if (n4 >= (size_t)4 * grid * kBlock) {
      hipLaunchKernelGGL(copy_f32x4_x4_kernel, dim3(grid), dim3(kBlock), 0,
                         stream, reinterpret_cast<const float4*>(src),
                         reinterpret_cast<float4*>(dst), n4);
    } else {
      hipLaunchKernelGGL(copy_f32x4_kernel, dim3(grid), dim3(kBlock), 0, stream,
                         reinterpret_cast<const float4*>(src),
                         reinterpret_cast<float4*>(dst), n4);
    }
  }
  size_t tail = n - n4 * 4;
  if (tail > 0) {
    hipLaunchKernelGGL(copy_f32_tail_kernel, dim3(1), dim3(kBlock), 0, stream,
                       src, dst, n, n4 * 4);
  }
}

// ---------------------------------------------------------------------------
// MFMA tiles
// ---------------------------------------------------------------------------

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

// Exact-f32 MFMA 16x16x4 (v_mfma_f32_16x16x4_f32). Operand map is documented
// in the CDNA4 guide §3: lane l supplies A[l&15][l>>4] and B[l>>4][l&15]
// (one f32 each); C/D: col = lane&15, row = (lane>>4)*4 + reg.
// K is tiled in steps of 4, accumulating in the same f32x4 (guide §3,
// "K-loop accumulator recipe"). One wave per 16x16 C tile.
__global__ void mfma_f32_16x16_kernel(const float* __restrict__ A,
                                      const float* __restrict__ B,
                                      float* __restrict__ C, int M, int N,
                                      int K) {
  int lane = threadIdx.x & 63;
  int wave = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  int tiles_n = N / 16;
  int tm = (wave / tiles_n) * 16;
  int tn = (wave % tiles_n) * 16;
  if (tm >= M) return;

  int row = lane & 15;
  int kk = lane >> 4;  // 0..3
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k0 = 0; k0 < K; k0 += 4) {
    float a = A[(tm + row) * K + (k0 + kk)];
    float b = B[(k0 + kk) * N + (tn + row)];
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }
  int crow = (lane >> 4) * 4;
  int ccol = lane & 15;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    C[(tm + crow + r) * N + (tn + ccol)] = acc[r];
  }
}

// bf16 MFMA 16x16x32 (gfx950 2xK form). Each lane holds 8 bf16 of A and B.
// Operand map: A[l&15][(l>>4)*8 + j], B[(l>>4)*8 + j][l&15]; C/D map as
// above (dtype-independent on gfx950, guide §3).
__global__ void mfma_bf16_16x16_kernel(const __hip_bfloat16* __restrict__ A,
                                       const __hip_bfloat16* __restrict__ B,
                                       float* __restrict__ C, int M, int N,
                                       int K) {
  int lane = threadIdx.x & 63;
  int wave = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  int tiles_n = N / 16;
  int tm = (wave / tiles_n) * 16;
  int tn = (wave % tiles_n) * 16;
  if (tm >= M) return;

  int row = lane & 15;
  int kbase = (lane >> 4) * 8;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int k0 = 0; k0 < K; k0 += 32) {
    bf16x8 a_frag, b_frag;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      a_frag[j] = *reinterpret_cast<const __bf16*>(&A[(tm + row) * K + k0 + kbase + j]);
      b_frag[j] = *reinterpret_cast<const __bf16*>(&B[(k0 + kbase + j) * N + tn + row]);
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc, 0, 0, 0);
  }
  int crow = (lane >> 4) * 4;
  int ccol = lane & 15;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    C[(tm + crow + r) * N + (tn + ccol)] = acc[r];
  }
}

// MFMA clock-ramp warm-up: dependent-chain f32 MFMA spin, one wave per block.
__global__ void mfma_warmup_kernel(float* __restrict__ sink, int iters) {
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  float a = (threadIdx.x & 15) * 0.001f + 1.0f;
  float b = (threadIdx.x >> 4) * 0.002f + 1.0f;
  for (int i = 0; i < iters; ++i) {
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }
  if (sink != nullptr && threadIdx.x == 0 && blockIdx.x == 0) sink[0] = acc[0];
}

}  // namespace  (reopened below — gemm lives in its own TU-style include)

#include "gemm_bf16.hip"
#include "gemm_bf16_8phase.hip"
#include "gemm_fp8_mx.hip"

namespace {

// LCG-hash fill: pseudorandom bf16 in [-1, 1) — zero-filled operands
// overstate GEMM throughput on this chip (guide §5.4 rule 25).
__global__ void fill_bf16_hash_kernel(__hip_bfloat16* p, size_t n, unsigned seed) {
  size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    unsigned h = (unsigned)i * 2654435761u + seed;
    h ^= h >> 15;
    h *= 2246822519u;
    h ^= h >> 13;
    float v = ((h & 0xFFFFFF) / 8388608.0f) - 1.0f;  // [-1, 1)
    p[i] = __hip_bfloat16(v);
  }
}

// ---------------------------------------------------------------------------
// Host-side probe machinery
// ---------------------------------------------------------------------------

struct DeviceBuf {
  float* ptr = nullptr;
  int device = -1;
  ~DeviceBuf() {
    if (ptr) {
      (void)hipSetDevice(device);
      (void)hipFree(ptr);
    }
  }
};

double time_kernel_ms(int device, std::function<void(hipStream_t)> body,
                      int iters) {
  HIP_CHECK(hipSetDevice(device));
  hipStream_t stream;
  HIP_CHECK(hipStreamCreate(&stream));
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  body(stream);  // one warm launch
  HIP_CHECK(hipStreamSynchronize(stream));
  HIP_CHECK(hipEventRecord(t0, stream));
  for (int i = 0; i < iters; ++i) body(stream);
  HIP_CHECK(hipEventRecord(t1, stream));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  HIP_CHECK(hipStreamDestroy(stream));
  return ms / iters;
}

void mfma_warmup(int device, int spins) {
  HIP_CHECK(hipSetDevice(device));
  hipLaunchKernelGGL(mfma_warmup_kernel, dim3(2048), dim3(256), 0, 0, nullptr,
                     spins);
  HIP_CHECK(hipDeviceSynchronize());
}

// Local HBM stream bandwidth: copy of `mib` MiB on one device; GB/s counts
// read + write bytes (achievable ceiling ≈6.3 TB/s on MI355X).
double stream_bandwidth_gbps(int device, int mib, int iters) {
  size_t n = (size_t)mib * 1024 * 1024 / sizeof(float);
  HIP_CHECK(hipSetDevice(device));
  mfma_warmup(device, 2000);
  DeviceBuf src, dst;
  src.device = dst.device = device;
  HIP_CHECK(hipMalloc(&src.ptr, n * sizeof(float)));
  HIP_CHECK(hipMalloc(&dst.ptr, n * sizeof(float)));
  HIP_CHECK(hipMemset(src.ptr, 1, n * sizeof(float)));
  double ms = time_kernel_ms(
      device,
      [&](hipStream_t s) { launch_copy_f32(src.ptr, dst.ptr, n, s); }, iters);
  return (2.0 * n * sizeof(float)) / (ms * 1e6);
}

// Reference point: SDMA/blit path via hipMemcpyAsync DtoD on one device.
double memcpy_bandwidth_gbps(int device, int mib, int iters) {
  size_t n = (size_t)mib * 1024 * 1024 / sizeof(float);
  HIP_CHECK(hipSetDevice(device));
  DeviceBuf src, dst;
  src.device = dst.device = device;
  HIP_CHECK(hipMalloc(&src.ptr, n * sizeof(float)));
  HIP_CHECK(hipMalloc(&dst.ptr, n * sizeof(float)));
  HIP_CHECK(hipMemset(src.ptr, 1, n * sizeof(float)));
  double ms = time_kernel_ms(
      device,
      [&](hipStream_t s) {
        HIP_CHECK(hipMemcpyAsync(dst.ptr, src.ptr, n * sizeof(float),
                                 hipMemcpyDeviceToDevice, s));
      },
      iters);
  return (2.0 * n * sizeof(float)) / (ms * 1e6);
}

// Peer bandwidth: dst-device kernel pulls from src-device memory (remote
// reads ride xGMI; per-link peak ≈153 GB/s, 7 links/GPU on an 8-GPU node).
// Falls back to hipMemcpyPeerAsync (SDMA path) when kernel p2p access is
// not available.
double p2p_bandwidth_gbps(int src_dev, int dst_dev, int mib, int iters) {
  if (src_dev == dst_dev) return stream_bandwidth_gbps(src_dev, mib, iters);
  size_t n = (size_t)mib * 1024 * 1024 / sizeof(float);

  int can_access = 0;
  HIP_CHECK(hipDeviceCanAccessPeer(&can_access, dst_dev, src_dev));

  DeviceBuf src, dst;
  src.device = src_dev;
  dst.device = dst_dev;
  HIP_CHECK(hipSetDevice(src_dev));
  HIP_CHECK(hipMalloc(&src.ptr, n * sizeof(float)));
  HIP_CHECK(hipMemset(src.ptr, 1, n * sizeof(float)));
  HIP_CHECK(hipSetDevice(dst_dev));
  HIP_CHECK(hipMalloc(&dst.ptr, n * sizeof(float)));

  if (can_access) {
    hipError_t e = hipDeviceEnablePeerAccess(src_dev, 0);
    if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) {
      (void)hipGetLastError();
      can_access = 0;
    } else {
      (void)hipGetLastError();
    }
  }
  mfma_warmup(dst_dev, 2000);
  double ms;
  if (can_access) {
    ms = time_kernel_ms(
        dst_dev,
        [&](hipStream_t s) { launch_copy_f32(src.ptr, dst.ptr, n, s); }, iters);
  } else {
    ms = time_kernel_ms(
        dst_dev,
        [&](hipStream_t s) {
          HIP_CHECK(hipMemcpyPeerAsync(dst.ptr, dst_dev, src.ptr, src_dev,
                                       n * sizeof(float), s));
        },
        iters);
  }
  // count bytes moved over the link once (n*4), not read+write
  return (double)(n * sizeof(float)) / (ms * 1e6);
}

// Full pairwise matrix in one call: per-device src/dst buffers allocated
// ONCE, peer access enabled once, then every ordered pair timed. On an
// 8-GPU node this is ~56 measurements; per-pair re-allocation would push
// daemon startup to minutes.
std::vector<std::vector<double>> p2p_matrix(int mib, int iters) {
  int n = 0;
  HIP_CHECK(hipGetDeviceCount(&n));
  std::vector<std::vector<double>> out(n, std::vector<double>(n, 0.0));
  if (n == 0) return out;
  size_t count = (size_t)mib * 1024 * 1024 / sizeof(float);

  std::vector<float*> src(n, nullptr), dst(n, nullptr);
  for (int i = 0; i < n; ++i) {
    HIP_CHECK(hipSetDevice(i));
    HIP_CHECK(hipMalloc(&src[i], count * sizeof(float)));
    HIP_CHECK(hipMalloc(&dst[i], count * sizeof(float)));
    HIP_CHECK(hipMemset(src[i], 1, count * sizeof(float)));
    mfma_warmup(i, 2000);
    for (int j = 0; j < n; ++j) {
      if (i == j) continue;
      int can = 0;
      HIP_CHECK(hipDeviceCanAccessPeer(&can, i, j));
      if (can) {
        hipError_t e = hipDeviceEnablePeerAccess(j, 0);
        if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled)
          (void)hipGetLastError();
        else
          (void)hipGetLastError();
      }
    }
  }
  for (int d = 0; d < n; ++d) {
    for (int s = 0; s < n; ++s) {
      if (s == d) {
        double ms = time_kernel_ms(
            d, [&](hipStream_t st) { launch_copy_f32(src[d], dst[d], count, st); },
            iters);
        out[s][d] = (2.0 * count * sizeof(float)) / (ms * 1e6);
        continue;
      }
      int can = 0;
      HIP_CHECK(hipDeviceCanAccessPeer(&can, d, s));
      double ms;
      if (can) {
        // dst-device kernel pulls from src-device memory over xGMI
        ms = time_kernel_ms(
            d, [&](hipStream_t st) { launch_copy_f32(src[s], dst[d], count, st); },
            iters);
      } else {
        ms = time_kernel_ms(
            d,
            [&](hipStream_t st) {
              HIP_CHECK(hipMemcpyPeerAsync(dst[d], d, src[s], s,
                                           count * sizeof(float), st));
            },
            iters);
      }
      out[s][d] = (double)(count * sizeof(float)) / (ms * 1e6);
    }
  }
  for (int i = 0; i < n; ++i) {
    (void)hipSetDevice(i);
    (void)hipFree(src[i]);
    (void)hipFree(dst[i]);
  }
  return out;
}

// ---------------------------------------------------------------------------
// Torch bindings
// ---------------------------------------------------------------------------

void copy_f32(torch::Tensor dst, torch::Tensor src) {
  TORCH_CHECK(src.is_cuda() && dst.is_cuda(), "tensors must be on GPU");
  TORCH_CHECK(src.scalar_type() == torch::kFloat32 &&
                  dst.scalar_type() == torch::kFloat32,
              "f32 only");
  TORCH_CHECK(src.is_contiguous() && dst.is_contiguous(), "contiguous only");
  TORCH_CHECK(src.numel() == dst.numel(), "size mismatch");
  auto stream = at::hip::getCurrentHIPStream();
  launch_copy_f32(src.data_ptr<float>(), dst.data_ptr<float>(), src.numel(),
                  stream.stream());
}

torch::Tensor mfma_f32_matmul(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda(), "GPU tensors required");
  TORCH_CHECK(A.scalar_type() == torch::kFloat32 &&
              B.scalar_type() == torch::kFloat32, "f32 only");
  A = A.contiguous();
  B = B.contiguous();
  int M = A.size(0), K = A.size(1), N = B.size(1);
  TORCH_CHECK(B.size(0) == K, "shape mismatch");
  TORCH_CHECK(M % 16 == 0 && N % 16 == 0 && K % 4 == 0,
              "M,N multiples of 16; K multiple of 4");
  auto C = torch::empty({M, N}, A.options());
  int waves = (M / 16) * (N / 16);
  constexpr int kWavesPerBlock = 4;  // 256 threads
  int grid = (waves + kWavesPerBlock - 1) / kWavesPerBlock;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_f32_16x16_kernel, dim3(grid),
                     dim3(kWavesPerBlock * 64), 0, stream.stream(),
                     A.data_ptr<float>(), B.data_ptr<float>(),
                     C.data_ptr<float>(), M, N, K);
  return C;
}

torch::Tensor mfma_bf16_matmul(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda(), "GPU tensors required");
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
              B.scalar_type() == torch::kBFloat16, "bf16 only");
  A = A.contiguous();
  B = B.contiguous();
  int M = A.size(0), K = A.size(1), N = B.size(1);
  TORCH_CHECK(B.size(0) == K, "shape mismatch");
  TORCH_CHECK(M % 16 == 0 && N % 16 == 0 && K % 32 == 0,
              "M,N multiples of 16; K multiple of 32");
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  int waves = (M / 16) * (N / 16);
  constexpr int kWavesPerBlock = 4;
  int grid = (waves + kWavesPerBlock - 1) / kWavesPerBlock;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_bf16_16x16_kernel, dim3(grid),
                     dim3(kWavesPerBlock * 64), 0, stream.stream(),
                     reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(B.data_ptr()),
                     C.data_ptr<float>(), M, N, K);
  return C;
}

torch::Tensor gemm_bf16_bt(torch::Tensor A, torch::Tensor Bt) {
  TORCH_CHECK(A.is_cuda() && Bt.is_cuda(), "GPU tensors required");
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
              Bt.scalar_type() == torch::kBFloat16, "bf16 only");
  A = A.contiguous();
  Bt = Bt.contiguous();
  int M = A.size(0), K = A.size(1), N = Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K, "Bt must be [N][K]");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 32 == 0,
              "M,N multiples of 128; K multiple of 32");
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  int grid = (M / 128) * (N / 128);
  auto stream = at::hip::getCurrentHIPStream();
  if (K % 64 == 0) {
    hipLaunchKernelGGL(gemm_bf16::gemm_bf16_tile_kernel<64>, dim3(grid),
                       dim3(256), 0, stream.stream(),
                       reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K, /*swizzle=*/0);
  } else {
    hipLaunchKernelGGL(gemm_bf16::gemm_bf16_tile_kernel<32>, dim3(grid),
                       dim3(256), 0, stream.stream(),
                       reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K, /*swizzle=*/0);
  }
  return C;
}

// Dense bf16 throughput on one device (the health-check burn-in number).
double gemm_bf16_tflops(int device, int size, int iters, int swizzle, int bk) {
  HIP_CHECK(hipSetDevice(device));
  mfma_warmup(device, 20000);
  size_t n = (size_t)size * size;
  __hip_bfloat16 *A = nullptr, *Bt = nullptr;
  float* C = nullptr;
  HIP_CHECK(hipMalloc(&A, n * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&Bt, n * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&C, n * sizeof(float)));
  hipLaunchKernelGGL(fill_bf16_hash_kernel, dim3(4096), dim3(256), 0, 0, A, n, 1u);
  hipLaunchKernelGGL(fill_bf16_hash_kernel, dim3(4096), dim3(256), 0, 0, Bt, n, 7u);
  HIP_CHECK(hipDeviceSynchronize());
  int grid = (size / 128) * (size / 128);
  double ms = time_kernel_ms(
      device,
      [&](hipStream_t s) {
        if (bk == 32) {
          hipLaunchKernelGGL(gemm_bf16::gemm_bf16_tile_kernel<32>, dim3(grid),
                             dim3(256), 0, s, A, Bt, C, size, size, size, swizzle);
        } else {
          hipLaunchKernelGGL(gemm_bf16::gemm_bf16_tile_kernel<64>, dim3(grid),
                             dim3(256), 0, s, A, Bt, C, size, size, size, swizzle);
        }
      },
      iters);
  (void)hipFree(A);
  (void)hipFree(Bt);
  (void)hipFree(C);
  return 2.0 * size * (double)size * size / (ms * 1e9);
}

// variant encoding: bit0 = XCD remap (T1), bits>=1 = schedule:
// 0/1 = <XCD,PIPE=0> round-1 schedule; 2/3 = <XCD,PIPE=1> phase-ahead;
// 4/5 = <XCD,PIPE=2> 2-tile-unrolled; 6/7 = <XCD,PIPE=3> unrolled,
// compiler-managed lgkm waits; 8/9 = <XCD,PIPE=4> merged phases 2+3;
// 10/11 = <XCD,PIPE=5> static-setprio younger half.
void launch_8ph(int variant, dim3 grid, hipStream_t s,
                const __hip_bfloat16* A, const __hip_bfloat16* Bt, float* C,
                int M, int N, int K) {
  switch (variant) {
    case 1:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<1, 0>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 2:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<0, 1>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 3:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<1, 1>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 4:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<0, 2>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 5:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<1, 2>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 6:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<0, 3>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 7:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<1, 3>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 8:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<0, 4>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 9:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<1, 4>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 10:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<0, 5>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 11:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<1, 5>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    case 12:  // PIPE=0 schedule + row-bit-3 LDS swizzle
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<0, 0, 1>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
    default:
      hipLaunchKernelGGL((gemm_bf16_8ph::gemm_bf16_8phase_kernel<0, 0>), grid,
                         dim3(512), 0, s, A, Bt, C, M, N, K);
      break;
  }
}

torch::Tensor gemm_bf16_8ph_bt(torch::Tensor A, torch::Tensor Bt, int variant) {
  TORCH_CHECK(A.is_cuda() && Bt.is_cuda(), "GPU tensors required");
  TORCH_CHECK(A.scalar_type() == torch::kBFloat16 &&
              Bt.scalar_type() == torch::kBFloat16, "bf16 only");
  A = A.contiguous();
  Bt = Bt.contiguous();
  int M = A.size(0), K = A.size(1), N = Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K, "Bt must be [N][K]");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 64 == 0 && K >= 192,
              "M,N multiples of 256; K multiple of 64, >= 192");
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  int grid = (M / 256) * (N / 256);
  auto stream = at::hip::getCurrentHIPStream();
  launch_8ph(variant, dim3(grid), stream.stream(),
             reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
             reinterpret_cast<const __hip_bfloat16*>(Bt.data_ptr()),
             C.data_ptr<float>(), M, N, K);
  return C;
}

double gemm_bf16_8ph_tflops(int device, int size, int iters, int variant) {
  HIP_CHECK(hipSetDevice(device));
  mfma_warmup(device, 20000);
  size_t n = (size_t)size * size;
  __hip_bfloat16 *A = nullptr, *Bt = nullptr;
  float* C = nullptr;
  HIP_CHECK(hipMalloc(&A, n * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&Bt, n * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&C, n * sizeof(float)));
  hipLaunchKernelGGL(fill_bf16_hash_kernel, dim3(4096), dim3(256), 0, 0, A, n, 1u);
  hipLaunchKernelGGL(fill_bf16_hash_kernel, dim3(4096), dim3(256), 0, 0, Bt, n, 7u);
  HIP_CHECK(hipDeviceSynchronize());
  int grid = (size / 256) * (size / 256);
  double ms = time_kernel_ms(
      device,
      [&](hipStream_t s) {
        launch_8ph(variant, dim3(grid), s, A, Bt, C, size, size, size);
      },
      iters);
  (void)hipFree(A);
  (void)hipFree(Bt);
  (void)hipFree(C);
  return 2.0 * size * (double)size * size / (ms * 1e9);
}

// Within-probe interleaved A/B over the template variants (guide §5.4
// rules 9/24: run-to-run noise ~±3%, so variants must be interleaved in one
// process). Returns {variant: [tflops per round]}.
py::dict gemm_bf16_8ph_ab(int device, int size, int iters, int rounds,
                          std::vector<int> variants) {
  HIP_CHECK(hipSetDevice(device));
  mfma_warmup(device, 20000);
  size_t n = (size_t)size * size;
  __hip_bfloat16 *A = nullptr, *Bt = nullptr;
  float* C = nullptr;
  HIP_CHECK(hipMalloc(&A, n * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&Bt, n * sizeof(__hip_bfloat16)));
  HIP_CHECK(hipMalloc(&C, n * sizeof(float)));
  hipLaunchKernelGGL(fill_bf16_hash_kernel, dim3(4096), dim3(256), 0, 0, A, n, 1u);
  hipLaunchKernelGGL(fill_bf16_hash_kernel, dim3(4096), dim3(256), 0, 0, Bt, n, 7u);
  HIP_CHECK(hipDeviceSynchronize());
  int grid = (size / 256) * (size / 256);
  std::map<int, std::vector<double>> out;
  for (int r = 0; r < rounds; ++r) {
    for (int v : variants) {
      double ms = time_kernel_ms(
          device,
          [&](hipStream_t s) {
            launch_8ph(v, dim3(grid), s, A, Bt, C, size, size, size);
          },
          iters);
      out[v].push_back(2.0 * size * (double)size * size / (ms * 1e9));
    }
  }
  (void)hipFree(A);
  (void)hipFree(Bt);
  (void)hipFree(C);
  py::dict d;
  for (auto& kv : out) d[py::int_(kv.first)] = kv.second;
  return d;
}

torch::Tensor gemm_fp8_mx_bt(torch::Tensor A, torch::Tensor Bt, int shape) {
  TORCH_CHECK(A.is_cuda() && Bt.is_cuda(), "GPU tensors required");
  TORCH_CHECK(A.scalar_type() == torch::kUInt8 &&
              Bt.scalar_type() == torch::kUInt8,
              "raw e4m3 bytes expected (uint8 view of float8_e4m3fn)");
  A = A.contiguous();
  Bt = Bt.contiguous();
  int M = A.size(0), K = A.size(1), N = Bt.size(0);
  TORCH_CHECK(Bt.size(1) == K, "Bt must be [N][K]");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 128 == 0 && K >= 256,
              "M,N multiples of 256; K multiple of 128, >= 256");
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  int grid = (M / 256) * (N / 256);
  auto stream = at::hip::getCurrentHIPStream();
  if (shape == 32)
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_mx32_kernel<0, 0>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else if (shape == 20)  // merged phases 2+3
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<0, 1, 0, 1>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else if (shape == 18)  // + quad-transpose dwordx4 epilogue
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<0, 1, 1>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else if (shape == 17)  // legacy row&7-only swizzle (A/B reference)
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<0, 0>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else  // default: row-bit-3 swizzle (conflict-free for the 2-chunk reads, +11%)
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<0, 1, 0, 0>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  return C;
}

torch::Tensor gemm_fp4_mx_bt(torch::Tensor A, torch::Tensor Bt, int K, int shape) {
  // A [M][K/2], Bt [N][K/2]: e2m1 nibble-packed (low nibble = even k)
  TORCH_CHECK(A.is_cuda() && Bt.is_cuda(), "GPU tensors required");
  TORCH_CHECK(A.scalar_type() == torch::kUInt8 &&
              Bt.scalar_type() == torch::kUInt8, "packed e2m1 bytes expected");
  A = A.contiguous();
  Bt = Bt.contiguous();
  int M = A.size(0), N = Bt.size(0);
  TORCH_CHECK((long)A.size(1) * 2 == K && (long)Bt.size(1) * 2 == K,
              "K must equal 2 x packed byte columns");
  TORCH_CHECK(M % 256 == 0 && N % 256 == 0 && K % 256 == 0 && K >= 512,
              "M,N multiples of 256; K multiple of 256, >= 512");
  auto C = torch::empty({M, N}, A.options().dtype(torch::kFloat32));
  int grid = (M / 256) * (N / 256);
  auto stream = at::hip::getCurrentHIPStream();
  if (shape == 32)
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_mx32_kernel<4, 0>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else if (shape == 19)  // merged phases 2+3
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 0, 0, 1>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else if (shape == 18)  // + quad-transpose dwordx4 epilogue
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 0, 1>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else if (shape == 17)  // 16x16 with the row-bit-3 swizzle
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 1>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else if (shape == 21)  // unmerged legacy (A/B reference)
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 0, 0, 0>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  else  // default: merged phases 2+3 (+1-2% measured both orders)
    hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 0, 0, 1>), dim3(grid), dim3(512),
                       0, stream.stream(),
                       reinterpret_cast<const unsigned char*>(A.data_ptr()),
                       reinterpret_cast<const unsigned char*>(Bt.data_ptr()),
                       C.data_ptr<float>(), M, N, K);
  return C;
}

double gemm_fp8_mx_tflops(int device, int size, int iters, int shape) {
  HIP_CHECK(hipSetDevice(device));
  mfma_warmup(device, 20000);
  size_t n = (size_t)size * size;
  unsigned char *A = nullptr, *Bt = nullptr;
  float* C = nullptr;
  HIP_CHECK(hipMalloc(&A, n));
  HIP_CHECK(hipMalloc(&Bt, n));
  HIP_CHECK(hipMalloc(&C, n * sizeof(float)));
  hipLaunchKernelGGL(gemm_fp8_mx::fill_e4m3_hash_kernel, dim3(4096), dim3(256),
                     0, 0, A, n, 1u);
  hipLaunchKernelGGL(gemm_fp8_mx::fill_e4m3_hash_kernel, dim3(4096), dim3(256),
                     0, 0, Bt, n, 7u);
  HIP_CHECK(hipDeviceSynchronize());
  int grid = (size / 256) * (size / 256);
  double ms = time_kernel_ms(
      device,
      [&](hipStream_t s) {
        if (shape == 32)
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_mx32_kernel<0, 0>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else if (shape == 17)  // legacy swizzle (A/B reference)
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<0, 0>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else if (shape == 20)  // merged phases 2+3
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<0, 1, 0, 1>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else if (shape == 18)  // + quad-transpose dwordx4 epilogue
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<0, 1, 1>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else  // default: row-bit-3 swizzle
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<0, 1>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
      },
      iters);
  (void)hipFree(A);
  (void)hipFree(Bt);
  (void)hipFree(C);
  return 2.0 * size * (double)size * size / (ms * 1e9);
}

double gemm_fp4_mx_tflops(int device, int size, int iters, int shape) {
  HIP_CHECK(hipSetDevice(device));
  mfma_warmup(device, 20000);
  size_t nb = (size_t)size * size / 2;  // nibble-packed
  unsigned char *A = nullptr, *Bt = nullptr;
  float* C = nullptr;
  HIP_CHECK(hipMalloc(&A, nb));
  HIP_CHECK(hipMalloc(&Bt, nb));
  HIP_CHECK(hipMalloc(&C, (size_t)size * size * sizeof(float)));
  hipLaunchKernelGGL(gemm_fp8_mx::fill_e2m1_hash_kernel, dim3(4096), dim3(256),
                     0, 0, A, nb, 1u);
  hipLaunchKernelGGL(gemm_fp8_mx::fill_e2m1_hash_kernel, dim3(4096), dim3(256),
                     0, 0, Bt, nb, 7u);
  HIP_CHECK(hipDeviceSynchronize());
  int grid = (size / 256) * (size / 256);
  double ms = time_kernel_ms(
      device,
      [&](hipStream_t s) {
        if (shape == 32)
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_mx32_kernel<4, 0>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else if (shape == 17)
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 1>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else if (shape == 19)  // merged phases 2+3
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 0, 0, 1>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else if (shape == 18)  // + quad-transpose dwordx4 epilogue
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 0, 1>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else if (shape == 21)  // unmerged legacy
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 0, 0, 0>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
        else  // default: merged phases 2+3
          hipLaunchKernelGGL((gemm_fp8_mx::gemm_fp8_mx_kernel<4, 0, 0, 1>), dim3(grid),
                             dim3(512), 0, s, A, Bt, C, size, size, size);
      },
      iters);
  (void)hipFree(A);
  (void)hipFree(Bt);
  (void)hipFree(C);
  return 2.0 * size * (double)size * size / (ms * 1e9);
}

int device_count() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

py::dict device_info(int device) {
  hipDeviceProp_t prop;
  HIP_CHECK(hipGetDeviceProperties(&prop, device));
  py::dict d;
  d["name"] = std::string(prop.name);
  d["gcnArchName"] = std::string(prop.gcnArchName);
  d["totalGlobalMem"] = (long long)prop.totalGlobalMem;
  d["multiProcessorCount"] = prop.multiProcessorCount;
  d["pciBusID"] = prop.pciBusID;
  d["pciDomainID"] = prop.pciDomainID;
  d["pciDeviceID"] = prop.pciDeviceID;
  size_t free_b = 0, total_b = 0;
  HIP_CHECK(hipSetDevice(device));
  HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
  d["freeMem"] = (long long)free_b;
  return d;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X native core: bandwidth probes + MFMA tiles";
  m.def("device_count", &device_count);
  m.def("device_info", &device_info, py::arg("device"));
  m.def("copy_f32", &copy_f32, py::arg("dst"), py::arg("src"));
  m.def("mfma_f32_matmul", &mfma_f32_matmul, py::arg("A"), py::arg("B"));
  m.def("mfma_bf16_matmul", &mfma_bf16_matmul, py::arg("A"), py::arg("B"));
  m.def("mfma_warmup", &mfma_warmup, py::arg("device") = 0,
        py::arg("spins") = 20000);
  m.def("gemm_bf16_bt", &gemm_bf16_bt, py::arg("A"), py::arg("Bt"));
  m.def("gemm_bf16_8ph", &gemm_bf16_8ph_bt, py::arg("A"), py::arg("Bt"),
        py::arg("variant") = 0);
  m.def("gemm_fp8_mx", &gemm_fp8_mx_bt, py::arg("A"), py::arg("Bt"),
        py::arg("shape") = 16);
  m.def("gemm_fp4_mx", &gemm_fp4_mx_bt, py::arg("A"), py::arg("Bt"), py::arg("K"),
        py::arg("shape") = 16);
  m.def("gemm_fp4_mx_tflops", &gemm_fp4_mx_tflops, py::arg("device") = 0,
        py::arg("size") = 4096, py::arg("iters") = 10, py::arg("shape") = 16,
        py::call_guard<py::gil_scoped_release>());
  m.def("gemm_fp8_mx_tflops", &gemm_fp8_mx_tflops, py::arg("device") = 0,
        py::arg("size") = 4096, py::arg("iters") = 10, py::arg("shape") = 16,
        py::call_guard<py::gil_scoped_release>());
  m.def("gemm_bf16_8ph_ab", &gemm_bf16_8ph_ab, py::arg("device") = 0,
        py::arg("size") = 4096, py::arg("iters") = 4, py::arg("rounds") = 3,
        py::arg("variants") = std::vector<int>{0, 1, 2, 3});
  m.def("gemm_bf16_8ph_tflops", &gemm_bf16_8ph_tflops, py::arg("device") = 0,
        py::arg("size") = 4096, py::arg("iters") = 10, py::arg("variant") = 0,
        py::call_guard<py::gil_scoped_release>());
  m.def("gemm_bf16_tflops", &gemm_bf16_tflops, py::arg("device") = 0,
        py::arg("size") = 4096, py::arg("iters") = 10, py::arg("swizzle") = 0,
        py::arg("bk") = 64, py::call_guard<py::gil_scoped_release>());
  m.def("stream_bandwidth_gbps", &stream_bandwidth_gbps, py::arg("device") = 0,
        py::arg("mib") = 1024, py::arg("iters") = 10,
        py::call_guard<py::gil_scoped_release>());
  m.def("memcpy_bandwidth_gbps", &memcpy_bandwidth_gbps, py::arg("device") = 0,
        py::arg("mib") = 1024, py::arg("iters") = 10,
        py::call_guard<py::gil_scoped_release>());
  m.def("p2p_bandwidth_gbps", &p2p_bandwidth_gbps, py::arg("src_dev"),
        py::arg("dst_dev"), py::arg("mib") = 512, py::arg("iters") = 10,
        py::call_guard<py::gil_scoped_release>());
  m.def("p2p_matrix", &p2p_matrix, py::arg("mib") = 256, py::arg("iters") = 5,
        py::call_guard<py::gil_scoped_release>());
}
