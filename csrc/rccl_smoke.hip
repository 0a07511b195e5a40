// RCCL all-reduce smoke test over a candidate GPU set.
//
// Native component #2 of SURVEY.md §2.4: validates a scheduler placement by
// running a real ring all-reduce over xGMI across the GPUs the daemon is
// about to hand to a container, and reports the achieved bus bandwidth
// (per-link bound on xGMI: ~153 GB/s/link peak). Single process, one thread
// per GPU via ncclCommInitAll + grouped calls — no MPI, no torchrun.
//
// Usage: rccl_smoke [ndev] [MiB]   (default: all visible devices, 64 MiB)
// Output: one JSON line, e.g.
//   {"ok": true, "world": 8, "mib": 64, "avg_ms": 1.2, "busbw_gbps": 93.1}
//
// Build: hipcc --offload-arch=gfx950 -O2 rccl_smoke.hip -lrccl -o rccl_smoke
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(cmd, what)                                                      \
  do {                                                                        \
    auto _e = (cmd);                                                          \
    if (_e != 0) {                                                            \
      std::printf("{\"ok\": false, \"error\": \"%s rc=%d\"}\n", what, (int)_e); \
      return 1;                                                               \
    }                                                                         \
  } while (0)

int main(int argc, char** argv) {
  int ndev = 0;
  CHECK(hipGetDeviceCount(&ndev), "hipGetDeviceCount");
  if (argc > 1) {
    int want = std::atoi(argv[1]);
    if (want > 0 && want <= ndev) ndev = want;
  }
  if (ndev < 1) {
    std::printf("{\"ok\": false, \"error\": \"no GPUs visible\"}\n");
    return 1;
  }
  size_t mib = 64;
  if (argc > 2) mib = (size_t)std::atoll(argv[2]);
  size_t count = mib * 1024 * 1024 / sizeof(float);

  std::vector<ncclComm_t> comms(ndev);
  std::vector<int> devs(ndev);
  for (int i = 0; i < ndev; ++i) devs[i] = i;
  CHECK(ncclCommInitAll(comms.data(), ndev, devs.data()), "ncclCommInitAll");

  std::vector<float*> send(ndev), recv(ndev);
  std::vector<hipStream_t> streams(ndev);
  std::vector<float> host(count);
  for (int i = 0; i < ndev; ++i) {
    CHECK(hipSetDevice(i), "hipSetDevice");
    CHECK(hipMalloc(&send[i], count * sizeof(float)), "hipMalloc");
    CHECK(hipMalloc(&recv[i], count * sizeof(float)), "hipMalloc");
    for (size_t j = 0; j < count; ++j) host[j] = (float)(i + 1);
    CHECK(hipMemcpy(send[i], host.data(), count * sizeof(float),
                    hipMemcpyHostToDevice), "hipMemcpy");
    CHECK(hipStreamCreate(&streams[i]), "hipStreamCreate");
  }

  auto allreduce_once = [&]() -> int {
    CHECK(ncclGroupStart(), "ncclGroupStart");
    for (int i = 0; i < ndev; ++i) {
      CHECK(ncclAllReduce(send[i], recv[i], count, ncclFloat, ncclSum,
                          comms[i], streams[i]), "ncclAllReduce");
    }
    CHECK(ncclGroupEnd(), "ncclGroupEnd");
    for (int i = 0; i < ndev; ++i) {
      CHECK(hipSetDevice(i), "hipSetDevice");
      CHECK(hipStreamSynchronize(streams[i]), "hipStreamSynchronize");
    }
    return 0;
  };

  // warm-up then timed iterations
  for (int w = 0; w < 3; ++w)
    if (allreduce_once() != 0) return 1;
  const int iters = 10;
  auto t0 = std::chrono::steady_clock::now();
  for (int it = 0; it < iters; ++it)
    if (allreduce_once() != 0) return 1;
  auto t1 = std::chrono::steady_clock::now();
  double avg_ms =
      std::chrono::duration<double, std::milli>(t1 - t0).count() / iters;

  // verify: every element must equal sum(1..ndev)
  float expect = 0.f;
  for (int i = 1; i <= ndev; ++i) expect += (float)i;
  bool ok = true;
  for (int i = 0; i < ndev && ok; ++i) {
    CHECK(hipSetDevice(i), "hipSetDevice");
    CHECK(hipMemcpy(host.data(), recv[i], count * sizeof(float),
                    hipMemcpyDeviceToHost), "hipMemcpy");
    for (size_t j = 0; j < count; j += count / 97 + 1) {
      if (host[j] != expect) {
        ok = false;
        break;
      }
    }
  }

  // ring all-reduce bus bandwidth: 2*(n-1)/n * bytes / time per GPU
  double bytes = (double)count * sizeof(float);
  double busbw =
      ndev > 1 ? (2.0 * (ndev - 1) / ndev) * bytes / (avg_ms * 1e6) : 0.0;

  for (int i = 0; i < ndev; ++i) {
    (void)hipSetDevice(i);
    (void)hipFree(send[i]);
    (void)hipFree(recv[i]);
    (void)hipStreamDestroy(streams[i]);
    (void)ncclCommDestroy(comms[i]);
  }
  std::printf(
      "{\"ok\": %s, \"world\": %d, \"mib\": %zu, \"avg_ms\": %.3f, "
      "\"busbw_gbps\": %.2f}\n",
      ok ? "true" : "false", ndev, mib, avg_ms, busbw);
  return ok ? 0 : 1;
}
