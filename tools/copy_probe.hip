// Standalone probe: which hipMemcpy D2H/H2D variants engage SDMA on this
// box? Blit-kernel copies measure ~25 GB/s and occupy CUs; SDMA measures
// ~55+ GB/s. Build: hipcc --offload-arch=gfx950 -O2 tools/copy_probe.hip -o copy_probe
#include <hip/hip_runtime.h>

#include <chrono>
#include <cstdio>
#include <thread>
#include <vector>

#define CHECK(x)                                                      \
  do {                                                                \
    hipError_t e = (x);                                               \
    if (e != hipSuccess) {                                            \
      printf("FAIL %s: %s\n", #x, hipGetErrorString(e));              \
      return;                                                         \
    }                                                                 \
  } while (0)

constexpr size_t N = 256ull << 20;  // 256 MiB
constexpr int ITERS = 10;

double now() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

void bench(const char* name, unsigned host_flags, bool with_stream_api,
           bool nonblocking_stream, bool sync_api = false) {
  void* dptr;
  CHECK(hipMalloc(&dptr, N));
  void* hptr;
  CHECK(hipHostMalloc(&hptr, N, host_flags));
  hipStream_t s;
  CHECK(hipStreamCreateWithFlags(&s, nonblocking_stream ? hipStreamNonBlocking : 0));
  // warmup
  CHECK(hipMemcpyAsync(hptr, dptr, N, hipMemcpyDeviceToHost, s));
  CHECK(hipStreamSynchronize(s));
  double t0 = now();
  for (int i = 0; i < ITERS; ++i) {
    if (sync_api) {
      CHECK(hipMemcpy(hptr, dptr, N, hipMemcpyDeviceToHost));
    } else if (with_stream_api) {
      CHECK(hipMemcpyWithStream(hptr, dptr, N, hipMemcpyDeviceToHost, s));
    } else {
      CHECK(hipMemcpyAsync(hptr, dptr, N, hipMemcpyDeviceToHost, s));
      CHECK(hipStreamSynchronize(s));
    }
  }
  if (!sync_api) CHECK(hipStreamSynchronize(s));
  double dt = now() - t0;
  printf("%-44s D2H %.1f GB/s\n", name, N * (double)ITERS / dt / 1e9);
  hipStreamDestroy(s);
  hipHostFree(hptr);
  hipFree(dptr);
}

void bench_threads(int nthreads) {
  std::vector<std::thread> ts;
  double agg_t0 = now();
  for (int t = 0; t < nthreads; ++t) {
    ts.emplace_back([] {
      void* dptr;
      if (hipMalloc(&dptr, N) != hipSuccess) return;
      void* hptr;
      if (hipHostMalloc(&hptr, N, hipHostMallocPortable) != hipSuccess) return;
      hipStream_t s;
      hipStreamCreateWithFlags(&s, hipStreamNonBlocking);
      for (int i = 0; i < ITERS; ++i) {
        hipMemcpyAsync(hptr, dptr, N, hipMemcpyDeviceToHost, s);
        hipStreamSynchronize(s);
      }
      hipStreamDestroy(s);
      hipHostFree(hptr);
      hipFree(dptr);
    });
  }
  for (auto& t : ts) t.join();
  double dt = now() - agg_t0;
  printf("%d threads x own stream (Portable)            D2H %.1f GB/s aggregate\n",
         nthreads, N * (double)ITERS * nthreads / dt / 1e9);
}

int main() {
  int count = 0;
  hipGetDeviceCount(&count);
  printf("devices: %d\n", count);
  bench("async + default-flag pinned + nb stream", hipHostMallocDefault, false, true);
  bench("async + Portable pinned + nb stream", hipHostMallocPortable, false, true);
  bench("async + Mapped|Portable pinned + nb stream",
        hipHostMallocMapped | hipHostMallocPortable, false, true);
  bench("async + NonCoherent pinned + nb stream", hipHostMallocNonCoherent, false,
        true);
  bench("async + default pinned + DEFAULT stream flag", hipHostMallocDefault, false,
        false);
  bench("WithStream + default pinned", hipHostMallocDefault, true, true);
  bench("sync hipMemcpy + default pinned", hipHostMallocDefault, false, true, true);
  bench_threads(4);
  return 0;
}
