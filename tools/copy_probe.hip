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

// Full-duplex: one thread streams D2H, another H2D, own streams + own
// buffers. Decides whether ~50 GB/s TOTAL in the offload bench is an
// orchestration gap or the platform's duplex ceiling (both directions are
// blit kernels here, not SDMA — see profiles/r01_offload_profile.md).
void bench_duplex(int streams_per_dir) {
  // smaller buffers + fewer iters than the one-direction benches: if the
  // two directions collapse when concurrent, a full-size run would take
  // minutes — the collapse itself is the finding
  constexpr size_t ND = 128ull << 20;
  constexpr int ITD = 5;
  struct Dir {
    double gbps = 0;
  } res[2];
  std::vector<std::thread> ts;
  double t0 = now();
  for (int dir = 0; dir < 2; ++dir) {
    ts.emplace_back([dir, streams_per_dir, &res] {
      std::vector<void*> d(streams_per_dir), h(streams_per_dir);
      std::vector<hipStream_t> s(streams_per_dir);
      for (int i = 0; i < streams_per_dir; ++i) {
        if (hipMalloc(&d[i], ND) != hipSuccess) return;
        if (hipHostMalloc(&h[i], ND, hipHostMallocPortable) != hipSuccess)
          return;
        hipStreamCreateWithFlags(&s[i], hipStreamNonBlocking);
      }
      auto kind = dir == 0 ? hipMemcpyDeviceToHost : hipMemcpyHostToDevice;
      // warmup
      for (int i = 0; i < streams_per_dir; ++i) {
        hipMemcpyAsync(dir == 0 ? h[i] : d[i], dir == 0 ? d[i] : h[i], ND,
                       kind, s[i]);
        hipStreamSynchronize(s[i]);
      }
      double t1 = now();
      for (int it = 0; it < ITD; ++it) {
        for (int i = 0; i < streams_per_dir; ++i)
          hipMemcpyAsync(dir == 0 ? h[i] : d[i], dir == 0 ? d[i] : h[i], ND,
                         kind, s[i]);
        for (int i = 0; i < streams_per_dir; ++i) hipStreamSynchronize(s[i]);
      }
      double dt = now() - t1;
      res[dir].gbps = ND * (double)ITD * streams_per_dir / dt / 1e9;
      for (int i = 0; i < streams_per_dir; ++i) {
        hipStreamDestroy(s[i]);
        hipHostFree(h[i]);
        hipFree(d[i]);
      }
    });
  }
  for (auto& t : ts) t.join();
  double wall = now() - t0;
  printf("duplex %d stream/dir: D2H %.1f + H2D %.1f = %.1f GB/s total "
         "(wall %.2fs)\n",
         streams_per_dir, res[0].gbps, res[1].gbps, res[0].gbps + res[1].gbps,
         wall);
}

void run_file_bench();
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
  bench_duplex(1);
  bench_duplex(2);
  run_file_bench();
  return 0;
}

// ---- file I/O section: tmpfs write/read from pinned vs malloc buffers ----
#include <fcntl.h>
#include <unistd.h>
#include <cstring>
#include <atomic>

static void file_bench(const char* name, void* buf, size_t file_sz, int nthreads) {
  std::vector<std::thread> ts;
  std::atomic<int> ctr{0};
  double t0 = now();
  for (int t = 0; t < nthreads; ++t) {
    ts.emplace_back([&, t] {
      char path[128];
      snprintf(path, sizeof(path), "/dev/shm/probe_%s_%d.bin", name, t);
      for (int i = 0; i < 8; ++i) {
        int fd = open(path, O_WRONLY | O_CREAT | O_TRUNC, 0644);
        size_t off = 0;
        while (off < file_sz) {
          ssize_t w = pwrite(fd, (char*)buf + (t * file_sz) + off, file_sz - off, off);
          if (w <= 0) break;
          off += w;
        }
        close(fd);
      }
      ctr++;
    });
  }
  for (auto& th : ts) th.join();
  double wt = now() - t0;
  printf("%-34s write %d thr: %.2f GB/s\n", name, nthreads,
         file_sz * 8.0 * nthreads / wt / 1e9);
  t0 = now();
  ts.clear();
  for (int t = 0; t < nthreads; ++t) {
    ts.emplace_back([&, t] {
      char path[128];
      snprintf(path, sizeof(path), "/dev/shm/probe_%s_%d.bin", name, t);
      for (int i = 0; i < 8; ++i) {
        int fd = open(path, O_RDONLY);
        size_t off = 0;
        while (off < file_sz) {
          ssize_t r = pread(fd, (char*)buf + (t * file_sz) + off, file_sz - off, off);
          if (r <= 0) break;
          off += r;
        }
        close(fd);
      }
    });
  }
  for (auto& th : ts) th.join();
  double rt = now() - t0;
  printf("%-34s read  %d thr: %.2f GB/s\n", name, nthreads,
         file_sz * 8.0 * nthreads / rt / 1e9);
  for (int t = 0; t < nthreads; ++t) {
    char path[128];
    snprintf(path, sizeof(path), "/dev/shm/probe_%s_%d.bin", name, t);
    unlink(path);
  }
}

void run_file_bench() {
  const size_t FSZ = 32ull << 20;
  const int NT = 16;
  void* mbuf = malloc(FSZ * NT);
  memset(mbuf, 1, FSZ * NT);
  file_bench("malloc", mbuf, FSZ, 1);
  file_bench("malloc", mbuf, FSZ, NT);
  free(mbuf);
  void* pbuf;
  if (hipHostMalloc(&pbuf, FSZ * NT, hipHostMallocPortable) == hipSuccess) {
    memset(pbuf, 1, FSZ * NT);
    file_bench("pinned-portable", pbuf, FSZ, 1);
    file_bench("pinned-portable", pbuf, FSZ, NT);
    hipHostFree(pbuf);
  }
  if (hipHostMalloc(&pbuf, FSZ * NT, 0) == hipSuccess) {
    memset(pbuf, 1, FSZ * NT);
    file_bench("pinned-default", pbuf, FSZ, 1);
    file_bench("pinned-default", pbuf, FSZ, NT);
    hipHostFree(pbuf);
  }
}
