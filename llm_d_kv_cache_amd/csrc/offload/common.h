// Shared plumbing for the offload data plane: HIP error handling, logging,
// NUMA discovery, pinned/device staging buffers.
//
// Every component here is dual-mode: on a GPU box the engine runs HIP
// streams/events/kernels; constructed against host memory (no GPU present,
// e.g. CPU-only CI) the same control logic runs with memcpy + malloc. The
// mode is explicit (Engine ctor argument), never a silent fallback: asking
// for GPU mode on a machine without a device throws.
#pragma once

#include <hip/hip_runtime.h>

#include <atomic>
#include <chrono>
#include <cstdarg>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace kvo {

// ---- errors -----------------------------------------------------------------

struct HipError : std::runtime_error {
  using std::runtime_error::runtime_error;
};

#define KVO_HIP_CHECK(expr)                                                   \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    if (_e != hipSuccess) {                                                   \
      throw ::kvo::HipError(std::string(#expr) + " failed: " +                \
                            hipGetErrorString(_e));                           \
    }                                                                         \
  } while (0)

// ---- env-leveled logger (parity with reference csrc logger.hpp) -------------

enum class LogLevel { kError = 0, kWarn = 1, kInfo = 2, kDebug = 3, kTrace = 4 };

inline LogLevel log_level() {
  static LogLevel lvl = [] {
    const char* e = std::getenv("KVC_LOG_LEVEL");
    if (!e) return LogLevel::kWarn;
    std::string s(e);
    if (s == "error") return LogLevel::kError;
    if (s == "warn") return LogLevel::kWarn;
    if (s == "info") return LogLevel::kInfo;
    if (s == "debug") return LogLevel::kDebug;
    if (s == "trace") return LogLevel::kTrace;
    return LogLevel::kWarn;
  }();
  return lvl;
}

inline void logf(LogLevel lvl, const char* fmt, ...) {
  if (lvl > log_level()) return;
  static const char* names[] = {"ERROR", "WARN", "INFO", "DEBUG", "TRACE"};
  char buf[1024];
  va_list args;
  va_start(args, fmt);
  vsnprintf(buf, sizeof(buf), fmt, args);
  va_end(args);
  fprintf(stderr, "[kvoffload %s] %s\n", names[static_cast<int>(lvl)], buf);
}

#define KVO_LOG_INFO(...) ::kvo::logf(::kvo::LogLevel::kInfo, __VA_ARGS__)
#define KVO_LOG_WARN(...) ::kvo::logf(::kvo::LogLevel::kWarn, __VA_ARGS__)
#define KVO_LOG_DEBUG(...) ::kvo::logf(::kvo::LogLevel::kDebug, __VA_ARGS__)
#define KVO_LOG_ERROR(...) ::kvo::logf(::kvo::LogLevel::kError, __VA_ARGS__)

inline double now_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

// ---- NUMA discovery ---------------------------------------------------------
// GPU -> host NUMA node via sysfs (PCIe BDF -> numa_node -> cpulist), so I/O
// threads and their pinned staging live on the GPU's host bridge domain.

inline int gpu_numa_node(int device) {
  char bus_id[64] = {0};
  if (hipDeviceGetPCIBusId(bus_id, sizeof(bus_id), device) != hipSuccess)
    return -1;
  for (char* p = bus_id; *p; ++p) *p = std::tolower(*p);
  std::string path = std::string("/sys/bus/pci/devices/") + bus_id + "/numa_node";
  FILE* f = fopen(path.c_str(), "r");
  if (!f) return -1;
  int node = -1;
  if (fscanf(f, "%d", &node) != 1) node = -1;
  fclose(f);
  return node;
}

inline std::vector<int> numa_node_cpus(int node) {
  std::vector<int> cpus;
  if (node < 0) return cpus;
  std::string path =
      "/sys/devices/system/node/node" + std::to_string(node) + "/cpulist";
  FILE* f = fopen(path.c_str(), "r");
  if (!f) return cpus;
  char buf[4096] = {0};
  if (fgets(buf, sizeof(buf), f)) {
    // format: "0-31,64-95"
    char* save = nullptr;
    for (char* tok = strtok_r(buf, ",\n", &save); tok;
         tok = strtok_r(nullptr, ",\n", &save)) {
      int lo, hi;
      if (sscanf(tok, "%d-%d", &lo, &hi) == 2) {
        for (int c = lo; c <= hi; ++c) cpus.push_back(c);
      } else if (sscanf(tok, "%d", &lo) == 1) {
        cpus.push_back(lo);
      }
    }
  }
  fclose(f);
  return cpus;
}

// ---- staging buffers --------------------------------------------------------

// Pinned host buffer (device-visible) in GPU mode; aligned malloc in host
// mode. 4 KiB alignment keeps O_DIRECT file I/O legal on either path.
class HostStaging {
 public:
  // mapped=true exposes a device pointer for zero-copy kernel writes;
  // mapped=false keeps the buffer SDMA-eligible for hipMemcpyAsync (a
  // mapped buffer makes the runtime treat D2H as device-to-device and run
  // a blit kernel at ~half SDMA speed while burning CUs — measured on
  // MI355X, see profiles/).
  HostStaging(size_t bytes, bool gpu_mode, bool mapped = false)
      : bytes_(bytes), gpu_mode_(gpu_mode) {
    if (bytes_ == 0) return;
    if (gpu_mode_) {
      unsigned flags = hipHostMallocPortable;
      if (mapped) flags |= hipHostMallocMapped;
      KVO_HIP_CHECK(hipHostMalloc(&ptr_, bytes_, flags));
      if (mapped) KVO_HIP_CHECK(hipHostGetDevicePointer(&dev_ptr_, ptr_, 0));
    } else {
      if (posix_memalign(&ptr_, 4096, bytes_) != 0)
        throw std::bad_alloc();
      dev_ptr_ = ptr_;
    }
  }
  ~HostStaging() {
    if (!ptr_) return;
    if (gpu_mode_)
      (void)hipHostFree(ptr_);
    else
      free(ptr_);
  }
  HostStaging(const HostStaging&) = delete;
  HostStaging& operator=(const HostStaging&) = delete;

  uint8_t* host() const { return static_cast<uint8_t*>(ptr_); }
  uint8_t* device() const { return static_cast<uint8_t*>(dev_ptr_); }
  size_t size() const { return bytes_; }

 private:
  size_t bytes_;
  bool gpu_mode_;
  void* ptr_ = nullptr;
  void* dev_ptr_ = nullptr;
};

// Device bounce buffer (HBM) for the staged copy path; null in host mode.
class DeviceStaging {
 public:
  DeviceStaging(size_t bytes, bool gpu_mode) : bytes_(bytes) {
    if (bytes_ == 0 || !gpu_mode) return;
    KVO_HIP_CHECK(hipMalloc(&ptr_, bytes_));
  }
  ~DeviceStaging() {
    if (ptr_) (void)hipFree(ptr_);
  }
  DeviceStaging(const DeviceStaging&) = delete;
  DeviceStaging& operator=(const DeviceStaging&) = delete;

  uint8_t* ptr() const { return static_cast<uint8_t*>(ptr_); }
  size_t size() const { return bytes_; }

 private:
  size_t bytes_;
  void* ptr_ = nullptr;
};

}  // namespace kvo
