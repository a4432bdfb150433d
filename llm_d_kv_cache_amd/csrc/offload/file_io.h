// POSIX storage backend: atomic tmp+rename writes, tail-seek reads, atime
// touch (the storage evictor's hotness signal).
//
// Capability parity with the reference FileIO
// (csrc/storage/backends/fs_io/file_io.cpp): same durability contract
// (<path>.<rand>.tmp + rename), same atime semantics (utimensat with
// mtime=UTIME_OMIT). Writes are a single pwrite of the contiguous staging
// buffer instead of a 1 MB-buffered ofstream loop.
#pragma once

#include <fcntl.h>
#include <sys/stat.h>
#include <sys/time.h>
#include <unistd.h>

#include <cerrno>
#include <cstdint>
#include <random>
#include <string>

#include "common.h"

namespace kvo {

struct FileIoError : std::runtime_error {
  using std::runtime_error::runtime_error;
};

inline bool file_exists(const std::string& path) {
  struct stat st;
  return ::stat(path.c_str(), &st) == 0;
}

inline int64_t file_size(const std::string& path) {
  struct stat st;
  if (::stat(path.c_str(), &st) != 0) return -1;
  return st.st_size;
}

// Refresh atime (PVC-evictor hotness), preserve mtime.
inline void touch_atime(const std::string& path) {
  struct timespec times[2];
  times[0].tv_sec = 0;
  times[0].tv_nsec = UTIME_NOW;   // atime = now
  times[1].tv_sec = 0;
  times[1].tv_nsec = UTIME_OMIT;  // mtime untouched
  ::utimensat(AT_FDCWD, path.c_str(), times, 0);
}

inline void make_parent_dirs(const std::string& path) {
  size_t pos = 0;
  while ((pos = path.find('/', pos + 1)) != std::string::npos) {
    std::string dir = path.substr(0, pos);
    if (!dir.empty()) ::mkdir(dir.c_str(), 0755);
  }
}

constexpr size_t kDirectIoAlign = 4096;

inline bool direct_io_eligible(const void* ptr, uint64_t off, size_t len) {
  return (reinterpret_cast<uintptr_t>(ptr) % kDirectIoAlign == 0) &&
         (off % kDirectIoAlign == 0) && (len % kDirectIoAlign == 0);
}

// Atomic whole-buffer write: <path>.<rand>.tmp then rename.
inline void write_file_atomic(const std::string& path, const uint8_t* data,
                              size_t len) {
  static thread_local std::mt19937_64 rng{std::random_device{}()};
  std::string tmp = path + "." + std::to_string(rng()) + ".tmp";
  int fd = ::open(tmp.c_str(), O_WRONLY | O_CREAT | O_EXCL, 0644);
  if (fd < 0 && errno == ENOENT) {
    make_parent_dirs(tmp);
    fd = ::open(tmp.c_str(), O_WRONLY | O_CREAT | O_EXCL, 0644);
  }
  if (fd < 0)
    throw FileIoError("open " + tmp + ": " + std::strerror(errno));
  size_t off = 0;
  while (off < len) {
    ssize_t w = ::pwrite(fd, data + off, len - off, static_cast<off_t>(off));
    if (w < 0) {
      if (errno == EINTR) continue;
      int err = errno;
      ::close(fd);
      ::unlink(tmp.c_str());
      throw FileIoError("pwrite " + tmp + ": " + std::strerror(err));
    }
    off += static_cast<size_t>(w);
  }
  ::close(fd);
  if (::rename(tmp.c_str(), path.c_str()) != 0) {
    int err = errno;
    ::unlink(tmp.c_str());
    throw FileIoError("rename " + path + ": " + std::strerror(err));
  }
}

// Incremental atomic writer: open <path>.<rand>.tmp, pwrite chunks, then
// commit() renames into place (abort unlinks). Enables overlapping file
// writes with the PCIe pipeline inside one transfer.
class AtomicFileWriter {
 public:
  // direct_io: request O_DIRECT (page-cache bypass — the GDS-substitute
  // mode for local NVMe; reference GdsFileIO role). Falls back to buffered
  // I/O when the filesystem refuses O_DIRECT (tmpfs) or when a write's
  // buffer/offset/length is not 4 KiB-aligned (e.g. fp8 tile records).
  explicit AtomicFileWriter(const std::string& path, bool direct_io = false)
      : path_(path) {
    static thread_local std::mt19937_64 rng{std::random_device{}()};
    tmp_ = path + "." + std::to_string(rng()) + ".tmp";
    int flags = O_WRONLY | O_CREAT | O_EXCL;
    if (direct_io) {
      fd_ = open_with_dirs(flags | O_DIRECT);
      direct_ = fd_ >= 0;
    }
    if (fd_ < 0) fd_ = open_with_dirs(flags);
    if (fd_ < 0)
      throw FileIoError("open " + tmp_ + ": " + std::strerror(errno));
  }

  ~AtomicFileWriter() {
    if (fd_ >= 0) {
      ::close(fd_);
      ::unlink(tmp_.c_str());
    }
  }
  AtomicFileWriter(const AtomicFileWriter&) = delete;

  void write_at(uint64_t offset, const uint8_t* data, size_t len) {
    if (direct_ && !direct_io_eligible(data, offset, len)) {
      // drop O_DIRECT for the rest of this file: mixed modes on one fd
      // risk short-write semantics differences across filesystems
      ::fcntl(fd_, F_SETFL, ::fcntl(fd_, F_GETFL) & ~O_DIRECT);
      direct_ = false;
    }
    size_t off = 0;
    while (off < len) {
      ssize_t w = ::pwrite(fd_, data + off, len - off,
                           static_cast<off_t>(offset + off));
      if (w < 0) {
        if (errno == EINTR) continue;
        if (direct_ && (errno == EINVAL)) {
          ::fcntl(fd_, F_SETFL, ::fcntl(fd_, F_GETFL) & ~O_DIRECT);
          direct_ = false;
          continue;
        }
        throw FileIoError("pwrite " + tmp_ + ": " + std::strerror(errno));
      }
      off += static_cast<size_t>(w);
    }
  }

  void commit() {
    ::close(fd_);
    fd_ = -1;
    if (::rename(tmp_.c_str(), path_.c_str()) != 0) {
      int err = errno;
      ::unlink(tmp_.c_str());
      throw FileIoError("rename " + path_ + ": " + std::strerror(err));
    }
  }

 private:
  int open_with_dirs(int flags) {
    int fd = ::open(tmp_.c_str(), flags, 0644);
    if (fd < 0 && errno == ENOENT) {
      make_parent_dirs(tmp_);
      fd = ::open(tmp_.c_str(), flags, 0644);
    }
    return fd;
  }

  std::string path_;
  std::string tmp_;
  int fd_ = -1;
  bool direct_ = false;
};

// RAII reader for chunked loads: one open per transfer.
class FileReader {
 public:
  explicit FileReader(const std::string& path, bool direct_io = false)
      : path_(path) {
    if (direct_io) {
      fd_ = ::open(path.c_str(), O_RDONLY | O_DIRECT);
      direct_ = fd_ >= 0;
    }
    if (fd_ < 0) fd_ = ::open(path.c_str(), O_RDONLY);
    if (fd_ < 0)
      throw FileIoError("open " + path + ": " + std::strerror(errno));
  }
  ~FileReader() {
    if (fd_ >= 0) ::close(fd_);
  }
  FileReader(const FileReader&) = delete;

  void read_at(uint64_t offset, uint8_t* buf, size_t len) {
    if (direct_ && !direct_io_eligible(buf, offset, len)) {
      ::fcntl(fd_, F_SETFL, ::fcntl(fd_, F_GETFL) & ~O_DIRECT);
      direct_ = false;
    }
    size_t got = 0;
    while (got < len) {
      ssize_t r = ::pread(fd_, buf + got, len - got,
                          static_cast<off_t>(offset + got));
      if (r < 0) {
        if (errno == EINTR) continue;
        if (direct_ && errno == EINVAL) {
          ::fcntl(fd_, F_SETFL, ::fcntl(fd_, F_GETFL) & ~O_DIRECT);
          direct_ = false;
          continue;
        }
        throw FileIoError("pread " + path_ + ": " + std::strerror(errno));
      }
      if (r == 0) throw FileIoError("short read from " + path_);
      got += static_cast<size_t>(r);
    }
  }

 private:
  std::string path_;
  int fd_ = -1;
  bool direct_ = false;
};

// Read [offset, offset+len) into buf; the file may be a head-partial
// (shorter than the nominal full span) — callers validate coverage first.
inline void read_file_range(const std::string& path, uint64_t offset,
                            uint8_t* buf, size_t len) {
  int fd = ::open(path.c_str(), O_RDONLY);
  if (fd < 0)
    throw FileIoError("open " + path + ": " + std::strerror(errno));
  size_t got = 0;
  while (got < len) {
    ssize_t r = ::pread(fd, buf + got, len - got, static_cast<off_t>(offset + got));
    if (r < 0) {
      if (errno == EINTR) continue;
      int err = errno;
      ::close(fd);
      throw FileIoError("pread " + path + ": " + std::strerror(err));
    }
    if (r == 0) {
      ::close(fd);
      throw FileIoError("short read from " + path);
    }
    got += static_cast<size_t>(r);
  }
  ::close(fd);
}

}  // namespace kvo
