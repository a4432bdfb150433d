// BlockCopier: standalone gather/scatter between paged KV tensors and a
// contiguous buffer — the packing primitive of the xGMI peer-migration
// path (peer rank gathers blocks into a send slab, RCCL ships it, the
// puller scatters into its own pages). Same kernels as the offload engine;
// host mode memcpys for CPU-only tests.
#pragma once

#include "common.h"
#include "engine.h"  // GroupDesc, kernel launcher decls

namespace kvo {

class BlockCopier {
 public:
  BlockCopier(std::vector<GroupDesc> groups, bool gpu_mode, int device)
      : groups_(std::move(groups)), gpu_mode_(gpu_mode) {
    if (gpu_mode_) {
      KVO_HIP_CHECK(hipSetDevice(device));
      for (auto& g : groups_) {
        void* dp = nullptr;
        KVO_HIP_CHECK(hipMalloc(&dp, g.layer_ptrs.size() * sizeof(void*)));
        KVO_HIP_CHECK(hipMemcpy(dp, g.layer_ptrs.data(),
                                g.layer_ptrs.size() * sizeof(void*),
                                hipMemcpyHostToDevice));
        dev_layer_ptrs_.push_back(static_cast<void**>(dp));
        void* ds = nullptr;
        KVO_HIP_CHECK(hipMalloc(&ds, g.layer_strides.size() * sizeof(uint64_t)));
        KVO_HIP_CHECK(hipMemcpy(ds, g.layer_strides.data(),
                                g.layer_strides.size() * sizeof(uint64_t),
                                hipMemcpyHostToDevice));
        dev_layer_strides_.push_back(static_cast<uint64_t*>(ds));
      }
    }
  }

  ~BlockCopier() {
    for (auto p : dev_layer_ptrs_) (void)hipFree(p);
    for (auto p : dev_layer_strides_) (void)hipFree(p);
  }

  size_t packed_bytes(int group, size_t n_blocks) const {
    const GroupDesc& g = groups_.at(group);
    return n_blocks * g.layer_ptrs.size() * g.block_bytes;
  }

  // fp8 e4m3 record size: block_bytes/2 payload + f32 scale per tile.
  size_t packed_bytes_fp8(int group, size_t n_blocks) const {
    const GroupDesc& g = groups_.at(group);
    return n_blocks * g.layer_ptrs.size() * (g.block_bytes / 2 + 4);
  }

  // Scratch floats the fp8 gather needs beyond the packed output
  // (partial-max slab; copy_grid caps slices at 2048/tiles + 1).
  size_t fp8_scratch_bytes(int group, size_t n_blocks) const {
    const GroupDesc& g = groups_.at(group);
    size_t tiles = n_blocks * g.layer_ptrs.size();
    return (tiles + 4096) * sizeof(float);
  }

  // Gather + quantize into dst (device ptr, GPU mode); scratch must hold
  // fp8_scratch_bytes(). Host mode uses the software codec.
  void gather_fp8(int group, const std::vector<int32_t>& ids, void* dst,
                  void* scratch, uintptr_t stream) {
    const GroupDesc& g = groups_.at(group);
    check_ids(ids, g);
    if (!gpu_mode_) {
      host_fp8(g, ids, static_cast<uint8_t*>(dst), /*quantize=*/true);
      return;
    }
    hipError_t err = kvc_launch_gather_fp8_split(
        const_cast<const void* const*>(dev_layer_ptrs_[group]),
        dev_layer_strides_[group], static_cast<int>(g.layer_ptrs.size()),
        g.block_bytes, ids.data(), static_cast<int>(ids.size()), nullptr,
        static_cast<uint8_t*>(dst), static_cast<float*>(scratch),
        reinterpret_cast<hipStream_t>(stream));
    if (err != hipSuccess) throw HipError(hipGetErrorString(err));
  }

  // Dequantize + scatter from an fp8 record slab.
  void scatter_fp8(int group, const std::vector<int32_t>& ids, const void* src,
                   uintptr_t stream) {
    const GroupDesc& g = groups_.at(group);
    check_ids(ids, g);
    if (!gpu_mode_) {
      host_fp8(g, ids,
               const_cast<uint8_t*>(static_cast<const uint8_t*>(src)),
               /*quantize=*/false);
      return;
    }
    hipError_t err = kvc_launch_scatter_fp8(
        const_cast<const void* const*>(dev_layer_ptrs_[group]),
        dev_layer_strides_[group], static_cast<int>(g.layer_ptrs.size()),
        g.block_bytes, ids.data(), static_cast<int>(ids.size()), nullptr,
        static_cast<const uint8_t*>(src), reinterpret_cast<hipStream_t>(stream));
    if (err != hipSuccess) throw HipError(hipGetErrorString(err));
  }

  // Gather block_ids of `group` into contiguous dst (device ptr in GPU
  // mode). Async on `stream`; caller synchronizes.
  void gather(int group, const std::vector<int32_t>& ids, void* dst,
              uintptr_t stream) {
    const GroupDesc& g = groups_.at(group);
    check_ids(ids, g);
    if (!gpu_mode_) {
      host_copy(g, ids, static_cast<uint8_t*>(dst), /*to_packed=*/true);
      return;
    }
    hipError_t err = kvc_launch_gather(
        const_cast<const void* const*>(dev_layer_ptrs_[group]),
        dev_layer_strides_[group], static_cast<int>(g.layer_ptrs.size()),
        g.block_bytes, ids.data(), static_cast<int>(ids.size()), nullptr,
        static_cast<uint8_t*>(dst), reinterpret_cast<hipStream_t>(stream));
    if (err != hipSuccess) throw HipError(hipGetErrorString(err));
  }

  void scatter(int group, const std::vector<int32_t>& ids, const void* src,
               uintptr_t stream) {
    const GroupDesc& g = groups_.at(group);
    check_ids(ids, g);
    if (!gpu_mode_) {
      host_copy(g, ids, const_cast<uint8_t*>(static_cast<const uint8_t*>(src)),
                /*to_packed=*/false);
      return;
    }
    hipError_t err = kvc_launch_scatter(
        const_cast<const void* const*>(dev_layer_ptrs_[group]),
        dev_layer_strides_[group], static_cast<int>(g.layer_ptrs.size()),
        g.block_bytes, ids.data(), static_cast<int>(ids.size()), nullptr,
        static_cast<const uint8_t*>(src), reinterpret_cast<hipStream_t>(stream));
    if (err != hipSuccess) throw HipError(hipGetErrorString(err));
  }

 private:
  static void check_ids(const std::vector<int32_t>& ids, const GroupDesc& g) {
    if (ids.empty() || ids.size() > kMaxBlocksPerFileHost)
      throw std::invalid_argument("block count must be in [1, 128]");
    if (g.num_blocks > 0) {
      for (int32_t id : ids)
        if (id < 0 || id >= g.num_blocks)
          throw std::invalid_argument("block id out of range");
    }
  }

  void host_copy(const GroupDesc& g, const std::vector<int32_t>& ids,
                 uint8_t* packed, bool to_packed) const {
    const size_t nl = g.layer_ptrs.size();
    for (size_t bi = 0; bi < ids.size(); ++bi) {
      for (size_t l = 0; l < nl; ++l) {
        uint8_t* page = static_cast<uint8_t*>(g.layer_ptrs[l]) +
                        static_cast<uint64_t>(ids[bi]) * g.layer_strides[l];
        uint8_t* slab = packed + (bi * nl + l) * g.block_bytes;
        if (to_packed)
          std::memcpy(slab, page, g.block_bytes);
        else
          std::memcpy(page, slab, g.block_bytes);
      }
    }
  }

  void host_fp8(const GroupDesc& g, const std::vector<int32_t>& ids,
                uint8_t* packed, bool quantize) const {
    // software codec shared with the engine's host mode
    StorageOffloadEngine::host_fp8_copy(g, ids, packed, quantize);
  }

  std::vector<GroupDesc> groups_;
  bool gpu_mode_;
  std::vector<void**> dev_layer_ptrs_;
  std::vector<uint64_t*> dev_layer_strides_;
};

}  // namespace kvo
