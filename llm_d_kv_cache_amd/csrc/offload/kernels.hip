// CDNA4 (gfx950) kernels for the KV-block data plane.
//
// Replaces the reference's byte-wise copy_blocks_kernel + per-layer launch
// loop (csrc/storage/tensor_copier_kernels.cu:54-151) with an MI355X-first
// design:
//   - one launch covers ALL (block, layer) tiles of a file transfer;
//   - 16 B/lane vectorized copies (dwordx4), wave64-shaped, grid-stride
//     within each tile so the chip is filled regardless of tile count;
//   - gather destination is contiguous staging, so the PCIe hop can run on
//     the SDMA engines (hipMemcpyAsync) with zero CU occupancy, or the
//     kernel can write device-mapped pinned host memory directly
//     (zero-copy mode) — the engine chooses per config.
//
// A batched prefix-hash kernel (FNV-64a over canonical CBOR, one lane per
// sequence) accelerates bulk block-key computation for event floods.
#include <hip/hip_runtime.h>

#include <cstdint>

namespace kvo {

constexpr int kMaxBlocksPerFile = 64;

struct BlockList {
  int32_t ids[kMaxBlocksPerFile];
};

// ---- gather / scatter -------------------------------------------------------
// Tile (bi, l): block_bytes contiguous bytes.
//   device side: layer_ptrs[l] + ids[bi] * layer_strides[l]
//   staging side: dst + (bi * num_layers + l) * block_bytes
// Grid: x = workgroups per tile (grid-stride inside the tile),
//       y = tile index. 16-byte vectors; block_bytes % 16 == 0 (host-checked).

__global__ __launch_bounds__(256) void kvc_gather_blocks(
    const void* const* __restrict__ layer_ptrs,
    const uint64_t* __restrict__ layer_strides, int num_layers,
    uint64_t block_bytes, BlockList blocks, uint8_t* __restrict__ dst) {
  const uint32_t tile = blockIdx.y;
  const int l = tile % num_layers;
  const int bi = tile / num_layers;
  const uint4* __restrict__ src = reinterpret_cast<const uint4*>(
      static_cast<const uint8_t*>(layer_ptrs[l]) +
      static_cast<uint64_t>(blocks.ids[bi]) * layer_strides[l]);
  uint4* __restrict__ out =
      reinterpret_cast<uint4*>(dst + static_cast<uint64_t>(tile) * block_bytes);
  const uint64_t nvec = block_bytes / 16;
  const uint64_t stride = static_cast<uint64_t>(gridDim.x) * blockDim.x;
  for (uint64_t v = static_cast<uint64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       v < nvec; v += stride)
    out[v] = src[v];
}

__global__ __launch_bounds__(256) void kvc_scatter_blocks(
    const void* const* __restrict__ layer_ptrs,
    const uint64_t* __restrict__ layer_strides, int num_layers,
    uint64_t block_bytes, BlockList blocks, const uint8_t* __restrict__ src) {
  const uint32_t tile = blockIdx.y;
  const int l = tile % num_layers;
  const int bi = tile / num_layers;
  uint4* __restrict__ out = reinterpret_cast<uint4*>(
      static_cast<uint8_t*>(const_cast<void*>(layer_ptrs[l])) +
      static_cast<uint64_t>(blocks.ids[bi]) * layer_strides[l]);
  const uint4* __restrict__ in = reinterpret_cast<const uint4*>(
      src + static_cast<uint64_t>(tile) * block_bytes);
  const uint64_t nvec = block_bytes / 16;
  const uint64_t stride = static_cast<uint64_t>(gridDim.x) * blockDim.x;
  for (uint64_t v = static_cast<uint64_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       v < nvec; v += stride)
    out[v] = in[v];
}

// ---- batched prefix hashing -------------------------------------------------
// One lane per sequence: walks the chain hash_{c+1} = FNV64a(CBOR([h_c,
// chunk, null])). Sequences share a flat token buffer with offsets.
// Text extras (multimodal) stay on the CPU path.

__device__ __forceinline__ uint64_t fnv_byte(uint64_t h, uint8_t b) {
  return (h ^ b) * 0x100000001b3ull;
}

__device__ __forceinline__ uint64_t fnv_uint_cbor(uint64_t h, uint64_t v,
                                                  uint8_t major) {
  const uint8_t m = major << 5;
  if (v < 24) return fnv_byte(h, m | static_cast<uint8_t>(v));
  if (v <= 0xff) return fnv_byte(fnv_byte(h, m | 24), static_cast<uint8_t>(v));
  if (v <= 0xffff) {
    h = fnv_byte(h, m | 25);
    h = fnv_byte(h, static_cast<uint8_t>(v >> 8));
    return fnv_byte(h, static_cast<uint8_t>(v));
  }
  if (v <= 0xffffffffull) {
    h = fnv_byte(h, m | 26);
    for (int s = 24; s >= 0; s -= 8) h = fnv_byte(h, static_cast<uint8_t>(v >> s));
    return h;
  }
  h = fnv_byte(h, m | 27);
  for (int s = 56; s >= 0; s -= 8) h = fnv_byte(h, static_cast<uint8_t>(v >> s));
  return h;
}

__global__ void kvc_prefix_hash(
    const uint32_t* __restrict__ tokens,   // flat token buffer
    const uint64_t* __restrict__ seq_off,  // [n_seq + 1] offsets into tokens
    const uint64_t* __restrict__ seeds,    // [n_seq] chain seeds (parent or model seed)
    uint64_t* __restrict__ keys,           // flat key buffer
    const uint64_t* __restrict__ key_off,  // [n_seq + 1] offsets into keys
    int block_size, int n_seq) {
  int s = blockIdx.x * blockDim.x + threadIdx.x;
  if (s >= n_seq) return;
  const uint32_t* t = tokens + seq_off[s];
  const uint64_t n_tokens = seq_off[s + 1] - seq_off[s];
  const uint64_t n_chunks = n_tokens / block_size;
  uint64_t* out = keys + key_off[s];
  uint64_t prefix = seeds[s];
  for (uint64_t c = 0; c < n_chunks; ++c) {
    uint64_t h = 0xcbf29ce484222325ull;
    h = fnv_byte(h, 0x83);                       // array(3)
    h = fnv_uint_cbor(h, prefix, 0);             // parent
    h = fnv_uint_cbor(h, block_size, 4);         // array(block_size)
    for (int i = 0; i < block_size; ++i)
      h = fnv_uint_cbor(h, t[c * block_size + i], 0);
    h = fnv_byte(h, 0xf6);                       // null extra
    prefix = h;
    out[c] = h;
  }
}

// ---- launchers --------------------------------------------------------------

inline dim3 copy_grid(uint32_t tiles, uint64_t block_bytes) {
  // Fill the chip: >= 2048 workgroups when the transfer is large enough.
  uint64_t vec_per_tile = block_bytes / 16;
  uint32_t max_wg_per_tile =
      static_cast<uint32_t>((vec_per_tile + 255) / 256);  // cap: 1 vec/lane
  uint32_t want = tiles >= 2048 ? 1 : (2048 + tiles - 1) / tiles;
  uint32_t wg_per_tile = want < max_wg_per_tile ? want : max_wg_per_tile;
  if (wg_per_tile == 0) wg_per_tile = 1;
  return dim3(wg_per_tile, tiles);
}

extern "C" hipError_t kvc_launch_gather(
    const void* const* layer_ptrs_dev, const uint64_t* layer_strides_dev,
    int num_layers, uint64_t block_bytes, const int32_t* block_ids,
    int num_blocks, uint8_t* dst, hipStream_t stream) {
  BlockList bl;
  for (int i = 0; i < num_blocks; ++i) bl.ids[i] = block_ids[i];
  dim3 grid = copy_grid(static_cast<uint32_t>(num_blocks) * num_layers, block_bytes);
  hipLaunchKernelGGL(kvc_gather_blocks, grid, dim3(256), 0, stream,
                     layer_ptrs_dev, layer_strides_dev, num_layers, block_bytes,
                     bl, dst);
  return hipGetLastError();
}

extern "C" hipError_t kvc_launch_scatter(
    const void* const* layer_ptrs_dev, const uint64_t* layer_strides_dev,
    int num_layers, uint64_t block_bytes, const int32_t* block_ids,
    int num_blocks, const uint8_t* src, hipStream_t stream) {
  BlockList bl;
  for (int i = 0; i < num_blocks; ++i) bl.ids[i] = block_ids[i];
  dim3 grid = copy_grid(static_cast<uint32_t>(num_blocks) * num_layers, block_bytes);
  hipLaunchKernelGGL(kvc_scatter_blocks, grid, dim3(256), 0, stream,
                     layer_ptrs_dev, layer_strides_dev, num_layers, block_bytes,
                     bl, src);
  return hipGetLastError();
}

extern "C" hipError_t kvc_launch_prefix_hash(
    const uint32_t* tokens, const uint64_t* seq_off, const uint64_t* seeds,
    uint64_t* keys, const uint64_t* key_off, int block_size, int n_seq,
    hipStream_t stream) {
  int threads = 256;
  int blocks = (n_seq + threads - 1) / threads;
  hipLaunchKernelGGL(kvc_prefix_hash, dim3(blocks), dim3(threads), 0, stream,
                     tokens, seq_off, seeds, keys, key_off, block_size, n_seq);
  return hipGetLastError();
}

}  // namespace kvo
