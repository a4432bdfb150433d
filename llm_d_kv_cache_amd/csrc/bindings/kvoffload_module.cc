// _kvoffload: pybind11 bindings for the GPU data plane (HIP/CDNA4).
//
// Tensors cross this boundary as raw (pointer, stride, bytes) descriptors —
// the extension has no libtorch dependency; Python computes them via
// torch.Tensor.data_ptr()/stride(). Streams are passed as the integer
// handle of torch.cuda.current_stream().
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "../offload/block_copier.h"
#include "../offload/engine.h"

namespace py = pybind11;
using namespace kvo;

extern "C" hipError_t kvc_launch_prefix_hash(const uint32_t*, const uint64_t*,
                                             const uint64_t*, uint64_t*,
                                             const uint64_t*, int, int,
                                             hipStream_t);

PYBIND11_MODULE(_kvoffload, m) {
  m.doc() = "llm-d-kv-cache-amd data plane (HIP gfx950)";

  py::class_<EngineStats>(m, "EngineStats")
      .def_readonly("stores_submitted", &EngineStats::stores_submitted)
      .def_readonly("loads_submitted", &EngineStats::loads_submitted)
      .def_readonly("files_written", &EngineStats::files_written)
      .def_readonly("files_deduped", &EngineStats::files_deduped)
      .def_readonly("files_read", &EngineStats::files_read)
      .def_readonly("writes_dropped", &EngineStats::writes_dropped)
      .def_readonly("tasks_cancelled", &EngineStats::tasks_cancelled)
      .def_readonly("host_cache_hits", &EngineStats::host_cache_hits)
      .def_readonly("host_cache_stores", &EngineStats::host_cache_stores)
      .def_readonly("writeback_flushes", &EngineStats::writeback_flushes)
      .def_readonly("errors", &EngineStats::errors)
      .def_readonly("avg_write_seconds", &EngineStats::avg_write_seconds)
      .def_readonly("bytes_stored", &EngineStats::bytes_stored)
      .def_readonly("bytes_loaded", &EngineStats::bytes_loaded)
      .def_readonly("t_gather_ms", &EngineStats::t_gather_ms)
      .def_readonly("t_d2h_ms", &EngineStats::t_d2h_ms)
      .def_readonly("t_write_ms", &EngineStats::t_write_ms)
      .def_readonly("t_read_ms", &EngineStats::t_read_ms)
      .def_readonly("t_h2d_ms", &EngineStats::t_h2d_ms)
      .def_readonly("t_scatter_ms", &EngineStats::t_scatter_ms)
      .def_readonly("d2h_lane_busy_ms", &EngineStats::d2h_lane_busy_ms)
      .def_readonly("h2d_lane_busy_ms", &EngineStats::h2d_lane_busy_ms)
      .def_readonly("d2h_lane_bytes", &EngineStats::d2h_lane_bytes)
      .def_readonly("h2d_lane_bytes", &EngineStats::h2d_lane_bytes);

  py::class_<StorageOffloadEngine>(m, "StorageOffloadEngine")
      .def(py::init([](std::vector<std::tuple<std::vector<uintptr_t>,
                                              std::vector<uint64_t>, uint64_t,
                                              int64_t>>
                           groups,
                       int io_threads, int gpu_blocks_per_file,
                       double read_preferring_ratio, double max_write_queued_seconds,
                       bool gpu_mode, int device, const std::string& copy_path,
                       const std::string& serialize, size_t host_cache_bytes,
                       const std::string& write_policy, bool direct_io) {
             EngineConfig cfg;
             cfg.io_threads = io_threads;
             cfg.gpu_blocks_per_file = gpu_blocks_per_file;
             cfg.read_preferring_ratio = read_preferring_ratio;
             cfg.max_write_queued_seconds = max_write_queued_seconds;
             cfg.gpu_mode = gpu_mode;
             cfg.device = device;
             if (copy_path == "staged")
               cfg.copy_path = CopyPath::kStaged;
             else if (copy_path == "zero_copy")
               cfg.copy_path = CopyPath::kZeroCopy;
             else if (copy_path == "host")
               cfg.copy_path = CopyPath::kHostMemcpy;
             else
               throw std::invalid_argument(
                   "copy_path must be staged|zero_copy|host");
             if (serialize == "raw")
               cfg.serialize = Serialize::kRaw;
             else if (serialize == "fp8_e4m3")
               cfg.serialize = Serialize::kFp8E4M3;
             else
               throw std::invalid_argument("serialize must be raw|fp8_e4m3");
             cfg.host_cache_bytes = host_cache_bytes;
             if (write_policy == "through")
               cfg.write_policy = WritePolicy::kThrough;
             else if (write_policy == "back")
               cfg.write_policy = WritePolicy::kBack;
             else
               throw std::invalid_argument("write_policy must be through|back");
             cfg.direct_io = direct_io;
             std::vector<GroupDesc> gs;
             for (auto& [ptrs, strides, block_bytes, num_blocks] : groups) {
               GroupDesc g;
               for (auto p : ptrs) g.layer_ptrs.push_back(reinterpret_cast<void*>(p));
               g.layer_strides = strides;
               g.block_bytes = block_bytes;
               g.num_blocks = num_blocks;
               gs.push_back(std::move(g));
             }
             py::gil_scoped_release rel;
             return std::make_unique<StorageOffloadEngine>(cfg, std::move(gs));
           }),
           py::arg("groups"), py::arg("io_threads") = 16,
           py::arg("gpu_blocks_per_file") = 16,
           py::arg("read_preferring_ratio") = 0.75,
           py::arg("max_write_queued_seconds") = 30.0, py::arg("gpu_mode") = false,
           py::arg("device") = 0, py::arg("copy_path") = "staged",
           py::arg("serialize") = "raw", py::arg("host_cache_bytes") = 0,
           py::arg("write_policy") = "through", py::arg("direct_io") = false)
      .def(
          "async_store",
          [](StorageOffloadEngine& e,
             std::vector<std::tuple<int, std::string, std::vector<int32_t>, int>>
                 files,
             uintptr_t caller_stream) {
            std::vector<FileTransfer> fts;
            for (auto& [group, path, ids, off] : files) {
              FileTransfer ft;
              ft.group = group;
              ft.path = std::move(path);
              ft.block_ids = std::move(ids);
              ft.slot_offset = off;
              fts.push_back(std::move(ft));
            }
            py::gil_scoped_release rel;
            return e.async_store(std::move(fts), caller_stream);
          },
          py::arg("files"), py::arg("caller_stream") = 0)
      .def(
          "async_load",
          [](StorageOffloadEngine& e,
             std::vector<std::tuple<int, std::string, std::vector<int32_t>, int>>
                 files) {
            std::vector<FileTransfer> fts;
            for (auto& [group, path, ids, off] : files) {
              FileTransfer ft;
              ft.group = group;
              ft.path = std::move(path);
              ft.block_ids = std::move(ids);
              ft.slot_offset = off;
              fts.push_back(std::move(ft));
            }
            py::gil_scoped_release rel;
            return e.async_load(std::move(fts));
          },
          py::arg("files"))
      .def("get_finished",
           [](StorageOffloadEngine& e) {
             py::gil_scoped_release rel;
             auto fin = e.get_finished();
             py::gil_scoped_acquire acq;
             py::list out;
             for (auto& f : fin)
               out.append(py::make_tuple(f.id, f.success, f.dropped));
             return out;
           })
      .def("wait_job", &StorageOffloadEngine::wait_job,
           py::call_guard<py::gil_scoped_release>(), py::arg("job_id"))
      .def("cancel_job", &StorageOffloadEngine::cancel_job, py::arg("job_id"))
      .def(
          "host_cache_read",
          [](kvo::StorageOffloadEngine& e, const std::string& path,
             uintptr_t dst, size_t cap) {
            py::gil_scoped_release rel;
            return e.host_cache_read(path, reinterpret_cast<uint8_t*>(dst),
                                     cap);
          },
          py::arg("path"), py::arg("dst"), py::arg("cap"))
      .def("stats", &StorageOffloadEngine::stats,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("pending_writes", &StorageOffloadEngine::pending_writes);

  py::class_<BlockCopier>(m, "BlockCopier")
      .def(py::init([](std::vector<py::tuple> groups, bool gpu_mode,
                       int device) {
             std::vector<GroupDesc> gs;
             for (auto& t : groups) {
               GroupDesc g;
               for (auto p : t[0].cast<std::vector<uintptr_t>>())
                 g.layer_ptrs.push_back(reinterpret_cast<void*>(p));
               g.layer_strides = t[1].cast<std::vector<uint64_t>>();
               g.block_bytes = t[2].cast<uint64_t>();
               if (t.size() > 3) g.num_blocks = t[3].cast<int64_t>();
               gs.push_back(std::move(g));
             }
             py::gil_scoped_release rel;
             return std::make_unique<BlockCopier>(std::move(gs), gpu_mode, device);
           }),
           py::arg("groups"), py::arg("gpu_mode") = false, py::arg("device") = 0)
      .def("packed_bytes", &BlockCopier::packed_bytes, py::arg("group"),
           py::arg("n_blocks"))
      .def("packed_bytes_fp8", &BlockCopier::packed_bytes_fp8, py::arg("group"),
           py::arg("n_blocks"))
      .def("fp8_scratch_bytes", &BlockCopier::fp8_scratch_bytes,
           py::arg("group"), py::arg("n_blocks"))
      .def(
          "gather_fp8",
          [](BlockCopier& c, int group, std::vector<int32_t> ids, uintptr_t dst,
             uintptr_t scratch, uintptr_t stream) {
            py::gil_scoped_release rel;
            c.gather_fp8(group, ids, reinterpret_cast<void*>(dst),
                         reinterpret_cast<void*>(scratch), stream);
          },
          py::arg("group"), py::arg("block_ids"), py::arg("dst"),
          py::arg("scratch") = 0, py::arg("stream") = 0)
      .def(
          "scatter_fp8",
          [](BlockCopier& c, int group, std::vector<int32_t> ids, uintptr_t src,
             uintptr_t stream) {
            py::gil_scoped_release rel;
            c.scatter_fp8(group, ids, reinterpret_cast<const void*>(src),
                          stream);
          },
          py::arg("group"), py::arg("block_ids"), py::arg("src"),
          py::arg("stream") = 0)
      .def(
          "gather",
          [](BlockCopier& c, int group, std::vector<int32_t> ids, uintptr_t dst,
             uintptr_t stream) {
            py::gil_scoped_release rel;
            c.gather(group, ids, reinterpret_cast<void*>(dst), stream);
          },
          py::arg("group"), py::arg("block_ids"), py::arg("dst"),
          py::arg("stream") = 0)
      .def(
          "scatter",
          [](BlockCopier& c, int group, std::vector<int32_t> ids, uintptr_t src,
             uintptr_t stream) {
            py::gil_scoped_release rel;
            c.scatter(group, ids, reinterpret_cast<const void*>(src), stream);
          },
          py::arg("group"), py::arg("block_ids"), py::arg("src"),
          py::arg("stream") = 0);

  m.def(
      "prefix_hash",
      [](uintptr_t tokens, uintptr_t seq_off, uintptr_t seeds, uintptr_t keys,
         uintptr_t key_off, int block_size, int n_seq, uintptr_t stream) {
        py::gil_scoped_release rel;
        hipError_t err = kvc_launch_prefix_hash(
            reinterpret_cast<const uint32_t*>(tokens),
            reinterpret_cast<const uint64_t*>(seq_off),
            reinterpret_cast<const uint64_t*>(seeds),
            reinterpret_cast<uint64_t*>(keys),
            reinterpret_cast<const uint64_t*>(key_off), block_size, n_seq,
            reinterpret_cast<hipStream_t>(stream));
        if (err != hipSuccess) throw HipError(hipGetErrorString(err));
      },
      py::arg("tokens"), py::arg("seq_off"), py::arg("seeds"), py::arg("keys"),
      py::arg("key_off"), py::arg("block_size"), py::arg("n_seq"),
      py::arg("stream") = 0,
      "Batched chained block hashing on device buffers (one lane per sequence)");
}
